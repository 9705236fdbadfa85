// Shared pybind helpers: GIL-safe py::object holders, FutureState, Queue.
//
// Capability parity with the reference's GIL machinery (src/pyutil.h
// GilWrapper/glock and moolib.cc FutureWrapper/QueueWrapper). Our rules:
//  - a C++ lock is never held while acquiring the GIL, and a binding entry
//    point always releases the GIL before blocking on a C++ lock or cv;
//  - py::objects captured in std::functions are wrapped in PyGuard so their
//    destruction from a non-Python thread acquires the GIL first.
#pragma once

#include <torch/extension.h>

#include <condition_variable>
#include <deque>
#include <memory>
#include <mutex>
#include <optional>

#include "common.h"

namespace mrl {

namespace py = pybind11;

// True while the interpreter can safely be entered from a non-Python thread.
// During finalization, acquiring the GIL from a C++ thread aborts the
// process, so every completion/cleanup path checks this first.
inline bool pyAlive() { return Py_IsInitialized() && !_Py_IsFinalizing(); }

struct PyGuard {
  py::object obj;
  explicit PyGuard(py::object o) : obj(std::move(o)) {}
  PyGuard(const PyGuard&) = delete;
  ~PyGuard() {
    if (obj.ptr() != nullptr) {
      if (pyAlive()) {
        py::gil_scoped_acquire gil;
        obj = py::object();
      } else {
        (void)obj.release();  // interpreter gone: leak instead of crash
      }
    }
  }
};
using PyGuardPtr = std::shared_ptr<PyGuard>;

// The registered RpcError exception type (set in pymodule.cc init).
py::object rpcErrorType();

struct FutureState {
  std::mutex mu;
  std::condition_variable cv;
  bool done = false;
  bool cancelled = false;
  bool hasError = false;
  std::string error;
  py::object value;                    // touched under GIL only
  std::vector<py::object> callbacks;   // touched under GIL only

  ~FutureState() {
    if ((value.ptr() || !callbacks.empty())) {
      if (pyAlive()) {
        py::gil_scoped_acquire gil;
        value = py::object();
        callbacks.clear();
      } else {
        (void)value.release();
        for (auto& c : callbacks) (void)c.release();
      }
    }
  }
};
using FutureStatePtr = std::shared_ptr<FutureState>;

// GIL must be held.
inline void futureComplete(const FutureStatePtr& st, py::object v) {
  std::vector<py::object> cbs;
  {
    std::lock_guard<std::mutex> lk(st->mu);
    if (st->done) return;
    st->value = std::move(v);
    st->done = true;
    cbs.swap(st->callbacks);
  }
  st->cv.notify_all();
  for (auto& cb : cbs) {
    try {
      cb();
    } catch (py::error_already_set& e) {
      e.discard_as_unraisable("moolib_amd future callback");
    }
  }
}

// GIL must be held.
inline void futureFail(const FutureStatePtr& st, const std::string& err) {
  std::vector<py::object> cbs;
  {
    std::lock_guard<std::mutex> lk(st->mu);
    if (st->done) return;
    st->hasError = true;
    st->error = err;
    st->done = true;
    cbs.swap(st->callbacks);
  }
  st->cv.notify_all();
  for (auto& cb : cbs) {
    try {
      cb();
    } catch (py::error_already_set& e) {
      e.discard_as_unraisable("moolib_amd future callback");
    }
  }
}

// Bound as both Future and AllReduce (same surface as the reference's
// FutureWrapper / AllReduceWrapper, moolib.cc:316-392, 2266-2284).
class PyFuture {
 public:
  PyFuture() : st_(std::make_shared<FutureState>()) {}
  explicit PyFuture(FutureStatePtr st) : st_(std::move(st)) {}
  FutureStatePtr state() const { return st_; }

  void waitNoTimeout() {
    py::gil_scoped_release rel;
    std::unique_lock<std::mutex> lk(st_->mu);
    st_->cv.wait(lk, [&] { return st_->done; });
  }

  void waitTimeout(double timeout) {
    py::gil_scoped_release rel;
    std::unique_lock<std::mutex> lk(st_->mu);
    if (!st_->cv.wait_for(lk, std::chrono::duration<double>(timeout), [&] { return st_->done; })) {
      throw RpcError("Future result wait timed out");
    }
  }

  py::object resultNoTimeout() {
    waitNoTimeout();
    return takeResult();
  }

  py::object resultTimeout(double timeout) {
    waitTimeout(timeout);
    return takeResult();
  }

  bool done() {
    std::lock_guard<std::mutex> lk(st_->mu);
    return st_->done;
  }

  void cancel() {
    std::vector<py::object> cbs;
    {
      std::lock_guard<std::mutex> lk(st_->mu);
      if (st_->done) return;
      st_->done = true;
      st_->cancelled = true;
      cbs.swap(st_->callbacks);
    }
    st_->cv.notify_all();
    for (auto& cb : cbs) {
      try {
        cb();
      } catch (py::error_already_set& e) {
        e.discard_as_unraisable("moolib_amd future callback");
      }
    }
  }

  py::object exception() {
    std::lock_guard<std::mutex> lk(st_->mu);
    if (!st_->done || (!st_->hasError && !st_->cancelled)) return py::none();
    return rpcErrorType()(st_->cancelled ? "cancelled" : st_->error);
  }

  void addDoneCallback(py::object cb) {
    bool callNow = false;
    {
      std::lock_guard<std::mutex> lk(st_->mu);
      if (st_->done) {
        callNow = true;
      } else {
        st_->callbacks.push_back(cb);
      }
    }
    if (callNow) cb();
  }

 private:
  py::object takeResult() {
    std::lock_guard<std::mutex> lk(st_->mu);
    if (st_->cancelled) throw RpcError("Future was cancelled");
    if (st_->hasError) throw RpcError(st_->error);
    return st_->value;
  }
  FutureStatePtr st_;
};

// Awaitable multi-producer multi-consumer queue (reference: QueueWrapper,
// moolib.cc:433-576).
class PyQueue {
 public:
  struct State {
    std::mutex mu;
    std::deque<py::object> items;          // GIL
    std::deque<FutureStatePtr> waiters;
    ~State() {
      // May be destroyed from a non-Python thread (e.g. a define_queue
      // handler closure cleared during Rpc::shutdown).
      if (!items.empty()) {
        if (pyAlive()) {
          py::gil_scoped_acquire gil;
          items.clear();
        } else {
          for (auto& o : items) (void)o.release();
        }
      }
    }
  };
  PyQueue() : st_(std::make_shared<State>()) {}

  void enqueue(py::object obj) {
    FutureStatePtr waiter;
    {
      std::lock_guard<std::mutex> lk(st_->mu);
      while (!st_->waiters.empty()) {
        auto w = st_->waiters.front();
        st_->waiters.pop_front();
        std::lock_guard<std::mutex> wk(w->mu);
        if (w->done) continue;  // cancelled waiter
        waiter = w;
        break;
      }
      if (!waiter) st_->items.push_back(std::move(obj));
    }
    if (waiter) futureComplete(waiter, std::move(obj));
  }

  PyFuture popFuture() {
    auto fs = std::make_shared<FutureState>();
    std::optional<py::object> item;
    {
      std::lock_guard<std::mutex> lk(st_->mu);
      if (!st_->items.empty()) {
        item = std::move(st_->items.front());
        st_->items.pop_front();
      } else {
        st_->waiters.push_back(fs);
      }
    }
    if (item) futureComplete(fs, std::move(*item));
    return PyFuture(fs);
  }

  size_t size() {
    std::lock_guard<std::mutex> lk(st_->mu);
    return st_->items.size();
  }

 private:
  std::shared_ptr<State> st_;
};

}  // namespace mrl
