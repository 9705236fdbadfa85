// Python bindings for moolib_amd._core.
//
// API parity with the reference module `moolib._C` (src/moolib.cc): Rpc,
// Future, Queue, Broker, Group, AllReduce, RpcDeferredReturn, RpcError,
// create_uid, set_log_level, set_logging, set_max_threads. Batching
// variants of define() are layered in Python (moolib_amd/__init__.py).
#include <torch/extension.h>

#include <atomic>

#include "accumulator.h"
#include "batcher.h"
#include "envpool.h"
#include "pybits.h"
#include "rpc.h"
#include "serde.h"
#include "services.h"

namespace mrl {

namespace {

py::object* g_rpcErrorType = new py::object();  // leaked: see serde.cc note

// Registry of live Rpc instances for atexit cleanup (reference keeps a leaked
// Rpc list and cleans it up at exit, moolib.cc:127-183).
std::mutex g_rpcsMu;
std::vector<std::weak_ptr<Rpc>> g_rpcs;

void registerRpc(const RpcPtr& r) {
  std::lock_guard<std::mutex> lk(g_rpcsMu);
  g_rpcs.push_back(r);
}

void shutdownAll() {
  std::vector<RpcPtr> live;
  {
    std::lock_guard<std::mutex> lk(g_rpcsMu);
    for (auto& w : g_rpcs) {
      if (auto p = w.lock()) live.push_back(p);
    }
    g_rpcs.clear();
  }
  for (auto& p : live) p->shutdown();
  globalScheduler().drain();
}

}  // namespace

py::object rpcErrorType() { return *g_rpcErrorType; }

// ------------------------------------------------------------ deferred

// Callable handed to deferred handlers; calling it sends the response.
class RpcDeferredReturn {
 public:
  RpcDeferredReturn() = default;
  explicit RpcDeferredReturn(RespondFn respond, bool ipcLocal = false)
      : respond_(std::move(respond)), ipcLocal_(ipcLocal) {}
  RpcDeferredReturn(const RpcDeferredReturn&) = delete;
  RpcDeferredReturn& operator=(const RpcDeferredReturn&) = delete;
  RpcDeferredReturn(RpcDeferredReturn&& o) noexcept
      : respond_(std::move(o.respond_)), called_(o.called_), ipcLocal_(o.ipcLocal_) {
    o.respond_ = nullptr;
    o.called_ = true;
  }
  ~RpcDeferredReturn() {
    if (respond_ && !called_) {
      auto r = std::move(respond_);
      r("deferred return dropped without a response", {}, true);
    }
  }
  void call(py::object value) {
    if (!respond_ || called_) throw RpcError("deferred return already used");
    called_ = true;
    std::vector<at::Tensor> tensors;
    std::string payload = serializeObject(value, tensors, ipcLocal_);
    auto r = std::move(respond_);
    py::gil_scoped_release rel;
    r(std::move(payload), std::move(tensors), false);
  }
  void error(const std::string& msg) {
    if (!respond_ || called_) throw RpcError("deferred return already used");
    called_ = true;
    auto r = std::move(respond_);
    py::gil_scoped_release rel;
    r(msg, {}, true);
  }

 private:
  RespondFn respond_;
  bool called_ = false;
  bool ipcLocal_ = false;  // requester is same-machine: CUDA replies go as hipIpc handles
};

// ----------------------------------------------------------------- Rpc

class RpcWrapper {
 public:
  RpcWrapper() {
    rpc_ = Rpc::create();
    registerRpc(rpc_);
  }
  explicit RpcWrapper(RpcPtr rpc) : rpc_(std::move(rpc)) {}
  RpcPtr rpc() const { return rpc_; }

  void setName(const std::string& n) { rpc_->setName(n); }
  std::string getName() { return rpc_->getName(); }
  void setTimeout(double t) { rpc_->setTimeout(t); }
  std::vector<std::string> listen(const std::string& addr) {
    py::gil_scoped_release rel;
    return rpc_->listen(addr);
  }
  void connect(const std::string& addr) {
    py::gil_scoped_release rel;
    rpc_->connect(addr);
  }
  std::string debugInfo() {
    py::gil_scoped_release rel;
    return rpc_->debugInfo();
  }
  std::vector<std::string> localAddrs() {
    py::gil_scoped_release rel;
    return rpc_->localAddrs();
  }
  void setTransports(py::object transports) {
    // Reference semantics (rpc.cc:324-336): named transports stay enabled,
    // the rest are disabled — no default listener, never dialed. Accept the
    // reference's aliases; "shared memory"/"infiniband" map to our unix
    // transport (memfd-backed buffers ride unix sockets; no IB verbs, same
    // as the reference's shipped code).
    bool tcp = false, unixSock = false;
    for (auto t : transports) {
      std::string s = py::cast<std::string>(t);
      if (s == "tcp/ip" || s == "tcp" || s == "uv") {
        tcp = true;
      } else if (s == "shared memory" || s == "unix" || s == "ipc" || s == "infiniband") {
        unixSock = true;
      } else {
        throw RpcError("unknown transport: " + s);
      }
    }
    rpc_->setTransports(tcp, unixSock);
  }

  // Reference ExceptionMode (src/rpc.h:201-205): what a define()d handler's
  // exceptions do. 2 = All (default here: the error text reaches the
  // caller), 1 = DeserializationOnly (argument-decode failures reach the
  // caller; handler exceptions are logged and the caller times out),
  // 0 = None (everything is only logged). The reference's C++ default is
  // DeserializationOnly; ours is All because Python callers expect remote
  // tracebacks — set_exception_mode restores the stricter modes.
  void setExceptionMode(const std::string& mode) {
    if (mode == "none") excMode_->store(0);
    else if (mode == "deserialization_only") excMode_->store(1);
    else if (mode == "all") excMode_->store(2);
    else throw RpcError("exception mode must be none|deserialization_only|all");
  }

  void define(const std::string& name, py::function fn) {
    auto g = std::make_shared<PyGuard>(fn);
    std::weak_ptr<Rpc> wr = rpc_;
    auto mode = excMode_;
    rpc_->define(name, [g, wr, mode](Frame f, const std::string& from, RespondFn respond) {
      if (!pyAlive()) return;
      auto rpc = wr.lock();
      bool local = rpc && rpc->peerIsLocal(from);
      std::string payload;
      std::vector<at::Tensor> tensors;
      bool isErr = false;
      bool suppress = false;
      {
        py::gil_scoped_acquire gil;
        int m = mode->load();
        try {
          auto [args, kwargs] = deserializeCall(f.payload, f.tensors);
          try {
            py::object r = g->obj(*args, **kwargs);
            payload = serializeObject(r, tensors, local);
          } catch (py::error_already_set& e) {
            payload = e.what();
            isErr = true;
            suppress = m < 2;
          } catch (const std::exception& e) {
            payload = e.what();
            isErr = true;
            suppress = m < 2;
          }
        } catch (py::error_already_set& e) {
          payload = e.what();
          isErr = true;
          suppress = m < 1;
        } catch (const std::exception& e) {
          payload = e.what();
          isErr = true;
          suppress = m < 1;
        }
      }
      if (isErr && suppress) {
        MRL_LOG_ERROR("handler error (suppressed by exception mode): %s", payload.c_str());
        return;  // caller times out, reference None/DeserializationOnly behavior
      }
      respond(std::move(payload), std::move(tensors), isErr);
    });
  }

  void defineDeferred(const std::string& name, py::function fn) {
    auto g = std::make_shared<PyGuard>(fn);
    std::weak_ptr<Rpc> wr = rpc_;
    rpc_->define(name, [g, wr](Frame f, const std::string& from, RespondFn respond) {
      if (!pyAlive()) return;
      auto rpc = wr.lock();
      bool local = rpc && rpc->peerIsLocal(from);
      py::gil_scoped_acquire gil;
      try {
        auto [args, kwargs] = deserializeCall(f.payload, f.tensors);
        py::object deferred = py::cast(RpcDeferredReturn(std::move(respond), local));
        g->obj(deferred, *args, **kwargs);
      } catch (py::error_already_set& e) {
        // The deferred return (if not moved from) responds with an error when
        // it is destroyed; report the handler failure too.
        MRL_LOG_ERROR("deferred handler raised: %s", e.what());
      } catch (const std::exception& e) {
        MRL_LOG_ERROR("deferred handler raised: %s", e.what());
      }
    });
  }

  PyQueue defineQueue(const std::string& name) {
    PyQueue q;
    std::weak_ptr<Rpc> wr = rpc_;
    rpc_->define(name, [q, wr](Frame f, const std::string& from, RespondFn respond) mutable {
      if (!pyAlive()) return;
      auto rpc = wr.lock();
      bool local = rpc && rpc->peerIsLocal(from);
      py::gil_scoped_acquire gil;
      try {
        auto [args, kwargs] = deserializeCall(f.payload, f.tensors);
        py::object deferred = py::cast(RpcDeferredReturn(std::move(respond), local));
        q.enqueue(py::make_tuple(deferred, args, kwargs));
      } catch (py::error_already_set& e) {
        MRL_LOG_ERROR("queue handler failed: %s", e.what());
      } catch (const std::exception& e) {
        MRL_LOG_ERROR("queue handler failed: %s", e.what());
      }
    });
    return q;
  }

  void undefine(const std::string& name) { rpc_->undefine(name); }

  PyFuture asyncCall(const std::string& peer, const std::string& func, py::args args,
                     py::kwargs kwargs) {
    std::vector<at::Tensor> tensors;
    std::string payload = serializeCall(args, kwargs, tensors, rpc_->peerIsLocal(peer));
    PyFuture fut;
    auto st = fut.state();
    py::gil_scoped_release rel;
    rpc_->sendRequest(peer, func, std::move(payload), std::move(tensors),
                      [st](Frame* resp, const std::string* err) {
                        if (!pyAlive()) return;
                        py::gil_scoped_acquire gil;
                        if (err) {
                          futureFail(st, *err);
                        } else {
                          try {
                            py::object v = deserializeObject(resp->payload, resp->tensors);
                            futureComplete(st, std::move(v));
                          } catch (py::error_already_set& e) {
                            futureFail(st, std::string("deserialize failed: ") + e.what());
                          } catch (const std::exception& e) {
                            futureFail(st, std::string("deserialize failed: ") + e.what());
                          }
                        }
                      });
    return fut;
  }

  void asyncCallback(const std::string& peer, const std::string& func, py::function cb,
                     py::args args, py::kwargs kwargs) {
    std::vector<at::Tensor> tensors;
    std::string payload = serializeCall(args, kwargs, tensors, rpc_->peerIsLocal(peer));
    auto g = std::make_shared<PyGuard>(cb);
    py::gil_scoped_release rel;
    rpc_->sendRequest(peer, func, std::move(payload), std::move(tensors),
                      [g](Frame* resp, const std::string* err) {
                        if (!pyAlive()) return;
                        py::gil_scoped_acquire gil;
                        try {
                          if (err) {
                            g->obj(py::none(), rpcErrorType()(*err));
                          } else {
                            py::object v = deserializeObject(resp->payload, resp->tensors);
                            g->obj(v, py::none());
                          }
                        } catch (py::error_already_set& e) {
                          e.discard_as_unraisable("moolib_amd async_callback");
                        }
                      });
  }

  py::object syncCall(const std::string& peer, const std::string& func, py::args args,
                      py::kwargs kwargs) {
    PyFuture fut = asyncCall(peer, func, args, kwargs);
    return fut.resultNoTimeout();
  }

 private:
  RpcPtr rpc_;
  std::shared_ptr<std::atomic<int>> excMode_ = std::make_shared<std::atomic<int>>(2);
};

// -------------------------------------------------------------- Broker

class BrokerWrapper {
 public:
  explicit BrokerWrapper(py::object rpc) {
    if (rpc.is_none()) {
      rpc_ = Rpc::create();
      rpc_->setName("broker");
      registerRpc(rpc_);
    } else {
      rpc_ = py::cast<RpcWrapper&>(rpc).rpc();
    }
    broker_ = std::make_unique<Broker>(rpc_);
  }
  void setName(const std::string& n) { rpc_->setName(n); }
  std::vector<std::string> listen(const std::string& addr) {
    py::gil_scoped_release rel;
    return rpc_->listen(addr);
  }
  void update() { broker_->update(); }
  RpcPtr rpc() { return rpc_; }

 private:
  RpcPtr rpc_;
  std::unique_ptr<Broker> broker_;
};

// --------------------------------------------------------------- Group

PyFold makePyFold(PyGuardPtr op) {
  return [op](ReduceValue& dst, ReduceValue& src) {
    if (!pyAlive()) throw RpcError("interpreter shutting down");
    py::gil_scoped_acquire gil;
    py::object a = deserializeObject(dst.payload, dst.tensors);
    py::object b = deserializeObject(src.payload, src.tensors);
    py::object r;
    if (op && op->obj.ptr() && !op->obj.is_none()) {
      r = op->obj(a, b);
      if (r.is_none()) r = a;  // ops that mutate dst in place and return None
    } else {
      PyObject* res = PyNumber_Add(a.ptr(), b.ptr());
      if (!res) throw py::error_already_set();
      r = py::reinterpret_steal<py::object>(res);
    }
    dst.tensors.clear();
    dst.payload = serializeObject(r, dst.tensors);
  };
}

class GroupWrapper {
 public:
  GroupWrapper(RpcWrapper& rpc, const std::string& name) {
    group_ = Group::create(rpc.rpc(), name);
  }
  std::shared_ptr<Group> group() { return group_; }

  void update() { group_->update(); }
  std::vector<std::string> members() { return group_->members(); }
  uint64_t syncId() { return group_->syncId(); }
  std::string name() { return group_->name(); }
  bool active() { return group_->active(); }
  void setBrokerName(const std::string& n) { group_->setBrokerName(n); }
  void setTimeout(double t) { group_->setTimeout(t); }
  void setSortOrder(int64_t o) { group_->setSortOrder(o); }

  PyFuture allReduce(const std::string& name, py::object value, py::object op) {
    ReduceValue v;
    v.kind = ReduceValue::pyObject;
    v.payload = serializeObject(value, v.tensors);
    auto opGuard = std::make_shared<PyGuard>(op);
    PyFuture fut;
    auto st = fut.state();
    PyFold fold = makePyFold(opGuard);
    py::gil_scoped_release rel;
    group_->allReduce(name, std::move(v), std::move(fold),
                      [st](ReduceValue* rv, const std::string* err) {
                        if (!pyAlive()) return;
                        py::gil_scoped_acquire gil;
                        if (err) {
                          futureFail(st, *err);
                        } else {
                          try {
                            futureComplete(st, deserializeObject(rv->payload, rv->tensors));
                          } catch (py::error_already_set& e) {
                            futureFail(st, std::string("deserialize failed: ") + e.what());
                          } catch (const std::exception& e) {
                            futureFail(st, e.what());
                          }
                        }
                      });
    return fut;
  }

 private:
  std::shared_ptr<Group> group_;
};

// --------------------------------------------------------- Accumulator

class AccumulatorWrapper {
 public:
  AccumulatorWrapper(const std::string& name, py::object parameters, py::object buffers,
                     py::object group) {
    std::vector<at::Tensor> ps, bs;
    for (auto h : parameters) ps.push_back(py::cast<at::Tensor>(h));
    for (auto h : buffers) bs.push_back(py::cast<at::Tensor>(h));
    if (group.is_none()) {
      ownRpc_ = Rpc::create();
      ownRpc_->setName(name + "-" + randomUid().substr(0, 8));
      registerRpc(ownRpc_);
      group_ = Group::create(ownRpc_, name + "_group");
    } else {
      group_ = py::cast<GroupWrapper&>(group).group();
    }
    acc_ = Accumulator::create(name, std::move(ps), std::move(bs), group_);
  }

  void connect(const std::string& addr) {
    py::gil_scoped_release rel;
    acc_->connect(addr);
  }
  void update() {
    py::gil_scoped_release rel;
    acc_->update();
  }
  bool connected() {
    py::gil_scoped_release rel;
    return acc_->connected();
  }
  bool wantsState() {
    py::gil_scoped_release rel;
    return acc_->wantsState();
  }
  bool hasNewState() {
    py::gil_scoped_release rel;
    return acc_->hasNewState();
  }
  void setState(py::object state) {
    std::vector<at::Tensor> tensors;
    std::string payload = serializeObject(state, tensors);
    py::gil_scoped_release rel;
    acc_->setState(std::move(payload), std::move(tensors));
  }
  py::object state() {
    std::pair<std::string, std::vector<at::Tensor>> st;
    {
      py::gil_scoped_release rel;
      st = acc_->state();
    }
    return deserializeObject(st.first, st.second);
  }
  bool wantsGradients() {
    py::gil_scoped_release rel;
    return acc_->wantsGradients();
  }
  bool hasGradients() {
    py::gil_scoped_release rel;
    return acc_->hasGradients();
  }
  void skipGradients() {
    py::gil_scoped_release rel;
    acc_->skipGradients();
  }
  void reduceGradients(int64_t batchSize) {
    py::gil_scoped_release rel;
    acc_->reduceGradients(batchSize);
  }
  void zeroGradients() {
    py::gil_scoped_release rel;
    acc_->zeroGradients();
  }
  int64_t modelVersion() {
    py::gil_scoped_release rel;
    return acc_->modelVersion();
  }
  void setModelVersion(int64_t v) {
    py::gil_scoped_release rel;
    acc_->setModelVersion(v);
  }
  std::string getLeader() {
    py::gil_scoped_release rel;
    return acc_->getLeader();
  }
  bool isLeader() {
    py::gil_scoped_release rel;
    return acc_->isLeader();
  }
  py::dict gradientStats() {
    std::unordered_map<std::string, int64_t> s;
    {
      py::gil_scoped_release rel;
      s = acc_->gradientStats();
    }
    py::dict d;
    for (auto& [k, v] : s) d[py::str(k)] = v;
    return d;
  }
  void setVirtualBatchSize(int64_t n) { acc_->setVirtualBatchSize(n); }
  void setParallelGradients(int64_t n) { acc_->setParallelGradients(n); }
  std::string debugState() {
    py::gil_scoped_release rel;
    return acc_->debugState();
  }

  void setLocalReduceHook(py::object fn) {
    if (fn.is_none()) {
      acc_->setLocalReduceHook(nullptr);
      return;
    }
    auto g = std::make_shared<PyGuard>(fn);
    acc_->setLocalReduceHook([g](at::Tensor& t) -> std::function<bool()> {
      if (!pyAlive()) throw RpcError("interpreter shutting down");
      py::gil_scoped_acquire gil;
      py::object poll = g->obj(py::cast(t));
      auto pg = std::make_shared<PyGuard>(poll);
      return [pg]() -> bool {
        if (!pyAlive()) return true;
        py::gil_scoped_acquire gil;
        return py::cast<bool>(pg->obj());
      };
    });
  }

 private:
  Accumulator::Ptr acc_;
  std::shared_ptr<Group> group_;
  RpcPtr ownRpc_;
};

// -------------------------------------------------------------- module

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "moolib_amd core runtime (MI355X-native distributed RL framework)";

  auto rpcError = py::register_exception<RpcError>(m, "RpcError", PyExc_RuntimeError);
  *g_rpcErrorType = rpcError;

  m.def("create_uid", [] { return randomUid(); });
  m.def("set_log_level", [](const std::string& level) {
    int v = level == "none"      ? 0
            : level == "error"   ? 1
            : level == "info"    ? 2
            : level == "verbose" ? 3
            : level == "debug"   ? 4
                                 : -1;
    if (v < 0) throw RpcError("unknown log level: " + level);
    g_logLevel.store(v);
  });
  m.def("set_logging", [](py::object sink) {
    if (sink.is_none()) {
      g_logSink = nullptr;
      return;
    }
    auto g = std::make_shared<PyGuard>(sink);
    g_logSink = [g](int level, const std::string& msg) {
      if (!pyAlive()) return;
      py::gil_scoped_acquire gil;
      try {
        g->obj(level, msg);
      } catch (py::error_already_set& e) {
        e.discard_as_unraisable("moolib_amd log sink");
      }
    };
  });
  m.def("set_max_threads", [](int n) { globalScheduler().setMaxThreads(n); });
  m.def("_shutdown_all", [] {
    py::gil_scoped_release rel;
    shutdownAll();
  });

  py::class_<PyFuture>(m, "Future")
      .def("result", &PyFuture::resultNoTimeout)
      .def("result", &PyFuture::resultTimeout, py::arg("timeout"))
      .def("wait", &PyFuture::waitNoTimeout)
      .def("wait", &PyFuture::waitTimeout, py::arg("timeout"))
      .def("done", &PyFuture::done)
      .def("cancel", &PyFuture::cancel)
      .def("exception", &PyFuture::exception)
      .def("_add_done_callback", &PyFuture::addDoneCallback);

  py::class_<PyQueue>(m, "Queue")
      .def(py::init<>())
      .def("enqueue", &PyQueue::enqueue)
      .def("size", &PyQueue::size)
      .def("_pop_future", &PyQueue::popFuture);

  py::class_<RpcDeferredReturn>(m, "RpcDeferredReturn")
      .def("__call__", &RpcDeferredReturn::call)
      .def("error", &RpcDeferredReturn::error);

  py::class_<RpcWrapper>(m, "Rpc")
      .def(py::init<>())
      .def("set_name", &RpcWrapper::setName, py::arg("name"))
      .def("get_name", &RpcWrapper::getName)
      .def("set_timeout", &RpcWrapper::setTimeout, py::arg("timeout"))
      .def("set_transports", &RpcWrapper::setTransports, py::arg("transports"))
      .def("set_exception_mode", &RpcWrapper::setExceptionMode, py::arg("mode"))
      .def("listen", &RpcWrapper::listen, py::arg("address"))
      .def("connect", &RpcWrapper::connect, py::arg("address"))
      .def("local_addrs", &RpcWrapper::localAddrs)
      .def("debug_info", &RpcWrapper::debugInfo)
      .def("_define_raw", &RpcWrapper::define)
      .def("define_deferred_raw", &RpcWrapper::defineDeferred)
      .def("define_queue_raw", &RpcWrapper::defineQueue)
      .def("undefine", &RpcWrapper::undefine, py::arg("name"))
      .def("async_", &RpcWrapper::asyncCall)
      .def("async_callback", &RpcWrapper::asyncCallback)
      .def("sync", &RpcWrapper::syncCall);

  py::class_<BrokerWrapper>(m, "Broker")
      .def(py::init<py::object>(), py::arg("rpc") = py::none())
      .def("set_name", &BrokerWrapper::setName, py::arg("name"))
      .def("listen", &BrokerWrapper::listen, py::arg("address"))
      .def("update", &BrokerWrapper::update, py::call_guard<py::gil_scoped_release>());

  py::class_<GroupWrapper>(m, "Group")
      .def(py::init<RpcWrapper&, const std::string&>(), py::arg("rpc"), py::arg("name"))
      .def("update", &GroupWrapper::update, py::call_guard<py::gil_scoped_release>())
      .def("members", &GroupWrapper::members, py::call_guard<py::gil_scoped_release>())
      .def("sync_id", &GroupWrapper::syncId, py::call_guard<py::gil_scoped_release>())
      .def("name", &GroupWrapper::name)
      .def("active", &GroupWrapper::active, py::call_guard<py::gil_scoped_release>())
      .def("set_broker_name", &GroupWrapper::setBrokerName)
      .def("set_timeout", &GroupWrapper::setTimeout)
      .def("set_sort_order", &GroupWrapper::setSortOrder)
      .def("all_reduce", &GroupWrapper::allReduce, py::arg("name"), py::arg("value"),
           py::arg("op") = py::none());

  py::class_<EnvStepperFuture>(m, "EnvStepperFuture")
      .def("result", &EnvStepperFuture::result);

  py::class_<EnvPool>(m, "EnvPool")
      .def(py::init<py::object, int, int, int, int64_t, const std::string&, bool>(),
           py::arg("create_env"), py::arg("num_processes"), py::arg("batch_size"),
           py::arg("num_batches"), py::arg("shared_memory_bytes") = 0,
           py::arg("shm_name") = std::string(), py::arg("external_workers") = false)
      .def("step", &EnvPool::step, py::arg("batch_index"), py::arg("action"))
      .def("poll", &EnvPool::poll, py::arg("batch_index"),
           "true iff result(batch_index) would return without blocking")
      .def("shared_buffer", &EnvPool::sharedBuffer)
      .def("running", &EnvPool::running)
      .def("num_workers_alive", &EnvPool::numWorkersAlive);

  py::class_<EnvRunner>(m, "EnvRunner")
      .def(py::init<py::object>(), py::arg("create_env"))
      .def("start", &EnvRunner::start, py::arg("shm_name"),
           "attach to the named EnvPool segment and serve one worker slot")
      .def("running", &EnvRunner::running);

  py::class_<Batcher>(m, "Batcher")
      .def(py::init<int64_t, py::object, int64_t>(), py::arg("size"),
           py::arg("device") = py::none(), py::arg("dim") = 0)
      .def("stack", &Batcher::stack, py::arg("tensors"))
      .def("cat", &Batcher::cat, py::arg("tensors"))
      .def("empty", &Batcher::empty)
      .def("size", &Batcher::size)
      .def("get", &Batcher::get)
      .def("_pop_future", &Batcher::popFuture);
  m.def("_set_batcher_fused_copy", &setBatcherFusedCopy,
        "register the _kernels.batched_copy fused slice-copy hook");

  py::class_<AccumulatorWrapper>(m, "Accumulator")
      .def(py::init<const std::string&, py::object, py::object, py::object>(), py::arg("name"),
           py::arg("parameters"), py::arg("buffers"), py::arg("group") = py::none())
      .def("connect", &AccumulatorWrapper::connect, py::arg("address"))
      .def("update", &AccumulatorWrapper::update)
      .def("connected", &AccumulatorWrapper::connected)
      .def("wants_state", &AccumulatorWrapper::wantsState)
      .def("has_new_state", &AccumulatorWrapper::hasNewState)
      .def("set_state", &AccumulatorWrapper::setState, py::arg("state"))
      .def("state", &AccumulatorWrapper::state)
      .def("wants_gradients", &AccumulatorWrapper::wantsGradients)
      .def("has_gradients", &AccumulatorWrapper::hasGradients)
      .def("skip_gradients", &AccumulatorWrapper::skipGradients)
      .def("reduce_gradients", &AccumulatorWrapper::reduceGradients, py::arg("batch_size"))
      .def("zero_gradients", &AccumulatorWrapper::zeroGradients)
      .def("model_version", &AccumulatorWrapper::modelVersion)
      .def("set_model_version", &AccumulatorWrapper::setModelVersion, py::arg("n"))
      .def("get_leader", &AccumulatorWrapper::getLeader)
      .def("is_leader", &AccumulatorWrapper::isLeader)
      .def("get_gradient_stats", &AccumulatorWrapper::gradientStats)
      .def("set_virtual_batch_size", &AccumulatorWrapper::setVirtualBatchSize, py::arg("n"))
      .def("set_parallel_gradients", &AccumulatorWrapper::setParallelGradients, py::arg("n"))
      .def("set_local_reduce_hook", &AccumulatorWrapper::setLocalReduceHook, py::arg("hook"))
      .def("debug_state", &AccumulatorWrapper::debugState);
}

}  // namespace mrl
