#include "scheduler.h"

#include <algorithm>

#include "common.h"

namespace mrl {

Scheduler::Scheduler(int maxThreads) {
  if (maxThreads <= 0) {
    unsigned hw = std::thread::hardware_concurrency();
    maxThreads = std::min<unsigned>(hw ? hw : 8, 32);
  }
  maxThreads_ = maxThreads;
}

Scheduler::~Scheduler() { shutdown(); }

void Scheduler::run(std::function<void()> f) {
  std::unique_lock<std::mutex> lk(mu_);
  if (stop_) return;
  queue_.push_back(std::move(f));
  if (idleThreads_ == 0 && static_cast<int>(threads_.size()) < maxThreads_) {
    threads_.emplace_back([this] { workerLoop(); });
  }
  cv_.notify_one();
}

void Scheduler::setMaxThreads(int n) {
  std::lock_guard<std::mutex> lk(mu_);
  if (n > 0) maxThreads_ = n;
}

void Scheduler::drain() {
  std::unique_lock<std::mutex> lk(mu_);
  idleCv_.wait(lk, [&] { return queue_.empty() && busyThreads_ == 0; });
}

void Scheduler::shutdown() {
  std::vector<std::thread> toJoin;
  {
    std::lock_guard<std::mutex> lk(mu_);
    if (stop_) return;
    stop_ = true;
    toJoin.swap(threads_);
  }
  cv_.notify_all();
  for (auto& t : toJoin) t.join();
}

void Scheduler::workerLoop() {
  std::unique_lock<std::mutex> lk(mu_);
  while (true) {
    while (queue_.empty()) {
      if (stop_) return;
      ++idleThreads_;
      if (busyThreads_ == 0) idleCv_.notify_all();
      cv_.wait(lk);
      --idleThreads_;
    }
    auto f = std::move(queue_.front());
    queue_.pop_front();
    ++busyThreads_;
    lk.unlock();
    try {
      f();
    } catch (const std::exception& e) {
      MRL_LOG_ERROR("scheduler task threw: %s", e.what());
    } catch (...) {
      MRL_LOG_ERROR("scheduler task threw unknown exception");
    }
    // Destroy the task's captures BEFORE re-locking: they may hold the last
    // reference to python objects / py-backed tensors, whose destruction
    // acquires the GIL — never block on the GIL while holding mu_.
    f = nullptr;
    lk.lock();
    --busyThreads_;
    if (queue_.empty() && busyThreads_ == 0) idleCv_.notify_all();
  }
}

Scheduler& globalScheduler() {
  static Scheduler* s = new Scheduler();  // intentionally leaked: outlive py atexit
  return *s;
}

}  // namespace mrl
