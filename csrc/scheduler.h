// Thread-pool scheduler for RPC callbacks and handler execution.
//
// Capability parity with the reference's SchedulerFifo (src/async.{h,cc}):
// lazily-spawned worker threads, bounded by set_max_threads. Our design is a
// plain mutex+condvar pool (the RPC data plane on MI355X moves tensors via
// RCCL, not via these threads, so lock-free mailboxes buy nothing here).
#pragma once

#include <condition_variable>
#include <deque>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

namespace mrl {

class Scheduler {
 public:
  explicit Scheduler(int maxThreads = 0);
  ~Scheduler();

  void run(std::function<void()> f);
  void setMaxThreads(int n);
  // Block until the queue is empty and all workers are idle (test helper).
  void drain();
  // Stop accepting work, join all threads. Safe to call multiple times.
  void shutdown();

 private:
  void workerLoop();

  std::mutex mu_;
  std::condition_variable cv_;
  std::condition_variable idleCv_;
  std::deque<std::function<void()>> queue_;
  std::vector<std::thread> threads_;
  int maxThreads_;
  int idleThreads_ = 0;
  int busyThreads_ = 0;
  bool stop_ = false;
};

// Global scheduler shared by all Rpc instances (reference: rpc::scheduler,
// src/rpc.cc:39). Lives for the process lifetime.
Scheduler& globalScheduler();

}  // namespace mrl
