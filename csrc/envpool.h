// EnvPool: batched gym-environment execution in forked worker processes over
// shared memory.
//
// Capability parity with the reference's src/env.{h,cc} + src/shm.h: a pool
// of forked workers steps `num_batches` double-buffered batches of
// `batch_size` environments; observations/reward/done live in shared memory
// and are returned to the client as zero-copy tensor views; environments
// auto-reset on episode end (returning the new episode's first observation
// with done=True and the final reward).
//
// Our design differs from the reference's (fork-server + shm_open + action
// atomics): the pool mmaps ONE anonymous MAP_SHARED segment before forking —
// children inherit the mapping, so there is no name management and no
// cleanup; per-worker work queues and per-(batch,worker) completion
// semaphores are process-shared sem_t's inside the segment; actions are
// plain int64 slots written by the client before the batch is queued.
#pragma once

#include <torch/extension.h>

#include <memory>
#include <string>
#include <vector>

#include "pybits.h"

namespace mrl {

class EnvPoolImpl;

class EnvStepperFuture {
 public:
  EnvStepperFuture() = default;
  EnvStepperFuture(std::shared_ptr<EnvPoolImpl> pool, int batchIndex)
      : pool_(std::move(pool)), batchIndex_(batchIndex) {}
  py::object result();  // blocks (GIL released while waiting)

 private:
  std::shared_ptr<EnvPoolImpl> pool_;
  int batchIndex_ = 0;
};

class EnvPool {
 public:
  // shmName nonempty: the segment is shm_open()-named so OTHER processes
  // can host the workers (see EnvRunner); externalWorkers then skips the
  // fork and waits for runners to claim the num_processes slots.
  EnvPool(py::object createEnv, int numProcesses, int batchSize, int numBatches,
          int64_t sharedMemoryBytes, const std::string& shmName = "",
          bool externalWorkers = false);
  ~EnvPool();
  EnvStepperFuture step(int batchIndex, py::object action);
  bool poll(int batchIndex);  // true iff result(batchIndex) would not block
  at::Tensor sharedBuffer();
  bool running();
  int numWorkersAlive();

 private:
  std::shared_ptr<EnvPoolImpl> impl_;
};

// Hosts env-stepping capacity for an EnvPool created with shm_name in
// ANOTHER process on this machine — the reference's EnvRunner role
// (src/env.h:363-453: attach to a named segment, claim a client slot, run
// the worker loop). Ours runs the loop on a thread of the calling process
// (so the runner can be any separately-launched python program).
class EnvRunner {
 public:
  explicit EnvRunner(py::object createEnv);
  ~EnvRunner();
  void start(const std::string& shmName);
  bool running();

 private:
  py::object createEnv_;
  std::shared_ptr<EnvPoolImpl> impl_;
};

}  // namespace mrl
