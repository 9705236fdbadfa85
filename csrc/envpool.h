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
  EnvPool(py::object createEnv, int numProcesses, int batchSize, int numBatches,
          int64_t sharedMemoryBytes);
  ~EnvPool();
  EnvStepperFuture step(int batchIndex, py::object action);
  at::Tensor sharedBuffer();
  bool running();
  int numWorkersAlive();

 private:
  std::shared_ptr<EnvPoolImpl> impl_;
};

}  // namespace mrl
