#include "envpool.h"

#include <fcntl.h>
#include <semaphore.h>
#include <signal.h>
#include <sys/mman.h>
#include <sys/prctl.h>
#include <sys/wait.h>
#include <time.h>
#include <unistd.h>

#include <torch/csrc/autograd/python_variable.h>

#include <atomic>
#include <cstring>
#include <thread>

#include "common.h"
#include "message.h"

namespace mrl {

namespace {

constexpr uint32_t kRingSize = 64;
constexpr uint32_t kMaxFields = 32;
constexpr uint32_t kMaxFieldNameLen = 63;
constexpr uint32_t kMaxDims = 7;

struct FieldDesc {
  char name[kMaxFieldNameLen + 1];
  uint8_t dtype;  // wire dtype code (message.h)
  uint8_t ndim;
  int64_t shape[kMaxDims];  // per-env shape
  uint64_t offset;          // within data region, per batch: [batchSize, *shape]
  uint64_t bytesPerEnv;
};

constexpr uint32_t kSegMagic = 0x6d726c45;  // 'mrlE'

struct Header {
  std::atomic<uint32_t> magicReady;  // kSegMagic once the creator's init is done
  std::atomic<uint32_t> layoutReady;
  std::atomic<uint32_t> workerFailed;  // worker index + 1
  std::atomic<uint32_t> claimedWorkers;  // external EnvRunners claim slots
  // Geometry, so a named-segment attacher can recompute the layout.
  int32_t numWorkers;
  int32_t batchSize;
  int32_t numBatches;
  int64_t segBytes;
  uint32_t numFields;
  uint64_t bytesPerBatch;
  FieldDesc fields[kMaxFields];
  char errorMsg[4096];
};

struct WorkerQueue {
  sem_t itemsSem;
  std::atomic<uint32_t> head;
  std::atomic<uint32_t> tail;
  uint32_t ring[kRingSize];
};

int64_t dtypeBytes(uint8_t code) {
  switch (code) {
    case 1: return 4;   // f32
    case 2: return 8;   // f64
    case 3: return 2;   // f16
    case 4: return 2;   // bf16
    case 5: return 8;   // i64
    case 6: return 4;   // i32
    case 7: return 2;   // i16
    case 8: return 1;   // i8
    case 9: return 1;   // u8
    case 10: return 1;  // bool
    default: throw RpcError("envpool: bad dtype");
  }
}

}  // namespace

class EnvPoolImpl : public std::enable_shared_from_this<EnvPoolImpl> {
 public:
  EnvPoolImpl(py::object createEnv, int numProcesses, int batchSize, int numBatches,
              int64_t segBytes, std::string shmName = "", bool externalWorkers = false)
      : createEnv_(std::move(createEnv)),
        numWorkers_(numProcesses),
        batchSize_(batchSize),
        numBatches_(numBatches),
        shmName_(shmPath(shmName)),
        externalWorkers_(externalWorkers) {
    if (numWorkers_ <= 0 || batchSize_ <= 0 || numBatches_ <= 0) {
      throw RpcError("envpool: sizes must be positive");
    }
    if (externalWorkers_ && shmName_.empty()) {
      throw RpcError("envpool: external workers need a shm_name to attach to");
    }
    if (numWorkers_ > batchSize_) numWorkers_ = batchSize_;

    computeLayout();
    if (segBytes < static_cast<int64_t>(dataOff_) + (64 << 20)) segBytes = dataOff_ + (512 << 20);
    segBytes_ = segBytes;

    if (shmName_.empty()) {
      base_ = static_cast<char*>(
          mmap(nullptr, segBytes_, PROT_READ | PROT_WRITE, MAP_SHARED | MAP_ANONYMOUS, -1, 0));
      if (base_ == MAP_FAILED) throw RpcError("envpool: mmap failed");
    } else {
      shmFd_ = shm_open(shmName_.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
      if (shmFd_ < 0) throw RpcError("envpool: shm_open(create " + shmName_ + ") failed");
      shmOwner_ = true;
      if (ftruncate(shmFd_, segBytes_) != 0) throw RpcError("envpool: ftruncate failed");
      base_ = static_cast<char*>(
          mmap(nullptr, segBytes_, PROT_READ | PROT_WRITE, MAP_SHARED, shmFd_, 0));
      if (base_ == MAP_FAILED) throw RpcError("envpool: mmap(shm) failed");
    }
    std::memset(base_, 0, dataOff_);

    new (header()) Header();
    Header* h = header();
    h->numWorkers = numWorkers_;
    h->batchSize = batchSize_;
    h->numBatches = numBatches_;
    h->segBytes = segBytes_;
    for (int w = 0; w < numWorkers_; ++w) {
      WorkerQueue* q = queue(w);
      sem_init(&q->itemsSem, 1, 0);
      q->head.store(0);
      q->tail.store(0);
    }
    for (int i = 0; i < numBatches_ * numWorkers_; ++i) sem_init(&doneSems()[i], 1, 0);
    h->magicReady.store(kSegMagic, std::memory_order_release);

    if (externalWorkers_) return;  // EnvRunner processes will claim the slots

    // ---- fork the workers (GIL is held: safe point for PyOS_AfterFork) ----
    for (int w = 0; w < numWorkers_; ++w) {
      pid_t pid = fork();
      if (pid < 0) throw RpcError("envpool: fork failed");
      if (pid == 0) {
        // ---- child ----
        prctl(PR_SET_PDEATHSIG, SIGKILL);
        signal(SIGINT, SIG_IGN);
        PyOS_AfterFork_Child();
        int code = 0;
        try {
          workerMain(w);
        } catch (...) {
          code = 1;
        }
        std::_Exit(code);
      }
      pids_.push_back(pid);
    }
  }

  // Attach to a named segment created by another process (EnvRunner side).
  struct AttachTag {};
  EnvPoolImpl(AttachTag, py::object createEnv, const std::string& shmName)
      : createEnv_(std::move(createEnv)), shmName_(shmPath(shmName)), attached_(true) {
    TimePoint t0 = now();
    while (true) {
      shmFd_ = shm_open(shmName_.c_str(), O_RDWR, 0600);
      if (shmFd_ >= 0) break;
      if (secondsSince(t0) > 60.0) {
        throw RpcError("envrunner: no segment named " + shmName_ + " appeared within 60s");
      }
      usleep(20000);
    }
    // Map the header, wait for init, read the geometry, then map fully.
    void* peek = mmap(nullptr, sizeof(Header), PROT_READ, MAP_SHARED, shmFd_, 0);
    if (peek == MAP_FAILED) throw RpcError("envrunner: header mmap failed");
    Header* ph = reinterpret_cast<Header*>(peek);
    while (ph->magicReady.load(std::memory_order_acquire) != kSegMagic) {
      if (secondsSince(t0) > 60.0) throw RpcError("envrunner: segment never initialized");
      usleep(1000);
    }
    numWorkers_ = ph->numWorkers;
    batchSize_ = ph->batchSize;
    numBatches_ = ph->numBatches;
    segBytes_ = ph->segBytes;
    munmap(peek, sizeof(Header));
    computeLayout();
    base_ = static_cast<char*>(
        mmap(nullptr, segBytes_, PROT_READ | PROT_WRITE, MAP_SHARED, shmFd_, 0));
    if (base_ == MAP_FAILED) throw RpcError("envrunner: mmap failed");
  }

  // Claim a worker slot and serve it on a host thread until terminated.
  void startRunnerThread() {
    int w = static_cast<int>(header()->claimedWorkers.fetch_add(1));
    if (w >= numWorkers_) {
      header()->claimedWorkers.fetch_sub(1);
      throw RpcError("envrunner: all " + std::to_string(numWorkers_) + " slots claimed");
    }
    runnerThread_ = std::thread([this, w] {
      py::gil_scoped_acquire gil;
      workerMain(w);
    });
  }

  ~EnvPoolImpl() {
    terminate_.store(true);
    if (runnerThread_.joinable()) {
      // The runner thread re-acquires the GIL between queue waits; joining
      // while holding it would deadlock.
      if (PyGILState_Check()) {
        py::gil_scoped_release rel;
        runnerThread_.join();
      } else {
        runnerThread_.join();
      }
    }
    for (pid_t p : pids_) kill(p, SIGKILL);
    for (pid_t p : pids_) waitpid(p, nullptr, 0);
    if (base_ && base_ != MAP_FAILED) munmap(base_, segBytes_);
    if (shmFd_ >= 0) close(shmFd_);
    if (shmOwner_) shm_unlink(shmName_.c_str());
  }

  bool runnerAlive() { return runnerThread_.joinable() && !terminate_.load(); }

 private:
  static std::string shmPath(const std::string& name) {
    return name.empty() ? name : "/mrl-env-" + name;
  }

  void computeLayout() {
    size_t off = 0;
    auto alloc = [&](size_t n, size_t align = 64) {
      off = (off + align - 1) & ~(align - 1);
      size_t at = off;
      off += n;
      return at;
    };
    headerOff_ = alloc(sizeof(Header));
    actionsOff_ = alloc(sizeof(int64_t) * numBatches_ * batchSize_);
    queuesOff_ = alloc(sizeof(WorkerQueue) * numWorkers_);
    doneSemsOff_ = alloc(sizeof(sem_t) * numBatches_ * numWorkers_);
    dataOff_ = alloc(1, 4096);
  }

 public:

  // ------------------------------------------------------------- client

  void step(int b, py::object action) {
    if (b < 0 || b >= numBatches_) throw RpcError("envpool: bad batch index");
    if (externalWorkers_ && !allClaimed_) {
      // Work posted before every slot has a runner would wait forever on
      // the unclaimed slots' completion semaphores.
      py::gil_scoped_release rel;
      TimePoint t0 = now();
      while (header()->claimedWorkers.load(std::memory_order_acquire) <
             static_cast<uint32_t>(numWorkers_)) {
        checkFailureNoGil();
        if (secondsSince(t0) > 120.0) {
          throw RpcError("envpool: only " +
                         std::to_string(header()->claimedWorkers.load()) + "/" +
                         std::to_string(numWorkers_) + " EnvRunner slots claimed after 120s");
        }
        usleep(5000);
      }
      allClaimed_ = true;
    }
    at::Tensor a = THPVariable_Unpack(action.ptr());
    if (!a.device().is_cpu()) a = a.to(at::kCPU);
    a = a.to(at::kLong).contiguous();
    if (a.numel() != batchSize_) throw RpcError("envpool: action size mismatch");
    std::memcpy(actions() + b * batchSize_, a.data_ptr<int64_t>(), sizeof(int64_t) * batchSize_);
    for (int w = 0; w < numWorkers_; ++w) {
      WorkerQueue* q = queue(w);
      uint32_t t = q->tail.load(std::memory_order_relaxed);
      q->ring[t % kRingSize] = static_cast<uint32_t>(b);
      q->tail.store(t + 1, std::memory_order_release);
      sem_post(&q->itemsSem);
    }
  }

  // Non-consuming completion check: true iff every worker has posted batch
  // b's completion semaphore (result(b) would return without blocking).
  bool poll(int b) {
    if (b < 0 || b >= numBatches_) return false;
    for (int w = 0; w < numWorkers_; ++w) {
      int v = 0;
      if (sem_getvalue(&doneSems()[b * numWorkers_ + w], &v) != 0 || v < 1) return false;
    }
    return true;
  }

  py::object result(int b) {
    {
      py::gil_scoped_release rel;
      for (int w = 0; w < numWorkers_; ++w) {
        while (true) {
          timespec ts;
          clock_gettime(CLOCK_REALTIME, &ts);
          ts.tv_sec += 2;
          if (sem_timedwait(&doneSems()[b * numWorkers_ + w], &ts) == 0) break;
          if (errno == EINTR) continue;
          checkFailureNoGil();
          checkChildrenAliveNoGil();
        }
      }
      checkFailureNoGil();
    }
    if (!tensorsBuilt_) buildTensors();
    return batchDicts_[b];
  }

  at::Tensor dataRegion() {
    // Whole obs/reward/done region as one uint8 view — the GPU path
    // hipHostRegisters it once so shm->HBM copies run as async DMA.
    return at::from_blob(base_ + dataOff_, {static_cast<int64_t>(segBytes_ - dataOff_)},
                         at::TensorOptions().dtype(at::kByte));
  }

  bool anyAlive() {
    if (externalWorkers_) return header()->claimedWorkers.load() > 0;
    for (pid_t p : pids_) {
      if (waitpid(p, nullptr, WNOHANG) == 0) return true;
    }
    return false;
  }
  int aliveCount() {
    if (externalWorkers_) return static_cast<int>(header()->claimedWorkers.load());
    int n = 0;
    for (pid_t p : pids_) {
      if (waitpid(p, nullptr, WNOHANG) == 0) ++n;
    }
    return n;
  }

 private:
  Header* header() { return reinterpret_cast<Header*>(base_ + headerOff_); }
  int64_t* actions() { return reinterpret_cast<int64_t*>(base_ + actionsOff_); }
  WorkerQueue* queue(int w) {
    return reinterpret_cast<WorkerQueue*>(base_ + queuesOff_) + w;
  }
  sem_t* doneSems() { return reinterpret_cast<sem_t*>(base_ + doneSemsOff_); }
  char* fieldPtr(const FieldDesc& f, int b) {
    return base_ + dataOff_ + static_cast<size_t>(b) * header()->bytesPerBatch + f.offset;
  }

  void checkFailureNoGil() {
    uint32_t wf = header()->workerFailed.load();
    if (wf != 0) {
      throw RpcError(std::string("envpool worker ") + std::to_string(wf - 1) +
                     " failed: " + header()->errorMsg);
    }
  }
  void checkChildrenAliveNoGil() {
    for (size_t i = 0; i < pids_.size(); ++i) {
      if (waitpid(pids_[i], nullptr, WNOHANG) != 0) {
        throw RpcError("envpool worker " + std::to_string(i) + " died");
      }
    }
  }

  void buildTensors() {
    // Wait for worker 0 to publish the field layout.
    {
      py::gil_scoped_release rel;
      TimePoint t0 = now();
      while (header()->layoutReady.load(std::memory_order_acquire) == 0) {
        checkFailureNoGil();
        if (secondsSince(t0) > 120.0) throw RpcError("envpool: layout never published");
        usleep(1000);
      }
    }
    Header* h = header();
    for (int b = 0; b < numBatches_; ++b) {
      py::dict d;
      for (uint32_t fi = 0; fi < h->numFields; ++fi) {
        FieldDesc& f = h->fields[fi];
        std::vector<int64_t> shape{batchSize_};
        for (int i = 0; i < f.ndim; ++i) shape.push_back(f.shape[i]);
        at::Tensor t = at::from_blob(fieldPtr(f, b), shape,
                                     at::TensorOptions().dtype(wireToDtype(f.dtype)));
        d[py::str(f.name)] = py::reinterpret_steal<py::object>(THPVariable_Wrap(t));
      }
      batchDicts_.push_back(std::move(d));
    }
    tensorsBuilt_ = true;
  }

  // ------------------------------------------------------------- worker

  struct EnvSlot {
    py::object env;
    bool started = false;
  };

  // Unpack both gym API generations.
  static py::object unpackReset(py::object r) {
    if (py::isinstance<py::tuple>(r)) {
      auto t = py::reinterpret_borrow<py::tuple>(r);
      if (t.size() == 2 && py::isinstance<py::dict>(t[1])) return t[0];
    }
    return r;
  }

  void workerMain(int w) {
    Header* h = header();
    try {
      // Create this worker's envs: indices {e : e % numWorkers_ == w} per batch.
      std::vector<int> owned;
      for (int e = w; e < batchSize_; e += numWorkers_) owned.push_back(e);
      std::vector<std::vector<EnvSlot>> envs(numBatches_);
      for (int b = 0; b < numBatches_; ++b) {
        for (size_t i = 0; i < owned.size(); ++i) {
          EnvSlot s;
          s.env = createEnv_();
          envs[b].push_back(std::move(s));
        }
      }

      py::object torchMod = py::module_::import("torch");
      py::object asTensor = torchMod.attr("as_tensor");

      // Field discovery (worker 0 publishes; others wait).
      auto obsToDict = [&](py::object obs) -> py::dict {
        if (py::isinstance<py::dict>(obs)) return py::reinterpret_borrow<py::dict>(obs);
        py::dict d;
        d["state"] = obs;
        return d;
      };

      auto publishLayout = [&](py::dict obsDict) {
        uint32_t nf = 0;
        uint64_t off = 0;
        auto addField = [&](const std::string& name, at::ScalarType dt,
                            const std::vector<int64_t>& shape) {
          if (nf >= kMaxFields) throw RpcError("envpool: too many observation fields");
          FieldDesc& f = h->fields[nf];
          std::snprintf(f.name, sizeof(f.name), "%s", name.c_str());
          f.dtype = dtypeToWire(dt);
          f.ndim = static_cast<uint8_t>(shape.size());
          int64_t bytes = dtypeBytes(f.dtype);
          for (size_t i = 0; i < shape.size(); ++i) {
            f.shape[i] = shape[i];
            bytes *= shape[i];
          }
          f.bytesPerEnv = bytes;
          off = (off + 63) & ~uint64_t(63);
          f.offset = off;
          off += f.bytesPerEnv * batchSize_;
          ++nf;
        };
        for (auto item : obsDict) {
          std::string name = py::cast<std::string>(item.first);
          at::Tensor t = py::cast<at::Tensor>(asTensor(item.second));
          std::vector<int64_t> shape(t.sizes().begin(), t.sizes().end());
          addField(name, t.scalar_type(), shape);
        }
        addField("reward", at::kFloat, {});
        addField("done", at::kBool, {});
        h->numFields = nf;
        h->bytesPerBatch = (off + 4095) & ~uint64_t(4095);
        if (dataOff_ + h->bytesPerBatch * numBatches_ > static_cast<uint64_t>(segBytes_)) {
          throw RpcError("envpool: observations too large for shared memory segment");
        }
        h->layoutReady.store(1, std::memory_order_release);
      };

      auto writeEnv = [&](int b, int e, py::dict obsDict, float reward, bool done) {
        for (uint32_t fi = 0; fi < h->numFields; ++fi) {
          FieldDesc& f = h->fields[fi];
          char* dst = fieldPtr(f, b) + f.bytesPerEnv * e;
          if (std::strcmp(f.name, "reward") == 0) {
            std::memcpy(dst, &reward, sizeof(float));
          } else if (std::strcmp(f.name, "done") == 0) {
            *reinterpret_cast<uint8_t*>(dst) = done ? 1 : 0;
          } else {
            py::object v = obsDict[py::str(f.name)];
            at::Tensor t = py::cast<at::Tensor>(asTensor(v)).contiguous();
            at::ScalarType want = wireToDtype(f.dtype);
            if (t.scalar_type() != want) t = t.to(want);
            if (t.nbytes() != static_cast<size_t>(f.bytesPerEnv)) {
              throw RpcError("envpool: observation field '" + std::string(f.name) +
                             "' changed shape");
            }
            std::memcpy(dst, t.data_ptr(), f.bytesPerEnv);
          }
        }
      };

      bool layoutDone = false;
      auto stepBatch = [&](int b) {
        for (size_t i = 0; i < owned.size(); ++i) {
          int e = owned[i];
          EnvSlot& slot = envs[b][i];
          py::dict obsDict;
          float reward = 0.f;
          bool done = false;
          if (!slot.started) {
            slot.started = true;
            obsDict = obsToDict(unpackReset(slot.env.attr("reset")()));
          } else {
            int64_t action = actions()[b * batchSize_ + e];
            py::tuple r = py::reinterpret_borrow<py::tuple>(slot.env.attr("step")(action));
            py::object obs = r[0];
            reward = py::cast<float>(py::float_(r[1]));
            if (r.size() >= 5) {
              done = py::cast<bool>(r[2]) || py::cast<bool>(r[3]);
            } else {
              done = py::cast<bool>(r[2]);
            }
            if (done) obs = unpackReset(slot.env.attr("reset")());
            obsDict = obsToDict(obs);
          }
          if (!layoutDone) {
            if (w == 0) {
              publishLayout(obsDict);
            } else {
              py::gil_scoped_release rel;
              while (h->layoutReady.load(std::memory_order_acquire) == 0) usleep(500);
            }
            layoutDone = true;
          }
          writeEnv(b, e, obsDict, reward, done);
        }
      };

      WorkerQueue* q = queue(w);
      while (true) {
        {
          py::gil_scoped_release rel;
          bool got = false;
          while (!got) {
            timespec ts;
            clock_gettime(CLOCK_REALTIME, &ts);
            ts.tv_nsec += 250 * 1000 * 1000;
            if (ts.tv_nsec >= 1000000000) {
              ts.tv_sec += 1;
              ts.tv_nsec -= 1000000000;
            }
            if (sem_timedwait(&q->itemsSem, &ts) == 0) {
              got = true;
            } else if (errno == ETIMEDOUT) {
              if (terminate_.load(std::memory_order_relaxed)) return;
            } else if (errno != EINTR) {
              throw RpcError("envpool: sem_wait failed");
            }
          }
        }
        uint32_t head = q->head.load(std::memory_order_relaxed);
        uint32_t b = q->ring[head % kRingSize];
        q->head.store(head + 1, std::memory_order_release);
        stepBatch(static_cast<int>(b));
        sem_post(&doneSems()[b * numWorkers_ + w]);
      }
    } catch (const std::exception& e) {
      std::snprintf(h->errorMsg, sizeof(h->errorMsg), "%s", e.what());
      h->workerFailed.store(static_cast<uint32_t>(w) + 1);
    } catch (...) {
      std::snprintf(h->errorMsg, sizeof(h->errorMsg), "unknown error");
      h->workerFailed.store(static_cast<uint32_t>(w) + 1);
    }
  }

  py::object createEnv_;
  int numWorkers_ = 0;
  int batchSize_ = 0;
  int numBatches_ = 0;
  int64_t segBytes_ = 0;
  char* base_ = nullptr;
  size_t headerOff_ = 0, actionsOff_ = 0, queuesOff_ = 0, doneSemsOff_ = 0, dataOff_ = 0;
  std::vector<pid_t> pids_;
  bool tensorsBuilt_ = false;
  std::vector<py::object> batchDicts_;
  // named-segment / external-worker mode
  std::string shmName_;
  bool externalWorkers_ = false;
  bool attached_ = false;
  bool allClaimed_ = false;
  bool shmOwner_ = false;
  int shmFd_ = -1;
  std::atomic<bool> terminate_{false};
  std::thread runnerThread_;
};

// ------------------------------------------------------------- wrappers

py::object EnvStepperFuture::result() {
  if (!pool_) throw RpcError("empty future");
  return pool_->result(batchIndex_);
}

EnvPool::EnvPool(py::object createEnv, int numProcesses, int batchSize, int numBatches,
                 int64_t sharedMemoryBytes, const std::string& shmName, bool externalWorkers) {
  impl_ = std::make_shared<EnvPoolImpl>(std::move(createEnv), numProcesses, batchSize, numBatches,
                                        sharedMemoryBytes, shmName, externalWorkers);
}

EnvPool::~EnvPool() = default;

EnvRunner::EnvRunner(py::object createEnv) : createEnv_(std::move(createEnv)) {}

EnvRunner::~EnvRunner() = default;

void EnvRunner::start(const std::string& shmName) {
  if (impl_) throw RpcError("envrunner: already started");
  // GIL stays held: the segment creator is another process, so the wait
  // does not depend on our Python threads.
  impl_ = std::make_shared<EnvPoolImpl>(EnvPoolImpl::AttachTag{}, createEnv_, shmName);
  impl_->startRunnerThread();
}

bool EnvRunner::running() { return impl_ && impl_->runnerAlive(); }

EnvStepperFuture EnvPool::step(int batchIndex, py::object action) {
  impl_->step(batchIndex, action);
  return EnvStepperFuture(impl_, batchIndex);
}

bool EnvPool::poll(int batchIndex) { return impl_->poll(batchIndex); }

at::Tensor EnvPool::sharedBuffer() { return impl_->dataRegion(); }
bool EnvPool::running() { return impl_->anyAlive(); }
int EnvPool::numWorkersAlive() { return impl_->aliveCount(); }

}  // namespace mrl
