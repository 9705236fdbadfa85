#include "accumulator.h"

#include <ATen/Functions.h>
#include <c10/core/GradMode.h>

#include <cstdlib>
#include <sstream>

namespace mrl {

namespace {
struct NoGrad {
  c10::AutoGradMode g{false};
};
}  // namespace

Accumulator::Ptr Accumulator::create(std::string name, std::vector<at::Tensor> params,
                                     std::vector<at::Tensor> buffers,
                                     std::shared_ptr<Group> group) {
  Ptr p(new Accumulator(std::move(name), std::move(params), std::move(buffers), std::move(group)));
  p->setup();
  return p;
}

Accumulator::Accumulator(std::string name, std::vector<at::Tensor> params,
                         std::vector<at::Tensor> buffers, std::shared_ptr<Group> group)
    : name_(std::move(name)),
      allParams_(std::move(params)),
      buffers_(std::move(buffers)),
      group_(std::move(group)) {
  rpc_ = group_->rpc();
  for (auto& p : allParams_) {
    if (p.requires_grad()) params_.push_back(p);
  }
  // Flat on-device buckets sized for every requires-grad parameter. On
  // MI355X these are the tensors RCCL reduces over xGMI; 288 GB HBM3E makes
  // resident buckets the right default even for very large models.
  int64_t total = 0;
  for (auto& p : params_) {
    offsets_.push_back(total);
    numels_.push_back(p.numel());
    total += p.numel();
  }
  slots_.resize(1);
  slots_[0].flat = makeFlatLocked();
  if (const char* e = getenv("MOOLIB_AMD_MODEL_BCAST_S")) {  // test hook
    modelBcastInterval_ = atof(e);
  }
}

at::Tensor Accumulator::makeFlatLocked() {
  NoGrad ng;
  int64_t total = offsets_.empty() ? 1 : offsets_.back() + numels_.back();
  auto opts = params_.empty()
                  ? at::TensorOptions().dtype(at::kFloat)
                  : at::TensorOptions().dtype(params_[0].scalar_type()).device(params_[0].device());
  return at::zeros({std::max<int64_t>(total, 1)}, opts);
}

Accumulator::~Accumulator() {
  if (rpc_ && !rpc_->isShutdown()) {
    rpc_->undefine(fn("reqmodel"));
    rpc_->undefine(fn("modelupd"));
    rpc_->undefine(fn("buffers"));
  }
}

void Accumulator::setup() {
  std::weak_ptr<Accumulator> weak = weak_from_this();

  rpc_->define(fn("reqmodel"), [weak](Frame f, const std::string& from, RespondFn respond) {
    if (auto self = weak.lock()) {
      std::lock_guard<std::mutex> lk(self->mu_);
      if (std::find(self->stateRequesters_.begin(), self->stateRequesters_.end(), from) ==
          self->stateRequesters_.end()) {
        self->stateRequesters_.push_back(from);
      }
    }
    respond("", {}, false);
  });

  rpc_->define(fn("modelupd"), [weak](Frame f, const std::string& from, RespondFn respond) {
    auto self = weak.lock();
    if (!self) {
      respond("accumulator gone", {}, true);
      return;
    }
    try {
      WireReader r(f.payload);
      int64_t version = r.i64();
      uint32_t nParams = r.u32();
      uint32_t nBuffers = r.u32();
      std::string statePayload(r.str());
      // 64-bit arithmetic: uint32 nParams + nBuffers can wrap on a corrupt
      // frame, turning the bound check into out-of-bounds iterator math.
      if (uint64_t(nParams) > f.tensors.size() || uint64_t(nBuffers) > f.tensors.size() ||
          uint64_t(nParams) + uint64_t(nBuffers) > f.tensors.size())
        throw RpcError("model update tensor shortfall");
      std::lock_guard<std::mutex> lk(self->mu_);
      self->pendingParams_.assign(f.tensors.begin(), f.tensors.begin() + nParams);
      self->pendingBuffers_.assign(f.tensors.begin() + nParams,
                                   f.tensors.begin() + nParams + nBuffers);
      self->pendingStateTensors_.assign(f.tensors.begin() + nParams + nBuffers, f.tensors.end());
      self->pendingStatePayload_ = std::move(statePayload);
      self->pendingVersion_ = version;
      self->havePendingModel_ = true;
      respond("", {}, false);
    } catch (const std::exception& e) {
      respond(e.what(), {}, true);
    }
  });

  rpc_->define(fn("buffers"), [weak](Frame f, const std::string& from, RespondFn respond) {
    if (auto self = weak.lock()) {
      std::lock_guard<std::mutex> lk(self->mu_);
      if (f.tensors.size() == self->buffers_.size()) {
        NoGrad ng;
        for (size_t i = 0; i < self->buffers_.size(); ++i) {
          if (self->buffers_[i].sizes() == f.tensors[i].sizes()) {
            self->buffers_[i].copy_(f.tensors[i]);
          }
        }
      }
    }
    respond("", {}, false);
  });
}

void Accumulator::connect(const std::string& addr) { rpc_->connect(addr); }

void Accumulator::resetLocked(const char* why) {
  MRL_LOG_INFO("accumulator '%s': reset (%s)", name_.c_str(), why);
  epoch_++;
  decided_ = false;
  hasGradients_ = false;
  results_.clear();
  slotCursor_ = 0;
  hookLaunchCounter_ = 0;
  hookLaunchedUpTo_ = 0;
  for (auto& s : slots_) {
    s.newBatch = s.newGrads = s.newSkipped = 0;
    s.totBatch = s.totGrads = s.totSkipped = 0;
    s.phase = GradPhase::wantDecision;
    s.hookPending = false;
    if (s.hookPoll) {
      // Never abandon an in-flight collective: peers' call sequences must
      // stay aligned. Drain it; its buffer (the old flat) belongs to the
      // closure now; the slot gets a fresh bucket.
      drains_.push_back(std::move(s.hookPoll));
      s.hookPoll = nullptr;
      s.flat = makeFlatLocked();
    } else {
      NoGrad ng;
      s.flat.zero_();
    }
  }
  havePendingModel_ = false;
  hasNewState_ = false;
  stateRequesters_.clear();
  modelRequestSent_ = false;
  leader_.clear();
  isLeader_ = false;
}

void Accumulator::failAndResyncLocked(const char* why) {
  MRL_LOG_INFO("accumulator '%s': %s; resync", name_.c_str(), why);
  phase_ = Phase::inactive;
  syncSeen_ = 0;
  group_->requestResync();
}

void Accumulator::startElectionLocked() {
  phase_ = Phase::electing;
  ReduceValue v;
  v.kind = ReduceValue::leaderTuple;
  v.version = modelVersion_;
  v.leaderName = group_->myName();
  uint64_t epoch = epoch_;
  std::weak_ptr<Accumulator> weak = weak_from_this();
  group_->allReduce(fn("elect"), std::move(v), nullptr,
                    [weak, epoch](ReduceValue* rv, const std::string* err) {
                      auto self = weak.lock();
                      if (!self) return;
                      std::lock_guard<std::mutex> lk(self->mu_);
                      if (epoch != self->epoch_ || self->phase_ != Phase::electing) return;
                      if (err) {
                        self->failAndResyncLocked(("election failed: " + *err).c_str());
                        return;
                      }
                      self->leader_ = rv->leaderName;
                      self->isLeader_ = (rv->leaderName == self->group_->myName());
                      MRL_LOG_INFO("accumulator '%s': leader is '%s' (version %lld)%s",
                                   self->name_.c_str(), self->leader_.c_str(),
                                   (long long)rv->version, self->isLeader_ ? " - that's me" : "");
                      if (self->isLeader_) {
                        self->phase_ = Phase::running;
                      } else {
                        self->phase_ = Phase::fetching;
                        self->modelRequestSent_ = false;
                      }
                    });
}

void Accumulator::update() {
  group_->update();
  bool active = group_->active();
  uint64_t sid = group_->syncId();

  // Hook poll closures capture python objects; their destruction can take
  // the GIL, so collect them here and let them die AFTER mu_ is released
  // (declared before lk: destroyed after it).
  std::vector<std::function<bool()>> pollTombs;
  std::unique_lock<std::mutex> lk(mu_);
  // Drain abandoned hook collectives regardless of phase.
  for (auto it = drains_.begin(); it != drains_.end();) {
    bool done = true;
    try {
      done = (*it)();
    } catch (const std::exception& e) {
      MRL_LOG_ERROR("abandoned hook drain failed: %s", e.what());
    }
    if (done) {
      pollTombs.push_back(std::move(*it));
      it = drains_.erase(it);
    } else {
      ++it;
    }
  }

  if (!active) {
    if (phase_ != Phase::inactive) resetLocked("group inactive");
    phase_ = Phase::inactive;
    syncSeen_ = 0;
    return;
  }
  if (sid != syncSeen_) {
    syncSeen_ = sid;
    resetLocked("membership changed");
    startElectionLocked();
    return;
  }

  if (phase_ == Phase::fetching) {
    if (!modelRequestSent_ || secondsSince(fetchStarted_) > 5.0) {
      modelRequestSent_ = true;
      fetchStarted_ = now();
      std::string leader = leader_;
      rpc_->sendRequest(leader, fn("reqmodel"), "", {},
                        [](Frame* resp, const std::string* err) {
                          if (err) MRL_LOG_VERBOSE("requestModel failed: %s", err->c_str());
                        },
                        10.0);
    }
  }

  if (havePendingModel_ && (phase_ == Phase::fetching || phase_ == Phase::running)) {
    NoGrad ng;
    bool shapesOk = pendingParams_.size() == allParams_.size() &&
                    pendingBuffers_.size() == buffers_.size();
    if (shapesOk) {
      for (size_t i = 0; i < allParams_.size(); ++i) {
        allParams_[i].copy_(pendingParams_[i], /*non_blocking=*/true);
      }
      for (size_t i = 0; i < buffers_.size(); ++i) {
        buffers_[i].copy_(pendingBuffers_[i], /*non_blocking=*/true);
      }
      modelVersion_ = pendingVersion_;
      // A bare weight broadcast (empty payload, no state tensors) adopts
      // silently; only a real state transfer surfaces through has_new_state.
      hasNewState_ = !pendingStatePayload_.empty() || !pendingStateTensors_.empty();
      phase_ = Phase::running;
      MRL_LOG_INFO("accumulator '%s': model adopted at version %lld", name_.c_str(),
                   (long long)modelVersion_);
    } else {
      MRL_LOG_ERROR("accumulator '%s': model update shape mismatch — ignored", name_.c_str());
    }
    havePendingModel_ = false;
    pendingParams_.clear();
    pendingBuffers_.clear();
  }

  if (phase_ == Phase::running) {
    // Watchdog: a wedged reduction forces a cluster-wide resync.
    for (auto& s : slots_) {
      if (s.phase == GradPhase::reducing && secondsSince(s.started) > 120.0) {
        MRL_LOG_ERROR("accumulator '%s': gradient reduce wedged >120s", name_.c_str());
        failAndResyncLocked("reduce wedged");
        return;
      }
    }
    GradSlot& cur = slots_[slotCursor_];
    if (cur.phase == GradPhase::wantDecision && decided_) {
      startCountRoundLocked();
    }
    tryLaunchHooksLocked();
    // Poll launched hook collectives.
    for (size_t si = 0; si < slots_.size(); ++si) {
      GradSlot& s = slots_[si];
      if (s.phase == GradPhase::reducing && s.hookPoll && !s.hookPending) {
        bool done = false;
        try {
          done = s.hookPoll();
        } catch (const std::exception& e) {
          pollTombs.push_back(std::move(s.hookPoll));
          s.hookPoll = nullptr;
          MRL_LOG_ERROR("local reduce hook poll failed: %s", e.what());
          failAndResyncLocked("hook poll failed");
          return;
        }
        if (done) {
          pollTombs.push_back(std::move(s.hookPoll));
          s.hookPoll = nullptr;
          NoGrad ng;
          at::Tensor result = s.flat.clone();
          completeSlotLocked(si, std::move(result));
        }
      }
    }
    applyPendingLocked();
    maybeSendModelUpdatesLocked();
  }
}

void Accumulator::startCountRoundLocked() {
  size_t si = slotCursor_;
  GradSlot& s = slots_[si];
  s.phase = GradPhase::counting;
  s.started = now();
  ReduceValue v;
  v.kind = ReduceValue::counts;
  v.batchSize = s.newBatch;
  v.numGradients = s.newGrads;
  v.numSkipped = s.newSkipped;
  s.newBatch = s.newGrads = s.newSkipped = 0;
  uint64_t epoch = epoch_;
  std::weak_ptr<Accumulator> weak = weak_from_this();
  group_->allReduce(
      fn("count") + "/" + std::to_string(si), std::move(v), nullptr,
      [weak, epoch, si](ReduceValue* rv, const std::string* err) {
        auto self = weak.lock();
        if (!self) return;
        std::lock_guard<std::mutex> lk(self->mu_);
        if (epoch != self->epoch_) return;
        GradSlot& s = self->slots_[si];
        if (s.phase != GradPhase::counting) return;
        if (err) {
          self->failAndResyncLocked(("count round failed: " + *err).c_str());
          return;
        }
        s.totBatch += rv->batchSize;
        s.totGrads += rv->numGradients;
        s.totSkipped += rv->numSkipped;
        self->decided_ = false;
        if (s.totBatch >= self->virtualBatchSize_ && s.totGrads > 0) {
          self->enterReducingLocked(si);
        } else {
          s.phase = GradPhase::wantDecision;
        }
      });
}

// The slot's virtual batch is full: start the gradient reduction and move
// the cursor to the next free slot (pipelining; reductions against up to
// slots_.size()-1 stale models, = moolib's set_parallel_gradients).
void Accumulator::enterReducingLocked(size_t si) {
  GradSlot& s = slots_[si];
  s.phase = GradPhase::reducing;
  s.started = now();
  slotCursor_ = (si + 1) % slots_.size();
  if (hook_) {
    // Launch order of collectives must be identical on every peer: slot
    // transitions are (count rounds are sequential cluster-wide), so a
    // ticket taken at transition time is globally consistent.
    s.launchSeq = hookLaunchCounter_++;
    s.hookPending = true;
    tryLaunchHooksLocked();
    return;
  }
  // RPC tree path: ship the bucket (staged to CPU) through the group
  // allreduce. Cross-node / elastic fallback.
  ReduceValue v;
  v.kind = ReduceValue::gradBundle;
  {
    NoGrad ng;
    v.tensors.push_back(s.flat.device().is_cpu() ? s.flat.clone() : s.flat.to(at::kCPU));
  }
  uint64_t epoch = epoch_;
  std::weak_ptr<Accumulator> weak = weak_from_this();
  group_->allReduce(
      fn("grads") + "/" + std::to_string(si), std::move(v), nullptr,
      [weak, epoch, si](ReduceValue* rv, const std::string* err) {
        auto self = weak.lock();
        if (!self) return;
        std::lock_guard<std::mutex> lk(self->mu_);
        if (epoch != self->epoch_) return;
        GradSlot& s = self->slots_[si];
        if (s.phase != GradPhase::reducing) return;
        if (err) {
          self->failAndResyncLocked(("gradient reduce failed: " + *err).c_str());
          return;
        }
        NoGrad ng;
        at::Tensor result = rv->tensors.at(0);
        if (!s.flat.device().is_cpu()) result = result.to(s.flat.device(), /*non_blocking=*/true);
        self->completeSlotLocked(si, std::move(result));
      });
}

void Accumulator::tryLaunchHooksLocked() {
  if (!hook_ || !drains_.empty()) return;
  // Launch in ticket order; one per call is fine (update runs every iter).
  for (size_t si = 0; si < slots_.size(); ++si) {
    GradSlot& s = slots_[si];
    if (s.phase == GradPhase::reducing && s.hookPending && s.launchSeq == hookLaunchedUpTo_) {
      s.hookPending = false;
      ++hookLaunchedUpTo_;
      try {
        s.hookPoll = hook_(s.flat);
      } catch (const std::exception& e) {
        MRL_LOG_ERROR("local reduce hook failed: %s", e.what());
        failAndResyncLocked("hook launch failed");
        return;
      }
    }
  }
}

void Accumulator::completeSlotLocked(size_t si, at::Tensor result) {
  GradSlot& s = slots_[si];
  results_.push_back(PendingResult{std::move(result), s.totBatch, s.totGrads, s.totSkipped});
  s.totBatch = s.totGrads = s.totSkipped = 0;
  s.phase = GradPhase::wantDecision;
  {
    NoGrad ng;
    s.flat.zero_();
  }
  // NOTE: the result is NOT applied here. Completion runs on a scheduler
  // thread at an arbitrary point in the user's iteration; writing .grad now
  // could clobber a backward() in progress (possible when parallel slots
  // keep wants_gradients() true while reductions are in flight). Results
  // apply only from update() / zero_gradients(), which the cooperative
  // loop calls outside its backward/reduce block.
}

void Accumulator::applyPendingLocked() {
  if (hasGradients_ || results_.empty()) return;
  PendingResult r = std::move(results_.front());
  results_.pop_front();
  NoGrad ng;
  int64_t n = std::max<int64_t>(r.grads, 1);
  r.flat.div_(static_cast<double>(n));
  for (size_t i = 0; i < params_.size(); ++i) {
    auto& p = params_[i];
    at::Tensor slice = r.flat.narrow(0, offsets_[i], numels_[i]).view(p.sizes());
    if (!p.grad().defined()) {
      p.mutable_grad() = slice.clone();
    } else {
      p.grad().copy_(slice, /*non_blocking=*/true);
    }
  }
  statBatch_ = r.batch;
  statGrads_ = r.grads;
  statSkipped_ = r.skipped;
  modelVersion_ += 1;
  hasGradients_ = true;
}

void Accumulator::maybeSendModelUpdatesLocked() {
  if (!isLeader_) return;
  if (!buffers_.empty() && secondsSince(lastBuffersBroadcast_) >= 10.0) {
    lastBuffersBroadcast_ = now();
    std::vector<at::Tensor> cpuBuffers;
    {
      NoGrad ng;
      for (auto& b : buffers_) cpuBuffers.push_back(b.detach().to(at::kCPU));
    }
    for (auto& m : group_->members()) {
      if (m == group_->myName()) continue;
      rpc_->sendRequest(m, fn("buffers"), "", cpuBuffers,
                        [](Frame*, const std::string*) {}, 10.0);
    }
  }
  // Periodic full-model broadcast (drift correction; the reference sends
  // full state every 600 s, accumulator.cc:744-768). Params+buffers only —
  // user state (optimizer etc.) still flows through the requestModel path.
  // An empty state payload means "adopt weights silently" on the receiver.
  if (lastModelBroadcast_.time_since_epoch().count() == 0) lastModelBroadcast_ = now();
  if (secondsSince(lastModelBroadcast_) >= modelBcastInterval_) {
    lastModelBroadcast_ = now();
    std::vector<at::Tensor> frameTensors;
    {
      NoGrad ng;
      for (auto& p : allParams_) frameTensors.push_back(p.detach().to(at::kCPU));
      for (auto& b : buffers_) frameTensors.push_back(b.detach().to(at::kCPU));
    }
    WireWriter w;
    w.i64(modelVersion_);
    w.u32(static_cast<uint32_t>(allParams_.size()));
    w.u32(static_cast<uint32_t>(buffers_.size()));
    w.str("");
    for (auto& m : group_->members()) {
      if (m == group_->myName()) continue;
      rpc_->sendRequest(m, fn("modelupd"), w.out, frameTensors,
                        [](Frame*, const std::string*) {}, 30.0);
    }
  }
}

bool Accumulator::connected() {
  std::lock_guard<std::mutex> lk(mu_);
  return phase_ == Phase::running;
}

bool Accumulator::wantsState() {
  std::lock_guard<std::mutex> lk(mu_);
  return phase_ == Phase::running && isLeader_ && !stateRequesters_.empty();
}

bool Accumulator::hasNewState() {
  std::lock_guard<std::mutex> lk(mu_);
  return hasNewState_;
}

void Accumulator::setState(std::string payload, std::vector<at::Tensor> tensors) {
  std::vector<std::string> requesters;
  int64_t version;
  std::vector<at::Tensor> frameTensors;
  uint32_t nParams, nBuffers;
  {
    std::lock_guard<std::mutex> lk(mu_);
    requesters.swap(stateRequesters_);
    version = modelVersion_;
    NoGrad ng;
    for (auto& p : allParams_) frameTensors.push_back(p.detach().to(at::kCPU));
    for (auto& b : buffers_) frameTensors.push_back(b.detach().to(at::kCPU));
    nParams = static_cast<uint32_t>(allParams_.size());
    nBuffers = static_cast<uint32_t>(buffers_.size());
  }
  for (auto& t : tensors) frameTensors.push_back(t.device().is_cpu() ? t : t.to(at::kCPU));
  WireWriter w;
  w.i64(version);
  w.u32(nParams);
  w.u32(nBuffers);
  w.str(payload);
  for (auto& r : requesters) {
    rpc_->sendRequest(r, fn("modelupd"), w.out, frameTensors,
                      [r](Frame*, const std::string* err) {
                        if (err) {
                          MRL_LOG_INFO("model update to '%s' failed: %s", r.c_str(), err->c_str());
                        }
                      },
                      30.0);
  }
}

std::pair<std::string, std::vector<at::Tensor>> Accumulator::state() {
  std::lock_guard<std::mutex> lk(mu_);
  hasNewState_ = false;
  return {pendingStatePayload_, pendingStateTensors_};
}

bool Accumulator::wantsGradients() {
  std::lock_guard<std::mutex> lk(mu_);
  if (phase_ != Phase::running || decided_) return false;
  if (slots_[slotCursor_].phase != GradPhase::wantDecision) return false;
  // Bound unapplied work: at most slots_.size() results outstanding.
  size_t pending = results_.size() + (hasGradients_ ? 1 : 0);
  return pending < slots_.size();
}

bool Accumulator::hasGradients() {
  std::lock_guard<std::mutex> lk(mu_);
  return hasGradients_;
}

void Accumulator::skipGradients() {
  std::lock_guard<std::mutex> lk(mu_);
  if (phase_ != Phase::running || decided_) return;
  GradSlot& s = slots_[slotCursor_];
  if (s.phase != GradPhase::wantDecision) return;
  decided_ = true;
  s.newSkipped += 1;
}

void Accumulator::reduceGradients(int64_t batchSize) {
  std::lock_guard<std::mutex> lk(mu_);
  GradSlot& s = slots_[slotCursor_];
  if (phase_ != Phase::running || s.phase != GradPhase::wantDecision || decided_) {
    throw RpcError("reduce_gradients called when wants_gradients() is false");
  }
  NoGrad ng;
  // Multi-tensor: one launch folds every param grad into its bucket slice
  // and one zeroes them (vs 2 launches per param — 40 tiny latency-bound
  // kernels each on the IMPALA model).
  std::vector<at::Tensor> slices, flatGrads, grads;
  slices.reserve(params_.size());
  flatGrads.reserve(params_.size());
  grads.reserve(params_.size());
  for (size_t i = 0; i < params_.size(); ++i) {
    auto& p = params_[i];
    if (p.grad().defined()) {
      slices.push_back(s.flat.narrow(0, offsets_[i], numels_[i]));
      flatGrads.push_back(p.grad().flatten());  // view for contiguous grads
      grads.push_back(p.grad());                // originals for the zero
    }
  }
  if (!grads.empty()) {
    at::_foreach_add_(slices, flatGrads);
    at::_foreach_zero_(grads);
  }
  decided_ = true;
  s.newBatch += batchSize;
  s.newGrads += 1;
}

void Accumulator::zeroGradients() {
  std::lock_guard<std::mutex> lk(mu_);
  NoGrad ng;
  std::vector<at::Tensor> grads;
  for (auto& p : params_) {
    if (p.grad().defined()) grads.push_back(p.grad());
  }
  if (!grads.empty()) at::_foreach_zero_(grads);
  hasGradients_ = false;
  applyPendingLocked();
}

int64_t Accumulator::modelVersion() {
  std::lock_guard<std::mutex> lk(mu_);
  return modelVersion_;
}

void Accumulator::setModelVersion(int64_t v) {
  std::lock_guard<std::mutex> lk(mu_);
  modelVersion_ = v;
}

std::string Accumulator::getLeader() {
  std::lock_guard<std::mutex> lk(mu_);
  return leader_;
}

bool Accumulator::isLeader() {
  std::lock_guard<std::mutex> lk(mu_);
  return isLeader_;
}

std::unordered_map<std::string, int64_t> Accumulator::gradientStats() {
  std::lock_guard<std::mutex> lk(mu_);
  return {{"batch_size", statBatch_}, {"num_gradients", statGrads_}, {"num_skipped", statSkipped_}};
}

void Accumulator::setVirtualBatchSize(int64_t n) {
  std::lock_guard<std::mutex> lk(mu_);
  virtualBatchSize_ = std::max<int64_t>(n, 1);
}

void Accumulator::setParallelGradients(int64_t n) {
  std::lock_guard<std::mutex> lk(mu_);
  n = std::max<int64_t>(n, 1);
  if (static_cast<size_t>(n) == slots_.size()) return;
  // Resize between rounds only; all members must configure the same value
  // before training (mirrors the reference's usage).
  slots_.resize(n);
  for (auto& s : slots_) {
    if (!s.flat.defined()) s.flat = makeFlatLocked();
  }
}

void Accumulator::setLocalReduceHook(LocalReduceHook h) {
  LocalReduceHook old;
  {
    std::lock_guard<std::mutex> lk(mu_);
    old = std::move(hook_);
    hook_ = std::move(h);
  }
  // old (py-capturing) destroyed here, outside mu_
}

std::string Accumulator::debugState() {
  std::lock_guard<std::mutex> lk(mu_);
  std::ostringstream os;
  os << "Accumulator '" << name_ << "' phase=" << static_cast<int>(phase_)
     << " leader='" << leader_ << "' version=" << modelVersion_ << " decided=" << decided_
     << " cursor=" << slotCursor_ << " hasGrads=" << hasGradients_
     << " results=" << results_.size() << " slots=[";
  for (auto& s : slots_) {
    os << static_cast<int>(s.phase) << "(tb=" << s.totBatch << ")";
  }
  os << "]";
  return os.str();
}

}  // namespace mrl
