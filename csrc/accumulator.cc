#include "accumulator.h"

#include <ATen/Functions.h>
#include <c10/core/GradMode.h>

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
  // One flat on-device bucket sized for every requires-grad parameter. On
  // MI355X this is the tensor RCCL reduces over xGMI; 288 GB HBM3E makes a
  // single resident bucket the right default even for very large models.
  int64_t total = 0;
  for (auto& p : params_) {
    offsets_.push_back(total);
    numels_.push_back(p.numel());
    total += p.numel();
  }
  if (!params_.empty()) {
    NoGrad ng;
    flat_ = at::zeros({std::max<int64_t>(total, 1)},
                      at::TensorOptions().dtype(params_[0].scalar_type()).device(params_[0].device()));
  } else {
    flat_ = at::zeros({1});
  }
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
      if (nParams + nBuffers > f.tensors.size()) throw RpcError("model update tensor shortfall");
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
  gradPhase_ = GradPhase::wantDecision;
  decided_ = false;
  newBatch_ = newGrads_ = newSkipped_ = 0;
  totBatch_ = totGrads_ = totSkipped_ = 0;
  hasGradients_ = false;
  hookStartPending_ = false;
  if (hookPoll_) {
    // Never abandon an in-flight collective: the peers' call sequences must
    // stay aligned. Keep polling it to completion (drain), then discard.
    hookAbandoned_ = true;
  } else {
    NoGrad ng;
    flat_.zero_();
  }
  havePendingModel_ = false;
  hasNewState_ = false;
  stateRequesters_.clear();
  modelRequestSent_ = false;
  leader_.clear();
  isLeader_ = false;
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
                        MRL_LOG_INFO("accumulator '%s': election failed (%s); will retry",
                                     self->name_.c_str(), err->c_str());
                        self->phase_ = Phase::inactive;
                        self->syncSeen_ = 0;
                        self->group_->requestResync();
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

  std::unique_lock<std::mutex> lk(mu_);
  // Drain an abandoned hook collective regardless of phase: the peers'
  // collective call sequences must stay aligned, so it is polled to
  // completion and its (mixed-round) result discarded.
  if (hookPoll_ && hookAbandoned_) {
    bool done = true;
    try {
      done = hookPoll_();
    } catch (const std::exception& e) {
      MRL_LOG_ERROR("abandoned hook drain failed: %s", e.what());
    }
    if (done) {
      hookPoll_ = nullptr;
      hookAbandoned_ = false;
      NoGrad ng;
      flat_.zero_();
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
      hasNewState_ = true;
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
    if (gradPhase_ == GradPhase::reducing &&
        secondsSince(gradPhaseStarted_) > 120.0) {
      MRL_LOG_ERROR("accumulator '%s': gradient reduce wedged >120s; forcing resync",
                    name_.c_str());
      if (hookPoll_) hookAbandoned_ = true;
      phase_ = Phase::inactive;
      syncSeen_ = 0;
      group_->requestResync();
      return;
    }
    if (gradPhase_ == GradPhase::wantDecision && decided_) {
      startCountRoundLocked();
    } else if (gradPhase_ == GradPhase::reducing && hookStartPending_ && !hookPoll_) {
      hookStartPending_ = false;
      try {
        hookPoll_ = hook_(flat_);
      } catch (const std::exception& e) {
        MRL_LOG_ERROR("local reduce hook failed: %s", e.what());
        phase_ = Phase::inactive;
        syncSeen_ = 0;
        group_->requestResync();
        return;
      }
    } else if (gradPhase_ == GradPhase::reducing && hookPoll_ && !hookAbandoned_) {
      bool done = false;
      try {
        done = hookPoll_();
      } catch (const std::exception& e) {
        MRL_LOG_ERROR("local reduce hook poll failed: %s", e.what());
        hookPoll_ = nullptr;
        phase_ = Phase::inactive;
        syncSeen_ = 0;
        group_->requestResync();
        return;
      }
      if (done) {
        hookPoll_ = nullptr;
        applyGradResultLocked(flat_);
      }
    }
    maybeSendModelUpdatesLocked();
  }
}

void Accumulator::startCountRoundLocked() {
  gradPhase_ = GradPhase::counting;
  gradPhaseStarted_ = now();
  ReduceValue v;
  v.kind = ReduceValue::counts;
  v.batchSize = newBatch_;
  v.numGradients = newGrads_;
  v.numSkipped = newSkipped_;
  newBatch_ = newGrads_ = newSkipped_ = 0;
  uint64_t epoch = epoch_;
  std::weak_ptr<Accumulator> weak = weak_from_this();
  group_->allReduce(fn("count"), std::move(v), nullptr,
                    [weak, epoch](ReduceValue* rv, const std::string* err) {
                      auto self = weak.lock();
                      if (!self) return;
                      std::lock_guard<std::mutex> lk(self->mu_);
                      if (epoch != self->epoch_ || self->gradPhase_ != GradPhase::counting) return;
                      if (err) {
                        MRL_LOG_INFO("accumulator '%s': count round failed (%s); resync",
                                     self->name_.c_str(), err->c_str());
                        self->phase_ = Phase::inactive;
                        self->syncSeen_ = 0;
                        self->group_->requestResync();
                        return;
                      }
                      self->totBatch_ += rv->batchSize;
                      self->totGrads_ += rv->numGradients;
                      self->totSkipped_ += rv->numSkipped;
                      self->decided_ = false;
                      if (self->totBatch_ >= self->virtualBatchSize_ && self->totGrads_ > 0) {
                        self->startGradReduceLocked();
                      } else {
                        self->gradPhase_ = GradPhase::wantDecision;
                      }
                    });
}

void Accumulator::startGradReduceLocked() {
  gradPhase_ = GradPhase::reducing;
  gradPhaseStarted_ = now();
  if (hook_) {
    if (hookPoll_) {
      // Previous (abandoned) collective still draining: start ours once it
      // completes (update() watches hookStartPending_).
      hookStartPending_ = true;
      return;
    }
    // MI355X fast path: in-place sum over the fixed torch.distributed world
    // (RCCL over xGMI). Peers that skipped contribute zeros.
    try {
      hookPoll_ = hook_(flat_);
    } catch (const std::exception& e) {
      MRL_LOG_ERROR("local reduce hook failed: %s", e.what());
      phase_ = Phase::inactive;
      syncSeen_ = 0;
      group_->requestResync();
    }
    return;
  }
  // RPC tree path: ship the flat bucket (staged to CPU) through the group
  // allreduce. Cross-node / elastic fallback.
  ReduceValue v;
  v.kind = ReduceValue::gradBundle;
  v.tensors.push_back(flat_.device().is_cpu() ? flat_.clone() : flat_.to(at::kCPU));
  uint64_t epoch = epoch_;
  std::weak_ptr<Accumulator> weak = weak_from_this();
  group_->allReduce(fn("grads"), std::move(v), nullptr,
                    [weak, epoch](ReduceValue* rv, const std::string* err) {
                      auto self = weak.lock();
                      if (!self) return;
                      std::lock_guard<std::mutex> lk(self->mu_);
                      if (epoch != self->epoch_ || self->gradPhase_ != GradPhase::reducing) return;
                      if (err) {
                        MRL_LOG_INFO("accumulator '%s': gradient reduce failed (%s); resync",
                                     self->name_.c_str(), err->c_str());
                        self->phase_ = Phase::inactive;
                        self->syncSeen_ = 0;
                        self->group_->requestResync();
                        return;
                      }
                      NoGrad ng;
                      at::Tensor result = rv->tensors.at(0);
                      if (!self->flat_.device().is_cpu()) {
                        self->flat_.copy_(result, /*non_blocking=*/true);
                        self->applyGradResultLocked(self->flat_);
                      } else {
                        // (Group::completeOp already cloned the tensors, so
                        // in-place mutation here is safe.)
                        self->applyGradResultLocked(result);
                      }
                    });
}

void Accumulator::applyGradResultLocked(at::Tensor flatResult) {
  NoGrad ng;
  int64_t n = std::max<int64_t>(totGrads_, 1);
  flatResult.div_(static_cast<double>(n));
  for (size_t i = 0; i < params_.size(); ++i) {
    auto& p = params_[i];
    at::Tensor slice = flatResult.narrow(0, offsets_[i], numels_[i]).view(p.sizes());
    if (!p.grad().defined()) {
      p.mutable_grad() = slice.clone();
    } else {
      p.grad().copy_(slice, /*non_blocking=*/true);
    }
  }
  statBatch_ = totBatch_;
  statGrads_ = totGrads_;
  statSkipped_ = totSkipped_;
  totBatch_ = totGrads_ = totSkipped_ = 0;
  modelVersion_ += 1;
  hasGradients_ = true;
  gradPhase_ = GradPhase::resultReady;
}

void Accumulator::maybeSendModelUpdatesLocked() {
  if (!isLeader_ || buffers_.empty()) return;
  if (secondsSince(lastBuffersBroadcast_) < 10.0) return;
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
  return phase_ == Phase::running && gradPhase_ == GradPhase::wantDecision && !decided_ &&
         !hasGradients_;
}

bool Accumulator::hasGradients() {
  std::lock_guard<std::mutex> lk(mu_);
  return hasGradients_;
}

void Accumulator::skipGradients() {
  std::lock_guard<std::mutex> lk(mu_);
  if (phase_ != Phase::running || gradPhase_ != GradPhase::wantDecision || decided_) return;
  decided_ = true;
  newSkipped_ += 1;
}

void Accumulator::reduceGradients(int64_t batchSize) {
  std::lock_guard<std::mutex> lk(mu_);
  if (phase_ != Phase::running || gradPhase_ != GradPhase::wantDecision || decided_) {
    throw RpcError("reduce_gradients called when wants_gradients() is false");
  }
  NoGrad ng;
  for (size_t i = 0; i < params_.size(); ++i) {
    auto& p = params_[i];
    if (p.grad().defined()) {
      flat_.narrow(0, offsets_[i], numels_[i]).add_(p.grad().flatten());
      p.grad().zero_();
    }
  }
  decided_ = true;
  newBatch_ += batchSize;
  newGrads_ += 1;
}

void Accumulator::zeroGradients() {
  std::lock_guard<std::mutex> lk(mu_);
  NoGrad ng;
  for (auto& p : params_) {
    if (p.grad().defined()) p.grad().zero_();
  }
  flat_.zero_();
  hasGradients_ = false;
  if (gradPhase_ == GradPhase::resultReady) gradPhase_ = GradPhase::wantDecision;
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
  parallelGradients_ = std::max<int64_t>(n, 1);
}

void Accumulator::setLocalReduceHook(LocalReduceHook h) {
  std::lock_guard<std::mutex> lk(mu_);
  hook_ = std::move(h);
}

std::string Accumulator::debugState() {
  std::lock_guard<std::mutex> lk(mu_);
  std::ostringstream os;
  os << "Accumulator '" << name_ << "' phase=" << static_cast<int>(phase_)
     << " gradPhase=" << static_cast<int>(gradPhase_) << " leader='" << leader_ << "'"
     << " version=" << modelVersion_ << " decided=" << decided_ << " totBatch=" << totBatch_
     << " hasGrads=" << hasGradients_;
  return os.str();
}

}  // namespace mrl
