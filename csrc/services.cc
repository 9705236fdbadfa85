#include "services.h"

#include <algorithm>

namespace mrl {

// --------------------------------------------------------- ReduceValue

void ReduceValue::encode(WireWriter& w) const {
  w.u8(static_cast<uint8_t>(kind));
  switch (kind) {
    case pyObject:
      w.str(payload);
      break;
    case counts:
      w.i64(batchSize);
      w.i64(numGradients);
      w.i64(numSkipped);
      break;
    case gradBundle:
      w.i64(batchSize);
      w.i64(numGradients);
      w.i64(numSkipped);
      break;
    case leaderTuple:
      w.i64(version);
      w.str(leaderName);
      break;
  }
}

ReduceValue ReduceValue::decode(WireReader& r, std::vector<at::Tensor> tensors) {
  ReduceValue v;
  v.kind = static_cast<Kind>(r.u8());
  switch (v.kind) {
    case pyObject:
      v.payload = std::string(r.str());
      v.tensors = std::move(tensors);
      break;
    case counts:
      v.batchSize = r.i64();
      v.numGradients = r.i64();
      v.numSkipped = r.i64();
      break;
    case gradBundle:
      v.batchSize = r.i64();
      v.numGradients = r.i64();
      v.numSkipped = r.i64();
      v.tensors = std::move(tensors);
      break;
    case leaderTuple:
      v.version = r.i64();
      v.leaderName = std::string(r.str());
      break;
  }
  return v;
}

void ReduceValue::fold(ReduceValue& src, const std::function<void(ReduceValue&, ReduceValue&)>& pyFold) {
  if (src.kind != kind) throw RpcError("allreduce: mismatched value kinds");
  switch (kind) {
    case pyObject:
      if (!pyFold) throw RpcError("allreduce: missing fold op for python values");
      pyFold(*this, src);
      break;
    case counts:
      batchSize += src.batchSize;
      numGradients += src.numGradients;
      numSkipped += src.numSkipped;
      break;
    case gradBundle:
      batchSize += src.batchSize;
      numGradients += src.numGradients;
      numSkipped += src.numSkipped;
      if (tensors.size() != src.tensors.size())
        throw RpcError("allreduce: gradient bundle tensor count mismatch");
      for (size_t i = 0; i < tensors.size(); ++i) {
        // add_ would silently BROADCAST a mismatched shape (e.g. a peer on
        // a different model revision) into corrupt gradients; refuse.
        if (!tensors[i].sizes().equals(src.tensors[i].sizes()))
          throw RpcError("allreduce: gradient tensor shape mismatch at index " + std::to_string(i));
        tensors[i].add_(src.tensors[i]);
      }
      break;
    case leaderTuple:
      if (std::tie(src.version, src.leaderName) > std::tie(version, leaderName)) {
        version = src.version;
        leaderName = src.leaderName;
      }
      break;
  }
}

// -------------------------------------------------------------- Broker

namespace {
constexpr const char* kBrokerPing = "__mrl_broker_ping";
constexpr const char* kBrokerResync = "__mrl_broker_resync";
}

void Broker::GroupState::resort() {
  if (!dirty) return;
  dirty = false;
  sortedCache.clear();
  std::vector<std::pair<std::pair<int64_t, uint64_t>, std::string>> v;
  for (auto& [name, m] : members) v.push_back({{m.sortOrder, m.joinSeq}, name});
  std::sort(v.begin(), v.end());
  for (auto& e : v) sortedCache.push_back(e.second);
}

Broker::Broker(RpcPtr rpc) : rpc_(std::move(rpc)) {
  rpc_->define(kBrokerResync, [this](Frame f, const std::string& fromPeer, RespondFn respond) {
    try {
      WireReader r(f.payload);
      std::string group(r.str());
      std::lock_guard<std::mutex> lk(mu_);
      // Rate-limit: concurrent failure reports need only one epoch bump.
      if (secondsSince(lastForcedResync_) > 0.5) {
        lastForcedResync_ = now();
        auto it = groups_.find(group);
        if (it != groups_.end()) {
          it->second.syncId++;
          it->second.dirty = true;
          it->second.resort();
          MRL_LOG_INFO("broker: forced resync of group '%s' by '%s' (syncId now %llu)",
                       group.c_str(), fromPeer.c_str(), (unsigned long long)it->second.syncId);
        }
      }
      respond("", {}, false);
    } catch (const std::exception& e) {
      respond(e.what(), {}, true);
    }
  });
  rpc_->define(kBrokerPing, [this](Frame f, const std::string& fromPeer, RespondFn respond) {
    try {
      WireReader r(f.payload);
      std::string group(r.str());
      std::string member(r.str());
      int64_t sortOrder = r.i64();
      double timeout = r.f64();
      (void)fromPeer;
      WireWriter w;
      {
        std::lock_guard<std::mutex> lk(mu_);
        GroupState& g = groups_[group];
        if (timeout > 0) g.timeout = timeout;
        evictStaleLocked(g);
        auto it = g.members.find(member);
        if (it == g.members.end()) {
          Member m;
          m.lastPing = now();
          m.sortOrder = sortOrder;
          m.joinSeq = g.joinCounter++;
          g.members[member] = m;
          g.syncId++;
          g.dirty = true;
          MRL_LOG_INFO("broker: '%s' joined group '%s' (syncId now %llu, %zu members)",
                       member.c_str(), group.c_str(), (unsigned long long)g.syncId,
                       g.members.size());
        } else {
          it->second.lastPing = now();
          if (it->second.sortOrder != sortOrder) {
            it->second.sortOrder = sortOrder;
            g.syncId++;
            g.dirty = true;
          }
        }
        g.resort();
        w.u64(g.syncId);
        w.u32(static_cast<uint32_t>(g.sortedCache.size()));
        for (auto& n : g.sortedCache) w.str(n);
      }
      respond(std::move(w.out), {}, false);
    } catch (const std::exception& e) {
      respond(std::string("broker error: ") + e.what(), {}, true);
    }
  });
}

Broker::~Broker() {
  if (rpc_ && !rpc_->isShutdown()) {
    rpc_->undefine(kBrokerPing);
    rpc_->undefine(kBrokerResync);
  }
}

void Broker::setName(const std::string& n) { rpc_->setName(n); }
void Broker::listen(const std::string& addr) { rpc_->listen(addr); }

void Broker::evictStaleLocked(GroupState& g) {
  for (auto it = g.members.begin(); it != g.members.end();) {
    if (secondsSince(it->second.lastPing) > g.timeout) {
      MRL_LOG_INFO("broker: evicting stale member '%s'", it->first.c_str());
      it = g.members.erase(it);
      g.syncId++;
      g.dirty = true;
    } else {
      ++it;
    }
  }
  g.resort();
}

void Broker::update() {
  std::lock_guard<std::mutex> lk(mu_);
  for (auto& [name, g] : groups_) evictStaleLocked(g);
}

// --------------------------------------------------------------- Group

std::shared_ptr<Group> Group::create(RpcPtr rpc, std::string name) {
  std::shared_ptr<Group> g(new Group(std::move(rpc), std::move(name)));
  g->setup();
  return g;
}

Group::Group(RpcPtr rpc, std::string name) : rpc_(std::move(rpc)), name_(std::move(name)) {}

void Group::setup() {
  std::weak_ptr<Group> weak = shared_from_this();
  rpc_->define(funcUp(), [weak](Frame f, const std::string& from, RespondFn respond) {
    if (auto g = weak.lock()) g->handleContribution(false, f, from, respond);
  });
  rpc_->define(funcDown(), [weak](Frame f, const std::string& from, RespondFn respond) {
    if (auto g = weak.lock()) g->handleContribution(true, f, from, respond);
  });
}

Group::~Group() {
  std::vector<AllReduceOpPtr> toFail;
  {
    std::lock_guard<std::mutex> lk(mu_);
    stopped_ = true;
    for (auto& [k, op] : ops_) {
      if (op->done && !op->completed) {
        op->completed = true;
        toFail.push_back(op);
      }
    }
    ops_.clear();
  }
  for (auto& op : toFail) {
    std::string err = "group destroyed";
    op->done(nullptr, &err);
  }
  if (rpc_ && !rpc_->isShutdown()) {
    rpc_->undefine(funcUp());
    rpc_->undefine(funcDown());
  }
}

std::vector<std::string> Group::members() {
  std::lock_guard<std::mutex> lk(mu_);
  return members_;
}
uint64_t Group::syncId() {
  std::lock_guard<std::mutex> lk(mu_);
  return syncId_;
}
bool Group::active() {
  std::lock_guard<std::mutex> lk(mu_);
  return active_;
}

void Group::update() {
  // 1. Ping the broker (rate-limited).
  bool sendPing = false;
  {
    std::lock_guard<std::mutex> lk(mu_);
    double interval = active_ ? 0.5 : 0.2;
    if (!pingInFlight_ && secondsSince(lastPing_) >= interval) {
      pingInFlight_ = true;
      lastPing_ = now();
      sendPing = true;
    }
  }
  if (sendPing) {
    WireWriter w;
    w.str(name_);
    w.str(myName());
    w.i64(sortOrder_.load());
    w.f64(timeout_.load());
    std::weak_ptr<Group> weak = shared_from_this();
    rpc_->sendRequest(
        brokerName_, kBrokerPing, std::move(w.out), {},
        [weak](Frame* resp, const std::string* error) {
          auto g = weak.lock();
          if (!g) return;
          std::vector<std::pair<AllReduceOpPtr, OpKey>> toAdvance;
          std::vector<AllReduceOpPtr> toFail;
          std::string failMsg;
          {
            std::lock_guard<std::mutex> lk(g->mu_);
            g->pingInFlight_ = false;
            if (error) {
              if (g->active_ && secondsSince(g->lastPingOk_) > g->timeout_.load()) {
                g->active_ = false;
                MRL_LOG_INFO("group '%s': lost contact with broker (%s)", g->name_.c_str(),
                             error->c_str());
              } else if (!g->active_ && g->lastPingOk_.time_since_epoch().count() == 0 &&
                         secondsSince(g->lastUnreachableWarn_) > 10.0) {
                // Never reached the broker at all: almost always a wrong
                // broker NAME (set_broker_name; the peer must be literally
                // named that) or address. Silent forever is undebuggable.
                g->lastUnreachableWarn_ = now();
                MRL_LOG_ERROR(
                    "group '%s': cannot reach broker peer '%s' (%s) — check the broker's "
                    "rpc name and connect() address",
                    g->name_.c_str(), g->brokerName_.c_str(), error->c_str());
              }
              return;
            }
            g->lastPingOk_ = now();
            WireReader r(resp->payload);
            uint64_t syncId = r.u64();
            uint32_t n = r.u32();
            std::vector<std::string> members;
            for (uint32_t i = 0; i < n; ++i) members.emplace_back(r.str());
            bool inGroup =
                std::find(members.begin(), members.end(), g->myName()) != members.end();
            if (syncId != g->syncId_) {
              MRL_LOG_INFO("group '%s': syncId %llu -> %llu (%u members)", g->name_.c_str(),
                           (unsigned long long)g->syncId_, (unsigned long long)syncId, n);
              // Fail ops from older epochs; attach membership to queued newer ops.
              for (auto it = g->ops_.begin(); it != g->ops_.end();) {
                if (it->first.syncId < syncId) {
                  auto op = it->second;
                  if (!op->completed) {
                    op->completed = true;
                    if (op->done) toFail.push_back(op);
                  }
                  it = g->ops_.erase(it);
                } else {
                  ++it;
                }
              }
              failMsg = "group membership changed (syncId " + std::to_string(g->syncId_) +
                        " -> " + std::to_string(syncId) + ")";
              g->syncId_ = syncId;
              g->members_ = members;
              g->seqByName_.clear();
              g->active_ = inGroup;
              // Ops that arrived early for this syncId: fill in membership.
              for (auto& [key, op] : g->ops_) {
                if (key.syncId == syncId && op->childrenExpected < 0) {
                  op->members = members;
                  auto mit = std::find(members.begin(), members.end(), g->myName());
                  op->myIndex = mit == members.end() ? -1 : int(mit - members.begin());
                  int n2 = static_cast<int>(members.size());
                  int c = 0;
                  if (op->myIndex >= 0) {
                    if (2 * op->myIndex + 1 < n2) ++c;
                    if (2 * op->myIndex + 2 < n2) ++c;
                  }
                  op->childrenExpected = c;
                  toAdvance.push_back({op, key});
                }
              }
            } else {
              g->active_ = inGroup;
            }
          }
          for (auto& op : toFail) op->done(nullptr, &failMsg);
          for (auto& [op, key] : toAdvance) g->advance(key, op);
        },
        timeout_.load());
  }
  // 2. Time out stale ops.
  std::vector<AllReduceOpPtr> timedOut;
  {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto it = ops_.begin(); it != ops_.end();) {
      if (!it->second->completed && now() >= it->second->deadline) {
        it->second->completed = true;
        if (it->second->done) timedOut.push_back(it->second);
        it = ops_.erase(it);
      } else {
        ++it;
      }
    }
  }
  std::string err = "allreduce timed out";
  for (auto& op : timedOut) op->done(nullptr, &err);
}

void Group::requestResync() {
  WireWriter w;
  w.str(name_);
  rpc_->sendRequest(brokerName_, kBrokerResync, std::move(w.out), {},
                    [](Frame*, const std::string*) {}, 10.0);
}

void Group::allReduce(const std::string& opName, ReduceValue value, PyFold fold, ReduceDone done) {
  OpKey key;
  AllReduceOpPtr op;
  {
    std::lock_guard<std::mutex> lk(mu_);
    if (!active_) {
      std::string err = "group not active";
      // Run completion inline — caller handles errors synchronously.
      op = nullptr;
      key = {};
      globalScheduler().run([done = std::move(done), err] { done(nullptr, &err); });
      return;
    }
    key.syncId = syncId_;
    key.opName = opName;
    key.seq = seqByName_[opName]++;
    auto& slot = ops_[key];
    if (!slot) slot = std::make_shared<AllReduceOp>();
    op = slot;
    op->localContributed = true;
    op->acc = std::move(value);
    op->pyFold = std::move(fold);
    op->done = std::move(done);
    op->deadline = now() + std::chrono::duration_cast<Clock::duration>(
                               std::chrono::duration<double>(timeout_.load()));
    op->members = members_;
    auto mit = std::find(members_.begin(), members_.end(), myName());
    op->myIndex = mit == members_.end() ? -1 : int(mit - members_.begin());
    int n = static_cast<int>(members_.size());
    int c = 0;
    if (op->myIndex >= 0) {
      if (2 * op->myIndex + 1 < n) ++c;
      if (2 * op->myIndex + 2 < n) ++c;
    }
    op->childrenExpected = c;
  }
  advance(key, op);
}

// Fold queued contributions and move the op forward. mu_ must NOT be held.
void Group::advance(const OpKey& key, AllReduceOpPtr op) {
  while (true) {
    std::vector<std::pair<std::string, ReduceValue>> batch;
    bool sendUpNow = false;
    bool completeNow = false;
    {
      std::lock_guard<std::mutex> lk(mu_);
      if (op->completed || op->folding || !op->localContributed || op->childrenExpected < 0) return;
      if (op->sentUp) return;  // late contribution: acc may be serializing on another thread
      if (!op->queued.empty()) {
        batch = std::move(op->queued);
        op->queued.clear();
        op->folding = true;
      } else if (op->childrenArrived >= op->childrenExpected && !op->sentUp) {
        op->sentUp = true;
        if (op->myIndex == 0) {
          completeNow = true;
        } else {
          sendUpNow = true;
        }
      } else {
        return;
      }
    }
    if (!batch.empty()) {
      std::string foldError;
      for (auto& [from, v] : batch) {
        try {
          op->acc->fold(v, op->pyFold);
        } catch (const std::exception& e) {
          foldError = e.what();
          break;
        }
      }
      {
        std::lock_guard<std::mutex> lk(mu_);
        op->folding = false;
        op->childrenArrived += static_cast<int>(batch.size());
      }
      if (!foldError.empty()) {
        completeOp(key, op, nullptr, "allreduce fold failed: " + foldError);
        return;
      }
      continue;  // loop: more queued / completion check
    }
    if (sendUpNow) {
      int parent = (op->myIndex - 1) / 2;
      sendValue(op->members[parent], false, key, *op->acc);
      return;
    }
    if (completeNow) {
      // Root: broadcast down, then complete locally.
      ReduceValue result = *op->acc;
      int n = static_cast<int>(op->members.size());
      if (2 * op->myIndex + 1 < n) sendValue(op->members[2 * op->myIndex + 1], true, key, result);
      if (2 * op->myIndex + 2 < n) sendValue(op->members[2 * op->myIndex + 2], true, key, result);
      completeOp(key, op, &result, "");
      return;
    }
  }
}

void Group::completeOp(const OpKey& key, AllReduceOpPtr op, ReduceValue* v, const std::string& err) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    if (op->completed) return;
    op->completed = true;
    ops_.erase(key);
  }
  if (op->done) {
    // Always deliver on a scheduler thread: an op can complete inline from
    // inside allReduce() (single-member group), and callers (Accumulator)
    // hold their own locks across allReduce().
    if (v) {
      auto val = std::make_shared<ReduceValue>(std::move(*v));
      // Clone tensors before delivery: the same storages may still be
      // referenced by in-flight zero-copy sends down the tree, and the
      // consumer is free to mutate the result in place.
      for (auto& t : val->tensors) t = t.clone();
      globalScheduler().run([done = std::move(op->done), val] { done(val.get(), nullptr); });
    } else {
      globalScheduler().run([done = std::move(op->done), err] { done(nullptr, &err); });
    }
  }
}

void Group::sendValue(const std::string& member, bool isDown, const OpKey& key,
                      const ReduceValue& v) {
  WireWriter w;
  w.u64(key.syncId);
  w.u64(key.seq);
  w.str(key.opName);
  v.encode(w);
  std::string groupName = name_;
  rpc_->sendRequest(
      member, isDown ? funcDown() : funcUp(), std::move(w.out), v.tensors,
      [groupName, member](Frame* resp, const std::string* error) {
        if (error) {
          MRL_LOG_VERBOSE("group '%s': send to '%s' failed: %s", groupName.c_str(),
                          member.c_str(), error->c_str());
        }
      },
      timeout_.load());
}

void Group::handleContribution(bool isDown, Frame& f, const std::string& from, RespondFn respond) {
  respond("", {}, false);  // transport-level ack; op completion flows via down messages
  try {
    WireReader r(f.payload);
    OpKey key;
    key.syncId = r.u64();
    key.seq = r.u64();
    key.opName = std::string(r.str());
    ReduceValue v = ReduceValue::decode(r, std::move(f.tensors));

    AllReduceOpPtr op;
    bool isRootResult = false;
    {
      std::lock_guard<std::mutex> lk(mu_);
      if (stopped_) return;
      if (key.syncId < syncId_) return;  // stale epoch: drop
      auto& slot = ops_[key];
      if (!slot) {
        slot = std::make_shared<AllReduceOp>();
        slot->deadline = now() + std::chrono::duration_cast<Clock::duration>(
                                     std::chrono::duration<double>(timeout_.load()));
        if (key.syncId == syncId_) {
          slot->members = members_;
          auto mit = std::find(members_.begin(), members_.end(), myName());
          slot->myIndex = mit == members_.end() ? -1 : int(mit - members_.begin());
          int n = static_cast<int>(members_.size());
          int c = 0;
          if (slot->myIndex >= 0) {
            if (2 * slot->myIndex + 1 < n) ++c;
            if (2 * slot->myIndex + 2 < n) ++c;
          }
          slot->childrenExpected = c;
        }  // else: future epoch — membership filled in when we adopt it
      }
      op = slot;
      if (!isDown) {
        op->queued.push_back({from, std::move(v)});
      } else {
        isRootResult = true;
      }
    }
    if (isRootResult) {
      // Forward down the tree, then complete.
      std::vector<std::string> fwd;
      {
        std::lock_guard<std::mutex> lk(mu_);
        if (op->completed) return;
        int n = static_cast<int>(op->members.size());
        if (op->myIndex >= 0) {
          if (2 * op->myIndex + 1 < n) fwd.push_back(op->members[2 * op->myIndex + 1]);
          if (2 * op->myIndex + 2 < n) fwd.push_back(op->members[2 * op->myIndex + 2]);
        }
      }
      for (auto& m : fwd) sendValue(m, true, key, v);
      completeOp(key, op, &v, "");
    } else {
      advance(key, op);
    }
  } catch (const std::exception& e) {
    MRL_LOG_ERROR("group '%s': bad contribution: %s", name_.c_str(), e.what());
  }
}

}  // namespace mrl
