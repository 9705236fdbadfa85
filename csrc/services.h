// Coordination services: Broker (membership directory), Group (synced member
// list with epoch syncId), AllReduce (binary-tree reduce+broadcast over RPC).
//
// Capability parity with the reference's src/broker.h and src/group.h.
// Protocol is our own: members poll the broker with pings (reply carries the
// current syncId + sorted member list) instead of the reference's push-based
// sync rounds; allreduce ops are keyed by (syncId, name, seq) and fold up a
// binary tree rooted at member 0, then broadcast down the same tree.
//
// Values reduced are either arbitrary Python objects (folded under the GIL
// with a user op) or C++-native bundles (counts / flat gradient buckets)
// folded without the GIL — the Accumulator's hot path never touches Python.
#pragma once

#include <optional>

#include "rpc.h"

namespace mrl {

// ------------------------------------------------------------------ value

// What travels through an allreduce.
struct ReduceValue {
  enum Kind : uint8_t { pyObject = 0, counts = 1, gradBundle = 2, leaderTuple = 3 };
  Kind kind = pyObject;
  // pyObject: serialized payload + tensors (fold needs GIL + user op)
  std::string payload;
  std::vector<at::Tensor> tensors;
  // counts / gradBundle:
  int64_t batchSize = 0;
  int64_t numGradients = 0;
  int64_t numSkipped = 0;
  // gradBundle: tensors[0] = flat gradient bucket (fold: add_)
  // leaderTuple:
  int64_t version = 0;
  std::string leaderName;

  void encode(WireWriter& w) const;
  static ReduceValue decode(WireReader& r, std::vector<at::Tensor> tensors);
  // Fold src into *this. For pyObject kind, pyFold must be provided (it
  // handles its own GIL acquisition); native kinds fold without the GIL.
  void fold(ReduceValue& src, const std::function<void(ReduceValue&, ReduceValue&)>& pyFold);
};

// User fold for pyObject values; impl acquires the GIL itself.
using PyFold = std::function<void(ReduceValue& dst, ReduceValue& src)>;

// Completion: exactly one of (value, error).
using ReduceDone = std::function<void(ReduceValue* value, const std::string* error)>;

// ---------------------------------------------------------------- broker

class Broker {
 public:
  explicit Broker(RpcPtr rpc);
  ~Broker();
  void setName(const std::string& n);
  void listen(const std::string& addr);
  void update();  // evict stale members, bump syncIds

 private:
  TimePoint lastForcedResync_{};
  struct Member {
    TimePoint lastPing{};
    int64_t sortOrder = 0;
    uint64_t joinSeq = 0;
  };
  struct GroupState {
    std::unordered_map<std::string, Member> members;
    uint64_t syncId = 1;
    uint64_t joinCounter = 0;
    double timeout = 6.0;
    std::vector<std::string> sortedCache;
    bool dirty = true;
    void resort();
  };
  void evictStaleLocked(GroupState& g);
  RpcPtr rpc_;
  std::mutex mu_;
  std::unordered_map<std::string, GroupState> groups_;
};

// ----------------------------------------------------------------- group

struct OpKey {
  uint64_t syncId;
  uint64_t seq;
  std::string opName;
  bool operator==(const OpKey& o) const {
    return syncId == o.syncId && seq == o.seq && opName == o.opName;
  }
};
struct OpKeyHash {
  size_t operator()(const OpKey& k) const {
    return std::hash<uint64_t>()(k.syncId * 1000003 + k.seq) ^ std::hash<std::string>()(k.opName);
  }
};

// One in-flight allreduce on one member.
struct AllReduceOp {
  std::optional<ReduceValue> acc;       // local value folded with arrived children
  int childrenArrived = 0;
  int childrenExpected = -1;            // -1: membership for this syncId unknown yet
  bool localContributed = false;
  bool sentUp = false;
  bool completed = false;
  bool folding = false;  // a thread is folding outside the lock
  std::vector<std::pair<std::string, ReduceValue>> queued;  // arrived before local start
  PyFold pyFold;
  ReduceDone done;
  TimePoint deadline{};
  // Snapshot of membership when the op became runnable:
  std::vector<std::string> members;
  int myIndex = -1;
};
using AllReduceOpPtr = std::shared_ptr<AllReduceOp>;

class Group : public std::enable_shared_from_this<Group> {
 public:
  static std::shared_ptr<Group> create(RpcPtr rpc, std::string name);
  ~Group();

  void update();  // ping broker, adopt membership, drive op timeouts
  void setBrokerName(const std::string& n) { brokerName_ = n; }
  void setTimeout(double t) { timeout_ = t; }
  void setSortOrder(int64_t o) { sortOrder_ = o; }
  std::vector<std::string> members();
  uint64_t syncId();
  const std::string& name() const { return name_; }
  bool active();

  // Start an allreduce. All members must start ops with the same name in the
  // same order. done runs on a scheduler thread (or inline).
  void allReduce(const std::string& opName, ReduceValue value, PyFold fold, ReduceDone done);

  // Ask the broker to bump this group's syncId so every member resets to a
  // fresh epoch (re-aligns allreduce sequence counters after a failure).
  void requestResync();

  RpcPtr rpc() { return rpc_; }
  std::string myName() { return rpc_->getName(); }

 private:
  Group(RpcPtr rpc, std::string name);
  void setup();

  void handleContribution(bool isDown, Frame& f, const std::string& from, RespondFn respond);
  void advance(const OpKey& key, AllReduceOpPtr op);  // mu_ must NOT be held
  void completeOp(const OpKey& key, AllReduceOpPtr op, ReduceValue* v, const std::string& err);
  void sendValue(const std::string& member, bool isDown, const OpKey& key, const ReduceValue& v);
  std::string funcUp() const { return "__mrl_group_up:" + name_; }
  std::string funcDown() const { return "__mrl_group_down:" + name_; }

  RpcPtr rpc_;
  std::string name_;
  std::string brokerName_ = "broker";
  std::atomic<double> timeout_{10.0};
  std::atomic<int64_t> sortOrder_{0};

  std::mutex mu_;
  std::vector<std::string> members_;
  uint64_t syncId_ = 0;
  bool active_ = false;
  bool pingInFlight_ = false;
  TimePoint lastPing_{};
  TimePoint lastPingOk_{};
  TimePoint lastUnreachableWarn_{};
  std::unordered_map<std::string, uint64_t> seqByName_;  // per-opName sequence
  std::unordered_map<OpKey, AllReduceOpPtr, OpKeyHash> ops_;
  bool stopped_ = false;
};

}  // namespace mrl
