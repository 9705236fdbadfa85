// Accumulator: elastic data-parallel gradient accumulation + model/optimizer
// state synchronization.
//
// Capability parity with the reference's src/accumulator.{h,cc}: the same
// cooperative protocol (update / wants_state / set_state / has_new_state /
// state / wants_gradients / reduce_gradients / skip_gradients /
// has_gradients / zero_gradients), leader election by allreduce-max of
// (modelVersion, name), model+optimizer-state transfer to (re)joining peers,
// and virtual-batch-size gating via count rounds.
//
// MI355X-native redesign of the data plane: gradients are packed once into a
// single FLAT bucket that lives on the parameters' device. When a
// local-reduce hook is installed (moolib_amd.parallel wires it to
// torch.distributed.all_reduce — RCCL over xGMI), the bucket is reduced
// on-device without ever visiting the CPU (the reference stages every
// gradient through pinned CPU memory and reduces over TCP,
// accumulator.cc:847-1003). Without a hook, the bucket rides the RPC tree
// allreduce (cross-node / elastic fallback).
#pragma once

#include <deque>

#include "services.h"

namespace mrl {

class Accumulator : public std::enable_shared_from_this<Accumulator> {
 public:
  using Ptr = std::shared_ptr<Accumulator>;
  // LocalReduceHook: called with the flat on-device bucket; must start an
  // in-place sum-allreduce over the fixed peer world and return a poll
  // function (called later; returns true when the collective is done).
  // Both run with the GIL NOT held; implementations manage their own GIL.
  using LocalReduceHook = std::function<std::function<bool()>(at::Tensor&)>;

  static Ptr create(std::string name, std::vector<at::Tensor> params,
                    std::vector<at::Tensor> buffers, std::shared_ptr<Group> group);
  ~Accumulator();

  void connect(const std::string& addr);
  void update();  // drives group + state machine; call every iteration
  bool connected();

  bool wantsState();
  bool hasNewState();
  // userState is an opaque serialized blob (payload + out-of-band tensors);
  // the Python layer runs serde around these.
  void setState(std::string payload, std::vector<at::Tensor> tensors);
  std::pair<std::string, std::vector<at::Tensor>> state();

  bool wantsGradients();
  bool hasGradients();
  void skipGradients();
  void reduceGradients(int64_t batchSize);
  void zeroGradients();

  int64_t modelVersion();
  void setModelVersion(int64_t v);
  std::string getLeader();
  bool isLeader();
  std::unordered_map<std::string, int64_t> gradientStats();

  void setVirtualBatchSize(int64_t n);
  void setParallelGradients(int64_t n);
  void setLocalReduceHook(LocalReduceHook h);

  std::shared_ptr<Group> group() { return group_; }
  std::string debugState();

 private:
  Accumulator(std::string name, std::vector<at::Tensor> params, std::vector<at::Tensor> buffers,
              std::shared_ptr<Group> group);
  void setup();

  enum class Phase { inactive, electing, fetching, running };
  enum class GradPhase { wantDecision, counting, reducing, resultReady };

  struct GradSlot {
    at::Tensor flat;  // on params' device; local accumulated gradient sum
    int64_t newBatch = 0, newGrads = 0, newSkipped = 0;   // since last count round
    int64_t totBatch = 0, totGrads = 0, totSkipped = 0;   // accumulated global totals
    GradPhase phase = GradPhase::wantDecision;
    std::function<bool()> hookPoll;
    uint64_t launchSeq = 0;     // hook-launch ordering ticket
    bool hookPending = false;   // waiting for its launch turn / drains
    TimePoint started{};
  };
  struct PendingResult {
    at::Tensor flat;  // globally summed gradients (on device)
    int64_t batch, grads, skipped;
  };

  // all called with mu_ held unless noted
  void resetLocked(const char* why);
  void startElectionLocked();
  void startCountRoundLocked();
  void enterReducingLocked(size_t si);
  void tryLaunchHooksLocked();
  void completeSlotLocked(size_t si, at::Tensor result);
  void applyPendingLocked();
  void failAndResyncLocked(const char* why);
  at::Tensor makeFlatLocked();
  void maybeSendModelUpdatesLocked();
  std::string fn(const char* suffix) const { return "__mrl_acc_" + std::string(suffix) + ":" + name_; }

  std::string name_;
  std::vector<at::Tensor> params_;        // with requires_grad
  std::vector<at::Tensor> allParams_;     // ctor order (for model sync)
  std::vector<at::Tensor> buffers_;
  std::shared_ptr<Group> group_;
  RpcPtr rpc_;

  std::mutex mu_;
  Phase phase_ = Phase::inactive;
  GradPhase gradPhase_ = GradPhase::wantDecision;
  uint64_t syncSeen_ = 0;
  uint64_t epoch_ = 0;  // bumped on every reset; stale async callbacks check it
  int64_t modelVersion_ = 0;
  std::string leader_;
  bool isLeader_ = false;
  TimePoint fetchStarted_{};
  bool modelRequestSent_ = false;

  // pending model update received from the leader
  bool havePendingModel_ = false;
  int64_t pendingVersion_ = 0;
  std::vector<at::Tensor> pendingParams_, pendingBuffers_;
  std::string pendingStatePayload_;
  std::vector<at::Tensor> pendingStateTensors_;
  bool hasNewState_ = false;

  // leader side
  std::vector<std::string> stateRequesters_;
  TimePoint lastBuffersBroadcast_{};
  TimePoint lastModelBroadcast_{};
  double modelBcastInterval_ = 600.0;  // MOOLIB_AMD_MODEL_BCAST_S overrides

  // gradient machine: parallelGradients_ staging slots, used round-robin.
  // Count rounds are strictly sequential cluster-wide (decided via shared
  // allreduce results), so every member walks the same slot/round sequence;
  // a slot entering `reducing` frees the cursor to the next slot, bounding
  // in-flight reductions to the slot count (moolib's set_parallel_gradients
  // pipelining, reference accumulator.cc:251-256).
  std::vector<GradSlot> slots_;
  size_t slotCursor_ = 0;
  std::vector<int64_t> offsets_, numels_;
  int64_t virtualBatchSize_ = 1;
  bool decided_ = false;
  bool hasGradients_ = false;
  int64_t statBatch_ = 0, statGrads_ = 0, statSkipped_ = 0;  // of the applied result
  std::deque<PendingResult> results_;
  LocalReduceHook hook_;
  uint64_t hookLaunchCounter_ = 0;   // tickets handed to reducing slots
  uint64_t hookLaunchedUpTo_ = 0;    // tickets actually launched
  // Collectives whose round was reset: polled to completion so the peers'
  // call sequences stay aligned; their (mixed-round) output buffers are
  // owned by the closure and discarded.
  std::vector<std::function<bool()>> drains_;
};

}  // namespace mrl
