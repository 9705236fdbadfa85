// Reliable peer-to-peer RPC.
//
// Capability parity with the reference's Rpc core (src/rpc.{h,cc}): named
// peers, listen/connect with auto-reconnect, request reliability across
// connection churn, duplicate suppression, peer discovery by gossip, call
// timeouts, keepalives. The design is new and simpler than the reference's:
//  - fids are computed locally by both sides (FNV-1a of the function name) —
//    no remote id-resolution protocol (reference: rpc.cc:2094-2142).
//  - reliability is resend-on-reconnect + receiver-side dedupe, instead of
//    the reference's ack/nack/poke machinery (rpc.cc:1106-1498): TCP already
//    guarantees delivery on a live connection, so the only loss mode is
//    connection death, which we handle by re-sending every outstanding call
//    on the next established connection and deduplicating at the receiver.
//  - one reactor (epoll thread) + one timer thread per Rpc; handler
//    execution on the shared scheduler pool.
#pragma once

#include <atomic>
#include <deque>
#include <memory>
#include <map>
#include <unordered_map>
#include <unordered_set>

#include "scheduler.h"
#include "socket.h"

namespace mrl {

class Rpc;
using RpcPtr = std::shared_ptr<Rpc>;

// Handler: receives the request frame; must eventually call respond exactly
// once (possibly from another thread). isError => payload is an error string.
using RespondFn = std::function<void(std::string payload, std::vector<at::Tensor> tensors, bool isError)>;
using Handler = std::function<void(Frame req, const std::string& fromPeer, RespondFn respond)>;

// Response callback: exactly one of (frame, error) is non-null.
using ResponseCallback = std::function<void(Frame* resp, const std::string* error)>;

class Rpc : public std::enable_shared_from_this<Rpc> {
 public:
  static RpcPtr create();
  ~Rpc();

  void setName(const std::string& name);
  std::string getName() const;
  const std::string& uid() const { return uid_; }

  std::vector<std::string> listen(const std::string& addr);
  void connect(const std::string& addr);

  void define(const std::string& name, Handler h);
  void undefine(const std::string& name);

  void setTimeout(double seconds) { defaultTimeout_.store(seconds); }
  double timeout() const { return defaultTimeout_.load(); }

  // Asynchronous call. cb runs exactly once on a scheduler thread.
  // timeoutOverride < 0 means "use default".
  void sendRequest(const std::string& peerName, const std::string& funcName, std::string payload,
                   std::vector<at::Tensor> tensors, ResponseCallback cb, double timeoutOverride = -1);

  // Best-effort fire-and-forget variant (no retries, no response tracking).
  std::string debugInfo();

  // Addresses this peer can be reached at (listeners; lazily created).
  std::vector<std::string> localAddrs();

  // Names of peers we currently have a live, greeted connection to.
  std::vector<std::string> connectedPeers();

  // True if `peerName` greeted us from this machine (machineId match) —
  // the gate for hipIpc zero-copy GPU-tensor serialization. Unknown or
  // not-yet-greeted peers report false (callers fall back to staging).
  bool peerIsLocal(const std::string& peerName);

  // Restrict transports (reference rpc.cc:324-336 semantics): disabled
  // transports get no default listener and are never dialed. Must be
  // called before the first listen/connect.
  void setTransports(bool tcp, bool unixSock);

  void shutdown();
  bool isShutdown() const { return stopping_.load(); }

 private:
  Rpc();
  void start();

  struct ConnInfo {
    std::string peerUid;   // empty until greeting received
    std::string peerName;
    std::string addr;      // dial address (outbound only; empty if inbound)
    bool ready = false;
    bool inbound = false;
    TimePoint lastRecv{};
    TimePoint established{};
    TimePoint lastKeepaliveSent{};
  };

  // Per-address transport model (reference rpc.cc:640 "bandit" transport
  // choice, MI355X-shaped: round-trip latency EMA per dial address plus a
  // decaying failure penalty; reconnects pick the argmin, unexplored
  // addresses get optimistic priors so every transport gets sampled).
  struct TransportStat {
    double ema = -1;          // seconds; <0 = never sampled
    uint64_t samples = 0;
    double failPenalty = 0;   // added to score, halves per connect cycle
  };

  struct PeerInfo {
    std::string uid;
    std::string machine;             // machineId from the greeting
    std::vector<std::string> addrs;  // candidate addresses
    ConnId activeConn = 0;           // ready connection
    ConnId connecting = 0;           // outbound connect in flight
    size_t nextAddr = 0;
    TimePoint lastConnectAttempt{};
    TimePoint lastFindBroadcast{};
    uint64_t sendCount = 0;
    uint64_t recvCount = 0;
    uint64_t bytesSent = 0;
    uint64_t bytesRecv = 0;
    double latencyEma = 0;  // seconds, over request->response round trips
    std::map<std::string, TransportStat> transport;  // keyed by dial addr
  };

  struct Outgoing {
    uint64_t rid;
    TimePoint sentAt{};
    std::string peerName;
    std::string funcName;
    Frame frame;  // full request for resend
    ConnId sentOn = 0;
    TimePoint deadline;
    ResponseCallback cb;
  };

  struct IncomingKey {
    std::string peerUid;
    uint64_t rid;
    bool operator==(const IncomingKey& o) const { return rid == o.rid && peerUid == o.peerUid; }
  };
  struct IncomingKeyHash {
    size_t operator()(const IncomingKey& k) const {
      return std::hash<std::string>()(k.peerUid) ^ std::hash<uint64_t>()(k.rid);
    }
  };
  struct Incoming {
    bool responded = false;
    bool acked = false;  // sender confirmed receipt: stored response freed
    Frame response;      // kept for re-send to duplicate requests until acked
    ConnId lastConn = 0;
    TimePoint created{};   // for reaping handlers that never respond
    TimePoint doneTime{};
  };

  struct Endpoint {  // persistent endpoint from connect()
    std::string addr;
    ConnId conn = 0;
    bool up = false;
    TimePoint lastAttempt{};
    double backoff = 0.25;
  };

  // --- reactor callbacks (epoll thread) ---
  void onFrame(ConnId id, Frame&& f);
  void onClosed(ConnId id, const std::string& reason);
  void onAccept(ConnId id);
  void onConnected(ConnId id);

  void handleGreeting(ConnId id, Frame& f);
  void handleRequest(ConnId id, Frame&& f);
  void handleResponse(ConnId id, Frame&& f, bool isError);
  void handleFindPeer(ConnId id, Frame& f);
  void handleResponseAck(ConnId id, Frame& f);
  void handlePeerInfo(ConnId id, Frame& f);

  void sendGreeting(ConnId id);
  // mu_ must be held:
  PeerInfo& getPeer(const std::string& name);
  void tryConnectPeerLocked(const std::string& name, PeerInfo& p);
  void flushPeerLocked(const std::string& name, PeerInfo& p);
  void broadcastFindPeerLocked(const std::string& name);
  void ensureListeningLocked();

  void timerLoop();
  void failOutgoing(uint64_t rid, const std::string& error);

  std::string name_;
  std::string uid_;
  std::string machineId_;
  std::atomic<double> defaultTimeout_{30.0};
  std::atomic<bool> stopping_{false};
  std::atomic<uint64_t> nextRid_{1};

  mutable std::mutex mu_;
  std::unordered_map<ConnId, ConnInfo> conns_;
  std::unordered_map<std::string, PeerInfo> peers_;
  std::unordered_map<uint64_t, Outgoing> outgoing_;
  std::unordered_map<IncomingKey, Incoming, IncomingKeyHash> incoming_;
  std::unordered_map<uint64_t, std::pair<std::string, Handler>> functions_;  // fid -> (name, h)
  std::vector<Endpoint> endpoints_;
  std::vector<std::string> listenAddrs_;
  bool defaultListenersCreated_ = false;
  bool tcpEnabled_ = true;
  bool unixEnabled_ = true;

  std::unique_ptr<SocketEngine> engine_;
  std::thread timerThread_;
  std::mutex timerMu_;
  std::condition_variable timerCv_;
};

std::string getMachineId();

}  // namespace mrl
