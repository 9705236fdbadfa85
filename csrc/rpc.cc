#include "rpc.h"

#include <fstream>
#include <sstream>

namespace mrl {

namespace {
constexpr uint32_t kMagic = 0x4d524c31;  // "MRL1"
constexpr uint8_t kVersion = 1;

std::string readMachineIdFile() {
  std::ifstream f("/proc/sys/kernel/random/boot_id");
  std::string s;
  std::getline(f, s);
  if (s.empty()) s = "unknown-machine";
  return s;
}
}  // namespace

std::string getMachineId() {
  static std::string id = readMachineIdFile();
  return id;
}

RpcPtr Rpc::create() {
  RpcPtr p(new Rpc());
  p->start();
  return p;
}

Rpc::Rpc() {
  uid_ = randomUid();
  name_ = uid_;
  machineId_ = getMachineId();
}

void Rpc::start() {
  SocketEngineCallbacks cbs;
  std::weak_ptr<Rpc> weak = shared_from_this();
  cbs.onFrame = [this](ConnId id, Frame&& f) { onFrame(id, std::move(f)); };
  cbs.onClosed = [this](ConnId id, const std::string& r) { onClosed(id, r); };
  cbs.onAccept = [this](ConnId id) { onAccept(id); };
  cbs.onConnected = [this](ConnId id) { onConnected(id); };
  engine_ = std::make_unique<SocketEngine>(std::move(cbs));
  timerThread_ = std::thread([this] { timerLoop(); });
}

Rpc::~Rpc() { shutdown(); }

void Rpc::shutdown() {
  if (stopping_.exchange(true)) return;
  timerCv_.notify_all();
  if (timerThread_.joinable()) timerThread_.join();
  engine_->shutdown();
  // Fail all outstanding calls.
  std::vector<ResponseCallback> cbs;
  {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& [rid, rec] : outgoing_) {
      if (rec.cb) cbs.push_back(std::move(rec.cb));
    }
    outgoing_.clear();
    incoming_.clear();
    functions_.clear();
  }
  for (auto& cb : cbs) {
    globalScheduler().run([cb = std::move(cb)] {
      std::string err = "rpc shutdown";
      cb(nullptr, &err);
    });
  }
}

void Rpc::setName(const std::string& name) {
  std::lock_guard<std::mutex> lk(mu_);
  name_ = name;
}

std::string Rpc::getName() const {
  std::lock_guard<std::mutex> lk(mu_);
  return name_;
}

void Rpc::ensureListeningLocked() {
  if (defaultListenersCreated_) return;
  defaultListenersCreated_ = true;
  if (tcpEnabled_) {
    try {
      auto tcp = engine_->listen("tcp://0.0.0.0:0");
      for (auto& a : tcp) listenAddrs_.push_back(a);
    } catch (const std::exception& e) {
      MRL_LOG_ERROR("default tcp listener failed: %s", e.what());
    }
  }
  if (unixEnabled_) {
    try {
      auto ux = engine_->listen("unix://" + machineId_ + "/" + uid_);
      for (auto& a : ux) listenAddrs_.push_back(a);
    } catch (const std::exception& e) {
      MRL_LOG_ERROR("default unix listener failed: %s", e.what());
    }
  }
}

void Rpc::setTransports(bool tcp, bool unixSock) {
  std::lock_guard<std::mutex> lk(mu_);
  if (defaultListenersCreated_) {
    throw RpcError("set_transports must be called before listen/connect");
  }
  if (!tcp && !unixSock) throw RpcError("at least one transport must stay enabled");
  tcpEnabled_ = tcp;
  unixEnabled_ = unixSock;
}

std::vector<std::string> Rpc::listen(const std::string& addr) {
  std::lock_guard<std::mutex> lk(mu_);
  bool isUnix = addr.rfind("unix://", 0) == 0;
  if ((isUnix && !unixEnabled_) || (!isUnix && !tcpEnabled_)) {
    throw RpcError("transport disabled by set_transports: " + addr);
  }
  auto bound = engine_->listen(addr);
  for (auto& a : bound) listenAddrs_.push_back(a);
  ensureListeningLocked();
  return bound;
}

void Rpc::connect(const std::string& addr) {
  std::lock_guard<std::mutex> lk(mu_);
  ensureListeningLocked();
  for (auto& e : endpoints_) {
    if (e.addr == addr) return;  // already connecting/connected
  }
  Endpoint e;
  e.addr = addr;
  e.lastAttempt = now();
  e.conn = engine_->connect(addr);
  ConnInfo ci;
  ci.addr = addr;
  ci.established = now();
  ci.lastRecv = now();
  conns_[e.conn] = ci;
  endpoints_.push_back(e);
}

void Rpc::define(const std::string& name, Handler h) {
  uint64_t fid = fnv1a64(name);
  std::lock_guard<std::mutex> lk(mu_);
  auto it = functions_.find(fid);
  if (it != functions_.end() && it->second.first != name) {
    throw RpcError("fid hash collision between '" + it->second.first + "' and '" + name + "'");
  }
  functions_[fid] = {name, std::move(h)};
}

void Rpc::undefine(const std::string& name) {
  std::lock_guard<std::mutex> lk(mu_);
  functions_.erase(fnv1a64(name));
}

std::vector<std::string> Rpc::localAddrs() {
  std::lock_guard<std::mutex> lk(mu_);
  ensureListeningLocked();
  return listenAddrs_;
}

std::vector<std::string> Rpc::connectedPeers() {
  std::lock_guard<std::mutex> lk(mu_);
  std::vector<std::string> out;
  for (auto& [name, p] : peers_) {
    if (p.activeConn != 0) out.push_back(name);
  }
  return out;
}

bool Rpc::peerIsLocal(const std::string& peerName) {
  if (peerName == name_) return true;  // self-calls dispatch locally
  std::lock_guard<std::mutex> lk(mu_);
  auto it = peers_.find(peerName);
  return it != peers_.end() && !it->second.machine.empty() && it->second.machine == machineId_;
}

// ----------------------------------------------------------- greeting

void Rpc::sendGreeting(ConnId id) {
  Frame f;
  f.kind = FrameKind::greeting;
  WireWriter w;
  std::lock_guard<std::mutex> lk(mu_);
  ensureListeningLocked();
  w.u32(kMagic);
  w.u8(kVersion);
  w.str(name_);
  w.str(uid_);
  w.str(machineId_);
  w.u32(static_cast<uint32_t>(listenAddrs_.size()));
  for (auto& a : listenAddrs_) w.str(a);
  f.payload = std::move(w.out);
  engine_->send(id, std::move(f));
}

void Rpc::onAccept(ConnId id) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    ConnInfo ci;
    ci.inbound = true;
    ci.lastRecv = now();
    ci.established = now();
    conns_[id] = ci;
  }
  sendGreeting(id);
}

void Rpc::onConnected(ConnId id) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = conns_.find(id);
    if (it == conns_.end()) conns_[id] = ConnInfo{};
    conns_[id].lastRecv = now();
    conns_[id].established = now();
  }
  sendGreeting(id);
}

void Rpc::onClosed(ConnId id, const std::string& reason) {
  std::lock_guard<std::mutex> lk(mu_);
  auto it = conns_.find(id);
  std::string peerName;
  std::string connAddr;
  bool wasReady = false;
  if (it != conns_.end()) {
    peerName = it->second.peerName;
    connAddr = it->second.addr;
    wasReady = it->second.ready;
    conns_.erase(it);
  }
  if (!connAddr.empty() && !wasReady && !peerName.empty()) {
    // dial failed before greeting: this transport is suspect
    auto pit = peers_.find(peerName);
    if (pit != peers_.end()) pit->second.transport[connAddr].failPenalty += 1.0;
  }
  if (!peerName.empty()) {
    auto pit = peers_.find(peerName);
    if (pit != peers_.end()) {
      if (pit->second.activeConn == id) pit->second.activeConn = 0;
      if (pit->second.connecting == id) pit->second.connecting = 0;
    }
  }
  for (auto& p : peers_) {
    if (p.second.connecting == id) p.second.connecting = 0;
    if (p.second.activeConn == id) p.second.activeConn = 0;
  }
  for (auto& e : endpoints_) {
    if (e.conn == id) {
      e.conn = 0;
      e.up = false;
    }
  }
  MRL_LOG_DEBUG("connection %llu closed: %s", static_cast<unsigned long long>(id), reason.c_str());
}

void Rpc::handleGreeting(ConnId id, Frame& f) {
  try {
    WireReader r(f.payload);
    if (r.u32() != kMagic || r.u8() != kVersion) {
      engine_->close(id);
      return;
    }
    std::string peerName(r.str());
    std::string peerUid(r.str());
    std::string peerMachine(r.str());
    uint32_t nAddrs = r.u32();
    std::vector<std::string> addrs;
    for (uint32_t i = 0; i < nAddrs && i < 64; ++i) addrs.emplace_back(r.str());

    std::lock_guard<std::mutex> lk(mu_);
    if (peerUid == uid_) {  // connected to ourselves
      engine_->close(id);
      return;
    }
    auto cit = conns_.find(id);
    if (cit == conns_.end()) return;
    cit->second.peerUid = peerUid;
    cit->second.peerName = peerName;
    cit->second.ready = true;
    cit->second.lastRecv = now();

    PeerInfo& p = getPeer(peerName);
    if (!p.uid.empty() && p.uid != peerUid) {
      MRL_LOG_INFO("peer '%s' has a new uid (restarted peer?); adopting", peerName.c_str());
      p.addrs.clear();
    }
    p.uid = peerUid;
    p.machine = peerMachine;
    for (auto& a : addrs) {
      // Unix addrs are only usable from the same machine.
      if (a.rfind("unix://", 0) == 0 && a.find(machineId_) == std::string::npos) continue;
      if (std::find(p.addrs.begin(), p.addrs.end(), a) == p.addrs.end()) p.addrs.push_back(a);
    }
    if (p.connecting == id) p.connecting = 0;
    p.activeConn = id;
    flushPeerLocked(peerName, p);
  } catch (const std::exception& e) {
    MRL_LOG_ERROR("bad greeting: %s", e.what());
    engine_->close(id);
  }
}

void Rpc::handleResponseAck(ConnId id, Frame& f) {
  // Destroying py-backed tensors can ACQUIRE THE GIL (torch pyobj decref);
  // doing that while holding mu_ on the reactor thread deadlocks against a
  // python thread that holds the GIL and wants mu_. Move the frame out
  // under the lock; let a scheduler thread (lock-free) destroy it.
  Frame tomb;
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto cit = conns_.find(id);
    if (cit == conns_.end() || cit->second.peerUid.empty()) return;
    auto it = incoming_.find(IncomingKey{cit->second.peerUid, f.rid});
    if (it == incoming_.end() || !it->second.responded) return;
    it->second.acked = true;
    tomb = std::move(it->second.response);
    it->second.response = Frame();  // marker stays for dedupe
  }
  if (!tomb.payload.empty() || !tomb.tensors.empty()) {
    globalScheduler().run([tomb = std::move(tomb)]() mutable {});
  }
}

// ----------------------------------------------------------- peer mgmt

Rpc::PeerInfo& Rpc::getPeer(const std::string& name) { return peers_[name]; }

void Rpc::tryConnectPeerLocked(const std::string& name, PeerInfo& p) {
  if (p.activeConn != 0 || p.connecting != 0 || p.addrs.empty()) return;
  if (secondsSince(p.lastConnectAttempt) < 0.25) return;
  p.lastConnectAttempt = now();
  // Latency-informed transport choice: score every candidate address by its
  // measured round-trip EMA plus a decaying failure penalty; unexplored
  // addresses get optimistic priors (unix beats tcp on the same machine
  // until data says otherwise). The resend-on-reconnect + receiver-dedupe
  // machinery makes switching transports between attempts safe.
  std::string addr;
  double best = 0;
  for (auto& cand : p.addrs) {
    bool isUnix = cand.rfind("unix://", 0) == 0;
    if ((isUnix && !unixEnabled_) || (!isUnix && !tcpEnabled_)) continue;
    TransportStat& st = p.transport[cand];
    st.failPenalty *= 0.5;  // dead transports get retried eventually
    double score = st.samples == 0 ? (isUnix ? -2.0 : -1.0) : st.ema;
    score += st.failPenalty;
    if (addr.empty() || score < best) {
      best = score;
      addr = cand;
    }
  }
  if (addr.empty()) return;
  p.connecting = engine_->connect(addr);
  ConnInfo ci;
  ci.peerName = name;
  ci.addr = addr;
  ci.established = now();
  ci.lastRecv = now();
  conns_[p.connecting] = ci;
}

void Rpc::flushPeerLocked(const std::string& name, PeerInfo& p) {
  if (p.activeConn == 0) return;
  for (auto& [rid, rec] : outgoing_) {
    if (rec.peerName == name && rec.sentOn != p.activeConn) {
      rec.sentOn = p.activeConn;
      engine_->send(p.activeConn, rec.frame);
      ++p.sendCount;
    }
  }
}

void Rpc::broadcastFindPeerLocked(const std::string& name) {
  PeerInfo& p = getPeer(name);
  if (secondsSince(p.lastFindBroadcast) < 0.5) return;
  p.lastFindBroadcast = now();
  Frame f;
  f.kind = FrameKind::findPeer;
  WireWriter w;
  w.str(name);
  f.payload = std::move(w.out);
  for (auto& [cid, ci] : conns_) {
    if (ci.ready) engine_->send(cid, f);
  }
}

void Rpc::handleFindPeer(ConnId id, Frame& f) {
  try {
    WireReader r(f.payload);
    std::string name(r.str());
    std::lock_guard<std::mutex> lk(mu_);
    std::string uid;
    std::vector<std::string> addrs;
    if (name == name_) {
      uid = uid_;
      addrs = listenAddrs_;
    } else {
      auto it = peers_.find(name);
      if (it == peers_.end() || it->second.addrs.empty()) return;  // unknown; stay silent
      uid = it->second.uid;
      addrs = it->second.addrs;
    }
    Frame out;
    out.kind = FrameKind::peerInfo;
    WireWriter w;
    w.str(name);
    w.str(uid);
    w.u32(static_cast<uint32_t>(addrs.size()));
    for (auto& a : addrs) w.str(a);
    out.payload = std::move(w.out);
    engine_->send(id, std::move(out));
  } catch (const std::exception& e) {
    MRL_LOG_ERROR("bad findPeer: %s", e.what());
  }
}

void Rpc::handlePeerInfo(ConnId id, Frame& f) {
  try {
    WireReader r(f.payload);
    std::string name(r.str());
    std::string uid(r.str());
    uint32_t n = r.u32();
    std::vector<std::string> addrs;
    for (uint32_t i = 0; i < n && i < 64; ++i) addrs.emplace_back(r.str());
    std::lock_guard<std::mutex> lk(mu_);
    PeerInfo& p = getPeer(name);
    if (!p.uid.empty() && !uid.empty() && p.uid != uid) p.addrs.clear();
    if (!uid.empty()) p.uid = uid;
    for (auto& a : addrs) {
      if (a.rfind("unix://", 0) == 0 && a.find(machineId_) == std::string::npos) continue;
      if (std::find(p.addrs.begin(), p.addrs.end(), a) == p.addrs.end()) p.addrs.push_back(a);
    }
    // If we have anything queued for this peer, connect now.
    bool pending = false;
    for (auto& [rid, rec] : outgoing_) {
      if (rec.peerName == name) {
        pending = true;
        break;
      }
    }
    if (pending) tryConnectPeerLocked(name, p);
  } catch (const std::exception& e) {
    MRL_LOG_ERROR("bad peerInfo: %s", e.what());
  }
}

// ------------------------------------------------------------ requests

void Rpc::sendRequest(const std::string& peerName, const std::string& funcName, std::string payload,
                      std::vector<at::Tensor> tensors, ResponseCallback cb, double timeoutOverride) {
  if (stopping_.load()) {
    globalScheduler().run([cb = std::move(cb)] {
      std::string err = "rpc shutdown";
      cb(nullptr, &err);
    });
    return;
  }
  uint64_t rid = nextRid_.fetch_add(1);
  double timeoutS = timeoutOverride >= 0 ? timeoutOverride : defaultTimeout_.load();
  Frame f;
  f.kind = FrameKind::request;
  f.rid = rid;
  f.fid = fnv1a64(funcName);
  f.payload = std::move(payload);
  f.tensors = std::move(tensors);
  for (auto& t : f.tensors) {
    if (!t.device().is_cpu()) t = t.cpu();
    if (!t.is_contiguous()) t = t.contiguous();
  }

  std::lock_guard<std::mutex> lk(mu_);
  ensureListeningLocked();
  Outgoing& rec = outgoing_[rid];
  rec.rid = rid;
  rec.sentAt = now();
  rec.peerName = peerName;
  rec.funcName = funcName;
  rec.frame = f;
  rec.deadline = now() + std::chrono::duration_cast<Clock::duration>(std::chrono::duration<double>(timeoutS));
  rec.cb = std::move(cb);
  if (peerName == name_) {
    // Self-call: a peer addressing itself by name dispatches locally — no
    // wire, no connection (the reference instead detects and closes
    // self-connections; local dispatch keeps "peer lists that include
    // yourself" uniform). The Outgoing record stays registered so the
    // normal timeout machinery still covers deferred handlers that never
    // respond; completion flows through handleResponse with a synthetic
    // frame (sentOn=0 -> no transport attribution).
    Handler handler;
    {
      auto fit = functions_.find(f.fid);
      if (fit != functions_.end()) handler = fit->second.second;
    }
    std::weak_ptr<Rpc> weak = weak_from_this();
    RespondFn respond = [weak, rid](std::string payload, std::vector<at::Tensor> tensors,
                                    bool isError) {
      auto self = weak.lock();
      if (!self || self->stopping_.load()) return;
      Frame resp;
      resp.kind = isError ? FrameKind::errorResponse : FrameKind::response;
      resp.rid = rid;
      resp.payload = std::move(payload);
      resp.tensors = std::move(tensors);
      self->handleResponse(0, std::move(resp), isError);
    };
    if (!handler) {
      uint64_t fid = f.fid;
      globalScheduler().run([respond, fid] {
        respond("unknown function id " + std::to_string(fid), {}, true);
      });
    } else {
      globalScheduler().run(
          [handler = std::move(handler), f = std::move(f), selfName = name_,
           respond = std::move(respond)]() mutable {
            try {
              handler(std::move(f), selfName, respond);
            } catch (const std::exception& e) {
              respond(std::string("handler exception: ") + e.what(), {}, true);
            }
          });
    }
    return;
  }
  PeerInfo& p = getPeer(peerName);
  uint64_t fbytes = f.payload.size();
  for (auto& t : f.tensors) fbytes += t.nbytes();
  p.bytesSent += fbytes;
  if (p.activeConn != 0) {
    rec.sentOn = p.activeConn;
    engine_->send(p.activeConn, std::move(f));
    ++p.sendCount;
  } else {
    tryConnectPeerLocked(peerName, p);
    if (p.addrs.empty()) broadcastFindPeerLocked(peerName);
  }
}

void Rpc::handleRequest(ConnId id, Frame&& f) {
  std::string peerUid, peerName;
  Handler handler;
  std::string funcName;
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto cit = conns_.find(id);
    if (cit == conns_.end() || !cit->second.ready) return;
    peerUid = cit->second.peerUid;
    peerName = cit->second.peerName;
    IncomingKey key{peerUid, f.rid};
    auto [iit, inserted] = incoming_.try_emplace(key);
    iit->second.lastConn = id;
    if (inserted) iit->second.created = now();
    if (!inserted) {
      if (iit->second.responded && !iit->second.acked) {
        engine_->send(id, iit->second.response);  // duplicate of a completed request
      }
      return;  // duplicate of an in-flight (or acked) request: drop
    }
    auto fit = functions_.find(f.fid);
    if (fit != functions_.end()) {
      funcName = fit->second.first;
      handler = fit->second.second;
    }
  }
  uint64_t rid = f.rid;
  std::weak_ptr<Rpc> weak = weak_from_this();
  RespondFn respond = [weak, peerUid, rid](std::string payload, std::vector<at::Tensor> tensors,
                                           bool isError) {
    auto self = weak.lock();
    if (!self || self->stopping_.load()) return;
    Frame resp;
    resp.kind = isError ? FrameKind::errorResponse : FrameKind::response;
    resp.rid = rid;
    resp.payload = std::move(payload);
    resp.tensors = std::move(tensors);
    for (auto& t : resp.tensors) {
      if (!t.device().is_cpu()) t = t.cpu();
      if (!t.is_contiguous()) t = t.contiguous();
    }
    std::lock_guard<std::mutex> lk(self->mu_);
    IncomingKey key{peerUid, rid};
    auto it = self->incoming_.find(key);
    if (it == self->incoming_.end()) return;
    it->second.responded = true;
    it->second.response = resp;
    it->second.doneTime = now();
    self->engine_->send(it->second.lastConn, std::move(resp));
  };
  if (!handler) {
    respond("unknown function id " + std::to_string(f.fid), {}, true);
    return;
  }
  globalScheduler().run(
      [handler = std::move(handler), f = std::move(f), peerName, respond = std::move(respond)]() mutable {
        try {
          handler(std::move(f), peerName, respond);
        } catch (const std::exception& e) {
          respond(std::string("handler exception: ") + e.what(), {}, true);
        }
      });
}

void Rpc::handleResponse(ConnId id, Frame&& f, bool isError) {
  ResponseCallback cb;
  std::string peerName, funcName;
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = outgoing_.find(f.rid);
    if (it == outgoing_.end()) return;  // duplicate/late response
    cb = std::move(it->second.cb);
    peerName = it->second.peerName;
    funcName = it->second.funcName;
    double lat = secondsSince(it->second.sentAt);
    PeerInfo& p = getPeer(peerName);
    p.latencyEma = p.latencyEma == 0 ? lat : p.latencyEma * 0.9 + lat * 0.1;
    auto cit = conns_.find(it->second.sentOn);
    if (cit != conns_.end() && !cit->second.addr.empty()) {
      TransportStat& st = p.transport[cit->second.addr];
      st.ema = st.ema < 0 ? lat : st.ema * 0.9 + lat * 0.1;
      ++st.samples;
    }
    ++p.recvCount;
    uint64_t fbytes = f.payload.size();
    for (auto& t : f.tensors) fbytes += t.nbytes();
    p.bytesRecv += fbytes;
    Frame reqTomb = std::move(it->second.frame);  // destroyed after unlock
    outgoing_.erase(it);
    if (!reqTomb.tensors.empty()) {
      globalScheduler().run([reqTomb = std::move(reqTomb)]() mutable {});
    }
    if (id != 0) {
      // Ack the response so the responder can free its stored copy (it
      // keeps full response frames — tensors included — for duplicate
      // requests until acked or 60 s; without acks a busy tensor-serving
      // peer retains every reply for the full minute).
      Frame ackf;
      ackf.kind = FrameKind::responseAck;
      ackf.rid = f.rid;
      engine_->send(id, std::move(ackf));
    }
  }
  if (!cb) return;
  globalScheduler().run([cb = std::move(cb), f = std::move(f), isError, peerName, funcName]() mutable {
    if (isError) {
      std::string err = f.payload.empty() ? "remote error" : f.payload;
      if (err.rfind("unknown function id", 0) == 0) {
        // Mirror the reference's error text for a missing remote function.
        err = "RPC remote function " + peerName + "::'" + funcName + "' does not exist";
      }
      cb(nullptr, &err);
    } else {
      cb(&f, nullptr);
    }
  });
}

void Rpc::onFrame(ConnId id, Frame&& f) {
  if (stopping_.load()) return;
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = conns_.find(id);
    if (it != conns_.end()) it->second.lastRecv = now();
  }
  switch (f.kind) {
    case FrameKind::greeting:
      handleGreeting(id, f);
      break;
    case FrameKind::request:
      handleRequest(id, std::move(f));
      break;
    case FrameKind::response:
      handleResponse(id, std::move(f), false);
      break;
    case FrameKind::errorResponse:
      handleResponse(id, std::move(f), true);
      break;
    case FrameKind::findPeer:
      handleFindPeer(id, f);
      break;
    case FrameKind::peerInfo:
      handlePeerInfo(id, f);
      break;
    case FrameKind::responseAck:
      handleResponseAck(id, f);
      break;
    case FrameKind::keepalive:
      break;
    default:
      break;
  }
}

// --------------------------------------------------------------- timer

void Rpc::failOutgoing(uint64_t rid, const std::string& error) {
  ResponseCallback cb;
  Frame reqTomb;
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = outgoing_.find(rid);
    if (it == outgoing_.end()) return;
    cb = std::move(it->second.cb);
    reqTomb = std::move(it->second.frame);  // destroy outside mu_ (GIL hazard)
    outgoing_.erase(it);
  }
  if (!reqTomb.tensors.empty()) {
    globalScheduler().run([reqTomb = std::move(reqTomb)]() mutable {});
  }
  if (!cb) return;
  globalScheduler().run([cb = std::move(cb), error] { cb(nullptr, &error); });
}

void Rpc::timerLoop() {
  while (!stopping_.load()) {
    {
      std::unique_lock<std::mutex> lk(timerMu_);
      timerCv_.wait_for(lk, std::chrono::milliseconds(100));
    }
    if (stopping_.load()) return;
    TimePoint t = now();
    std::vector<std::pair<uint64_t, std::string>> failures;
    std::vector<Frame> tombs;  // frames destroyed after mu_ is released
    {
      std::lock_guard<std::mutex> lk(mu_);
      // Reconnect persistent endpoints.
      for (auto& e : endpoints_) {
        if (e.conn == 0 && secondsSince(e.lastAttempt) >= e.backoff) {
          e.lastAttempt = t;
          e.backoff = std::min(e.backoff * 1.6, 2.0);
          e.conn = engine_->connect(e.addr);
          ConnInfo ci;
          ci.addr = e.addr;
          ci.established = now();
          ci.lastRecv = now();
          conns_[e.conn] = ci;
        }
      }
      // Outgoing: timeouts + resend-on-reconnect + discovery retries.
      for (auto& [rid, rec] : outgoing_) {
        if (t >= rec.deadline) {
          // Message format mirrors the reference's timeout error text.
          failures.push_back({rid, "Call (" + rec.peerName + "::" + rec.funcName + ") timed out"});
          continue;
        }
        PeerInfo& p = getPeer(rec.peerName);
        if (p.activeConn != 0) {
          if (rec.sentOn != p.activeConn) {
            rec.sentOn = p.activeConn;
            engine_->send(p.activeConn, rec.frame);
          }
        } else {
          tryConnectPeerLocked(rec.peerName, p);
          if (p.connecting == 0 && rec.peerName != name_) broadcastFindPeerLocked(rec.peerName);
        }
      }
      // Incoming GC. Stored response frames may hold the last reference to
      // py-backed tensors; their destruction can take the GIL, so it must
      // happen outside mu_ (tombs vector, cleared after the lock scope).
      for (auto it = incoming_.begin(); it != incoming_.end();) {
        bool reap = false;
        if (it->second.responded) {
          reap = secondsSince(it->second.doneTime) > 60.0;
        } else {
          // A deferred handler that never calls respond would otherwise pin
          // the Incoming record (and its dedupe key) forever; the reference
          // expires these via its timeout machinery (rpc.cc:1667-1760).
          // A respond() after the reap is a silent no-op (record lookup
          // fails), which matches a caller whose own timeout fired long ago.
          reap = secondsSince(it->second.created) > 120.0;
        }
        if (reap) {
          tombs.push_back(std::move(it->second.response));
          it = incoming_.erase(it);
        } else {
          ++it;
        }
      }
      // Keepalives + dead connection detection.
      for (auto it = conns_.begin(); it != conns_.end();) {
        ConnId cid = it->first;
        ConnInfo& ci = it->second;
        double idle = secondsSince(ci.lastRecv);
        if (ci.ready && idle > 2.0 && secondsSince(ci.lastKeepaliveSent) >= 2.0) {
          ci.lastKeepaliveSent = t;
          Frame ka;
          ka.kind = FrameKind::keepalive;
          engine_->send(cid, std::move(ka));
        }
        if ((ci.ready && idle > 30.0) || (!ci.ready && secondsSince(ci.established) > 20.0)) {
          engine_->close(cid);  // onClosed will clean up maps
        }
        ++it;
      }
    }
    tombs.clear();  // may take the GIL (py-backed tensor decref) — no locks held
    for (auto& [rid, err] : failures) failOutgoing(rid, err);
  }
}

std::string Rpc::debugInfo() {
  std::ostringstream os;
  std::lock_guard<std::mutex> lk(mu_);
  os << "Rpc '" << name_ << "' uid=" << uid_ << "\n";
  os << "  connections: " << conns_.size() << "\n";
  for (auto& [id, ci] : conns_) {
    os << "    #" << id << " peer='" << ci.peerName << "' ready=" << ci.ready
       << " inbound=" << ci.inbound << " idle=" << secondsSince(ci.lastRecv) << "s\n";
  }
  os << "  peers: " << peers_.size() << "\n";
  for (auto& [name, p] : peers_) {
    os << "    '" << name << "' conn=" << p.activeConn << " reqs=" << p.sendCount << "/"
       << p.recvCount << " bytes tx/rx=" << p.bytesSent << "/" << p.bytesRecv
       << " latency_ema=" << p.latencyEma * 1000 << "ms addrs=[";
    for (auto& a : p.addrs) os << a << ",";
    os << "] transport={";
    for (auto& [a, st] : p.transport) {
      os << a << ": ema=" << st.ema * 1000 << "ms n=" << st.samples
         << " fail=" << st.failPenalty << ", ";
    }
    os << "}\n";
  }
  os << "  outgoing in flight: " << outgoing_.size() << "\n";
  os << "  incoming tracked: " << incoming_.size() << "\n";
  return os.str();
}

}  // namespace mrl
