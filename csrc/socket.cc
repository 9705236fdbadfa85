#include "socket.h"

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <ifaddrs.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <sys/uio.h>
#include <sys/un.h>
#include <pthread.h>
#include <unistd.h>

#include <cstring>
#include <deque>
#include <set>
#include <unordered_map>

namespace mrl {

std::string Addr::str() const {
  if (scheme == "unix") return "unix://" + name;
  return "tcp://" + host + ":" + std::to_string(port);
}

Addr parseAddr(const std::string& s) {
  Addr a;
  std::string rest = s;
  if (rest.rfind("unix://", 0) == 0 || rest.rfind("ipc://", 0) == 0) {
    a.scheme = "unix";
    a.name = rest.substr(rest.find("//") + 2);
    return a;
  }
  if (rest.rfind("tcp://", 0) == 0) rest = rest.substr(6);
  a.scheme = "tcp";
  auto colon = rest.rfind(':');
  if (colon == std::string::npos) {
    // bare port
    a.host = "0.0.0.0";
    a.port = std::atoi(rest.c_str());
  } else {
    a.host = rest.substr(0, colon);
    if (a.host.empty()) a.host = "0.0.0.0";
    a.port = std::atoi(rest.c_str() + colon + 1);
  }
  if (a.port <= 0 && a.port != 0) throw RpcError("bad address: " + s);
  return a;
}

std::vector<std::string> localIpv4Addresses() {
  std::vector<std::string> out{"127.0.0.1"};
  struct ifaddrs* ifs = nullptr;
  if (getifaddrs(&ifs) == 0) {
    for (auto* p = ifs; p; p = p->ifa_next) {
      if (!p->ifa_addr || p->ifa_addr->sa_family != AF_INET) continue;
      auto* sin = reinterpret_cast<sockaddr_in*>(p->ifa_addr);
      char buf[INET_ADDRSTRLEN];
      if (inet_ntop(AF_INET, &sin->sin_addr, buf, sizeof(buf))) {
        std::string s(buf);
        if (s != "127.0.0.1") out.push_back(s);
      }
    }
    freeifaddrs(ifs);
  }
  return out;
}

namespace {

void setNonBlocking(int fd) {
  int flags = fcntl(fd, F_GETFL, 0);
  fcntl(fd, F_SETFL, flags | O_NONBLOCK);
}

void setTcpOpts(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

// Large tensor frames want deep socket buffers (loopback included): the
// defaults (~200 KB) throttle 64 MB gradient/model messages to well under
// a GB/s.
void setBufSizes(int fd) {
  int sz = 8 << 20;
  setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sz, sizeof(sz));
  setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sz, sizeof(sz));
}

sockaddr_un makeUnixAddr(const std::string& name, socklen_t* len) {
  sockaddr_un sa{};
  sa.sun_family = AF_UNIX;
  // Abstract namespace: leading NUL byte (Linux-only, no filesystem residue).
  std::string path = "moolib-amd-" + name;
  if (path.size() + 1 > sizeof(sa.sun_path)) throw RpcError("unix addr too long");
  sa.sun_path[0] = '\0';
  std::memcpy(sa.sun_path + 1, path.data(), path.size());
  *len = offsetof(sockaddr_un, sun_path) + 1 + path.size();
  return sa;
}

constexpr size_t kMaxFrameLen = size_t(1) << 36;  // 64 GiB sanity cap

}  // namespace

struct OutMsg {
  std::string head;                 // 12-byte prefix + frame head
  std::vector<at::Tensor> tensors;  // blobs (cpu contiguous)
  size_t totalBytes() const {
    size_t n = head.size();
    for (auto& t : tensors) n += t.nbytes();
    return n;
  }
};

struct Conn {
  int fd = -1;
  bool connecting = false;  // outbound connect in flight
  bool wantWrite = false;
  bool isUnix = false;
  // ---- read state machine ----
  enum class RState { prefix, head, blobs };
  RState rstate = RState::prefix;
  char prefix[12];
  size_t prefixGot = 0;
  uint64_t totalLen = 0;
  uint32_t headLen = 0;
  std::string headBuf;
  size_t headGot = 0;
  Frame pending;
  size_t blobIdx = 0;
  size_t blobOff = 0;
  // ---- write state ----
  std::deque<OutMsg> writeQ;
  size_t frontOffset = 0;  // bytes of writeQ.front() already sent
};

// Fork hygiene: EnvPool forks worker processes that never exec, so every
// engine fd (sockets, epoll, eventfd) would stay open in the children and
// keep ports/peer connections alive after the parent dies. A pthread_atfork
// child handler closes all registered engine fds. The fd set has its own
// mutex (locked across the fork) because the conn maps are epoll-thread-
// owned and may be mid-mutation at fork time.
namespace {
std::mutex* forkRegMutex() {
  static std::mutex* m = new std::mutex();  // leaked: outlives exit order
  return m;
}
std::set<std::set<int>*>* forkRegSet() {
  static std::set<std::set<int>*>* s = new std::set<std::set<int>*>();
  return s;
}
void forkPrepare() { forkRegMutex()->lock(); }
void forkParent() { forkRegMutex()->unlock(); }
void forkChild() {
  for (std::set<int>* fds : *forkRegSet()) {
    for (int fd : *fds) ::close(fd);
    fds->clear();
  }
  forkRegMutex()->unlock();
}
void registerForkHygiene(std::set<int>* fds) {
  static std::once_flag once;
  std::call_once(once, [] { pthread_atfork(forkPrepare, forkParent, forkChild); });
  std::lock_guard<std::mutex> lk(*forkRegMutex());
  forkRegSet()->insert(fds);
}
void unregisterForkHygiene(std::set<int>* fds) {
  std::lock_guard<std::mutex> lk(*forkRegMutex());
  forkRegSet()->erase(fds);
}
}  // namespace

struct SocketEngine::Impl {
  SocketEngineCallbacks cbs;
  int epfd = -1;
  int wakeFd = -1;
  std::set<int> ownedFds;  // guarded by the fork-hygiene mutex

  void trackFd(int fd) {
    std::lock_guard<std::mutex> lk(*forkRegMutex());
    ownedFds.insert(fd);
  }
  void untrackFd(int fd) {
    std::lock_guard<std::mutex> lk(*forkRegMutex());
    ownedFds.erase(fd);
  }
  std::thread thread;
  std::atomic<bool> stopping{false};
  std::atomic<uint64_t> nextId{1};

  std::mutex cmdMu;
  std::vector<std::function<void()>> cmds;
  bool stopped = false;

  // Owned by epoll thread only:
  std::unordered_map<ConnId, Conn> conns;
  std::unordered_map<int, ConnId> fdToConn;
  std::unordered_map<int, bool> listeners;  // fd -> isUnix

  void wake() {
    uint64_t one = 1;
    [[maybe_unused]] ssize_t r = write(wakeFd, &one, sizeof(one));
  }

  void post(std::function<void()> f) {
    {
      std::lock_guard<std::mutex> lk(cmdMu);
      if (stopped) return;
      cmds.push_back(std::move(f));
    }
    wake();
  }

  void epollCtl(int op, int fd, uint32_t events, uint64_t data) {
    epoll_event ev{};
    ev.events = events;
    ev.data.u64 = data;
    if (epoll_ctl(epfd, op, fd, &ev) != 0 && op != EPOLL_CTL_DEL) {
      MRL_LOG_ERROR("epoll_ctl failed: %s", strerror(errno));
    }
  }

  void closeConn(ConnId id, const std::string& reason) {
    auto it = conns.find(id);
    if (it == conns.end()) return;
    int fd = it->second.fd;
    epollCtl(EPOLL_CTL_DEL, fd, 0, 0);
    ::close(fd);
    untrackFd(fd);
    fdToConn.erase(fd);
    conns.erase(it);
    if (cbs.onClosed && !stopping.load()) cbs.onClosed(id, reason);
  }

  void updateEvents(ConnId id, Conn& c) {
    uint32_t ev = EPOLLIN;
    if (c.wantWrite || c.connecting) ev |= EPOLLOUT;
    epollCtl(EPOLL_CTL_MOD, c.fd, ev, id);
  }

  // ------------------------------------------------------------- writing
  void flushWrites(ConnId id, Conn& c) {
    while (!c.writeQ.empty()) {
      // Build iovec batch from the queue front.
      iovec iov[64];
      int niov = 0;
      size_t skip = c.frontOffset;
      for (auto& m : c.writeQ) {
        if (niov >= 60) break;
        auto addSeg = [&](const void* base, size_t len) {
          if (len == 0) return;
          if (skip >= len) {
            skip -= len;
            return;
          }
          if (niov < 60) {
            iov[niov].iov_base = const_cast<char*>(static_cast<const char*>(base)) + skip;
            iov[niov].iov_len = len - skip;
            ++niov;
          }
          skip = 0;
        };
        addSeg(m.head.data(), m.head.size());
        for (auto& t : m.tensors) addSeg(t.data_ptr(), t.nbytes());
        if (niov >= 60) break;
        skip = 0;  // only the first message has an offset
      }
      if (niov == 0) break;
      ssize_t n = writev(c.fd, iov, niov);
      if (n < 0) {
        if (errno == EAGAIN || errno == EWOULDBLOCK) break;
        if (errno == EINTR) continue;
        closeConn(id, std::string("write error: ") + strerror(errno));
        return;
      }
      // Consume n bytes from the queue front.
      size_t left = static_cast<size_t>(n) + c.frontOffset;
      while (!c.writeQ.empty()) {
        size_t msgLen = c.writeQ.front().totalBytes();
        if (left >= msgLen) {
          left -= msgLen;
          c.writeQ.pop_front();
        } else {
          break;
        }
      }
      c.frontOffset = left;
    }
    bool want = !c.writeQ.empty();
    if (want != c.wantWrite) {
      c.wantWrite = want;
      updateEvents(id, c);
    }
  }

  // ------------------------------------------------------------- reading
  // Returns false if the connection died.
  bool handleReadable(ConnId id, Conn& c) {
    while (true) {
      if (c.rstate == Conn::RState::prefix) {
        ssize_t n = read(c.fd, c.prefix + c.prefixGot, 12 - c.prefixGot);
        if (n == 0) {
          closeConn(id, "peer closed connection");
          return false;
        }
        if (n < 0) {
          if (errno == EAGAIN || errno == EWOULDBLOCK) return true;
          if (errno == EINTR) continue;
          closeConn(id, std::string("read error: ") + strerror(errno));
          return false;
        }
        c.prefixGot += n;
        if (c.prefixGot < 12) continue;
        std::memcpy(&c.totalLen, c.prefix, 8);
        std::memcpy(&c.headLen, c.prefix + 8, 4);
        if (c.totalLen > kMaxFrameLen || c.headLen > c.totalLen - 4) {
          closeConn(id, "malformed frame header");
          return false;
        }
        c.headBuf.resize(c.headLen);
        c.headGot = 0;
        c.rstate = Conn::RState::head;
      } else if (c.rstate == Conn::RState::head) {
        ssize_t n = read(c.fd, c.headBuf.data() + c.headGot, c.headLen - c.headGot);
        if (n == 0) {
          closeConn(id, "peer closed connection");
          return false;
        }
        if (n < 0) {
          if (errno == EAGAIN || errno == EWOULDBLOCK) return true;
          if (errno == EINTR) continue;
          closeConn(id, std::string("read error: ") + strerror(errno));
          return false;
        }
        c.headGot += n;
        if (c.headGot < c.headLen) continue;
        try {
          c.pending = decodeFrameHead(c.headBuf);
        } catch (const std::exception& e) {
          closeConn(id, std::string("bad frame: ") + e.what());
          return false;
        }
        uint64_t blobBytes = 0;
        for (auto& t : c.pending.tensors) blobBytes += t.nbytes();
        if (blobBytes != c.totalLen - 4 - c.headLen) {
          closeConn(id, "frame blob length mismatch");
          return false;
        }
        c.blobIdx = 0;
        c.blobOff = 0;
        c.rstate = Conn::RState::blobs;
      } else {
        // blobs: readv directly into tensor storage
        while (c.blobIdx < c.pending.tensors.size() &&
               c.pending.tensors[c.blobIdx].nbytes() == 0) {
          ++c.blobIdx;
        }
        if (c.blobIdx >= c.pending.tensors.size()) {
          deliver(id, c);
          continue;
        }
        iovec iov[16];
        int niov = 0;
        for (size_t i = c.blobIdx; i < c.pending.tensors.size() && niov < 16; ++i) {
          auto& t = c.pending.tensors[i];
          size_t off = (i == c.blobIdx) ? c.blobOff : 0;
          iov[niov].iov_base = static_cast<char*>(t.data_ptr()) + off;
          iov[niov].iov_len = t.nbytes() - off;
          ++niov;
        }
        ssize_t n = readv(c.fd, iov, niov);
        if (n == 0) {
          closeConn(id, "peer closed connection");
          return false;
        }
        if (n < 0) {
          if (errno == EAGAIN || errno == EWOULDBLOCK) return true;
          if (errno == EINTR) continue;
          closeConn(id, std::string("read error: ") + strerror(errno));
          return false;
        }
        size_t got = static_cast<size_t>(n);
        while (got > 0) {
          auto& t = c.pending.tensors[c.blobIdx];
          size_t rem = t.nbytes() - c.blobOff;
          if (got >= rem) {
            got -= rem;
            ++c.blobIdx;
            c.blobOff = 0;
            while (c.blobIdx < c.pending.tensors.size() &&
                   c.pending.tensors[c.blobIdx].nbytes() == 0) {
              ++c.blobIdx;
            }
          } else {
            c.blobOff += got;
            got = 0;
          }
        }
        if (c.blobIdx >= c.pending.tensors.size()) deliver(id, c);
      }
    }
  }

  void deliver(ConnId id, Conn& c) {
    Frame f = std::move(c.pending);
    c.pending = Frame();
    c.rstate = Conn::RState::prefix;
    c.prefixGot = 0;
    if (cbs.onFrame) cbs.onFrame(id, std::move(f));
  }

  void handleAccept(int listenFd, bool isUnix) {
    while (true) {
      int fd = accept4(listenFd, nullptr, nullptr, SOCK_NONBLOCK | SOCK_CLOEXEC);
      if (fd < 0) {
        if (errno == EAGAIN || errno == EWOULDBLOCK || errno == EINTR) return;
        MRL_LOG_ERROR("accept failed: %s", strerror(errno));
        return;
      }
      if (!isUnix) setTcpOpts(fd);
      setBufSizes(fd);
      trackFd(fd);
      ConnId id = nextId.fetch_add(1);
      Conn& c = conns[id];
      c.fd = fd;
      c.isUnix = isUnix;
      fdToConn[fd] = id;
      epollCtl(EPOLL_CTL_ADD, fd, EPOLLIN, id);
      if (cbs.onAccept) cbs.onAccept(id);
    }
  }

  void loop() {
    epoll_event evs[128];
    while (!stopping.load(std::memory_order_acquire)) {
      int n = epoll_wait(epfd, evs, 128, 200);
      if (n < 0) {
        if (errno == EINTR) continue;
        MRL_LOG_ERROR("epoll_wait failed: %s", strerror(errno));
        break;
      }
      // Run queued commands first.
      std::vector<std::function<void()>> batch;
      {
        std::lock_guard<std::mutex> lk(cmdMu);
        batch.swap(cmds);
      }
      for (auto& f : batch) f();
      for (int i = 0; i < n; ++i) {
        uint64_t data = evs[i].data.u64;
        if (data == 0) {  // wake eventfd
          uint64_t junk;
          while (read(wakeFd, &junk, sizeof(junk)) > 0) {
          }
          continue;
        }
        if (data & (uint64_t(1) << 63)) {  // listener
          int lfd = static_cast<int>(data & 0x7fffffff);
          auto lit = listeners.find(lfd);
          if (lit != listeners.end()) handleAccept(lfd, lit->second);
          continue;
        }
        ConnId id = data;
        auto it = conns.find(id);
        if (it == conns.end()) continue;
        Conn& c = it->second;
        uint32_t e = evs[i].events;
        if (c.connecting) {
          if (e & (EPOLLOUT | EPOLLERR | EPOLLHUP)) {
            int err = 0;
            socklen_t len = sizeof(err);
            getsockopt(c.fd, SOL_SOCKET, SO_ERROR, &err, &len);
            if (err != 0) {
              closeConn(id, std::string("connect failed: ") + strerror(err));
              continue;
            }
            c.connecting = false;
            updateEvents(id, c);
            if (cbs.onConnected) cbs.onConnected(id);
            if (conns.count(id)) flushWrites(id, conns[id]);
          }
          continue;
        }
        if (e & (EPOLLERR | EPOLLHUP)) {
          closeConn(id, "connection error/hangup");
          continue;
        }
        if (e & EPOLLIN) {
          if (!handleReadable(id, c)) continue;
        }
        if (e & EPOLLOUT) {
          auto it2 = conns.find(id);
          if (it2 != conns.end()) flushWrites(id, it2->second);
        }
      }
    }
  }
};

SocketEngine::SocketEngine(SocketEngineCallbacks cbs) : impl_(new Impl()) {
  impl_->cbs = std::move(cbs);
  impl_->epfd = epoll_create1(EPOLL_CLOEXEC);
  impl_->wakeFd = eventfd(0, EFD_NONBLOCK | EFD_CLOEXEC);
  if (impl_->epfd < 0 || impl_->wakeFd < 0) throw RpcError("epoll/eventfd creation failed");
  impl_->epollCtl(EPOLL_CTL_ADD, impl_->wakeFd, EPOLLIN, 0);
  impl_->trackFd(impl_->epfd);
  impl_->trackFd(impl_->wakeFd);
  registerForkHygiene(&impl_->ownedFds);
  impl_->thread = std::thread([this] { impl_->loop(); });
}

SocketEngine::~SocketEngine() { shutdown(); }

void SocketEngine::shutdown() {
  if (impl_->stopping.exchange(true)) return;
  {
    std::lock_guard<std::mutex> lk(impl_->cmdMu);
    impl_->stopped = true;
    impl_->cmds.clear();
  }
  impl_->wake();
  if (impl_->thread.joinable()) impl_->thread.join();
  for (auto& [id, c] : impl_->conns) ::close(c.fd);
  impl_->conns.clear();
  for (auto& [fd, isUnix] : impl_->listeners) ::close(fd);
  impl_->listeners.clear();
  ::close(impl_->epfd);
  ::close(impl_->wakeFd);
  unregisterForkHygiene(&impl_->ownedFds);
  {
    std::lock_guard<std::mutex> lk(*forkRegMutex());
    impl_->ownedFds.clear();
  }
}

ConnId SocketEngine::connect(const std::string& addrStr) {
  Addr a = parseAddr(addrStr);
  ConnId id = impl_->nextId.fetch_add(1);
  // Resolve in the caller's thread (blocking DNS never stalls the reactor).
  sockaddr_storage ss{};
  socklen_t slen = 0;
  bool isUnix = a.scheme == "unix";
  if (isUnix) {
    auto sa = makeUnixAddr(a.name, &slen);
    std::memcpy(&ss, &sa, sizeof(sa));
  } else {
    addrinfo hints{};
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    addrinfo* res = nullptr;
    std::string portStr = std::to_string(a.port);
    int rc = getaddrinfo(a.host.c_str(), portStr.c_str(), &hints, &res);
    if (rc != 0 || !res) {
      // Report failure asynchronously via onClosed so callers see a uniform path.
      impl_->post([this, id, a] {
        if (impl_->cbs.onClosed) impl_->cbs.onClosed(id, "dns resolution failed for " + a.host);
      });
      return id;
    }
    std::memcpy(&ss, res->ai_addr, res->ai_addrlen);
    slen = res->ai_addrlen;
    freeaddrinfo(res);
  }
  impl_->post([this, id, ss, slen, isUnix] {
    int fd = ::socket(isUnix ? AF_UNIX : AF_INET, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
    if (fd < 0) {
      if (impl_->cbs.onClosed) impl_->cbs.onClosed(id, "socket() failed");
      return;
    }
    if (!isUnix) setTcpOpts(fd);
    setBufSizes(fd);
    int rc = ::connect(fd, reinterpret_cast<const sockaddr*>(&ss), slen);
    if (rc != 0 && errno != EINPROGRESS) {
      std::string reason = std::string("connect failed: ") + strerror(errno);
      ::close(fd);
      if (impl_->cbs.onClosed) impl_->cbs.onClosed(id, reason);
      return;
    }
    impl_->trackFd(fd);
    Conn& c = impl_->conns[id];
    c.fd = fd;
    c.isUnix = isUnix;
    c.connecting = (rc != 0);
    impl_->fdToConn[fd] = id;
    impl_->epollCtl(EPOLL_CTL_ADD, fd, EPOLLIN | (c.connecting ? EPOLLOUT : 0), id);
    if (!c.connecting) {
      if (impl_->cbs.onConnected) impl_->cbs.onConnected(id);
      auto it = impl_->conns.find(id);
      if (it != impl_->conns.end()) impl_->flushWrites(id, it->second);
    }
  });
  return id;
}

std::vector<std::string> SocketEngine::listen(const std::string& addrStr) {
  Addr a = parseAddr(addrStr);
  int fd = -1;
  std::vector<std::string> bound;
  if (a.scheme == "unix") {
    fd = ::socket(AF_UNIX, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
    if (fd < 0) throw RpcError("socket() failed");
    socklen_t slen;
    auto sa = makeUnixAddr(a.name, &slen);
    if (bind(fd, reinterpret_cast<sockaddr*>(&sa), slen) != 0) {
      ::close(fd);
      throw RpcError("bind failed for " + addrStr + ": " + strerror(errno));
    }
    bound.push_back("unix://" + a.name);
  } else {
    fd = ::socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
    if (fd < 0) throw RpcError("socket() failed");
    int one = 1;
    setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in sin{};
    sin.sin_family = AF_INET;
    sin.sin_port = htons(static_cast<uint16_t>(a.port));
    if (inet_pton(AF_INET, a.host.c_str(), &sin.sin_addr) != 1) {
      ::close(fd);
      throw RpcError("bad listen host: " + a.host);
    }
    if (bind(fd, reinterpret_cast<sockaddr*>(&sin), sizeof(sin)) != 0) {
      ::close(fd);
      throw RpcError("bind failed for " + addrStr + ": " + strerror(errno));
    }
    sockaddr_in got{};
    socklen_t glen = sizeof(got);
    getsockname(fd, reinterpret_cast<sockaddr*>(&got), &glen);
    int port = ntohs(got.sin_port);
    if (a.host == "0.0.0.0") {
      for (auto& ip : localIpv4Addresses()) bound.push_back("tcp://" + ip + ":" + std::to_string(port));
    } else {
      bound.push_back("tcp://" + a.host + ":" + std::to_string(port));
    }
  }
  if (::listen(fd, 512) != 0) {
    ::close(fd);
    throw RpcError(std::string("listen failed: ") + strerror(errno));
  }
  bool isUnix = a.scheme == "unix";
  impl_->trackFd(fd);
  impl_->post([this, fd, isUnix] {
    impl_->listeners[fd] = isUnix;
    impl_->epollCtl(EPOLL_CTL_ADD, fd, EPOLLIN, (uint64_t(1) << 63) | static_cast<uint32_t>(fd));
  });
  return bound;
}

void SocketEngine::send(ConnId id, Frame f) {
  // Build the serialized head in the caller's thread; ensure blobs are CPU
  // contiguous so writev can take their pointers directly.
  OutMsg m;
  for (auto& t : f.tensors) {
    if (!t.device().is_cpu()) throw RpcError("socket send: tensor must be on CPU");
    if (!t.is_contiguous()) t = t.contiguous();
  }
  m.head = encodeFrameHead(f);
  m.tensors = std::move(f.tensors);
  impl_->post([this, id, m = std::move(m)]() mutable {
    auto it = impl_->conns.find(id);
    if (it == impl_->conns.end()) return;  // connection is gone; reliability layer resends
    Conn& c = it->second;
    c.writeQ.push_back(std::move(m));
    if (!c.connecting) impl_->flushWrites(id, c);
  });
}

void SocketEngine::close(ConnId id) {
  impl_->post([this, id] { impl_->closeConn(id, "closed locally"); });
}

}  // namespace mrl
