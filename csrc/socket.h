// Epoll socket engine: nonblocking TCP + abstract unix-domain sockets with
// frame-level zero-copy tensor iovecs.
//
// Capability parity with the reference's transport layer
// (src/transports/socket.{h,cc} + ipc.{h,cc}); our design is a
// single-threaded reactor per Rpc instance: one epoll thread owns every
// socket structure, all external entry points (connect/listen/send/close)
// post closures to a command queue and wake the reactor via eventfd — no
// per-socket locking at all.
#pragma once

#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "message.h"

namespace mrl {

using ConnId = uint64_t;

struct SocketEngineCallbacks {
  // All callbacks run on the epoll thread; keep them lean (the Rpc layer
  // forwards work to the scheduler).
  std::function<void(ConnId, Frame&&)> onFrame;
  std::function<void(ConnId, const std::string& reason)> onClosed;
  std::function<void(ConnId)> onAccept;     // inbound connection established
  std::function<void(ConnId)> onConnected;  // outbound connect completed
};

// Parsed address: scheme "tcp" or "unix".
struct Addr {
  std::string scheme;  // "tcp" | "unix"
  std::string host;    // tcp only
  int port = 0;        // tcp only
  std::string name;    // unix only (abstract namespace)
  std::string str() const;
};
Addr parseAddr(const std::string& s);

class SocketEngine {
 public:
  explicit SocketEngine(SocketEngineCallbacks cbs);
  ~SocketEngine();

  // Asynchronous connect; onConnected/onClosed fires later.
  ConnId connect(const std::string& addr);
  // Listen; returns the bound addresses (port resolved if 0). Throws on error.
  std::vector<std::string> listen(const std::string& addr);
  // Thread-safe enqueue; silently drops if conn is gone (onClosed already ran).
  void send(ConnId id, Frame f);
  void close(ConnId id);
  void shutdown();

 private:
  struct Impl;
  std::unique_ptr<Impl> impl_;
};

// Enumerate local IPv4 addresses ("a.b.c.d"), loopback first.
std::vector<std::string> localIpv4Addresses();

}  // namespace mrl
