// moolib_amd core runtime — foundation utilities.
//
// MI355X-native re-implementation of the capabilities of moolib's L0 layer
// (reference: src/logging.h, src/util.h, src/synchronization.h). Brand-new
// design: std::mutex/condition_variable based (no x86 pause-spin idioms),
// FNV-1a function ids, steady-clock timers.
#pragma once

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <functional>
#include <mutex>
#include <random>
#include <stdexcept>
#include <string>
#include <string_view>
#include <vector>

namespace mrl {

// ---------------------------------------------------------------- logging

enum class LogLevel : int { none = 0, error = 1, info = 2, verbose = 3, debug = 4 };

extern std::atomic<int> g_logLevel;
// Optional sink installed by the Python layer (routes into `logging`).
extern std::function<void(int, const std::string&)> g_logSink;
extern std::mutex g_logMutex;

void logLine(LogLevel level, const char* fmt, ...) __attribute__((format(printf, 2, 3)));

#define MRL_LOG_ERROR(...) ::mrl::logLine(::mrl::LogLevel::error, __VA_ARGS__)
#define MRL_LOG_INFO(...) ::mrl::logLine(::mrl::LogLevel::info, __VA_ARGS__)
#define MRL_LOG_VERBOSE(...) ::mrl::logLine(::mrl::LogLevel::verbose, __VA_ARGS__)
#define MRL_LOG_DEBUG(...) ::mrl::logLine(::mrl::LogLevel::debug, __VA_ARGS__)

// ----------------------------------------------------------------- errors

// RPC-level error surfaced to Python as moolib_amd.RpcError.
struct RpcError : std::runtime_error {
  explicit RpcError(std::string msg) : std::runtime_error(std::move(msg)) {}
};

[[noreturn]] inline void fatal(const std::string& msg) {
  std::fprintf(stderr, "moolib_amd fatal: %s\n", msg.c_str());
  std::abort();
}

// ------------------------------------------------------------------ time

using Clock = std::chrono::steady_clock;
using TimePoint = Clock::time_point;

inline TimePoint now() { return Clock::now(); }
inline double secondsSince(TimePoint t) {
  return std::chrono::duration<double>(now() - t).count();
}
inline double toSeconds(Clock::duration d) { return std::chrono::duration<double>(d).count(); }

// ------------------------------------------------------------------- ids

// 64-bit FNV-1a — function name -> fid mapping (both sides compute it
// locally; no remote id-resolution round trip needed, unlike the
// reference's murmur3+__reqFindFunction scheme, rpc.cc:1766, 2094).
inline uint64_t fnv1a64(std::string_view s) {
  uint64_t h = 1469598103934665603ull;
  for (unsigned char c : s) {
    h ^= c;
    h *= 1099511628211ull;
  }
  return h;
}

std::string randomUid();  // hex string, cryptographically-random-ish
uint64_t randomU64();

// ------------------------------------------------------------ wire codec

// Little-endian append-only writer over std::string. All wire structures in
// the transport/rpc layers use this (design is ours; the reference uses a
// two-pass templated serializer, src/serialization.h:143-459).
struct WireWriter {
  std::string out;
  void raw(const void* p, size_t n) { out.append(reinterpret_cast<const char*>(p), n); }
  template <typename T>
  void pod(T v) {
    static_assert(std::is_trivially_copyable_v<T>);
    raw(&v, sizeof(T));
  }
  void u8(uint8_t v) { pod(v); }
  void u32(uint32_t v) { pod(v); }
  void u64(uint64_t v) { pod(v); }
  void i64(int64_t v) { pod(v); }
  void f64(double v) { pod(v); }
  void str(std::string_view s) {
    u32(static_cast<uint32_t>(s.size()));
    raw(s.data(), s.size());
  }
};

struct WireReader {
  const char* p;
  const char* end;
  WireReader(const void* data, size_t n)
      : p(reinterpret_cast<const char*>(data)), end(reinterpret_cast<const char*>(data) + n) {}
  explicit WireReader(std::string_view s) : WireReader(s.data(), s.size()) {}
  void need(size_t n) const {
    if (static_cast<size_t>(end - p) < n) throw RpcError("wire: truncated message");
  }
  void raw(void* dst, size_t n) {
    need(n);
    std::memcpy(dst, p, n);
    p += n;
  }
  template <typename T>
  T pod() {
    T v;
    raw(&v, sizeof(T));
    return v;
  }
  uint8_t u8() { return pod<uint8_t>(); }
  uint32_t u32() { return pod<uint32_t>(); }
  uint64_t u64() { return pod<uint64_t>(); }
  int64_t i64() { return pod<int64_t>(); }
  double f64() { return pod<double>(); }
  std::string_view str() {
    uint32_t n = u32();
    need(n);
    std::string_view s(p, n);
    p += n;
    return s;
  }
  size_t remaining() const { return end - p; }
};

}  // namespace mrl
