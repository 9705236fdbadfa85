#include "common.h"

#include <cinttypes>

namespace mrl {

std::atomic<int> g_logLevel{static_cast<int>(LogLevel::error)};
std::function<void(int, const std::string&)> g_logSink;
std::mutex g_logMutex;

void logLine(LogLevel level, const char* fmt, ...) {
  if (static_cast<int>(level) > g_logLevel.load(std::memory_order_relaxed)) return;
  char buf[4096];
  va_list ap;
  va_start(ap, fmt);
  std::vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  std::lock_guard<std::mutex> lk(g_logMutex);
  if (g_logSink) {
    g_logSink(static_cast<int>(level), buf);
  } else {
    std::fprintf(stderr, "[moolib_amd %s] %s\n",
                 level == LogLevel::error     ? "E"
                 : level == LogLevel::info    ? "I"
                 : level == LogLevel::verbose ? "V"
                                              : "D",
                 buf);
  }
}

uint64_t randomU64() {
  static thread_local std::mt19937_64 rng = [] {
    std::random_device rd;
    std::seed_seq seq{rd(), rd(), rd(), rd(),
                      static_cast<unsigned>(std::chrono::system_clock::now().time_since_epoch().count()),
                      static_cast<unsigned>(reinterpret_cast<uintptr_t>(&rng))};
    return std::mt19937_64(seq);
  }();
  return rng();
}

std::string randomUid() {
  char buf[33];
  std::snprintf(buf, sizeof(buf), "%016" PRIx64 "%016" PRIx64, randomU64(), randomU64());
  return std::string(buf);
}

}  // namespace mrl
