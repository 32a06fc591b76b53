#include "util.hpp"
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <sstream>

namespace accl {

u64 wallclock_host_ns() {
  return u64(std::chrono::duration_cast<std::chrono::nanoseconds>(
                 std::chrono::steady_clock::now().time_since_epoch())
                 .count());
}

bool debug_enabled() {
  static const bool on = std::getenv("ACCL_DEBUG") != nullptr;
  return on;
}

void debug_log(const std::string& msg) {
  if (debug_enabled()) std::fprintf(stderr, "[accl] %s\n", msg.c_str());
}

std::string error_to_string(u32 bits) {
  if (bits == 0) return "OK";
  static const char* names[] = {
      "TIMEOUT", "MATCH", "SEGMENT", "COMPRESSION", "ARITH", "INVALID_OP",
      "ENGINE_DOWN", "RNDZV", "CREDIT", "TRANSPORT", "INVALID_ARG",
      "INFLIGHT_OVERFLOW", "COMM"};
  std::ostringstream os;
  bool first = true;
  for (u32 i = 0; i < sizeof(names) / sizeof(names[0]); ++i) {
    if (bits & (1u << i)) {
      if (!first) os << "|";
      os << names[i];
      first = false;
    }
  }
  u32 known = (1u << (sizeof(names) / sizeof(names[0]))) - 1;
  if (bits & ~known) os << (first ? "" : "|") << "UNKNOWN(0x" << std::hex
                        << (bits & ~known) << ")";
  return os.str();
}

}  // namespace accl
