#pragma once
#include <stdexcept>
#include <string>
#include "../common/types.hpp"

namespace accl {

u64 wallclock_host_ns();

class accl_error : public std::runtime_error {
 public:
  explicit accl_error(const std::string& what, u32 bits = 0)
      : std::runtime_error(what), bits_(bits) {}
  u32 bits() const { return bits_; }
 private:
  u32 bits_;
};

// decode an error bitmask (reference: ACCL::check_return_value /
// error_code_to_string, driver/xrt/src/accl.cpp:1210-1234)
std::string error_to_string(u32 bits);

// ACCL_DEBUG-gated host-side debug log (reference: debug() stderr logging
// under ACCL_DEBUG, driver/xrt/include/accl/common.hpp:38-58)
bool debug_enabled();
void debug_log(const std::string& msg);

}  // namespace accl
