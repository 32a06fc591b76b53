// First-fit heap allocator over the arena's buffer heap (host-side only).
// The analogue of the reference's buffer factories carving device memory
// (reference: driver/xrt/include/accl/buffer.hpp:32-203; 4 KiB alignment rule
// ACCL_FPGA_ALIGNMENT, driver/xrt/include/accl/common.hpp:29).
#pragma once
#include <map>
#include <mutex>
#include <stdexcept>
#include "../common/types.hpp"

namespace accl {

class HeapAlloc {
 public:
  static constexpr u64 ALIGN = 4096;
  void init(u64 base, u64 bytes) {
    base_ = (base + ALIGN - 1) & ~(ALIGN - 1);
    end_ = base + bytes;
    free_.clear();
    used_.clear();
    if (base_ < end_) free_[base_] = end_ - base_;
  }
  u64 alloc(u64 bytes) {
    std::lock_guard<std::mutex> lk(mu_);
    bytes = (bytes + ALIGN - 1) & ~(ALIGN - 1);
    if (!bytes) bytes = ALIGN;
    for (auto it = free_.begin(); it != free_.end(); ++it) {
      if (it->second >= bytes) {
        u64 off = it->first, sz = it->second;
        free_.erase(it);
        if (sz > bytes) free_[off + bytes] = sz - bytes;
        used_[off] = bytes;
        return off;
      }
    }
    throw std::runtime_error("accl: arena heap exhausted");
  }
  void free_block(u64 off) {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = used_.find(off);
    if (it == used_.end()) return;
    u64 sz = it->second;
    used_.erase(it);
    // coalesce with neighbours
    auto nx = free_.upper_bound(off);
    if (nx != free_.end() && off + sz == nx->first) {
      sz += nx->second;
      nx = free_.erase(nx);
    }
    if (nx != free_.begin()) {
      auto pv = std::prev(nx);
      if (pv->first + pv->second == off) {
        pv->second += sz;
        return;
      }
    }
    free_[off] = sz;
  }
  u64 bytes_free() const {
    std::lock_guard<std::mutex> lk(mu_);
    u64 t = 0;
    for (auto& kv : free_) t += kv.second;
    return t;
  }

 private:
  u64 base_ = 0, end_ = 0;
  mutable std::mutex mu_;
  std::map<u64, u64> free_;   // offset -> size
  std::map<u64, u64> used_;
};

}  // namespace accl
