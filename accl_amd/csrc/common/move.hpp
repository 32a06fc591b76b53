// accl_amd move instruction — the unit of data-plane work.
//
// Analogue of the reference's dma_mover instruction stream (reference:
// kernels/cclo/hls/dma_mover/dma_mover.cpp:355-421 fetch, :433-703 decode):
// the scheduler (control plane) emits MoveDescs; movers (GPU workgroups /
// emulator worker loop) execute them. Each move is an elementwise N-ary
// operation dst[i] = f(cvt(src0[i]), ..., cvt(srcN-1[i])) with per-source
// dtype conversion (the arith + compression lanes of the reference,
// kernels/plugins/reduce_ops/reduce_ops.cpp:31-107 and
// hp_compression/hp_compression.cpp:72-144, fused into one pass).
#pragma once
#include "types.hpp"

namespace accl {

constexpr u32 MOVE_MAX_SRC = 8;

struct alignas(128) MoveDesc {
  u64 src[MOVE_MAX_SRC];  // source addresses (engine address space)
  u64 dst;
  u64 count;              // elements
  u32 nsrc;               // 1 = copy/cast, >=2 = reduce
  u32 func;               // ReduceFunction (ignored for nsrc==1)
  u8 src_dt[MOVE_MAX_SRC];// DataType per source
  u8 dst_dt;
  u8 inline_done;         // GPU: executed inline by the scheduler WG's small
                          // mover wave; fleet must skip (bookkeeping only)
  u8 tile_log2;           // per-move tile size override (0 = default 128KB);
                          // lets perf sweeps tune tiling without rebuilds
  u8 _pad[5];
  u64 epoch;              // published last (GPU queue); emulator ignores
};
static_assert(sizeof(MoveDesc) == 128, "");

// Completion state per in-flight move slot (GPU: device memory, agent scope).
struct alignas(64) MoveState {
  u32 tiles_total;
  u32 tiles_claimed;      // atomicAdd ticket
  u32 tiles_done;         // atomicAdd, release per tile
  u32 _pad[13];
};
static_assert(sizeof(MoveState) == 64, "");

constexpr u32 MOVE_RING = 64;          // in-flight move slots per engine
constexpr u64 MOVE_TILE_BYTES = 1u << 17;  // 128 KiB per mover tile

ACCL_HD inline u64 move_bytes(const MoveDesc& m) {
  return m.count * dtype_size(DataType(m.dst_dt));
}
ACCL_HD inline u64 move_tile_bytes(const MoveDesc& m) {
  return m.tile_log2 ? (1ull << m.tile_log2) : MOVE_TILE_BYTES;
}

ACCL_HD inline u32 move_tiles(const MoveDesc& m) {
  // tile over DST elements; sources are index-aligned
  u64 tb = move_tile_bytes(m);
  u64 bytes = move_bytes(m);
  u64 t = (bytes + tb - 1) / tb;
  return t ? u32(t) : 1;
}

ACCL_HD inline u64 move_tile_elems(const MoveDesc& m) {
  u32 dsz = dtype_size(DataType(m.dst_dt));
  return move_tile_bytes(m) / (dsz ? dsz : 1);
}

}  // namespace accl

#include "arith.hpp"

namespace accl {

// Scalar reference execution of one move over dst elements [lo, hi).
// The CPU emulator's whole data plane, and the GPU movers' fallback for
// dtype/function combinations without a vectorized path.
ACCL_HD inline void execute_move_range(const MoveDesc& m, u64 lo, u64 hi) {
  const DataType ddt = DataType(m.dst_dt);
  if (m.nsrc == 1) {
    const DataType sdt = DataType(m.src_dt[0]);
    const char* s = (const char*)m.src[0] + lo * dtype_size(sdt);
    char* d = (char*)m.dst + lo * dtype_size(ddt);
    convert_range(s, sdt, d, ddt, hi - lo);
    return;
  }
  // n-ary reduce: accumulate pairwise left-to-right (fixed order so every
  // engine — CPU or GPU — produces the same result for the same schedule).
  const ReduceFunction f = ReduceFunction(m.func);
  bool ints = dtype_is_int(ddt);
  for (u32 k = 0; k < m.nsrc; ++k) ints = ints && dtype_is_int(DataType(m.src_dt[k]));
  for (u64 i = lo; i < hi; ++i) {
    if (ints) {
      i64 acc = 0;
      for (u32 k = 0; k < m.nsrc; ++k) {
        const DataType st = DataType(m.src_dt[k]);
        i64 v = (st == DataType::int64)   ? ((const i64*)m.src[k])[i]
                : (st == DataType::int32) ? i64(((const i32*)m.src[k])[i])
                                          : i64(((const signed char*)m.src[k])[i]);
        acc = (k == 0) ? v : ((f == ReduceFunction::SUM) ? acc + v : (acc > v ? acc : v));
      }
      if (ddt == DataType::int64) ((i64*)m.dst)[i] = acc;
      else if (ddt == DataType::int32) ((i32*)m.dst)[i] = i32(acc);
      else ((signed char*)m.dst)[i] = (signed char)acc;
    } else {
      double acc = 0.0;
      for (u32 k = 0; k < m.nsrc; ++k) {
        double v = load_as_f64((const void*)m.src[k], i, DataType(m.src_dt[k]));
        acc = (k == 0) ? v : ((f == ReduceFunction::SUM) ? acc + v : (acc > v ? acc : v));
      }
      store_from_f64((void*)m.dst, i, ddt, acc);
    }
  }
}

}  // namespace accl
