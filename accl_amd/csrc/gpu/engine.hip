// The persistent CCLO engine for MI355X (gfx950): TWO kernels on separate
// non-blocking streams (docs/DESIGN.md "Persistent engine").
//
//  * accl_scheduler_kernel — one 128-thread WG: wave 0 lane 0 runs the
//    collective scheduler (common/sched.hpp, the microcode analogue of
//    reference ccl_offload_control.c) with the Cclo state staged in LDS;
//    wave 1 is the inline small-mover (sub-32KB moves over an LDS mailbox).
//  * accl_mover_kernel — the mover fleet (~10 of 16 wave slots per CU):
//    static rotated tile partitioning, replicated packed doorbells,
//    non-temporal software-pipelined copy / cast / n-ary reduce tiles
//    between local HBM and peer HBM over xGMI (the data plane: reference
//    dma_mover + reduce_ops + hp_compression, kernels/cclo/hls, plugins).
//
// Memory-ordering discipline (MI355X_MICROARCH.md §Workgroup dispatch):
//  * movers: payload stores -> s_waitcnt vmcnt(0) (asm) -> release fence
//    (skippable when the tile used only non-temporal accesses) ->
//    tiles_done release-add (agent) — the release chain the scheduler
//    extends to peers when it publishes slot headers (system release).
//  * all polls are relaxed + s_sleep; one acquire fence after a match.
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <cstdlib>
#include "engine.hpp"

namespace accl {

#define AGENT __HIP_MEMORY_SCOPE_AGENT

// ------------------------------------------------------------- GpuMover
__device__ u32 GpuMover::submit(const MoveDesc& m) {
  u64 h = head_cache;
  u32 slot = u32(h % MOVE_RING);
  if (h >= MOVE_RING) {
    // wait for the slot's previous occupant to finish (movers never block,
    // so this always terminates)
    MoveState& prev = st[slot];
    while (__hip_atomic_load(&prev.tiles_done, __ATOMIC_RELAXED, AGENT) <
           prev.tiles_total)
      __builtin_amdgcn_s_sleep(1);
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  u64 mbytes = move_bytes(m);
  bool small = small_mb && mbytes <= (small_max ? small_max : SMALL_INLINE_MAX);
  MoveDesc& d = ring[slot];
  MoveState& s = st[slot];
  // Tile-size policy (measured, profiles/r2: 256 MiB 856->1199 GB/s at
  // 512 KiB tiles, 1 GiB 1486->2016 at 1 MiB): fewer, fatter tiles beat
  // maximum fan-out for big moves — per-wave wake/fence/desc overhead
  // amortizes over more streaming. ACCL_TILE_KB overrides.
  u8 tl = u8(tile_log2);
  if (!tl) {
    if (mbytes >= (768ull << 20)) tl = 20;      // 1 MiB tiles
    else if (mbytes >= (96ull << 20)) tl = 19;  // 512 KiB tiles
  }
  const_cast<MoveDesc&>(m).tile_log2 = tl;
  if (small) {
    // kick the sibling wave over LDS FIRST, overlap the ring bookkeeping
    // with its copy, then block until its system-release completes
    // (a sub-32KB move is ~2 us — a fleet handoff costs 10x that)
    if (dbg) dbg[5] = wallclock();
    SmallMb* mb = (SmallMb*)small_mb;
    mb->d = m;
    // sibling executes tile 0 only: pick a tile size covering the WHOLE
    // move (the default 128 KiB covers any cutoff up to ACCL_INLINE_KB=128)
    u8 cov = 17;
    while ((1ull << cov) < mbytes) ++cov;
    mb->d.tile_log2 = cov;
    u64 sq = ++small_seq;
    __hip_atomic_store(&mb->seq, sq, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_WORKGROUP);
    d = m;
    d.inline_done = 1;
    // epoch published LAST with release: lagging movers validate descs
    // seqlock-style against it (fields -> release -> epoch)
    __hip_atomic_store(&d.epoch, h + 1, __ATOMIC_RELEASE, AGENT);
    s.tiles_total = 0;
    __hip_atomic_store(&s.tiles_done, 0u, __ATOMIC_RELAXED, AGENT);
    // bounded wait: if the sibling wave is wedged/slow, execute the move
    // here (scalar). Double execution of a NON-aliasing elementwise move is
    // idempotent (same inputs, same outputs); an ALIASING fused reduce
    // (dst is also an operand, e.g. dst += slot) is NOT — re-execution or a
    // racing partial sibling pass would double-accumulate, so for those we
    // only ever wait (the sibling shares our workgroup: if it is dead the
    // kernel is gone anyway, and the host's bounded waits still fire).
    bool aliased = false;
    u64 dbytes = m.count * dtype_size(DataType(m.dst_dt));
    for (u32 k = 0; k < m.nsrc; ++k) {
      u64 sb = m.count * dtype_size(DataType(m.src_dt[k]));
      if (m.src[k] < m.dst + dbytes && m.dst < m.src[k] + sb) aliased = true;
    }
    u64 dl = wallclock() + 100000;  // 1 ms
    for (;;) {
      if (__hip_atomic_load(&mb->done, __ATOMIC_ACQUIRE,
                            __HIP_MEMORY_SCOPE_WORKGROUP) == sq)
        break;
      if (!aliased && wallclock() > dl) {
        execute_move_range(m, 0, m.count);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        break;
      }
      __builtin_amdgcn_s_sleep(1);
    }
    if (dbg) { dbg[6] = wallclock(); dbg[7]++; }
    // no fleet doorbell: movers discover this slot (and skip it) when the
    // next fleet move advances the packed head past it; the ring/desc
    // stores above are ordered by that submit's release fence
    head_cache = h + 1;
    if (dbg) dbg[13] = wallclock();
    return u32(h) | INLINE_TOKEN;
  }
  d = m;
  d.inline_done = 0;
  __hip_atomic_store(&d.epoch, h + 1, __ATOMIC_RELEASE, AGENT);
  s.tiles_total = move_tiles(m);
  __hip_atomic_store(&s.tiles_claimed, 0u, __ATOMIC_RELAXED, AGENT);
  __hip_atomic_store(&s.tiles_done, 0u, __ATOMIC_RELAXED, AGENT);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (dbg) dbg[0] = wallclock();
  __hip_atomic_store(head, h + 1, __ATOMIC_RELAXED, AGENT);
  // doorbell word packs (head, latest move's tile count): a wave that is
  // caught up decides ownership without touching the descriptor line, so a
  // 1-tile move wakes 1 wave's worth of desc traffic, not the whole fleet's.
  // ATOMIC relaxed stores (-> global_store sc0 sc1, write-through): a plain
  // store stays dirty in THIS XCD's L2, which the other XCDs' L2-served
  // polls never see until a later fence writes it back — the final move of
  // a flow set then never executes and the engine times out (guideline-16
  // "plain flag" invalid form; root cause of the round-1 fresh-box hang).
  u64 packed = ((h + 1) << 24) | (s.tiles_total & 0xFFFFFFu);
  for (u32 i = 0; i < DOORBELL_REPS; ++i)
    __hip_atomic_store(&rep[i][0], packed, __ATOMIC_RELAXED, AGENT);
  head_cache = h + 1;
  if (dbg) dbg[13] = wallclock();
  return u32(h);
}

__device__ bool GpuMover::poll(u32 token) {
  if (token & INLINE_TOKEN) return true;  // completed synchronously in submit
  u32 slot = token % MOVE_RING;
  // slot recycled past this token => long complete
  if (u32(ring[slot].epoch) != token + 1) return true;
  MoveState& s = st[slot];
  u32 done = __hip_atomic_load(&s.tiles_done, __ATOMIC_RELAXED, AGENT);
  if (done < s.tiles_total) return false;
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  return true;
}

// ------------------------------------------------------------ mover tiles
template <typename T>
struct SumOp { __device__ static T apply(T a, T b) { return a + b; } };
template <typename T>
struct MaxOp { __device__ static T apply(T a, T b) { return a > b ? a : b; } };

__device__ __forceinline__ float ld_f32(const void* p, u64 i, DataType dt) {
  switch (dt) {
    case DataType::float32: return ((const float*)p)[i];
    case DataType::float16: return __half2float(((const __half*)p)[i]);
    case DataType::bfloat16: {
      u16 h = ((const u16*)p)[i];
      u32 b = u32(h) << 16;
      return __uint_as_float(b);
    }
    default: return 0.f;
  }
}
__device__ __forceinline__ void st_f32(void* p, u64 i, DataType dt, float v) {
  switch (dt) {
    case DataType::float32: ((float*)p)[i] = v; break;
    case DataType::float16: ((__half*)p)[i] = __float2half(v); break;
    case DataType::bfloat16: {
      u32 x = __float_as_uint(v);
      if ((x & 0x7F800000u) == 0x7F800000u && (x & 0x7FFFFFu)) {
        ((u16*)p)[i] = u16((x >> 16) | 0x40);
      } else {
        x += 0x7FFFu + ((x >> 16) & 1);
        ((u16*)p)[i] = u16(x >> 16);
      }
      break;
    }
    default: break;
  }
}

__device__ __forceinline__ bool aligned16(const void* p) {
  return (u64(p) & 15) == 0;
}

// Explicit global address space for the data plane: generic ("flat") pointers
// compile to flat_load/flat_store, which tick BOTH vmcnt and lgkmcnt — and
// lgkmcnt retires out of order, so every consumer needs s_waitcnt lgkmcnt(0),
// serializing the whole software pipeline. addrspace(1) pointers compile to
// global_load/global_store with in-order vmcnt-only tracking.
#define GAS __attribute__((address_space(1)))
// Builtin 16B vectors: HIP_vector_type ctors are not address-space-aware;
// clang ext_vector_type works across address spaces with no ctors
typedef u32 U4 __attribute__((ext_vector_type(4)));
typedef float F4 __attribute__((ext_vector_type(4)));

// pure same-dtype copy, 16B vectorized when aligned. 8-deep software
// pipeline: 8 loads in flight per lane (8 KiB per wave) so HBM (~1 us over
// xGMI) latency is covered by MLP, not occupancy alone.
__device__ void tile_copy(const MoveDesc& m, u64 lo, u64 hi, int lane) {
  u32 esz = dtype_size(DataType(m.dst_dt));
  u64 bytes = (hi - lo) * esz;
  const char* s = (const char*)m.src[0] + lo * esz;
  char* d = (char*)m.dst + lo * esz;
  if (aligned16(s) && aligned16(d) && (bytes & 15) == 0) {
    GAS const U4* s4 = (GAS const U4*)s;
    GAS U4* d4 = (GAS U4*)d;
    u64 n = bytes / 16;
    u64 i = lane;
    // non-temporal: payload streams bypass L1/L2 (no dirty lines -> the
    // mover needs no L2-writeback release fence, and no cache pollution)
    for (; i + 7 * 64 < n; i += 8 * 64) {
      U4 v0 = __builtin_nontemporal_load(&s4[i]);
      U4 v1 = __builtin_nontemporal_load(&s4[i + 64]);
      U4 v2 = __builtin_nontemporal_load(&s4[i + 2 * 64]);
      U4 v3 = __builtin_nontemporal_load(&s4[i + 3 * 64]);
      U4 v4 = __builtin_nontemporal_load(&s4[i + 4 * 64]);
      U4 v5 = __builtin_nontemporal_load(&s4[i + 5 * 64]);
      U4 v6 = __builtin_nontemporal_load(&s4[i + 6 * 64]);
      U4 v7 = __builtin_nontemporal_load(&s4[i + 7 * 64]);
      __builtin_nontemporal_store(v0, &d4[i]);
      __builtin_nontemporal_store(v1, &d4[i + 64]);
      __builtin_nontemporal_store(v2, &d4[i + 2 * 64]);
      __builtin_nontemporal_store(v3, &d4[i + 3 * 64]);
      __builtin_nontemporal_store(v4, &d4[i + 4 * 64]);
      __builtin_nontemporal_store(v5, &d4[i + 5 * 64]);
      __builtin_nontemporal_store(v6, &d4[i + 6 * 64]);
      __builtin_nontemporal_store(v7, &d4[i + 7 * 64]);
    }
    if (i < n) {
      // tail: clamp load indices (duplicate loads are free) and predicate
      // stores, so all remaining loads issue together instead of a
      // one-load-one-store dependent chain (a 4KB move IS this tail)
      u64 last = n - 1;
      u64 i1 = i + 64 < n ? i + 64 : last, i2 = i + 128 < n ? i + 128 : last;
      u64 i3 = i + 192 < n ? i + 192 : last, i4 = i + 256 < n ? i + 256 : last;
      u64 i5 = i + 320 < n ? i + 320 : last, i6 = i + 384 < n ? i + 384 : last;
      U4 v0 = __builtin_nontemporal_load(&s4[i]);
      U4 v1 = __builtin_nontemporal_load(&s4[i1]);
      U4 v2 = __builtin_nontemporal_load(&s4[i2]);
      U4 v3 = __builtin_nontemporal_load(&s4[i3]);
      U4 v4 = __builtin_nontemporal_load(&s4[i4]);
      U4 v5 = __builtin_nontemporal_load(&s4[i5]);
      U4 v6 = __builtin_nontemporal_load(&s4[i6]);
      __builtin_nontemporal_store(v0, &d4[i]);
      if (i + 64 < n) __builtin_nontemporal_store(v1, &d4[i1]);
      if (i + 128 < n) __builtin_nontemporal_store(v2, &d4[i2]);
      if (i + 192 < n) __builtin_nontemporal_store(v3, &d4[i3]);
      if (i + 256 < n) __builtin_nontemporal_store(v4, &d4[i4]);
      if (i + 320 < n) __builtin_nontemporal_store(v5, &d4[i5]);
      if (i + 384 < n) __builtin_nontemporal_store(v6, &d4[i6]);
    }
  } else if ((u64(s) & 3) == 0 && (u64(d) & 3) == 0 && (bytes & 3) == 0) {
    GAS const u32* s1 = (GAS const u32*)s;
    GAS u32* d1 = (GAS u32*)d;
    u64 n = bytes / 4;
    u64 i = lane;
    for (; i + 7 * 64 < n; i += 8 * 64) {
      u32 v0 = s1[i], v1 = s1[i + 64], v2 = s1[i + 2 * 64], v3 = s1[i + 3 * 64];
      u32 v4 = s1[i + 4 * 64], v5 = s1[i + 5 * 64], v6 = s1[i + 6 * 64],
          v7 = s1[i + 7 * 64];
      d1[i] = v0; d1[i + 64] = v1; d1[i + 2 * 64] = v2; d1[i + 3 * 64] = v3;
      d1[i + 4 * 64] = v4; d1[i + 5 * 64] = v5; d1[i + 6 * 64] = v6;
      d1[i + 7 * 64] = v7;
    }
    for (; i < n; i += 64) d1[i] = s1[i];
  } else {
    GAS const char* sc = (GAS const char*)s;
    GAS char* dc = (GAS char*)d;
    for (u64 i = lane; i < bytes; i += 64) dc[i] = sc[i];
  }
}

// f32 n-ary reduce, 16B vectorized (the reduce_ops hot path:
// reference kernels/plugins/reduce_ops/reduce_ops.cpp:83-106)
template <template <class> class OP>
__device__ void tile_reduce_f32(const MoveDesc& m, u64 lo, u64 hi, int lane) {
  float* d = (float*)m.dst + lo;
  const float* s0 = (const float*)m.src[0] + lo;
  const float* s1 = (const float*)m.src[1] + lo;
  u64 n = hi - lo;
  bool v16 = aligned16(d) && aligned16(s0) && aligned16(s1) && (n & 3) == 0 &&
             m.nsrc == 2;
  if (v16) {
    GAS const F4* a = (GAS const F4*)s0;
    GAS const F4* b = (GAS const F4*)s1;
    GAS F4* o = (GAS F4*)d;
    u64 n4 = n / 4;
    u64 i = lane;
#define ACCL_NTL(p_) __builtin_nontemporal_load(p_)
#define ACCL_R4(a_, b_) (F4){OP<float>::apply(a_.x, b_.x), \
    OP<float>::apply(a_.y, b_.y), OP<float>::apply(a_.z, b_.z), \
    OP<float>::apply(a_.w, b_.w)}
    // 4-deep pipeline x 2 non-temporal operand streams (8 loads in flight);
    // nt stores leave no dirty L2 so the mover can skip the writeback fence
    for (; i + 3 * 64 < n4; i += 4 * 64) {
      F4 x0 = ACCL_NTL(&a[i]), x1 = ACCL_NTL(&a[i + 64]);
      F4 x2 = ACCL_NTL(&a[i + 2 * 64]), x3 = ACCL_NTL(&a[i + 3 * 64]);
      F4 y0 = ACCL_NTL(&b[i]), y1 = ACCL_NTL(&b[i + 64]);
      F4 y2 = ACCL_NTL(&b[i + 2 * 64]), y3 = ACCL_NTL(&b[i + 3 * 64]);
      __builtin_nontemporal_store(ACCL_R4(x0, y0), &o[i]);
      __builtin_nontemporal_store(ACCL_R4(x1, y1), &o[i + 64]);
      __builtin_nontemporal_store(ACCL_R4(x2, y2), &o[i + 2 * 64]);
      __builtin_nontemporal_store(ACCL_R4(x3, y3), &o[i + 3 * 64]);
    }
    if (i < n4) {  // clamped-tail (see tile_copy): all loads in flight
      u64 last = n4 - 1;
      u64 i1 = i + 64 < n4 ? i + 64 : last, i2 = i + 128 < n4 ? i + 128 : last;
      u64 i3 = i + 192 < n4 ? i + 192 : last;
      F4 x0 = ACCL_NTL(&a[i]), x1 = ACCL_NTL(&a[i1]);
      F4 x2 = ACCL_NTL(&a[i2]), x3 = ACCL_NTL(&a[i3]);
      F4 y0 = ACCL_NTL(&b[i]), y1 = ACCL_NTL(&b[i1]);
      F4 y2 = ACCL_NTL(&b[i2]), y3 = ACCL_NTL(&b[i3]);
      __builtin_nontemporal_store(ACCL_R4(x0, y0), &o[i]);
      if (i + 64 < n4) __builtin_nontemporal_store(ACCL_R4(x1, y1), &o[i1]);
      if (i + 128 < n4) __builtin_nontemporal_store(ACCL_R4(x2, y2), &o[i2]);
      if (i + 192 < n4) __builtin_nontemporal_store(ACCL_R4(x3, y3), &o[i3]);
    }
#undef ACCL_R4
#undef ACCL_NTL
    return;
  }
  for (u64 i = lane; i < n; i += 64) {
    float acc = ((GAS const float*)m.src[0])[lo + i];
    for (u32 k = 1; k < m.nsrc; ++k)
      acc = OP<float>::apply(acc, ((GAS const float*)m.src[k])[lo + i]);
    ((GAS float*)m.dst)[lo + i] = acc;
  }
}

// Packed 16-bit reduce (the bf16/f16 grad hot path, BASELINE config 4):
// 16B vectors = 8 elements per load; f16 pairs use v_pk (__half2) math,
// bf16 pairs convert through f32 with RNE repack (no native bf16 pk-add on
// CDNA4's VALU path we rely on). Non-temporal streams like the f32 path.
template <bool MAX_, bool BF16>
__device__ __forceinline__ u32 h2_op(u32 pa, u32 pb) {
  if (BF16) {
    float a0 = __uint_as_float(pa << 16), a1 = __uint_as_float(pa & 0xFFFF0000u);
    float b0 = __uint_as_float(pb << 16), b1 = __uint_as_float(pb & 0xFFFF0000u);
    float r0 = MAX_ ? (a0 > b0 ? a0 : b0) : a0 + b0;
    float r1 = MAX_ ? (a1 > b1 ? a1 : b1) : a1 + b1;
    auto pack = [](float v) -> u32 {
      u32 x = __float_as_uint(v);
      if ((x & 0x7F800000u) == 0x7F800000u && (x & 0x7FFFFFu))
        return (x >> 16) | 0x40;
      x += 0x7FFFu + ((x >> 16) & 1);
      return x >> 16;
    };
    return pack(r0) | (pack(r1) << 16);
  }
  __half2 a = *(__half2*)&pa, b = *(__half2*)&pb;
  __half2 r;
  if (MAX_) {
    r.x = __hgt(a.x, b.x) ? a.x : b.x;
    r.y = __hgt(a.y, b.y) ? a.y : b.y;
  } else {
    r = __hadd2(a, b);  // v_pk_add_f16
  }
  return *(u32*)&r;
}

template <bool MAX_, bool BF16>
__device__ void tile_reduce_h16(const MoveDesc& m, u64 lo, u64 hi, int lane) {
  GAS const U4* a = (GAS const U4*)((const u16*)m.src[0] + lo);
  GAS const U4* b = (GAS const U4*)((const u16*)m.src[1] + lo);
  GAS U4* o = (GAS U4*)((u16*)m.dst + lo);
  u64 n8 = (hi - lo) / 8;  // 8 halves per 16B vector
#define ACCL_H8(x_, y_) (U4){h2_op<MAX_, BF16>(x_.x, y_.x), \
    h2_op<MAX_, BF16>(x_.y, y_.y), h2_op<MAX_, BF16>(x_.z, y_.z), \
    h2_op<MAX_, BF16>(x_.w, y_.w)}
  u64 i = lane;
  for (; i + 3 * 64 < n8; i += 4 * 64) {
    U4 x0 = __builtin_nontemporal_load(&a[i]);
    U4 x1 = __builtin_nontemporal_load(&a[i + 64]);
    U4 x2 = __builtin_nontemporal_load(&a[i + 2 * 64]);
    U4 x3 = __builtin_nontemporal_load(&a[i + 3 * 64]);
    U4 y0 = __builtin_nontemporal_load(&b[i]);
    U4 y1 = __builtin_nontemporal_load(&b[i + 64]);
    U4 y2 = __builtin_nontemporal_load(&b[i + 2 * 64]);
    U4 y3 = __builtin_nontemporal_load(&b[i + 3 * 64]);
    __builtin_nontemporal_store(ACCL_H8(x0, y0), &o[i]);
    __builtin_nontemporal_store(ACCL_H8(x1, y1), &o[i + 64]);
    __builtin_nontemporal_store(ACCL_H8(x2, y2), &o[i + 2 * 64]);
    __builtin_nontemporal_store(ACCL_H8(x3, y3), &o[i + 3 * 64]);
  }
  if (i < n8) {
    u64 last = n8 - 1;
    u64 i1 = i + 64 < n8 ? i + 64 : last, i2 = i + 128 < n8 ? i + 128 : last;
    u64 i3 = i + 192 < n8 ? i + 192 : last;
    U4 x0 = __builtin_nontemporal_load(&a[i]);
    U4 x1 = __builtin_nontemporal_load(&a[i1]);
    U4 x2 = __builtin_nontemporal_load(&a[i2]);
    U4 x3 = __builtin_nontemporal_load(&a[i3]);
    U4 y0 = __builtin_nontemporal_load(&b[i]);
    U4 y1 = __builtin_nontemporal_load(&b[i1]);
    U4 y2 = __builtin_nontemporal_load(&b[i2]);
    U4 y3 = __builtin_nontemporal_load(&b[i3]);
    __builtin_nontemporal_store(ACCL_H8(x0, y0), &o[i]);
    if (i + 64 < n8) __builtin_nontemporal_store(ACCL_H8(x1, y1), &o[i1]);
    if (i + 128 < n8) __builtin_nontemporal_store(ACCL_H8(x2, y2), &o[i2]);
    if (i + 192 < n8) __builtin_nontemporal_store(ACCL_H8(x3, y3), &o[i3]);
  }
#undef ACCL_H8
}

// ---- vectorized compression lanes (reference: hp_compression 2:1 width
// converter, kernels/plugins/hp_compression/hp_compression.cpp:72-144) ----
__device__ __forceinline__ u16 bf16_rne(float v) {
  u32 x = __float_as_uint(v);
  if ((x & 0x7F800000u) == 0x7F800000u && (x & 0x7FFFFFu))
    return u16((x >> 16) | 0x40);
  x += 0x7FFFu + ((x >> 16) & 1);
  return u16(x >> 16);
}
template <bool BF16>
__device__ __forceinline__ void unpack8(U4 v, F4& a, F4& b) {
  if (BF16) {
    a = (F4){__uint_as_float(v.x << 16), __uint_as_float(v.x & 0xFFFF0000u),
             __uint_as_float(v.y << 16), __uint_as_float(v.y & 0xFFFF0000u)};
    b = (F4){__uint_as_float(v.z << 16), __uint_as_float(v.z & 0xFFFF0000u),
             __uint_as_float(v.w << 16), __uint_as_float(v.w & 0xFFFF0000u)};
  } else {
    u32 wx = v.x, wy = v.y, wz = v.z, ww = v.w;
    float2 f0 = __half22float2(*(__half2*)&wx);
    float2 f1 = __half22float2(*(__half2*)&wy);
    float2 f2 = __half22float2(*(__half2*)&wz);
    float2 f3 = __half22float2(*(__half2*)&ww);
    a = (F4){f0.x, f0.y, f1.x, f1.y};
    b = (F4){f2.x, f2.y, f3.x, f3.y};
  }
}
template <bool BF16>
__device__ __forceinline__ U4 pack8(F4 a, F4 b) {
  U4 o;
  if (BF16) {
    o.x = u32(bf16_rne(a.x)) | (u32(bf16_rne(a.y)) << 16);
    o.y = u32(bf16_rne(a.z)) | (u32(bf16_rne(a.w)) << 16);
    o.z = u32(bf16_rne(b.x)) | (u32(bf16_rne(b.y)) << 16);
    o.w = u32(bf16_rne(b.z)) | (u32(bf16_rne(b.w)) << 16);
  } else {
    __half2 h0 = __floats2half2_rn(a.x, a.y);
    __half2 h1 = __floats2half2_rn(a.z, a.w);
    __half2 h2 = __floats2half2_rn(b.x, b.y);
    __half2 h3 = __floats2half2_rn(b.z, b.w);
    o.x = *(u32*)&h0; o.y = *(u32*)&h1; o.z = *(u32*)&h2; o.w = *(u32*)&h3;
  }
  return o;
}

// pure width conversion: UP = h16 -> f32 (decompress), else f32 -> h16.
// 2x unrolled: 2-4 independent nt loads in flight per lane.
template <bool UP, bool BF16>
__device__ void tile_cast(const MoveDesc& m, u64 lo, u64 hi, int lane) {
  u64 n8 = (hi - lo) / 8;
  if (UP) {
    GAS const U4* s = (GAS const U4*)((const u16*)m.src[0] + lo);
    GAS F4* d = (GAS F4*)((float*)m.dst + lo);
    u64 i = lane;
    for (; i + 64 < n8; i += 2 * 64) {
      U4 v0 = __builtin_nontemporal_load(&s[i]);
      U4 v1 = __builtin_nontemporal_load(&s[i + 64]);
      F4 a0, b0, a1, b1;
      unpack8<BF16>(v0, a0, b0);
      unpack8<BF16>(v1, a1, b1);
      __builtin_nontemporal_store(a0, &d[2 * i]);
      __builtin_nontemporal_store(b0, &d[2 * i + 1]);
      __builtin_nontemporal_store(a1, &d[2 * (i + 64)]);
      __builtin_nontemporal_store(b1, &d[2 * (i + 64) + 1]);
    }
    if (i < n8) {
      U4 v = __builtin_nontemporal_load(&s[i]);
      F4 a, b;
      unpack8<BF16>(v, a, b);
      __builtin_nontemporal_store(a, &d[2 * i]);
      __builtin_nontemporal_store(b, &d[2 * i + 1]);
    }
  } else {
    GAS const F4* s = (GAS const F4*)((const float*)m.src[0] + lo);
    GAS U4* d = (GAS U4*)((u16*)m.dst + lo);
    u64 i = lane;
    for (; i + 64 < n8; i += 2 * 64) {
      F4 a0 = __builtin_nontemporal_load(&s[2 * i]);
      F4 b0 = __builtin_nontemporal_load(&s[2 * i + 1]);
      F4 a1 = __builtin_nontemporal_load(&s[2 * (i + 64)]);
      F4 b1 = __builtin_nontemporal_load(&s[2 * (i + 64) + 1]);
      __builtin_nontemporal_store(pack8<BF16>(a0, b0), &d[i]);
      __builtin_nontemporal_store(pack8<BF16>(a1, b1), &d[i + 64]);
    }
    if (i < n8) {
      F4 a = __builtin_nontemporal_load(&s[2 * i]);
      F4 b = __builtin_nontemporal_load(&s[2 * i + 1]);
      __builtin_nontemporal_store(pack8<BF16>(a, b), &d[i]);
    }
  }
}

// fused decompress + reduce (rx lane of a compressed collective:
// dst_f32 = op(cvt(wire_h16), dst_f32)); src0 = wire, src1 = f32 operand
template <bool MAX_, bool BF16>
__device__ void tile_cast_reduce(const MoveDesc& m, u64 lo, u64 hi, int lane) {
  u64 n8 = (hi - lo) / 8;
  GAS const U4* s0 = (GAS const U4*)((const u16*)m.src[0] + lo);
  GAS const F4* s1 = (GAS const F4*)((const float*)m.src[1] + lo);
  GAS F4* d = (GAS F4*)((float*)m.dst + lo);
  for (u64 i = lane; i < n8; i += 64) {
    U4 w = __builtin_nontemporal_load(&s0[i]);
    F4 x0 = __builtin_nontemporal_load(&s1[2 * i]);
    F4 x1 = __builtin_nontemporal_load(&s1[2 * i + 1]);
    F4 a, b;
    unpack8<BF16>(w, a, b);
    F4 r0, r1;
    if (MAX_) {
      r0 = (F4){a.x > x0.x ? a.x : x0.x, a.y > x0.y ? a.y : x0.y,
                a.z > x0.z ? a.z : x0.z, a.w > x0.w ? a.w : x0.w};
      r1 = (F4){b.x > x1.x ? b.x : x1.x, b.y > x1.y ? b.y : x1.y,
                b.z > x1.z ? b.z : x1.z, b.w > x1.w ? b.w : x1.w};
    } else {
      r0 = a + x0;
      r1 = b + x1;
    }
    __builtin_nontemporal_store(r0, &d[2 * i]);
    __builtin_nontemporal_store(r1, &d[2 * i + 1]);
  }
}

// n-ary f32 reduce (3..8 sources — the direct fan-in hot path): one nt
// load per source per 16B vector, fold, one nt store. 8 concurrent source
// streams give the MLP; dst may alias src[0].
template <template <class> class OP>
__device__ void tile_reduce_f32_n(const MoveDesc& m, u64 lo, u64 hi,
                                  int lane) {
  u64 n4 = (hi - lo) / 4;
  GAS F4* o = (GAS F4*)((float*)m.dst + lo);
  GAS const F4* s[MOVE_MAX_SRC];
  for (u32 k = 0; k < m.nsrc; ++k)
    s[k] = (GAS const F4*)((const float*)m.src[k] + lo);
  const u32 ns = m.nsrc;
  for (u64 i = lane; i < n4; i += 64) {
    F4 v[MOVE_MAX_SRC];
#pragma unroll
    for (u32 k = 0; k < MOVE_MAX_SRC; ++k)
      if (k < ns) v[k] = __builtin_nontemporal_load(&s[k][i]);
    F4 acc = v[0];
#pragma unroll
    for (u32 k = 1; k < MOVE_MAX_SRC; ++k)
      if (k < ns)
        acc = (F4){OP<float>::apply(acc.x, v[k].x),
                   OP<float>::apply(acc.y, v[k].y),
                   OP<float>::apply(acc.z, v[k].z),
                   OP<float>::apply(acc.w, v[k].w)};
    __builtin_nontemporal_store(acc, &o[i]);
  }
}

// n-ary packed 16-bit reduce (bf16/f16 config-4 fan-in)
template <bool MAX_, bool BF16>
__device__ void tile_reduce_h16_n(const MoveDesc& m, u64 lo, u64 hi,
                                  int lane) {
  u64 n8 = (hi - lo) / 8;
  GAS U4* o = (GAS U4*)((u16*)m.dst + lo);
  GAS const U4* s[MOVE_MAX_SRC];
  for (u32 k = 0; k < m.nsrc; ++k)
    s[k] = (GAS const U4*)((const u16*)m.src[k] + lo);
  const u32 ns = m.nsrc;
  for (u64 i = lane; i < n8; i += 64) {
    U4 v[MOVE_MAX_SRC];
#pragma unroll
    for (u32 k = 0; k < MOVE_MAX_SRC; ++k)
      if (k < ns) v[k] = __builtin_nontemporal_load(&s[k][i]);
    U4 acc = v[0];
#pragma unroll
    for (u32 k = 1; k < MOVE_MAX_SRC; ++k)
      if (k < ns)
        acc = (U4){h2_op<MAX_, BF16>(acc.x, v[k].x),
                   h2_op<MAX_, BF16>(acc.y, v[k].y),
                   h2_op<MAX_, BF16>(acc.z, v[k].z),
                   h2_op<MAX_, BF16>(acc.w, v[k].w)};
    __builtin_nontemporal_store(acc, &o[i]);
  }
}

// n-ary same-dtype reduce over 16-byte vectors for the remaining reduce_ops
// dtypes (i32: 4/vec, i64/f64: 2/vec) — reference reduce_ops.cpp:31-107
// supports f32/f64/i32/i64/f16; ELT is the element type, EPV elems/vector.
template <typename ELT, u32 EPV, bool MAX_>
__device__ void tile_reduce_wide(const MoveDesc& m, u64 lo, u64 hi,
                                 int lane) {
  u64 nv = (hi - lo) / EPV;
  GAS U4* o = (GAS U4*)((ELT*)m.dst + lo);
  GAS const U4* s[MOVE_MAX_SRC];
  for (u32 k = 0; k < m.nsrc; ++k)
    s[k] = (GAS const U4*)((const ELT*)m.src[k] + lo);
  const u32 ns = m.nsrc;
  for (u64 i = lane; i < nv; i += 64) {
    U4 v[MOVE_MAX_SRC];
#pragma unroll
    for (u32 k = 0; k < MOVE_MAX_SRC; ++k)
      if (k < ns) v[k] = __builtin_nontemporal_load(&s[k][i]);
    ELT acc[EPV];
#pragma unroll
    for (u32 e = 0; e < EPV; ++e) acc[e] = ((ELT*)&v[0])[e];
#pragma unroll
    for (u32 k = 1; k < MOVE_MAX_SRC; ++k) {
      if (k >= ns) break;
#pragma unroll
      for (u32 e = 0; e < EPV; ++e) {
        ELT b = ((ELT*)&v[k])[e];
        acc[e] = MAX_ ? (acc[e] > b ? acc[e] : b) : ELT(acc[e] + b);
      }
    }
    U4 out;
#pragma unroll
    for (u32 e = 0; e < EPV; ++e) ((ELT*)&out)[e] = acc[e];
    __builtin_nontemporal_store(out, &o[i]);
  }
}

// float-domain path for any f32/f16/bf16 mix (cast + reduce fused — the
// hp_compression + reduce_ops lanes in one pass)
__device__ void tile_float_generic(const MoveDesc& m, u64 lo, u64 hi, int lane) {
  const bool is_max = m.nsrc >= 2 && ReduceFunction(m.func) == ReduceFunction::MAX;
  for (u64 i = lo + lane; i < hi; i += 64) {
    float acc = ld_f32((const void*)m.src[0], i, DataType(m.src_dt[0]));
    for (u32 k = 1; k < m.nsrc; ++k) {
      float v = ld_f32((const void*)m.src[k], i, DataType(m.src_dt[k]));
      acc = is_max ? (acc > v ? acc : v) : acc + v;
    }
    st_f32((void*)m.dst, i, DataType(m.dst_dt), acc);
  }
}

__device__ bool dtype_is_floatish(DataType d) {
  return d == DataType::float32 || d == DataType::float16 ||
         d == DataType::bfloat16;
}

// Returns true if the executed path used only non-temporal accesses (no
// dirty L2 lines -> the caller may skip the L2-writeback release fence).
__device__ bool run_tile(const MoveDesc& m, u32 t, int lane) {
  u64 te = move_tile_elems(m);
  u64 lo = u64(t) * te;
  u64 hi = lo + te;
  if (hi > m.count) hi = m.count;
  if (lo >= hi) return true;
  // dispatch
  if (m.nsrc == 1 && m.src_dt[0] == m.dst_dt) {
    u32 esz = dtype_size(DataType(m.dst_dt));
    u64 bytes = (hi - lo) * esz;
    bool v16 = aligned16((const void*)(m.src[0] + lo * esz)) &&
               aligned16((const void*)(m.dst + lo * esz)) && (bytes & 15) == 0;
    tile_copy(m, lo, hi, lane);
    return v16;
  }
  bool all_f32 = m.dst_dt == u8(DataType::float32);
  bool floatish = dtype_is_floatish(DataType(m.dst_dt));
  for (u32 k = 0; k < m.nsrc; ++k) {
    all_f32 = all_f32 && m.src_dt[k] == u8(DataType::float32);
    floatish = floatish && dtype_is_floatish(DataType(m.src_dt[k]));
  }
  const bool mx = ReduceFunction(m.func) == ReduceFunction::MAX;
  if (all_f32 && m.nsrc == 2) {
    bool v16 = aligned16((const void*)(m.dst + lo * 4)) &&
               aligned16((const void*)(m.src[0] + lo * 4)) &&
               aligned16((const void*)(m.src[1] + lo * 4)) &&
               ((hi - lo) & 3) == 0;
    if (!mx) tile_reduce_f32<SumOp>(m, lo, hi, lane);
    else tile_reduce_f32<MaxOp>(m, lo, hi, lane);
    return v16;
  }
  if (all_f32 && m.nsrc > 2) {
    bool v16 = aligned16((const void*)(m.dst + lo * 4)) &&
               ((hi - lo) & 3) == 0;
    for (u32 k = 0; k < m.nsrc; ++k)
      v16 = v16 && aligned16((const void*)(m.src[k] + lo * 4));
    if (v16) {
      // n-ary fan-in (direct reduce/reduce_scatter stage fold)
      if (!mx) tile_reduce_f32_n<SumOp>(m, lo, hi, lane);
      else tile_reduce_f32_n<MaxOp>(m, lo, hi, lane);
      return true;
    }
    tile_float_generic(m, lo, hi, lane);
    return false;
  }
  // pure width conversion (compression lanes): f32 <-> f16/bf16
  if (m.nsrc == 1 && ((hi - lo) & 7) == 0) {
    DataType sd = DataType(m.src_dt[0]), dd = DataType(m.dst_dt);
    bool s16 = sd == DataType::float16 || sd == DataType::bfloat16;
    bool d16 = dd == DataType::float16 || dd == DataType::bfloat16;
    if (s16 && dd == DataType::float32 &&
        aligned16((const void*)(m.src[0] + lo * 2)) &&
        aligned16((const void*)(m.dst + lo * 4))) {
      if (sd == DataType::bfloat16) tile_cast<true, true>(m, lo, hi, lane);
      else tile_cast<true, false>(m, lo, hi, lane);
      return true;
    }
    if (d16 && sd == DataType::float32 &&
        aligned16((const void*)(m.src[0] + lo * 4)) &&
        aligned16((const void*)(m.dst + lo * 2))) {
      if (dd == DataType::bfloat16) tile_cast<false, true>(m, lo, hi, lane);
      else tile_cast<false, false>(m, lo, hi, lane);
      return true;
    }
  }
  // fused decompress + reduce: wire h16 segment folded into an f32 operand
  if (m.nsrc == 2 && m.dst_dt == u8(DataType::float32) &&
      m.src_dt[1] == u8(DataType::float32) &&
      (m.src_dt[0] == u8(DataType::float16) ||
       m.src_dt[0] == u8(DataType::bfloat16)) &&
      ((hi - lo) & 7) == 0 &&
      aligned16((const void*)(m.src[0] + lo * 2)) &&
      aligned16((const void*)(m.src[1] + lo * 4)) &&
      aligned16((const void*)(m.dst + lo * 4))) {
    bool bf = m.src_dt[0] == u8(DataType::bfloat16);
    if (bf) {
      if (mx) tile_cast_reduce<true, true>(m, lo, hi, lane);
      else tile_cast_reduce<false, true>(m, lo, hi, lane);
    } else {
      if (mx) tile_cast_reduce<true, false>(m, lo, hi, lane);
      else tile_cast_reduce<false, false>(m, lo, hi, lane);
    }
    return true;
  }
  // packed same-dtype f16/bf16 reduce, 2..8 sources (config-4 hot path +
  // the bf16 direct fan-in stage fold)
  {
    bool h16 = (m.dst_dt == u8(DataType::float16) ||
                m.dst_dt == u8(DataType::bfloat16)) &&
               m.nsrc >= 2 && ((hi - lo) & 7) == 0 &&
               aligned16((const void*)(m.dst + lo * 2));
    for (u32 k = 0; k < m.nsrc && h16; ++k)
      h16 = m.src_dt[k] == m.dst_dt &&
            aligned16((const void*)(m.src[k] + lo * 2));
    if (h16) {
      bool bf = m.dst_dt == u8(DataType::bfloat16);
      if (m.nsrc == 2) {
        if (bf) {
          if (mx) tile_reduce_h16<true, true>(m, lo, hi, lane);
          else tile_reduce_h16<false, true>(m, lo, hi, lane);
        } else {
          if (mx) tile_reduce_h16<true, false>(m, lo, hi, lane);
          else tile_reduce_h16<false, false>(m, lo, hi, lane);
        }
      } else {
        if (bf) {
          if (mx) tile_reduce_h16_n<true, true>(m, lo, hi, lane);
          else tile_reduce_h16_n<false, true>(m, lo, hi, lane);
        } else {
          if (mx) tile_reduce_h16_n<true, false>(m, lo, hi, lane);
          else tile_reduce_h16_n<false, false>(m, lo, hi, lane);
        }
      }
      return true;
    }
  }
  // same-dtype i32/i64/f64 reduce, 2..8 sources, 16-byte vectors
  {
    DataType dd = DataType(m.dst_dt);
    bool wide = (dd == DataType::int32 || dd == DataType::int64 ||
                 dd == DataType::float64) &&
                m.nsrc >= 2;
    u32 esz2 = dtype_size(dd);
    u32 epv = esz2 ? 16 / esz2 : 0;
    wide = wide && epv && ((hi - lo) % epv) == 0 &&
           aligned16((const void*)(m.dst + lo * esz2));
    for (u32 k = 0; k < m.nsrc && wide; ++k)
      wide = m.src_dt[k] == m.dst_dt &&
             aligned16((const void*)(m.src[k] + lo * esz2));
    if (wide) {
      if (dd == DataType::int32) {
        if (mx) tile_reduce_wide<i32, 4, true>(m, lo, hi, lane);
        else tile_reduce_wide<i32, 4, false>(m, lo, hi, lane);
      } else if (dd == DataType::int64) {
        if (mx) tile_reduce_wide<i64, 2, true>(m, lo, hi, lane);
        else tile_reduce_wide<i64, 2, false>(m, lo, hi, lane);
      } else {
        if (mx) tile_reduce_wide<double, 2, true>(m, lo, hi, lane);
        else tile_reduce_wide<double, 2, false>(m, lo, hi, lane);
      }
      return true;
    }
  }
  if (floatish) {
    tile_float_generic(m, lo, hi, lane);
    return false;
  }
  // exact scalar fallback (int8 / mixed dtypes)
  for (u64 i = lo + lane; i < hi; i += 64) execute_move_range(m, i, i + 1);
  return false;
}

// ------------------------------------------------------------- mover main
// Static tile partitioning: move mi's tile t belongs to wave
// (t + rot(mi)) % nwaves == gw — no claim atomics, no shared hot line.
// Completion = ONE tiles_done add per participating wave. The doorbell is
// replicated (DOORBELL_REPS lines, ~nwaves/64 pollers each) so noticing a
// new move costs ~L2-hit latency, not a fleet-wide serialized cacheline.
__device__ void mover_main(GpuEngineState* S) {
  const int lane = int(threadIdx.x) & 63;
  const u32 gw = blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
  const u32 nwaves = gridDim.x * (blockDim.x / 64);
  MoveDesc* ring = S->mover.ring;
  MoveState* st = S->mover.st;
  volatile u64* my_rep = &S->head_rep[gw % DOORBELL_REPS][0];
  u64 cursor = 0;
  u32 idle = 0;
  for (;;) {
    u64 packed = __hip_atomic_load((const u64*)&my_rep[0], __ATOMIC_RELAXED, AGENT);
    u64 h = packed >> 24;
    u32 latest_total = u32(packed & 0xFFFFFFu);
    if (cursor == h) {
      if (__hip_atomic_load((const u64*)&my_rep[1], __ATOMIC_RELAXED, AGENT))
        return;
      // exponential idle backoff: cheap wake-up when busy, low L2/issue
      // pressure on a quiet engine (co-resident compute kernels, config 5)
      idle = idle < 240 ? idle + 8 : 240;
      if (idle < 64) __builtin_amdgcn_s_sleep(8);
      else if (idle < 160) __builtin_amdgcn_s_sleep(32);
      else if (idle < 240) __builtin_amdgcn_s_sleep(64);
      else {
        __builtin_amdgcn_s_sleep(127);  // deep idle: ~3 us wake worst-case
        // liveness net: consult the true head (atomic, always visible) in
        // case this wave's doorbell replica was ever missed
        u64 hv = __hip_atomic_load(S->mover.head, __ATOMIC_RELAXED, AGENT);
        if (hv > cursor) { h = hv; latest_total = 0xFFFFFFu; goto have_work; }
      }
      continue;
    }
  have_work:
    idle = 0;
    // ONE SYSTEM acquire per wake batch: invalidates this CU's L1 and the
    // XCD L2's stale lines. Covers the descs AND every payload of the
    // moves <= h — peer payload writes happen-before the scheduler's
    // doorbell store (peer release -> scheduler acquire at hdr match ->
    // desc release -> doorbell), so acquiring after the doorbell read is
    // transitively sufficient. Persistent kernels never get the implicit
    // launch-boundary invalidate ordinary kernels rely on, and slot
    // payloads are peer-written: without this, an XCD L2 line cached from
    // a slot's previous occupancy reads stale.
    if (!S->no_acq) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
    else __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    while (cursor < h) {
      u32 slot = u32(cursor % MOVE_RING);
      u32 rot = u32(cursor * 37) % nwaves;
      u32 first = (gw + nwaves - rot) % nwaves;
      // when caught up, the doorbell word tells us our share without a
      // descriptor read (the common small-move fast path)
      if (cursor == h - 1 && first >= latest_total) { cursor++; continue; }
      const MoveDesc& m = ring[slot];
      // desc publish happens-before the doorbell store; sanity-check epoch.
      // Epoch AHEAD of cursor+1 = the slot was recycled past us by a burst
      // of >MOVE_RING inline submits (which never ring the fleet doorbell):
      // the occupant we missed completed inline, so step over it. Only an
      // OLDER epoch means the desc is not visible yet.
      if (u32(m.epoch) != u32(cursor + 1)) {
        if (i32(u32(m.epoch) - u32(cursor + 1)) > 0) { cursor++; continue; }
        break;  // not published yet: retry via doorbell
      }
      u32 total = m.inline_done ? 0 : move_tiles(m);
      if (first < total) {
        // seqlock re-check: the slot may have been recycled mid-read by an
        // inline burst (>MOVE_RING moves with no fleet doorbell); a torn
        // desc must never execute. Same epoch on both sides of the field
        // reads => consistent (writer publishes epoch last, release).
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        if (u32(__hip_atomic_load(&ring[slot].epoch, __ATOMIC_RELAXED,
                                  AGENT)) != u32(cursor + 1)) {
          cursor++;
          continue;
        }
        if (lane == 0 && first == 0) S->dbg[1] = wallclock();
        u32 cnt = 0;
        for (u32 t = first; t < total; t += nwaves, ++cnt) run_tile(m, t, lane);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        if (lane == 0) {
          S->wave_tiles[gw & 4095] += cnt;
          __hip_atomic_fetch_add(&st[slot].tiles_done, cnt, __ATOMIC_RELEASE,
                                 AGENT);
          if (first == 0) S->dbg[2] = wallclock();
        }
      }
      cursor++;
    }
  }
}

// --------------------------------------------------------- scheduler main
// C lives in LDS: the control loop's flow/seq state reads cost ~30ns instead
// of ~0.5us per dependent global-memory load (it dominated small-op latency).
__device__ void scheduler_main(GpuEngineState* S, Cclo<GpuMover>& C) {
  C.mv = &S->mover;
  C.cold = &S->cold;
  S->mover.ring = S->mq;
  S->mover.st = S->mst;
  S->mover.head = &S->mq_head;
  S->mover.stop = &S->stop;
  S->mover.dbg = S->dbg;
  S->mover.rep = S->head_rep;
  CtrlPage* ctrl = S->ctrl;
  st_sys(&ctrl->engine_up, 1);
  u64 consumed = 0, dev_consumed = 0, cached_gen = 0, beat = 0;
  auto publish = [&](u64 idx, u32 e, u64 t0) {
    RetEntry& r = S->rets[idx % RING_CAP];
    r.errcode = e;
    r.t_start = t0;
    r.t_end = wallclock();
    fence_release_sys();
    st_sys32(&r.seq, u32(idx + 1));
  };
  // dev_ring: host writes descs + doorbell into HBM over the large BAR —
  // local poll instead of a PCIe fetch per iteration
  volatile u64* door = S->dev_ring ? &S->ddoorbell : &ctrl->doorbell;
  CallDesc* descs = S->dev_ring ? S->dring : S->descs;
  bool halted = false;
  for (;;) {
    u64 db = ld_sys(door);
    bool dev_pending = C.device_call_pending(dev_consumed);
    if (consumed == db && !dev_pending && !C.nparked) {
      if (ld_sys(&ctrl->shutdown)) break;
      if ((++beat & 0x3FF) == 0) st_sys(&ctrl->heartbeat, beat);
      if (S->dev_ring) fence_acquire_sys();  // BAR-written doorbell/descs
      __builtin_amdgcn_s_sleep(16);
      continue;
    }
    fence_acquire_sys();
    // communicator cache refresh precedes BOTH rings (device calls name
    // communicators too); only paid when there is work (PCIe read)
    u64 gen = ld_sys(&ctrl->comm_gen);
    if (gen != cached_gen) {
      C.ncomms = u32(ld_sys(&ctrl->ncomms));
      for (u32 i = 0; i < C.ncomms; ++i) C.comms[i] = S->comm_mirror[i];
      cached_gen = gen;
    }
    // retry parked calls (the CMD_CALL_RETRY queue analogue)
    for (int pi; (pi = C.retry_parked()) >= 0;)
      publish(C.done_ring_idx, C.done_err, C.done_t0);
    if (dev_pending) {
      C.poll_device_calls(dev_consumed);
      continue;
    }
    if (consumed < db) {
      CallDesc d = descs[consumed % RING_CAP];
      u64 t0 = wallclock();
      if (Op(d.scenario) == Op::halt) {
        publish(consumed, E_OK, t0);
        consumed++;
        halted = true;
        break;
      }
      u32 e = C.serve_desc(d, consumed, consumed + 1 < db);
      if (!(e & E_NOT_READY)) publish(consumed, e, t0);
      consumed++;
    } else if (C.nparked) {
      // only parked work: light retry cadence; the acquire drops stale L2
      // lines so the next probe round re-reads peer-written records fresh
      fence_acquire_sys();
      __builtin_amdgcn_s_sleep(8);
    }
  }
  (void)halted;
  // engine exiting: fail anything still parked so host waits return
  for (int pi; (pi = C.fail_parked()) >= 0;)
    publish(C.done_ring_idx, C.done_err, C.done_t0);
  // tell movers to exit, then leave
  __hip_atomic_store(S->mover.stop, 1u, __ATOMIC_RELEASE, AGENT);
  for (u32 i = 0; i < DOORBELL_REPS; ++i)
    __hip_atomic_store(&S->head_rep[i][1], 1ull, __ATOMIC_RELEASE, AGENT);
}

// Small-mover wave: sibling wave of the scheduler, same workgroup. Executes
// sub-32KB moves on an LDS handshake. The payload release is SYSTEM scope so
// the scheduler's subsequent slot-header publish happens-after it.
__device__ void small_mover(SmallMb* mb, int lane) {
  u64 last = 0;
  for (;;) {
    u64 sq = __hip_atomic_load(&mb->seq, __ATOMIC_ACQUIRE,
                               __HIP_MEMORY_SCOPE_WORKGROUP);
    if (sq == last) {
      if (__hip_atomic_load(&mb->quit, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_WORKGROUP))
        return;
      __builtin_amdgcn_s_sleep(1);
      continue;
    }
    bool nt = run_tile(mb->d, 0, lane);  // a small move fits one tile
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    if (!nt) {
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    if (lane == 0)
      __hip_atomic_store(&mb->done, sq, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_WORKGROUP);
    last = sq;
  }
}

// Two kernels: the scheduler WG (control wave + small-mover wave — one CU)
// and the mover fleet (lean copy/reduce waves, ~10 of the 16 wave slots
// per CU so co-resident compute kernels always have room — config 5).
__global__ void __launch_bounds__(128, 1) accl_scheduler_kernel(GpuEngineState* S) {
  __shared__ SmallMb mb;
  __shared__ Cclo<GpuMover> C;
  // the per-workgroup LDS budget on CDNA is 64 KB: exceeding it makes the
  // kernel silently unlaunchable (engine never comes up)
  static_assert(sizeof(Cclo<GpuMover>) + sizeof(SmallMb) <= 60 * 1024,
                "scheduler LDS state too large — move cold tables to "
                "ColdState (device-global)");
  // both waves cooperatively stage the host-initialized Cclo into LDS
  {
    const u64* src = (const u64*)&S->cclo;
    u64* dst = (u64*)&C;
    for (u32 i = threadIdx.x; i < sizeof(Cclo<GpuMover>) / 8; i += blockDim.x)
      dst[i] = src[i];
  }
  if (threadIdx.x == 0) { mb.seq = 0; mb.done = 0; mb.quit = 0; }
  __syncthreads();
  if (threadIdx.x == 0) {
    S->mover.small_mb = (void*)&mb;
    scheduler_main(S, C);
    __hip_atomic_store(&mb.quit, 1ull, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_WORKGROUP);
  } else if (threadIdx.x >= 64) {
    small_mover(&mb, int(threadIdx.x) & 63);
  }
}

__global__ void __launch_bounds__(256, 2) accl_mover_kernel(GpuEngineState* S) {
  // bring-up telemetry: count mover WGs that actually started (dbg[15])
  if (threadIdx.x == 0)
    __hip_atomic_fetch_add(&S->dbg[15], 1ull, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
  mover_main(S);
}

// Bring-up probe: ONE thread stores a magic + ack mask into every peer
// arena through the IPC imports — validates the SHADER write path into
// each mapping before the persistent engine relies on it (a lazily
// enabled peer mapping has been observed to silently drop shader writes
// when two processes share one GPU; hipMemcpy probes would not catch it).
__global__ void accl_probe_kernel(ProbeArgs a) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  for (u32 r = 0; r < a.nranks; ++r) {
    if (r == a.me || !a.probe[r]) continue;
    __hip_atomic_store((u64*)a.probe[r], PROBE_MAGIC | a.me, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_SYSTEM);
    __hip_atomic_store((u64*)a.ack[r], a.seen_mask, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_SYSTEM);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

void gpu_probe_launch(const ProbeArgs& a, void* stream) {
  hipLaunchKernelGGL(accl_probe_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, a);
}

void gpu_engine_launch(GpuEngineState* state_dev, int n_wgs, void* sched_stream,
                       void* mover_stream) {
  // scheduler FIRST: its single WG must never be starved behind the
  // 640-WG fleet dispatch (ACCL_MOVER_FIRST restores the old order for
  // comparison)
  if (std::getenv("ACCL_MOVER_FIRST")) {
    hipLaunchKernelGGL(accl_mover_kernel, dim3(n_wgs), dim3(256), 0,
                       (hipStream_t)mover_stream, state_dev);
    hipLaunchKernelGGL(accl_scheduler_kernel, dim3(1), dim3(128), 0,
                       (hipStream_t)sched_stream, state_dev);
  } else {
    hipLaunchKernelGGL(accl_scheduler_kernel, dim3(1), dim3(128), 0,
                       (hipStream_t)sched_stream, state_dev);
    hipLaunchKernelGGL(accl_mover_kernel, dim3(n_wgs), dim3(256), 0,
                       (hipStream_t)mover_stream, state_dev);
  }
}

}  // namespace accl
