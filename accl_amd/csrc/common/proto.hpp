// accl_amd transport protocol — the arena layout + publish/consume primitives
// shared by the host runtime, the CPU emulator and the GPU persistent engine.
//
// Semantics reproduced from the reference's eager/rendezvous dual transport:
//   - eager rx slot rings   ~ rx-buffer table + packetizer/depacketizer + rxbuf
//     offload engines (reference: kernels/cclo/hls/rxbuf_offload/*.cpp,
//     eth_intf/udp_packetizer.cpp:24-83; table layout ccl_offload_control.h:267-295)
//   - rendezvous addr/done  ~ RNDZV machinery (reference:
//     ccl_offload_control.c:142-408, rdma_sq_handler.cpp:23-142)
// but re-designed for xGMI peer HBM: the sender writes payload directly into the
// receiver's arena (no wire, no reassembly); ordering comes from system-scope
// release/acquire publication instead of in-order AXI streams.
//
// Compiles as C++17 host code and as HIP device code (single-source property,
// reference: ccl_offload_control.h:229-264 MB_FW_EMULATION).
#pragma once
#include "types.hpp"

namespace accl {

// ------------------------------------------------------------ atomics shim
// Host side uses GCC builtins (shm across processes); device side uses HIP
// scoped atomics. System scope everywhere: peers are other GPUs (xGMI) or
// other processes (shm).
// Each shim is ACCL_HD with the body selected per compilation pass — the
// device pass uses HIP scoped atomics / gfx950 fences, every host pass the
// GCC builtins (standard single-source pattern).
u64 wallclock_host_ns();  // defined in core/util.cpp (steady_clock)

ACCL_HD inline u64 ld_sys(const volatile u64* p) {
#if defined(__HIP_DEVICE_COMPILE__)
  return __hip_atomic_load((const u64*)p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
#else
  return __atomic_load_n((const u64*)p, __ATOMIC_RELAXED);
#endif
}
ACCL_HD inline u32 ld_sys32(const volatile u32* p) {
#if defined(__HIP_DEVICE_COMPILE__)
  return __hip_atomic_load((const u32*)p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
#else
  return __atomic_load_n((const u32*)p, __ATOMIC_RELAXED);
#endif
}
ACCL_HD inline void st_sys(volatile u64* p, u64 v) {
#if defined(__HIP_DEVICE_COMPILE__)
  __hip_atomic_store((u64*)p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
#else
  __atomic_store_n((u64*)p, v, __ATOMIC_RELAXED);
#endif
}
ACCL_HD inline u64 afadd_sys(volatile u64* p, u64 v) {
#if defined(__HIP_DEVICE_COMPILE__)
  return __hip_atomic_fetch_add((u64*)p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
#else
  return __atomic_fetch_add((u64*)p, v, __ATOMIC_RELAXED);
#endif
}
ACCL_HD inline void st_sys32(volatile u32* p, u32 v) {
#if defined(__HIP_DEVICE_COMPILE__)
  __hip_atomic_store((u32*)p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
#else
  __atomic_store_n((u32*)p, v, __ATOMIC_RELAXED);
#endif
}
// Release all prior (plain) stores, then publish a flag. ROCm 7.2 can drop
// the s_waitcnt after buffer_wbl2 when the wave's scoreboard looks empty —
// restate it in asm, always (MI355X_MICROARCH.md §Workgroup dispatch,
// compiler hazard).
ACCL_HD inline void fence_release_sys() {
#if defined(__HIP_DEVICE_COMPILE__)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#else
  __atomic_thread_fence(__ATOMIC_RELEASE);
#endif
}
ACCL_HD inline void fence_acquire_sys() {
#if defined(__HIP_DEVICE_COMPILE__)
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
#else
  __atomic_thread_fence(__ATOMIC_ACQUIRE);
#endif
}
ACCL_HD inline void cpu_pause() {
#if defined(__HIP_DEVICE_COMPILE__)
  __builtin_amdgcn_s_sleep(8);
#elif defined(__x86_64__)
  __builtin_ia32_pause();
#endif
}
ACCL_HD inline u64 wallclock() {  // 100 MHz ticks on both engines
#if defined(__HIP_DEVICE_COMPILE__)
  return __builtin_amdgcn_s_memrealtime();
#else
  return wallclock_host_ns() / 10;
#endif
}
constexpr u64 TICKS_PER_US = 100;

// -------------------------------------------------------------- parameters
// Runtime-configurable (ACCL: rx-buffer count/size + max eager/rendezvous
// sizes, accl.hpp:102-104); these are the compile-time maxima / defaults.
struct ProtoConfig {
  u32 n_slots;        // eager rx slots per (src,dst) pair        [default 8]
  u32 slot_bytes;     // payload bytes per slot                   [default 1 MiB]
  u32 n_rndzv;        // rendezvous ring depth per pair           [default 8]
  u32 n_stream;       // device-stream ring slots per rank        [default 8]
  u32 stream_bytes;   // device-stream slot payload
  u64 max_eager;      // bytes; above this send/recv go rendezvous when legal
  u64 timeout_us;     // spin bound for every wait
  u32 nranks;
  u32 rank;
};

inline ProtoConfig default_proto_config(u32 nranks, u32 rank) {
  ProtoConfig c{};
  c.n_slots = 8; c.slot_bytes = 1u << 20; c.n_rndzv = 8;
  c.n_stream = 8; c.stream_bytes = 1u << 20;
  c.max_eager = 4u << 20;      // beyond this, arena-resident transfers go direct
  c.timeout_us = 10u * 1000 * 1000;
  c.nranks = nranks; c.rank = rank;
  return c;
}

// ------------------------------------------------------------ arena layout
// One arena per rank. Peers map the whole arena (hipIpc / shm) and address
// everything as arena_base[peer] + offset. All control words are 8-byte
// aligned and written by exactly one rank at a time (SPSC rings per pair).
//
// Layout (offsets in ArenaHdr, computed by arena_layout()):
//   [0]                ArenaHdr
//   hdr.eager_off      per src-peer EagerChan control (headers + credit)
//   hdr.rndzv_off      per peer RndzvRing pair (addr ring + done ring)
//   hdr.stream_off     per src-peer stream rings (stream_put / device-
//                      initiated collectives): ctl (credit) + SlotHdr[n] +
//                      payload, mirror of the eager channel but consumed by
//                      the APPLICATION (host or device kernel), not by a
//                      posted recv — the depacketizer-bypass path of the
//                      reference (udp_depacketizer.cpp:135-148 strm TDEST)
//   hdr.slots_off      eager payload slots [src_peer][slot]
//   hdr.heap_off       buffer heap until hdr.arena_bytes

struct alignas(64) SlotHdr {
  u64 seq;        // published last; seq is 1-based message-segment number
  u32 tag;
  u32 bytes;      // payload bytes in this slot (wire bytes)
  u64 msg_count;  // total element count of the message (first segment only)
  u32 arith;      // packed wire dtype info (same packing as CallDesc.arith)
  u32 flags;      // SEG_FIRST / SEG_LAST
  u64 _pad[4];
};
static_assert(sizeof(SlotHdr) == 64, "");

enum SegFlags : u32 { SEG_FIRST = 1, SEG_LAST = 2 };

struct alignas(64) RndzvRec {
  u64 seq;        // published last, 1-based per (pair, ring)
  u32 tag;
  u32 arith;
  u64 offset;     // destination arena offset (addr ring) / echo (done ring)
  u64 count;      // elements
  u64 prog_idx;   // addr ring: progress-word index the sender must write
                  // (allocated from the receiver's per-pair pool — NOT the
                  // ring slot, which recycles while windows are pending)
  u64 _pad[3];
};
static_assert(sizeof(RndzvRec) == 64, "");

// Control block per ordered pair, resident in the CONSUMER's arena so every
// poll is a local read. Writers are remote (xGMI / peer shm).
struct alignas(64) EagerChanCtl {
  // written by the sender (remote), polled by receiver:
  //   SlotHdr hdr[n_slots] follows this struct.
  // written by the receiver of the OPPOSITE direction (remote), polled locally
  // by this rank acting as sender towards `peer`:
  u64 credit;     // number of slots this rank's messages to `peer`... see note
  u64 tx_ctr;     // STREAM lanes only: shared tx-seq allocator so the engine,
                  // the host and device kernels can produce into one channel
  // rendezvous-ring flow control (same placement rule as `credit`): the
  // POSTER of addr/done records polls these locally; the ring CONSUMER
  // advances them (cumulative consumed seq). Without them a poster can
  // overwrite unconsumed records once parked ops leave windows
  // outstanding long enough for the n_rndzv ring to wrap.
  u64 addr_ret;   // addr records I posted to `peer` that it has consumed
  u64 done_ret;   // done records I posted to `peer` that it has consumed
  u64 _pad[4];
};
// NOTE on credit placement: for channel (s -> r), slot headers+payload live in
// r's arena at index [s]; the credit word lives in s's arena at index [r]
// (field `credit` of s's EagerChanCtl[r]) and is advanced by r. Each side
// polls only its own HBM.

// Device-initiated call ring, resident in the OWNER's arena: device kernels
// (or test harnesses) allocate a slot via atomic head, publish a CallDesc by
// seq, and the engine consumes it alongside the host ring — the analogue of
// the reference's client_arbiter merging PL-kernel and host command streams
// (kernels/plugins/client_arbiter/client_arbiter.cpp:21-51).
constexpr u32 DEVCALL_RING = 32;
struct alignas(64) DevCallRing {
  u64 head;                 // atomic allocator (producers fetch_add)
  u64 _pad[7];
  // CallDesc[DEVCALL_RING] follows, then RetEntry-style u64 seq+err pairs
};
struct alignas(64) DevCallSlot {
  u64 seq;                  // published last by the producer (idx+1)
  u64 _pad0[7];
  CallDesc d;
};
struct alignas(64) DevCallRet {
  u64 seq;                  // published by the engine when done (idx+1)
  u64 errcode;
  u64 t_start, t_end;
  u64 _pad[4];
};

struct alignas(64) ArenaHdr {
  u32 magic;            // 'ACCL'
  u32 version;
  u32 rank, nranks;
  u32 n_slots, slot_bytes, n_rndzv, n_stream;
  u32 stream_bytes, _pad0;
  u64 arena_bytes;
  u64 eager_off;        // EagerChanCtl[nranks] + SlotHdr[nranks][n_slots]
  u64 rndzv_addr_off;   // RndzvRec[nranks][n_rndzv]  (this rank = sender side)
  u64 rndzv_done_off;   // RndzvRec[nranks][n_rndzv]  (this rank = receiver side)
  u64 stream_off;       // stream rings
  u64 slots_off;        // payload [nranks][n_slots][slot_bytes]
  u64 heap_off;
  u64 barrier_off;      // per-peer barrier tokens u64[nranks]
  u64 direct_off;       // per-peer cumulative direct-write progress u64[nranks]
  u64 spare_off;        // staging region for non-arena rendezvous targets
  u64 spare_bytes;      //   (reference: spare buffers, accl.cpp:1174-1196)
  u64 devcall_off;      // DevCallRing + slots + rets (device-initiated calls)
  u64 dbg_off;          // engine flow-state dump on timeout (debug_dump)
};
constexpr u32 ARENA_MAGIC = 0x4143434Cu;  // "ACCL"

struct ArenaLayout {
  u64 eager_off, rndzv_addr_off, rndzv_done_off, stream_off, slots_off,
      barrier_off, direct_off, spare_off, spare_bytes, devcall_off, dbg_off,
      heap_off, total_ctl_bytes;
};

constexpr u64 DBG_DUMP_BYTES = 8192;  // flow-state dump region size

// per-pair pool of window progress words. Sized so it can never exhaust
// structurally: MAX_INFLIGHT parked recvs x 2 windows + an active direct
// collective's 2 banks < 128.
constexpr u32 N_PROG = 128;

inline ArenaLayout arena_layout(const ProtoConfig& c, u64 spare_bytes = 32u << 20) {
  auto align_up = [](u64 x, u64 a) { return (x + a - 1) & ~(a - 1); };
  ArenaLayout L{};
  u64 off = align_up(sizeof(ArenaHdr), 256);
  L.eager_off = off;
  off += u64(c.nranks) * (sizeof(EagerChanCtl) + u64(c.n_slots) * sizeof(SlotHdr));
  L.rndzv_addr_off = off = align_up(off, 256);
  off += u64(c.nranks) * c.n_rndzv * sizeof(RndzvRec);
  L.rndzv_done_off = off = align_up(off, 256);
  off += u64(c.nranks) * c.n_rndzv * sizeof(RndzvRec);
  L.stream_off = off = align_up(off, 4096);
  // per-src-peer: EagerChanCtl (credit) + SlotHdr[n_stream] + payload slots
  off += u64(c.nranks) *
         (sizeof(EagerChanCtl) + u64(c.n_stream) * sizeof(SlotHdr) +
          u64(c.n_stream) * c.stream_bytes);
  L.barrier_off = off = align_up(off, 256);
  off += u64(c.nranks) * sizeof(u64);
  L.direct_off = off = align_up(off, 256);
  off += u64(c.nranks) * N_PROG * sizeof(u64);
  L.devcall_off = off = align_up(off, 256);
  off += sizeof(DevCallRing) + u64(DEVCALL_RING) * sizeof(DevCallSlot) +
         u64(DEVCALL_RING) * sizeof(DevCallRet);
  L.dbg_off = off = align_up(off, 256);
  off += DBG_DUMP_BYTES;
  L.slots_off = off = align_up(off, 4096);
  off += u64(c.nranks) * c.n_slots * u64(c.slot_bytes);
  L.spare_off = off = align_up(off, 4096);
  L.spare_bytes = spare_bytes;
  off += spare_bytes;
  L.heap_off = off = align_up(off, 4096);
  L.total_ctl_bytes = off;
  return L;
}

// --------------------------------------------------- per-rank runtime view
// Everything an engine (GPU or CPU) needs to run the transport: its own arena
// plus the mapped base pointer of every peer's arena. Filled by the host at
// initialize() (the accl_network_utils analogue — SURVEY §3.1).
struct TransportView {
  ProtoConfig cfg;
  char* arena[MAX_RANKS];   // arena[r] = base of rank r's arena in MY address
                            // space (own rank: local pointer). null if absent.
  ACCL_HD ArenaHdr* hdr(u32 r) const { return (ArenaHdr*)arena[r]; }
  // eager channel (s -> r): control in r's arena, indexed by s
  ACCL_HD EagerChanCtl* chan_ctl(u32 r, u32 s) const {
    return (EagerChanCtl*)(arena[r] + hdr(r)->eager_off +
                           u64(s) * (sizeof(EagerChanCtl) + u64(hdr(r)->n_slots) * sizeof(SlotHdr)));
  }
  ACCL_HD SlotHdr* slot_hdr(u32 r, u32 s, u32 slot) const {
    return (SlotHdr*)((char*)chan_ctl(r, s) + sizeof(EagerChanCtl)) + slot;
  }
  ACCL_HD char* slot_payload(u32 r, u32 s, u32 slot) const {
    const ArenaHdr* h = hdr(r);
    return arena[r] + h->slots_off +
           (u64(s) * h->n_slots + slot) * u64(h->slot_bytes);
  }
  // rendezvous addr ring for pair (sender = s): lives in s's arena, lane [r]
  // (r = the receiver that posts into it).
  ACCL_HD RndzvRec* rndzv_addr(u32 s, u32 r, u32 i) const {
    return (RndzvRec*)(arena[s] + hdr(s)->rndzv_addr_off) +
           u64(r) * hdr(s)->n_rndzv + i;
  }
  // rendezvous done ring: lives in receiver r's arena, lane [s].
  ACCL_HD RndzvRec* rndzv_done(u32 r, u32 s, u32 i) const {
    return (RndzvRec*)(arena[r] + hdr(r)->rndzv_done_off) +
           u64(s) * hdr(r)->n_rndzv + i;
  }
  ACCL_HD u64* barrier_word(u32 r, u32 peer) const {
    return (u64*)(arena[r] + hdr(r)->barrier_off) + peer;
  }
  // per-WINDOW direct-write progress words: in r's arena, lane [s] word [i]
  // (i = RndzvRec.prog_idx, allocated from r's per-pair pool when posting —
  // decoupled from the addr-ring slot, which recycles while parked windows
  // are still pending). The sender of a window writes cumulative bytes
  // WITHIN the window; the receiver zeroes the word when posting and polls
  // it locally. Window-scoped (not pair-cumulative) so out-of-order matched
  // rendezvous ops never corrupt each other.
  ACCL_HD volatile u64* direct_word(u32 r, u32 s, u32 i) const {
    return (volatile u64*)(arena[r] + hdr(r)->direct_off) +
           u64(s) * N_PROG + i;
  }
  ACCL_HD char* heap_ptr(u32 r, u64 off) const { return arena[r] + off; }

  // ---- device-call ring (own arena only) ----
  ACCL_HD DevCallRing* devcall_ring(u32 r) const {
    return (DevCallRing*)(arena[r] + hdr(r)->devcall_off);
  }
  ACCL_HD DevCallSlot* devcall_slot(u32 r, u32 i) const {
    return (DevCallSlot*)((char*)devcall_ring(r) + sizeof(DevCallRing)) + i;
  }
  ACCL_HD DevCallRet* devcall_ret(u32 r, u32 i) const {
    return (DevCallRet*)((char*)devcall_slot(r, 0) +
                         u64(DEVCALL_RING) * sizeof(DevCallSlot)) + i;
  }

  // ---- stream channel (s -> r): ctl+hdrs+payload in r's arena lane [s];
  // credit word in s's arena lane [r] (advanced by the consumer at r).
  ACCL_HD u64 stream_lane_bytes(u32 r) const {
    const ArenaHdr* h = hdr(r);
    return sizeof(EagerChanCtl) + u64(h->n_stream) * sizeof(SlotHdr) +
           u64(h->n_stream) * h->stream_bytes;
  }
  ACCL_HD EagerChanCtl* stream_ctl(u32 r, u32 s) const {
    return (EagerChanCtl*)(arena[r] + hdr(r)->stream_off +
                           u64(s) * stream_lane_bytes(r));
  }
  ACCL_HD SlotHdr* stream_hdr(u32 r, u32 s, u32 slot) const {
    return (SlotHdr*)((char*)stream_ctl(r, s) + sizeof(EagerChanCtl)) + slot;
  }
  ACCL_HD char* stream_payload(u32 r, u32 s, u32 slot) const {
    const ArenaHdr* h = hdr(r);
    return (char*)stream_hdr(r, s, 0) + u64(h->n_stream) * sizeof(SlotHdr) +
           u64(slot) * h->stream_bytes;
  }
};

// Per-pair sequence state, PRIVATE to one engine (not shared): lives in the
// engine's own state block. Mirrors the communicator's inbound/outbound seq
// counters (reference: communicator.hpp:34-39 rank_t session/seq).
struct PairSeq {
  u64 eager_tx[MAX_RANKS];     // segments sent to peer
  u64 eager_rx[MAX_RANKS];     // segments consumed from peer
  u64 credit_ret[MAX_RANKS];   // last credit value written back to peer
  u64 rndzv_addr_tx[MAX_RANKS];
  u64 rndzv_addr_rx[MAX_RANKS];
  u64 rndzv_done_tx[MAX_RANKS];
  u64 rndzv_done_rx[MAX_RANKS];
  u64 stream_tx[MAX_RANKS];    // stream segments sent to peer
  u64 stream_fed_rx[MAX_RANKS];// segments the ENGINE consumed as a stream-fed
                               // op source (one consumer per lane)
  u64 barrier_epoch[MAX_RANKS];// per-PAIR barrier epoch (must match both ends)
};

}  // namespace accl
