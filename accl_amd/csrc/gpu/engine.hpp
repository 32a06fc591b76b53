// GPU engine state — shared between gpudevice.cpp (host setup) and
// engine.hip (device code). Both TUs are compiled by hipcc.
//
// The persistent-kernel analogue of the CCLO block design: scheduler lane =
// MicroBlaze control loop (ccl_offload_control.c run()), mover waves = the
// DMA/arith/segmenter data plane (reference: kernels/cclo/hls/dma_mover/,
// kernels/plugins/reduce_ops/), the pinned descriptor ring = hostctrl
// (kernels/plugins/hostctrl/hostctrl.cpp:22-63).
#pragma once
#include "../common/sched.hpp"
#include "../core/backend.hpp"

namespace accl {

// LDS mailbox between the scheduler wave and its sibling small-mover wave
// (same workgroup): sub-32KB moves run on a ~100ns LDS handshake instead of
// the fleet doorbell (which costs ~10us of wake/handoff).
struct SmallMb {
  u64 seq;    // published by scheduler (workgroup release)
  u64 done;   // published by small mover after system-release of payload
  u64 quit;
  u64 _pad;
  MoveDesc d;
};
constexpr u64 SMALL_INLINE_MAX = 32u << 10;  // bytes (default; the
                                             // ACCL_INLINE_KB env knob
                                             // overrides via small_max)
constexpr u32 INLINE_TOKEN = 0x80000000u;    // poll(): done at submit time

// Device-side mover handle. Methods (submit/poll) are device-only and live
// in engine.hip; the POD fields are set up by the host.
struct GpuMover {
  MoveDesc* ring;          // device, MOVE_RING entries
  MoveState* st;           // device
  u64* head;               // device: published move count (scheduler writes)
  u64 head_cache;          // scheduler-private mirror
  u32* stop;               // device: scheduler tells movers to exit
  u64* dbg;                // debug timeline (GpuEngineState::dbg)
  // replicated doorbell lines: [i][0]=head, [i][1]=stop. Each mover wave
  // polls replica [gw%64] so no single cacheline serves the whole fleet.
  u64 (*rep)[8];
  void* small_mb;          // LDS SmallMb (set by the scheduler kernel)
  u64 small_seq;           // scheduler-private inline-move counter
  u32 tile_log2;           // 0 = default tile; ACCL_TILE_KB env override
  u32 small_max;           // inline-path cutoff bytes (ACCL_INLINE_KB)

#if defined(__HIPCC__)
  __device__ u32 submit(const MoveDesc& m);
  __device__ bool poll(u32 token);
#endif
};

// Doorbell replicas are published with ATOMIC relaxed agent stores (they
// lower to global_store sc0 sc1 — write-through, line dropped from the
// producer XCD's L2). A PLAIN store would sit dirty in the scheduler's XCD
// L2 (per-XCD L2s are not coherent) and only become visible to the other 7
// XCDs' movers when a LATER release fence wrote it back — so the last move
// of a flow set was invisible to ~7/8 of the fleet and the engine hung when
// its owner waves were cross-XCD (round-1 fresh-box allgather_rs timeout).
constexpr u32 DOORBELL_REPS = 16;

struct GpuEngineState {
  Cclo<GpuMover> cclo;     // trivially-copyable; host fills, device runs
  Cclo<GpuMover>::ColdState cold;  // match/park tables (device-global, not
                                   // LDS: per-WG LDS budget is 64 KB)
  u32 no_acq;              // ACCL_NO_ACQ=1: skip the mover wake-batch system
                           // acquire (measurement only — UNSOUND for peer-
                           // written payload, see mover_main)
  u32 dev_ring;            // descriptor ring lives in DEVICE memory (host
                           // writes over the large BAR): scheduler polls
                           // HBM instead of fetching descs over PCIe
  // device-resident descriptor ring (dev_ring == 1)
  alignas(64) CallDesc dring[RING_CAP];
  alignas(64) u64 ddoorbell;
  u64 _dpad[7];
  GpuMover mover;
  MoveDesc mq[MOVE_RING];
  MoveState mst[MOVE_RING];
  u64 mq_head;
  u32 stop;
  u32 _pad;
  alignas(64) u64 head_rep[DOORBELL_REPS][8];
  // pinned-host pointers (device-accessible):
  CallDesc* descs;
  RetEntry* rets;
  CtrlPage* ctrl;
  CommView* comm_mirror;
  // debug: tiles executed per mover wave (plain per-wave stores)
  u32 wave_tiles[4096];
  // debug timeline of the most recent move: [0]=submit, [1]=first claim,
  // [2]=first tile done, [3]=poll observed complete (wallclock ticks)
  u64 dbg[16];
};

// launches the persistent engine kernels (defined in engine.hip):
// mover fleet on mover_stream, scheduler on sched_stream
void gpu_engine_launch(GpuEngineState* state_dev, int n_wgs, void* sched_stream,
                       void* mover_stream);

// bring-up handshake probe (shader-store into every peer mapping)
constexpr u64 PROBE_MAGIC = 0x50524F4245000000ull;  // "PROBE" | rank
struct ProbeArgs {
  u64* probe[MAX_RANKS];  // peer arenas: my probe word in rank r's arena
  u64* ack[MAX_RANKS];    // peer arenas: my ack word in rank r's arena
  u32 me, nranks;
  u64 seen_mask;
};
void gpu_probe_launch(const ProbeArgs& a, void* stream);

}  // namespace accl
