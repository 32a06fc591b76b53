// accl_amd device-side API — collectives driven from INSIDE user HIP
// kernels, without host involvement.
//
// The MI355X-native analogue of the reference's HLS device bindings
// (reference: driver/hls/accl_hls.h — ACCLCommand :134-500 / ACCLData
// :503-543, demo kernels/plugins/vadd_put/vadd_put.cpp:25-87): a producer
// kernel pushes payload segments straight into a peer's stream ring over
// xGMI (peer-mapped arena), a consumer kernel pops segments from its own
// ring. The host-side twin of this protocol is ACCL::stream_put /
// ACCL::pop_stream (core/accl.cpp), and the engine's op_stream_put uses the
// same rings, counters and credits — all three producers interoperate.
//
// All functions are WAVE-collective: call them from one 64-lane wavefront
// (lane = threadIdx.x & 63); lane 0 performs the control-word traffic.
#pragma once
#include <hip/hip_runtime.h>
#include "../common/proto.hpp"

namespace accl {
namespace device_api {

#define ACCL_DEV_SYS __HIP_MEMORY_SCOPE_SYSTEM

struct StreamChan {
  // channel (me -> peer): payload ring in the PEER's arena, credit +
  // tx-counter in MY arena (see proto.hpp "NOTE on credit placement")
  char* my_arena;
  char* peer_arena;
  u32 me, peer;
  u32 n_stream, stream_bytes;
  u64 ctl_off_mine;   // my arena: EagerChanCtl of lane [peer] (stream region)
  u64 hdr_off_peer;   // peer arena: SlotHdr[0] of lane [me]
  u64 pay_off_peer;   // peer arena: payload[0] of lane [me]
};

__device__ inline u64 _stream_lane_bytes(const ArenaHdr* h) {
  return sizeof(EagerChanCtl) + u64(h->n_stream) * sizeof(SlotHdr) +
         u64(h->n_stream) * h->stream_bytes;
}

// Build the channel view from the two mapped arena base pointers.
__device__ inline StreamChan stream_chan(char* my_arena, char* peer_arena,
                                         u32 me, u32 peer) {
  const ArenaHdr* mh = (const ArenaHdr*)my_arena;
  const ArenaHdr* ph = (const ArenaHdr*)peer_arena;
  StreamChan c{};
  c.my_arena = my_arena;
  c.peer_arena = peer_arena;
  c.me = me;
  c.peer = peer;
  c.n_stream = ph->n_stream;
  c.stream_bytes = ph->stream_bytes;
  c.ctl_off_mine = mh->stream_off + u64(peer) * _stream_lane_bytes(mh);
  u64 lane = ph->stream_off + u64(me) * _stream_lane_bytes(ph);
  c.hdr_off_peer = lane + sizeof(EagerChanCtl);
  c.pay_off_peer = c.hdr_off_peer + u64(ph->n_stream) * sizeof(SlotHdr);
  return c;
}

// Push one segment (bytes <= stream_bytes) into the peer's ring.
// Wave-collective; returns the segment's sequence number.
__device__ inline u64 stream_push(const StreamChan& c, const void* data,
                                  u32 bytes, u32 tag, u32 msg_flags = SEG_FIRST | SEG_LAST) {
  EagerChanCtl* ctl = (EagerChanCtl*)(c.my_arena + c.ctl_off_mine);
  const int lane = int(threadIdx.x) & 63;
  u64 seq = 0;
  if (lane == 0) {
    // tx counter shared with the engine and the host
    seq = __hip_atomic_fetch_add(&ctl->tx_ctr, 1ull, __ATOMIC_RELAXED,
                                 ACCL_DEV_SYS) + 1;
    // credit gate: slot free once consumer advanced past seq - n_stream
    // (in-loop acquire: the consumer may be another process — see
    // stream_pop's staleness note)
    while (__hip_atomic_load(&ctl->credit, __ATOMIC_RELAXED, ACCL_DEV_SYS) +
               c.n_stream < seq) {
      __builtin_amdgcn_s_sleep(8);
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
    }
  }
  seq = u64(__shfl(int(seq & 0xFFFFFFFF), 0, 64)) |
        (u64(u32(__shfl(int(seq >> 32), 0, 64))) << 32);
  u32 slot = u32((seq - 1) % c.n_stream);
  // payload: wave-cooperative copy into the peer slot over xGMI
  char* dst = c.peer_arena + c.pay_off_peer + u64(slot) * c.stream_bytes;
  const char* src = (const char*)data;
  for (u32 i = lane * 4; i + 3 < bytes; i += 64 * 4)
    *(u32*)(dst + i) = *(const u32*)(src + i);
  if (lane == 0)
    for (u32 i = bytes & ~3u; i < bytes; ++i) dst[i] = src[i];  // byte tail
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (lane == 0) {
    SlotHdr* h = (SlotHdr*)(c.peer_arena + c.hdr_off_peer) + slot;
    h->tag = tag;
    h->bytes = bytes;
    h->msg_count = bytes;
    h->arith = 0;
    h->flags = msg_flags;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __hip_atomic_store(&h->seq, seq, __ATOMIC_RELAXED, ACCL_DEV_SYS);
  }
  return seq;
}

struct StreamRx {
  // channel (peer -> me): ring in MY arena, credit word in PEER's arena
  char* my_arena;
  char* peer_arena;
  u32 n_stream, stream_bytes;
  u64 hdr_off_mine, pay_off_mine;
  u64 ctl_off_peer;
};

__device__ inline StreamRx stream_rx(char* my_arena, char* peer_arena,
                                     u32 me, u32 peer) {
  const ArenaHdr* mh = (const ArenaHdr*)my_arena;
  const ArenaHdr* ph = (const ArenaHdr*)peer_arena;
  StreamRx r{};
  r.my_arena = my_arena;
  r.peer_arena = peer_arena;
  r.n_stream = mh->n_stream;
  r.stream_bytes = mh->stream_bytes;
  u64 lane = mh->stream_off + u64(peer) * _stream_lane_bytes(mh);
  r.hdr_off_mine = lane + sizeof(EagerChanCtl);
  r.pay_off_mine = r.hdr_off_mine + u64(mh->n_stream) * sizeof(SlotHdr);
  r.ctl_off_peer = ph->stream_off + u64(me) * _stream_lane_bytes(ph);
  return r;
}

// Pop the next segment (sequence `seq`, 1-based caller-tracked) into dst.
// Wave-collective; returns payload bytes, fills *tag. Spins until arrival.
__device__ inline u32 stream_pop(const StreamRx& r, u64 seq, void* dst,
                                 u32 max_bytes, u32* tag) {
  const int lane = int(threadIdx.x) & 63;
  u32 slot = u32((seq - 1) % r.n_stream);
  SlotHdr* h = (SlotHdr*)(r.my_arena + r.hdr_off_mine) + slot;
  // every lane spins convergently under the exec mask (no workgroup barrier:
  // this function is wave-collective and may be called from ONE wave of a
  // multi-wave workgroup — an s_barrier here would hang the sibling waves).
  // The in-loop acquire drops stale L2 lines: the producer may be another
  // PROCESS (IPC) whose stores do not invalidate this XCD's L2 copies.
  while (__hip_atomic_load(&h->seq, __ATOMIC_RELAXED, ACCL_DEV_SYS) != seq) {
    __builtin_amdgcn_s_sleep(8);
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
  }
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
  u32 bytes = h->bytes;
  if (tag && lane == 0) *tag = h->tag;
  u32 n = bytes < max_bytes ? bytes : max_bytes;
  const char* src = r.my_arena + r.pay_off_mine + u64(slot) * r.stream_bytes;
  for (u32 i = lane * 4; i + 3 < n; i += 64 * 4)
    *(u32*)((char*)dst + i) = *(const u32*)(src + i);
  if (lane == 0)
    for (u32 i = n & ~3u; i < n; ++i) ((char*)dst)[i] = src[i];
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (lane == 0) {
    // return credit (word lives in the SENDER's arena)
    EagerChanCtl* ctl = (EagerChanCtl*)(r.peer_arena + r.ctl_off_peer);
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
    __hip_atomic_store(&ctl->credit, seq, __ATOMIC_RELAXED, ACCL_DEV_SYS);
  }
  return n;
}

// ---------------- device-initiated collective calls ----------------
// A kernel enqueues a full collective on its OWN engine (the reference's
// ACCLCommand::start_call path, driver/hls/accl_hls.h:134-188, arbitrated
// with host calls like client_arbiter.cpp:21-51). Lane-0-only (call from
// one lane or guard with lane==0); returns a token for device_call_wait.
__device__ inline u64 device_call(char* my_arena, const CallDesc& d) {
  const ArenaHdr* h = (const ArenaHdr*)my_arena;
  DevCallRing* ring = (DevCallRing*)(my_arena + h->devcall_off);
  u64 idx = __hip_atomic_fetch_add(&ring->head, 1ull, __ATOMIC_RELAXED,
                                   ACCL_DEV_SYS);
  DevCallSlot* s = (DevCallSlot*)((char*)ring + sizeof(DevCallRing)) +
                   (idx % DEVCALL_RING);
  // slot reuse guard: wait until the previous occupant's ret was published
  if (idx >= DEVCALL_RING) {
    DevCallRet* prev =
        (DevCallRet*)((char*)ring + sizeof(DevCallRing) +
                      u64(DEVCALL_RING) * sizeof(DevCallSlot)) +
        (idx % DEVCALL_RING);
    while (__hip_atomic_load(&prev->seq, __ATOMIC_RELAXED, ACCL_DEV_SYS) <
           idx + 1 - DEVCALL_RING)
      __builtin_amdgcn_s_sleep(16);
  }
  s->d = d;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __hip_atomic_store(&s->seq, idx + 1, __ATOMIC_RELAXED, ACCL_DEV_SYS);
  return idx;
}

// Poll completion of a device_call; returns errcode (0 = OK). Blocking.
__device__ inline u32 device_call_wait(char* my_arena, u64 token) {
  const ArenaHdr* h = (const ArenaHdr*)my_arena;
  DevCallRing* ring = (DevCallRing*)(my_arena + h->devcall_off);
  DevCallRet* r = (DevCallRet*)((char*)ring + sizeof(DevCallRing) +
                                u64(DEVCALL_RING) * sizeof(DevCallSlot)) +
                  (token % DEVCALL_RING);
  while (__hip_atomic_load(&r->seq, __ATOMIC_RELAXED, ACCL_DEV_SYS) < token + 1)
    __builtin_amdgcn_s_sleep(16);
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
  return u32(r->errcode);
}

#undef ACCL_DEV_SYS

}  // namespace device_api
}  // namespace accl
