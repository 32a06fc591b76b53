#include "gpudevice.hpp"
#include <hip/hip_runtime.h>
#include <unistd.h>
#include <cstdlib>
#include <cstring>
#include "engine.hpp"

namespace accl {

static void hip_check(hipError_t e, const char* what) {
  if (e != hipSuccess)
    throw accl_error(std::string("hip error in ") + what + ": " +
                     hipGetErrorString(e));
}

struct GpuBlob {
  hipIpcMemHandle_t handle;
  u64 arena_bytes;
  u32 rank;
  u32 device;
  i32 pid;
  u32 _pad;
};

static bool dbg_connect() { return std::getenv("ACCL_DEBUG_CONNECT"); }
#define CSTAGE(msg) do { if (dbg_connect()) { fprintf(stderr, "[accl connect r%u] %s\n", cfg_.rank, msg); fflush(stderr); } } while (0)

GpuDevice::GpuDevice(u32 nranks, u32 rank, int device_index,
                     const ProtoConfig* cfg_override, u64 heap_bytes,
                     int engine_wgs) {
  dev_ = device_index;
  cfg_ = cfg_override ? *cfg_override : default_proto_config(nranks, rank);
  cfg_.nranks = nranks;
  cfg_.rank = rank;
  if (const char* e = std::getenv("ACCL_ENGINE_WGS")) engine_wgs = atoi(e);
  if (engine_wgs > 0) engine_wgs_ = engine_wgs;

  hip_check(hipSetDevice(dev_), "hipSetDevice");
  layout_ = arena_layout(cfg_);
  arena_bytes_ = layout_.total_ctl_bytes + heap_bytes;

  // Coarse-grained HBM by default: the standard hipMalloc+IPC path (what
  // RCCL uses) — cross-agent visibility comes from system-scope atomics and
  // release fences, and measured bandwidth is identical to fine-grained.
  // ACCL_FINE_ARENA=1 opts into fine-grained allocation.
  bool fine = std::getenv("ACCL_FINE_ARENA") != nullptr;
  hipError_t e = hipErrorUnknown;
  if (fine) {
    e = hipExtMallocWithFlags((void**)&arena_base_, arena_bytes_,
                              hipDeviceMallocFinegrained);
    fine_grained_ = (e == hipSuccess);
  }
  if (!fine || e != hipSuccess) {
    hip_check(hipMalloc((void**)&arena_base_, arena_bytes_), "hipMalloc arena");
    fine_grained_ = false;
  }
  hip_check(hipMemset(arena_base_, 0, layout_.total_ctl_bytes), "memset ctl");

  ArenaHdr h{};
  h.version = 1;
  h.rank = rank; h.nranks = nranks;
  h.n_slots = cfg_.n_slots; h.slot_bytes = cfg_.slot_bytes;
  h.n_rndzv = cfg_.n_rndzv; h.n_stream = cfg_.n_stream;
  h.stream_bytes = cfg_.stream_bytes;
  h.arena_bytes = arena_bytes_;
  h.eager_off = layout_.eager_off;
  h.rndzv_addr_off = layout_.rndzv_addr_off;
  h.rndzv_done_off = layout_.rndzv_done_off;
  h.stream_off = layout_.stream_off;
  h.slots_off = layout_.slots_off;
  h.heap_off = layout_.heap_off;
  h.barrier_off = layout_.barrier_off;
  h.direct_off = layout_.direct_off;
  h.spare_off = layout_.spare_off;
  h.spare_bytes = layout_.spare_bytes;
  h.devcall_off = layout_.devcall_off;
  h.dbg_off = layout_.dbg_off;
  h.magic = ARENA_MAGIC;
  hip_check(hipMemcpy(arena_base_, &h, sizeof(h), hipMemcpyHostToDevice),
            "write ArenaHdr");

  heap_.init(layout_.heap_off, arena_bytes_ - layout_.heap_off);

  hip_check(hipHostMalloc(&ring_pinned_, sizeof(RingPage), 0), "hipHostMalloc ring");
  std::memset(ring_pinned_, 0, sizeof(RingPage));
  ring_ = (RingPage*)ring_pinned_;

  hip_check(hipMalloc(&state_dev_, sizeof(GpuEngineState)), "hipMalloc state");
  hipStream_t s;
  hip_check(hipStreamCreateWithFlags(&s, hipStreamNonBlocking), "stream");
  stream_ = s;
  hip_check(hipStreamCreateWithFlags(&s, hipStreamNonBlocking), "mover stream");
  mover_stream_ = s;
}

GpuDevice::~GpuDevice() {
  try { shutdown(); } catch (...) {}
  for (u32 r = 0; r < cfg_.nranks; ++r)
    if (peer_base_[r] && r != cfg_.rank) (void)hipIpcCloseMemHandle(peer_base_[r]);
  if (state_dev_) (void)hipFree(state_dev_);
  if (arena_base_) (void)hipFree(arena_base_);
  if (ring_pinned_) (void)hipHostFree(ring_pinned_);
  if (stream_) (void)hipStreamDestroy((hipStream_t)stream_);
  if (mover_stream_) (void)hipStreamDestroy((hipStream_t)mover_stream_);
}

std::vector<char> GpuDevice::local_blob() {
  GpuBlob b{};
  hip_check(hipIpcGetMemHandle(&b.handle, arena_base_), "hipIpcGetMemHandle");
  b.arena_bytes = arena_bytes_;
  b.rank = cfg_.rank;
  b.device = u32(dev_);
  b.pid = i32(getpid());
  std::vector<char> out(sizeof(b));
  std::memcpy(out.data(), &b, sizeof(b));
  return out;
}

void GpuDevice::connect(const std::vector<std::vector<char>>& blobs) {
  if (blobs.size() != cfg_.nranks) throw accl_error("gpu: blob count != nranks");
  std::vector<GpuBlob> pb(cfg_.nranks);
  for (u32 r = 0; r < cfg_.nranks; ++r) {
    if (r == cfg_.rank) { peer_base_[r] = arena_base_; continue; }
    GpuBlob& b = pb[r];
    std::memcpy(&b, blobs[r].data(), sizeof(b));
    if (b.arena_bytes != arena_bytes_)
      throw accl_error("gpu: rank " + std::to_string(r) +
                       " arena size mismatch (configs must agree)");
    void* p = nullptr;
    hip_check(hipIpcOpenMemHandle(&p, b.handle, hipIpcMemLazyEnablePeerAccess),
              "hipIpcOpenMemHandle");
    peer_base_[r] = (char*)p;
  }

  // ---- bring-up handshake: every peer mapping must pass SHADER writes
  // before the persistent engine depends on it. A lazily enabled IPC peer
  // mapping has been observed (2 procs / 1 GPU) to silently drop shader
  // stores in one direction, wedging the first collective; the handshake
  // detects the dead direction at init and re-imports the handle.
  if (cfg_.nranks > 1) {
    const u64 probe_base = layout_.dbg_off + DBG_DUMP_BYTES - 2048;
    const u64 ack_base = layout_.dbg_off + DBG_DUMP_BYTES - 1024;
    const u32 me = cfg_.rank;
    const u64 full = (cfg_.nranks >= 64) ? ~0ull
                                         : ((1ull << cfg_.nranks) - 1);
    u64 seen = 1ull << me;
    u64 t0 = wallclock_host_ns(), last_change = t0;
    int reimports = 0;
    auto read_word = [&](u64 off) {
      u64 v = 0;
      read_arena(off, &v, sizeof(v));
      return v;
    };
    auto launch_probe = [&]() {
      ProbeArgs a{};
      a.me = me;
      a.nranks = cfg_.nranks;
      a.seen_mask = seen;
      for (u32 r = 0; r < cfg_.nranks; ++r) {
        if (r == me) continue;
        a.probe[r] = (u64*)(peer_base_[r] + probe_base + u64(me) * 8);
        a.ack[r] = (u64*)(peer_base_[r] + ack_base + u64(me) * 8);
      }
      gpu_probe_launch(a, stream_);
      hip_check(hipStreamSynchronize((hipStream_t)stream_), "probe sync");
    };
    for (;;) {
      launch_probe();
      usleep(2000);
      u64 ns = seen;
      bool all_acked = true;
      for (u32 r = 0; r < cfg_.nranks; ++r) {
        if (r == me) continue;
        if (read_word(probe_base + u64(r) * 8) == (PROBE_MAGIC | r))
          ns |= 1ull << r;
        if (!(read_word(ack_base + u64(r) * 8) & (1ull << me)))
          all_acked = false;
      }
      bool progressed = ns != seen;
      seen = ns;
      u64 now = wallclock_host_ns();
      if (seen == full && all_acked) {
        launch_probe();  // final ack broadcast with the complete seen mask
        break;
      }
      if (progressed) last_change = now;
      if (now - last_change > 1500ull * 1000 * 1000) {
        // a direction is dead: re-import every peer handle and retry
        for (u32 r = 0; r < cfg_.nranks; ++r) {
          if (r == me) continue;
          (void)hipIpcCloseMemHandle(peer_base_[r]);
          void* p = nullptr;
          hip_check(hipIpcOpenMemHandle(&p, pb[r].handle,
                                        hipIpcMemLazyEnablePeerAccess),
                    "hipIpcOpenMemHandle (reimport)");
          peer_base_[r] = (char*)p;
        }
        reimports++;
        last_change = now;
      }
      if (now - t0 > 60ull * 1000 * 1000 * 1000)
        throw accl_error("gpu: peer-mapping handshake never completed "
                         "(dead IPC direction persisted)");
    }
    if (reimports)
      fprintf(stderr,
              "accl gpu rank %u: peer mapping recovered after %d "
              "re-import(s)\n", me, reimports);
  }

  CSTAGE("peers mapped + handshake done");
  // build the engine state on host, copy to device
  auto* st = new GpuEngineState();
  std::memset((void*)st, 0, sizeof(GpuEngineState));
  Cclo<GpuMover>& C = st->cclo;
  C.cfg = cfg_;
  C.tv.cfg = cfg_;
  for (u32 r = 0; r < cfg_.nranks; ++r) C.tv.arena[r] = peer_base_[r];
  C.timeout_ticks = cfg_.timeout_us * TICKS_PER_US;
  C.max_eager_bytes = cfg_.max_eager;
  // device pointers inside the state buffer itself
  auto* dstate = (GpuEngineState*)state_dev_;
  st->mover.ring = dstate->mq;          // address arithmetic on device ptr
  st->mover.st = dstate->mst;
  st->mover.head = &dstate->mq_head;
  st->mover.stop = &dstate->stop;
  st->mover.dbg = dstate->dbg;
  st->no_acq = std::getenv("ACCL_NO_ACQ") ? 1u : 0u;
  if (const char* ik = std::getenv("ACCL_INLINE_KB"))
    st->mover.small_max = u32(strtoul(ik, nullptr, 10)) << 10;
  if (const char* t = std::getenv("ACCL_TILE_KB")) {
    u64 kb = strtoull(t, nullptr, 10);
    u32 lg = 0;
    while ((1ull << (lg + 1)) <= kb * 1024) ++lg;
    st->mover.tile_log2 = lg;
  }
  st->mover.rep = dstate->head_rep;
  // pinned pointers as seen by the device
  RingPage* rp = (RingPage*)ring_pinned_;
  void* dev_ptr = nullptr;
  hip_check(hipHostGetDevicePointer(&dev_ptr, rp, 0), "hostGetDevicePointer");
  auto* rp_dev = (RingPage*)dev_ptr;
  st->descs = rp_dev->descs;
  st->rets = rp_dev->rets;
  st->ctrl = &rp_dev->ctrl;
  st->comm_mirror = rp_dev->comm_mirror;
  // Device-resident descriptor ring (small-op latency): with a large BAR
  // the host can store descs + doorbell straight into HBM, so the
  // scheduler polls local memory instead of fetching over PCIe each
  // iteration. Verified by write+readback; ACCL_NO_DEV_RING disables.
  CSTAGE("state built");
  bool dev_ring_ok = false;
  if (!std::getenv("ACCL_NO_DEV_RING")) {
    int large_bar = 0;
    (void)hipDeviceGetAttribute(&large_bar, hipDeviceAttributeIsLargeBar,
                                dev_);
    if (large_bar) {
      volatile u64* bar_door = &dstate->ddoorbell;
      *bar_door = 0xC0FFEEull;
      __sync_synchronize();
      u64 chk = 0;
      if (hipMemcpy(&chk, (void*)bar_door, 8, hipMemcpyDeviceToHost) ==
              hipSuccess &&
          chk == 0xC0FFEEull)
        dev_ring_ok = true;
    }
  }
  CSTAGE(dev_ring_ok ? "dev_ring ON" : "dev_ring off");
  st->dev_ring = dev_ring_ok ? 1u : 0u;
  hip_check(hipMemcpy(state_dev_, st, sizeof(GpuEngineState),
                      hipMemcpyHostToDevice), "state upload");
  if (dev_ring_ok) {  // host-side submit now targets the device ring
    desc_ring_ = dstate->dring;
    door_ = &dstate->ddoorbell;
  }
  delete st;

  // global communicator 0 (before launch: engine reads mirror at gen bump)
  std::vector<u32> members(cfg_.nranks);
  for (u32 i = 0; i < cfg_.nranks; ++i) members[i] = i;
  // write mirror directly (engine not yet running; add_comm would quiesce)
  CommView& c = ring_->comm_mirror[0];
  c.id = 0; c.rank = cfg_.rank; c.size = cfg_.nranks;
  for (u32 i = 0; i < cfg_.nranks; ++i) c.members[i] = i;
  __atomic_store_n((u64*)&ring_->ctrl.ncomms, 1, __ATOMIC_RELEASE);
  __atomic_store_n((u64*)&ring_->ctrl.comm_gen, 1, __ATOMIC_RELEASE);

  CSTAGE("state uploaded; launching engine");
  gpu_engine_launch((GpuEngineState*)state_dev_, engine_wgs_, stream_,
                    mover_stream_);
  hip_check(hipGetLastError(), "engine kernel launch");
  launched_ = true;
  CSTAGE("kernels launched; waiting engine_up");
  // wait for the engine to come up
  u64 t0 = wallclock_host_ns();
  while (!__atomic_load_n((u64*)&ring_->ctrl.engine_up, __ATOMIC_ACQUIRE)) {
    if (wallclock_host_ns() - t0 > 30ull * 1000000000) {
      u64 movers_started = 0;
      (void)hipMemcpy(&movers_started,
                      (char*)state_dev_ + offsetof(GpuEngineState, dbg) +
                          15 * sizeof(u64),
                      sizeof(u64), hipMemcpyDeviceToHost);
      throw accl_error("gpu: engine kernel never came up (mover WGs "
                       "started: " + std::to_string(movers_started) + "/" +
                       std::to_string(engine_wgs_) + ")");
    }
    usleep(100);
  }
}

void GpuDevice::shutdown() {
  if (!launched_) return;
  __atomic_store_n((u64*)&ring_->ctrl.shutdown, 1, __ATOMIC_RELEASE);
  // bounded: a wedged engine must not hang the process forever (process
  // teardown destroys the HIP context and with it the kernels)
  u64 t0 = wallclock_host_ns();
  hipError_t e = hipErrorUnknown, e2 = hipErrorUnknown;
  for (;;) {
    e = hipStreamQuery((hipStream_t)stream_);
    e2 = hipStreamQuery((hipStream_t)mover_stream_);
    if (e != hipErrorNotReady && e2 != hipErrorNotReady) break;
    if (wallclock_host_ns() - t0 > 15ull * 1000000000) {
      launched_ = false;
      throw accl_error("gpu: engine did not acknowledge shutdown (wedged)");
    }
    usleep(200);
  }
  launched_ = false;
  hip_check(e, "engine shutdown");
  hip_check(e2, "mover shutdown");
}

std::vector<u64> GpuDevice::debug_timeline() {
  std::vector<u64> v(16);
  hip_check(hipMemcpy(v.data(),
                      (char*)state_dev_ + offsetof(GpuEngineState, dbg),
                      sizeof(u64) * v.size(), hipMemcpyDeviceToHost),
            "read dbg");
  return v;
}

std::vector<u32> GpuDevice::debug_wave_tiles() {
  std::vector<u32> v(4096);
  hip_check(hipMemcpy(v.data(),
                      (char*)state_dev_ + offsetof(GpuEngineState, wave_tiles),
                      sizeof(u32) * v.size(), hipMemcpyDeviceToHost),
            "read wave_tiles");
  return v;
}

void GpuDevice::write_arena(u64 off, const void* src, u64 bytes) {
  hip_check(hipMemcpy(arena_base_ + off, src, bytes, hipMemcpyHostToDevice),
            "write_arena");
}
void GpuDevice::write_peer(u32 rank, u64 off, const void* src, u64 bytes) {
  if (rank >= cfg_.nranks || !peer_base_[rank])
    throw accl_error("gpu: write_peer to unmapped rank");
  hip_check(hipMemcpy(peer_base_[rank] + off, src, bytes,
                      hipMemcpyHostToDevice), "write_peer");
}

void GpuDevice::read_arena(u64 off, void* dst, u64 bytes) {
  hip_check(hipMemcpy(dst, arena_base_ + off, bytes, hipMemcpyDeviceToHost),
            "read_arena");
}

}  // namespace accl
