#include "emudevice.hpp"
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>
#include <cstdio>
#include <cstring>

namespace accl {

struct EmuDevice::Engine {
  Cclo<CpuMover> cclo;
  Cclo<CpuMover>::ColdState cold{};
  CpuMover mover;
  u64 cached_comm_gen = 0;
};

EmuDevice::EmuDevice(u32 nranks, u32 rank, const std::string& job,
                     const ProtoConfig* cfg_override, u64 heap_bytes) {
  job_ = job;
  cfg_ = cfg_override ? *cfg_override : default_proto_config(nranks, rank);
  cfg_.nranks = nranks;
  cfg_.rank = rank;
  layout_ = arena_layout(cfg_);
  arena_bytes_ = layout_.total_ctl_bytes + heap_bytes;

  // create + map my shm arena
  std::string name = shm_name(rank);
  shm_unlink(name.c_str());
  shm_fd_ = shm_open(name.c_str(), O_CREAT | O_RDWR | O_EXCL, 0600);
  if (shm_fd_ < 0) throw accl_error("emu: shm_open failed for " + name);
  if (ftruncate(shm_fd_, off_t(arena_bytes_)) != 0)
    throw accl_error("emu: ftruncate failed");
  arena_base_ = (char*)mmap(nullptr, arena_bytes_, PROT_READ | PROT_WRITE,
                            MAP_SHARED, shm_fd_, 0);
  if (arena_base_ == MAP_FAILED) throw accl_error("emu: mmap failed");
  std::memset(arena_base_, 0, layout_.total_ctl_bytes);

  ArenaHdr* h = (ArenaHdr*)arena_base_;
  h->version = 1;
  h->rank = rank; h->nranks = nranks;
  h->n_slots = cfg_.n_slots; h->slot_bytes = cfg_.slot_bytes;
  h->n_rndzv = cfg_.n_rndzv; h->n_stream = cfg_.n_stream;
  h->stream_bytes = cfg_.stream_bytes;
  h->arena_bytes = arena_bytes_;
  h->eager_off = layout_.eager_off;
  h->rndzv_addr_off = layout_.rndzv_addr_off;
  h->rndzv_done_off = layout_.rndzv_done_off;
  h->stream_off = layout_.stream_off;
  h->slots_off = layout_.slots_off;
  h->heap_off = layout_.heap_off;
  h->barrier_off = layout_.barrier_off;
  h->direct_off = layout_.direct_off;
  h->spare_off = layout_.spare_off;
  h->spare_bytes = layout_.spare_bytes;
  h->devcall_off = layout_.devcall_off;
  h->dbg_off = layout_.dbg_off;
  __atomic_store_n(&h->magic, ARENA_MAGIC, __ATOMIC_RELEASE);

  heap_.init(layout_.heap_off, arena_bytes_ - layout_.heap_off);
  ring_store_.reset(new RingPage());
  std::memset((void*)ring_store_.get(), 0, sizeof(RingPage));
  ring_ = ring_store_.get();
}

EmuDevice::~EmuDevice() {
  shutdown();
  for (u32 r = 0; r < cfg_.nranks; ++r) {
    if (peer_base_[r] && peer_base_[r] != arena_base_)
      munmap(peer_base_[r], arena_bytes_);
  }
  if (arena_base_) munmap(arena_base_, arena_bytes_);
  if (shm_fd_ >= 0) {
    close(shm_fd_);
    shm_unlink(shm_name(cfg_.rank).c_str());
  }
}

std::string EmuDevice::shm_name(u32 rank) const {
  return "/" + job_ + "_r" + std::to_string(rank);
}

std::vector<char> EmuDevice::local_blob() {
  std::string n = shm_name(cfg_.rank);
  return std::vector<char>(n.begin(), n.end());
}

void EmuDevice::connect(const std::vector<std::vector<char>>& blobs) {
  if (blobs.size() != cfg_.nranks)
    throw accl_error("emu: blob count != nranks");
  for (u32 r = 0; r < cfg_.nranks; ++r) {
    if (r == cfg_.rank) { peer_base_[r] = arena_base_; continue; }
    std::string name(blobs[r].begin(), blobs[r].end());
    int fd = -1;
    u64 t0 = wallclock_host_ns();
    for (;;) {
      fd = shm_open(name.c_str(), O_RDWR, 0600);
      if (fd >= 0) break;
      if (wallclock_host_ns() - t0 > 30ull * 1000000000)
        throw accl_error("emu: peer arena " + name + " never appeared");
      usleep(1000);
    }
    char* p = (char*)mmap(nullptr, arena_bytes_, PROT_READ | PROT_WRITE,
                          MAP_SHARED, fd, 0);
    close(fd);
    if (p == MAP_FAILED) throw accl_error("emu: peer mmap failed");
    // wait for the peer's header to be initialized
    u64 t1 = wallclock_host_ns();
    while (__atomic_load_n(&((ArenaHdr*)p)->magic, __ATOMIC_ACQUIRE) != ARENA_MAGIC) {
      if (wallclock_host_ns() - t1 > 30ull * 1000000000)
        throw accl_error("emu: peer arena " + name + " never became ready");
      usleep(1000);
    }
    // ranks must agree on the arena geometry (a size mismatch would SIGBUS
    // on access past the peer's shm object, not error)
    if (((ArenaHdr*)p)->arena_bytes != arena_bytes_)
      throw accl_error("emu: rank " + std::to_string(r) +
                       " arena size mismatch (configs must agree)");
    peer_base_[r] = p;
  }

  // engine state
  eng_.reset(new Engine());
  auto& C = eng_->cclo;
  std::memset((void*)&C, 0, sizeof(C));
  C.cfg = cfg_;
  for (u32 r = 0; r < cfg_.nranks; ++r) C.tv.arena[r] = peer_base_[r];
  C.tv.cfg = cfg_;
  C.mv = &eng_->mover;
  C.cold = &eng_->cold;
  C.timeout_ticks = cfg_.timeout_us * TICKS_PER_US;
  C.max_eager_bytes = cfg_.max_eager;
  // global communicator 0
  std::vector<u32> members(cfg_.nranks);
  for (u32 i = 0; i < cfg_.nranks; ++i) members[i] = i;
  add_comm(members, cfg_.rank);

  thread_ = std::thread([this] { engine_main(); });
  // wait for engine_up
  while (!__atomic_load_n((u64*)&ring_->ctrl.engine_up, __ATOMIC_ACQUIRE))
    usleep(100);
}

void EmuDevice::engine_main() {
  auto& C = eng_->cclo;
  CtrlPage& ctrl = ring_->ctrl;
  __atomic_store_n((u64*)&ctrl.engine_up, 1, __ATOMIC_RELEASE);
  u64 consumed = 0, dev_consumed = 0;
  for (;;) {
    // refresh communicator cache (reference: run() re-caches the
    // communicator per call, ccl_offload_control.c:2308-2360); must precede
    // device-ring calls too, which name communicators
    u64 gen = __atomic_load_n((u64*)&ctrl.comm_gen, __ATOMIC_ACQUIRE);
    if (gen != eng_->cached_comm_gen) {
      C.ncomms = u32(__atomic_load_n((u64*)&ctrl.ncomms, __ATOMIC_ACQUIRE));
      for (u32 i = 0; i < C.ncomms; ++i) C.comms[i] = ring_->comm_mirror[i];
      eng_->cached_comm_gen = gen;
    }
    u64 db = __atomic_load_n((u64*)&ctrl.doorbell, __ATOMIC_ACQUIRE);
    auto publish = [&](u64 idx, u32 e, u64 t0) {
      RetEntry& r = ring_->rets[idx % RING_CAP];
      r.t_start = t0;
      r.t_end = wallclock();
      r.errcode = e;
      __atomic_store_n(&r.seq, u32(idx + 1), __ATOMIC_RELEASE);
    };
    // retry parked calls first (CMD_CALL_RETRY analogue)
    for (int pi; (pi = C.retry_parked()) >= 0;)
      publish(C.done_ring_idx, C.done_err, C.done_t0);
    if (consumed == db) {
      if (C.poll_device_calls(dev_consumed)) continue;
      if (__atomic_load_n((u64*)&ctrl.shutdown, __ATOMIC_RELAXED)) break;
      if (!C.nparked)
        __atomic_fetch_add((u64*)&ctrl.heartbeat, 1, __ATOMIC_RELAXED);
      usleep(20);
      continue;
    }
    while (consumed < db) {
      const CallDesc d = ring_->descs[consumed % RING_CAP];
      u64 t0 = wallclock();
      if (Op(d.scenario) == Op::halt) {
        publish(consumed, E_OK, t0);
        consumed++;
        goto out;
      }
      u32 e = C.serve_desc(d, consumed, consumed + 1 < db);
      if (!(e & E_NOT_READY)) publish(consumed, e, t0);
      consumed++;
      if (C.nparked) break;  // interleave: give parked calls a retry round
    }
  }
out:
  // engine exiting with calls parked: fail them so host waits return
  for (int pi; (pi = eng_->cclo.fail_parked()) >= 0;) {
    RetEntry& r = ring_->rets[eng_->cclo.done_ring_idx % RING_CAP];
    r.t_start = eng_->cclo.done_t0;
    r.t_end = wallclock();
    r.errcode = eng_->cclo.done_err;
    __atomic_store_n(&r.seq, u32(eng_->cclo.done_ring_idx + 1),
                     __ATOMIC_RELEASE);
  }
}

void EmuDevice::shutdown() {
  if (!thread_.joinable()) return;
  __atomic_store_n((u64*)&ring_->ctrl.shutdown, 1, __ATOMIC_RELEASE);
  thread_.join();
}

void EmuDevice::write_arena(u64 off, const void* src, u64 bytes) {
  std::memcpy(arena_base_ + off, src, bytes);
  __atomic_thread_fence(__ATOMIC_RELEASE);
}

void EmuDevice::write_peer(u32 rank, u64 off, const void* src, u64 bytes) {
  if (rank >= cfg_.nranks || !peer_base_[rank])
    throw accl_error("emu: write_peer to unmapped rank");
  std::memcpy(peer_base_[rank] + off, src, bytes);
  __atomic_thread_fence(__ATOMIC_RELEASE);
}
void EmuDevice::read_arena(u64 off, void* dst, u64 bytes) {
  __atomic_thread_fence(__ATOMIC_ACQUIRE);
  std::memcpy(dst, arena_base_ + off, bytes);
}

}  // namespace accl
