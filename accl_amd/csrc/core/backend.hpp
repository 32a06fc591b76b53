// Backend = one rank's device engine as seen by the host runtime: a call
// descriptor ring (the hostctrl/CMD_CALL analogue, reference:
// kernels/plugins/hostctrl/hostctrl.cpp:22-63 and driver/xrt/src/
// xrtdevice.cpp:36-192), a return ring (RETVAL readback,
// ccl_offload_control.c:2291-2306), and the arena (exchange memory + rx
// buffers + heap). Concrete backends: EmuDevice (CPU engine thread over shm)
// and GpuDevice (persistent HIP kernel over HBM).
#pragma once
#include <cstdio>
#include <cstring>
#include <memory>
#include <mutex>
#include <string>
#include <vector>
#include "../common/proto.hpp"
#include "alloc.hpp"
#include "util.hpp"

namespace accl {

// Control words shared host<->engine (pinned host memory on GPU).
struct alignas(64) CtrlPage {
  volatile u64 doorbell;   // count of descriptors published by host
  volatile u64 shutdown;   // host sets 1; engine exits its loop
  volatile u64 comm_gen;   // bump after rewriting the comm mirror
  volatile u64 ncomms;
  volatile u64 heartbeat;  // engine bumps when idle (liveness)
  volatile u64 engine_up;  // engine sets 1 once running
  u64 _pad[2];
};

constexpr u32 RING_CAP = 64;

struct RingPage {
  CtrlPage ctrl;
  CommView comm_mirror[MAX_COMMS];
  CallDesc descs[RING_CAP];
  RetEntry rets[RING_CAP];
};

class Backend {
 public:
  virtual ~Backend() = default;

  // --- bring-up (two-phase; the caller runs the bootstrap allgather) ---
  virtual std::vector<char> local_blob() = 0;
  virtual void connect(const std::vector<std::vector<char>>& blobs) = 0;
  virtual void shutdown() = 0;
  virtual bool is_gpu() const = 0;

  const ProtoConfig& cfg() const { return cfg_; }
  char* arena_local() const { return arena_base_; }

  // --- calls ---
  // thread-safe in-order ring (reference: FPGAQueue serializes calls from
  // any host thread to the single engine, acclrequest.hpp:153-211)
  u64 submit(CallDesc d) {
    std::lock_guard<std::mutex> lk(ring_mu_);
    u64 seq = head_;
    while (seq - tail_retired_locked() >= RING_CAP) cpu_pause();
    d.seq = u32(seq);
    // desc/doorbell destination is normally the pinned ring; the GPU
    // backend may point these at a DEVICE-resident ring written over the
    // large BAR (engine then polls HBM instead of PCIe)
    CallDesc* descs = desc_ring_ ? desc_ring_ : ring_->descs;
    volatile u64* door = door_ ? door_ : &ring_->ctrl.doorbell;
    std::memcpy((void*)&descs[seq % RING_CAP], &d, sizeof(d));
    head_ = seq + 1;
    __sync_synchronize();  // drain WC/UC buffers before the doorbell
    __atomic_store_n((u64*)door, head_, __ATOMIC_RELEASE);
    return seq;
  }
  bool test(u64 seq, RetEntry* out) {
    RetEntry& r = ring_->rets[seq % RING_CAP];
    if (__atomic_load_n(&r.seq, __ATOMIC_ACQUIRE) != u32(seq + 1)) return false;
    if (out) *out = r;
    return true;
  }
  // returns error bits; throws on host-side timeout
  u32 wait(u64 seq, RetEntry* out = nullptr, u64 timeout_ms = 120000) {
    RetEntry r{};
    u64 t0 = wallclock_host_ns();
    while (!test(seq, &r)) {
      cpu_pause();
      if (wallclock_host_ns() - t0 > timeout_ms * 1000000ull)
        throw accl_error("accl: host wait timeout on call seq " +
                         std::to_string(seq));
    }
    if (out) *out = r;
    return r.errcode;
  }
  u64 call(const CallDesc& d, RetEntry* out = nullptr) {
    u64 s = submit(d);
    u32 e = wait(s, out);
    if (e) {
      std::string msg = "accl call failed: " + error_to_string(e);
      if (e & 1u /*E_TIMEOUT*/) msg += "\n" + timeout_dump_str();
      throw accl_error(msg, e);
    }
    return s;
  }

  // Formatted engine flow-state snapshot written by Cclo::dump_timeout —
  // what each live flow was waiting on when the deadline fired.
  std::string timeout_dump_str() {
    ArenaLayout L = arena_layout(cfg_);
    std::vector<u64> w(8 + 24 * 16);
    read_arena(L.dbg_off, w.data(), w.size() * sizeof(u64));
    if (!w[0]) return "(no engine flow dump)";
    static const char* kinds[] = {"IDLE", "LOCAL", "TX", "RX", "TX_DIRECT",
                                  "RX_DIRECT"};
    char buf[256];
    snprintf(buf, sizeof(buf),
             "engine flow dump #%llu: scenario=%llu err=0x%llx rank=%llu/%llu "
             "flows=%llu",
             (unsigned long long)w[0], (unsigned long long)(w[1] & 0xFFFFFFFF),
             (unsigned long long)(w[1] >> 32),
             (unsigned long long)(w[4] & 0xFFFFFFFF),
             (unsigned long long)(w[4] >> 32), (unsigned long long)w[2]);
    std::string out = buf;
    if (w[5]) {  // last non-flow waitpoint (wait_addr / wait_done spin)
      snprintf(buf, sizeof(buf),
               "\n  waitpoint: %s peer=%llu tag=0x%llx await_seq=%llu "
               "head_seq=%llu",
               w[5] == 1 ? "wait_addr" : w[5] == 2 ? "wait_done" : "?",
               (unsigned long long)(w[6] & 0xFFFFFFFF),
               (unsigned long long)(w[6] >> 32),
               (unsigned long long)(w[7] & 0xFFFFFFFF),
               (unsigned long long)(w[7] >> 32));
      out += buf;
    }
    if ((w[1] & 0xFFFFFFFF) == 13 && w[2] == 0) {  // barrier diagnosis
      for (u32 g = 0; g < cfg_.nranks && g < 24; ++g) {
        if (w[8 + 2 * g] == ~0ull) continue;
        snprintf(buf, sizeof(buf), "  barrier pair %u: token=%llu expect=%llu\n",
                 g, (unsigned long long)w[8 + 2 * g],
                 (unsigned long long)w[8 + 2 * g + 1]);
        out += buf;
      }
    }
    for (u64 i = 0; i < w[2] && i < 24; ++i) {
      const u64* f = &w[8 + i * 16];
      u32 kind = u32(f[0] & 0xFF);
      snprintf(buf, sizeof(buf),
               "\n  [%llu] %s peer=%llu func=%llu tag=0x%llx cnt=%llu "
               "sub=%llu done=%llu ph/pt=%llu/%llu gate=%lld",
               (unsigned long long)i, kind < 6 ? kinds[kind] : "?",
               (unsigned long long)(f[0] >> 32),
               (unsigned long long)((f[0] >> 16) & 0xFF),
               (unsigned long long)(f[1] & 0xFFFFFFFF),
               (unsigned long long)f[2], (unsigned long long)f[3],
               (unsigned long long)f[4], (unsigned long long)(f[5] & 0xFFFFFFFF),
               (unsigned long long)(f[5] >> 32), (long long)f[6]);
      out += buf;
      if (kind == 3) {  // RX: awaited seq vs observed slot header
        snprintf(buf, sizeof(buf),
                 " await_seq=%llu hdr_seq=%llu hdr_tag=0x%llx hdr_bytes=%llu",
                 (unsigned long long)f[7], (unsigned long long)f[8],
                 (unsigned long long)(f[9] & 0xFFFFFFFF),
                 (unsigned long long)(f[9] >> 32));
        out += buf;
      } else if (kind == 2) {  // TX: sent vs credit
        snprintf(buf, sizeof(buf), " sent=%llu credit=%llu",
                 (unsigned long long)f[7], (unsigned long long)f[8]);
        out += buf;
      } else if (kind == 5) {  // RX_DIRECT: progress word
        snprintf(buf, sizeof(buf), " prog_base=%llu prog=%llu",
                 (unsigned long long)f[7], (unsigned long long)f[8]);
        out += buf;
      }
      if ((f[5] & 0xFFFFFFFF) != (f[5] >> 32)) {
        snprintf(buf, sizeof(buf), " head_pend{tok=%llu done=%llu elems=%llu}",
                 (unsigned long long)(f[11] & 0xFFFFFFFF),
                 (unsigned long long)(f[11] >> 32), (unsigned long long)f[12]);
        out += buf;
      }
    }
    return out;
  }

  // --- arena memory ---
  u64 alloc(u64 bytes) { return heap_.alloc(bytes); }
  void free_block(u64 off) { heap_.free_block(off); }
  virtual void write_arena(u64 off, const void* src, u64 bytes) = 0;
  virtual void read_arena(u64 off, void* dst, u64 bytes) = 0;
  // write into a PEER rank's arena (credit returns for stream consumption;
  // xGMI/IPC on GPU, shm on the emulator)
  virtual void write_peer(u32 rank, u64 off, const void* src, u64 bytes) = 0;

  // --- communicators (quiesce, rewrite mirror, bump generation) ---
  // reference: Communicator rank-table write, driver/xrt/src/
  // communicator.cpp:25-52; subgroup creation accl.cpp:955-962.
  u32 add_comm(const std::vector<u32>& members, u32 my_local) {
    quiesce();
    u32 id = u32(__atomic_load_n((u64*)&ring_->ctrl.ncomms, __ATOMIC_RELAXED));
    if (id >= MAX_COMMS) throw accl_error("accl: too many communicators");
    CommView& c = ring_->comm_mirror[id];
    c.id = id;
    c.rank = my_local;
    c.size = u32(members.size());
    for (u32 i = 0; i < members.size(); ++i) c.members[i] = members[i];
    __atomic_store_n((u64*)&ring_->ctrl.ncomms, u64(id + 1), __ATOMIC_RELEASE);
    __atomic_fetch_add((u64*)&ring_->ctrl.comm_gen, 1, __ATOMIC_RELEASE);
    return id;
  }
  void quiesce() {
    if (head_) wait(head_ - 1);
  }

  // liveness/debug view of the control page (reference: exchange-memory
  // debug dumps, accl.cpp:964-1048; heartbeat has no direct analogue — the
  // persistent engine replaces the MicroBlaze's implicit liveness)
  struct CtrlView {
    u64 doorbell, shutdown, comm_gen, ncomms, heartbeat, engine_up;
    u64 submitted, retired;
  };
  CtrlView ctrl_view() {
    CtrlView v{};
    v.doorbell = __atomic_load_n((u64*)&ring_->ctrl.doorbell, __ATOMIC_RELAXED);
    v.shutdown = ring_->ctrl.shutdown;
    v.comm_gen = ring_->ctrl.comm_gen;
    v.ncomms = ring_->ctrl.ncomms;
    v.heartbeat = ring_->ctrl.heartbeat;
    v.engine_up = ring_->ctrl.engine_up;
    v.submitted = head_;
    v.retired = tail_retired();
    return v;
  }
  const CommView& comm_view(u32 id) const { return ring_->comm_mirror[id]; }

 protected:
  u64 tail_retired() {
    std::lock_guard<std::mutex> lk(ring_mu_);
    return tail_retired_locked();
  }
  u64 tail_retired_locked() {
    // ring slots free once their RetEntry is published
    while (retired_ < head_ && test(retired_, nullptr)) retired_++;
    return retired_;
  }

  ProtoConfig cfg_{};
  char* arena_base_ = nullptr;
  RingPage* ring_ = nullptr;   // host-visible (pinned on GPU)
  CallDesc* desc_ring_ = nullptr;   // optional device-resident desc ring
  volatile u64* door_ = nullptr;    //   (host-writable over large BAR)
  HeapAlloc heap_;
  u64 head_ = 0, retired_ = 0;
  std::mutex ring_mu_;
};

}  // namespace accl
