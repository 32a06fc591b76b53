// EmuDevice — the CPU emulator backend: the same collective scheduler
// (common/sched.hpp) running as a host thread per rank, with POSIX-shm
// arenas standing in for peer-mapped HBM.
//
// Analogue of the reference's software CCLO emulator, where the identical
// firmware source runs natively against hlslib FIFOs and a ZMQ "ethernet"
// (reference: test/model/emulator/cclo_emu.cpp:268-506, zmq transport
// test/model/zmq/zmq_server.cpp:107-190). Here the single-source property is
// the same — sched.hpp compiles for CPU and GPU — and the transport is the
// same protocol over shm instead of a different wire.
#pragma once
#include <atomic>
#include <thread>
#include "../common/sched.hpp"
#include "../core/backend.hpp"

namespace accl {

struct CpuMover {
  u32 next_token = 0;
  u32 submit(const MoveDesc& m) {
    execute_move_range(m, 0, m.count);
    fence_release_sys();
    return next_token++;
  }
  bool poll(u32) { return true; }
};

class EmuDevice : public Backend {
 public:
  // job: unique per launch (all ranks agree); e.g. "accl<pid-of-rank0>".
  EmuDevice(u32 nranks, u32 rank, const std::string& job,
            const ProtoConfig* cfg_override = nullptr, u64 heap_bytes = 256u << 20);
  ~EmuDevice() override;

  std::vector<char> local_blob() override;
  void connect(const std::vector<std::vector<char>>& blobs) override;
  void shutdown() override;
  bool is_gpu() const override { return false; }

  void write_arena(u64 off, const void* src, u64 bytes) override;
  void read_arena(u64 off, void* dst, u64 bytes) override;
  void write_peer(u32 rank, u64 off, const void* src, u64 bytes) override;

 private:
  void engine_main();
  std::string shm_name(u32 rank) const;

  std::string job_;
  u64 arena_bytes_ = 0;
  ArenaLayout layout_{};
  int shm_fd_ = -1;
  char* peer_base_[MAX_RANKS] = {};
  std::unique_ptr<RingPage> ring_store_;
  struct Engine;                   // holds Cclo<CpuMover> (large)
  std::unique_ptr<Engine> eng_;
  std::thread thread_;
};

}  // namespace accl
