// GpuDevice — host-side backend owning one MI355X GPU: fine-grained HBM
// arena (IPC-shared with peer ranks over xGMI), pinned host descriptor ring,
// and the persistent engine kernel.
//
// Analogue of the reference's XRTDevice + accl_network_utils bring-up
// (reference: driver/xrt/src/xrtdevice.cpp:36-303;
// driver/utils/accl_network_utils/ — POE configuration becomes IPC handle
// exchange + peer mapping).
#pragma once
#include "../core/backend.hpp"

namespace accl {

// plugin launchers (gpu/plugins.hip)
void launch_vadd_put(const void* in, u64 count, u32 tag, void* my_arena,
                     void* peer_arena, u32 me, u32 peer, u32 seg_bytes,
                     float addv, void* stream);
void launch_stream_drain(void* out, u64 max_elems, u32 nseg, void* my_arena,
                         void* peer_arena, u32 me, u32 peer, u64 start_seq,
                         void* stream);
void launch_vadd_devicecall(const void* in, void* scratch, u64 scratch_off,
                            u64 count, u32 tag, void* my_arena, u32 dst_rank,
                            float addv, void* stream);

class GpuDevice : public Backend {
 public:
  GpuDevice(u32 nranks, u32 rank, int device_index,
            const ProtoConfig* cfg_override = nullptr,
            u64 heap_bytes = 8ull << 30, int engine_wgs = 0);
  ~GpuDevice() override;

  std::vector<char> local_blob() override;
  void connect(const std::vector<std::vector<char>>& blobs) override;
  void shutdown() override;
  bool is_gpu() const override { return true; }

  void write_arena(u64 off, const void* src, u64 bytes) override;
  void read_arena(u64 off, void* dst, u64 bytes) override;
  void write_peer(u32 rank, u64 off, const void* src, u64 bytes) override;

  int device_index() const { return dev_; }
  u64 arena_bytes() const { return arena_bytes_; }
  // debug: per-wave executed-tile counters from the engine state
  std::vector<u32> debug_wave_tiles();
  std::vector<u64> debug_timeline();
  char* peer_base(u32 r) const { return r < MAX_RANKS ? peer_base_[r] : nullptr; }
  void* op_stream() const { return stream_; }  // for plugin launches

 private:
  int dev_ = 0;
  int engine_wgs_ = 640;  // 256-thr WGs, ~2.5 per CU (mover VGPR use caps
                          // occupancy at 3 WGs/CU; wave slots and >70
                          // VGPRs/SIMD stay free for co-resident kernels)
  u64 arena_bytes_ = 0;
  ArenaLayout layout_{};
  bool fine_grained_ = true;
  void* mover_stream_ = nullptr;
  char* peer_base_[MAX_RANKS] = {};
  void* ring_pinned_ = nullptr;     // RingPage, hipHostMalloc
  void* state_dev_ = nullptr;       // GpuEngineState, hipMalloc
  void* stream_ = nullptr;          // hipStream_t
  bool launched_ = false;
};

}  // namespace accl
