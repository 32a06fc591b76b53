// Example "plugin" kernels — device-initiated collectives built on
// device_api.hpp. The canonical demo mirrors the reference's vadd_put
// (kernels/plugins/vadd_put/vadd_put.cpp:25-87): a compute kernel reads a
// float vector, adds a constant, and STREAMS the result to a peer rank from
// inside the kernel — no host involvement per segment. The remote side pops
// it with ACCL::pop_stream (host) or device_api::stream_pop (kernel).
#include <hip/hip_runtime.h>
#include "device_api.hpp"

namespace accl {

// single-wave demo kernel: stage each segment in LDS, push over xGMI
__global__ void __launch_bounds__(64, 1) vadd_put_kernel(
    const float* in, u64 count, u32 tag, char* my_arena, char* peer_arena,
    u32 me, u32 peer, u32 seg_bytes, float addv) {
  __shared__ float stage[8192];  // 32 KiB staging
  auto chan = device_api::stream_chan(my_arena, peer_arena, me, peer);
  const int lane = int(threadIdx.x) & 63;
  u32 seg_elems = seg_bytes / 4;
  if (seg_elems > 8192) seg_elems = 8192;
  for (u64 off = 0; off < count; off += seg_elems) {
    u32 n = u32(count - off < seg_elems ? count - off : seg_elems);
    for (u32 i = lane; i < n; i += 64) stage[i] = in[off + i] + addv;
    __syncthreads();  // LDS visibility within the wave/WG
    device_api::stream_push(chan, stage, n * 4, tag);
    __syncthreads();
  }
}

// consumer demo: pop `nseg` segments into out (contiguous), device-side
__global__ void __launch_bounds__(64, 1) stream_drain_kernel(
    float* out, u64 max_elems, u32 nseg, char* my_arena, char* peer_arena,
    u32 me, u32 peer, u64 start_seq) {
  auto rx = device_api::stream_rx(my_arena, peer_arena, me, peer);
  u64 done = 0;
  for (u32 s = 0; s < nseg; ++s) {
    u32 nb = device_api::stream_pop(
        rx, start_seq + s, out + done, u32((max_elems - done) * 4), nullptr);
    done += nb / 4;
  }
}

// Full reference-flow demo: the kernel computes into an ARENA buffer and
// then issues the engine's own stream_put through the device-call ring —
// kernel -> device_call -> scheduler -> movers (vadd_put.cpp's
// ACCLCommand::stream_put flow, arbitrated with host calls).
__global__ void __launch_bounds__(64, 1) vadd_devicecall_kernel(
    const float* in, float* scratch_arena_buf, u64 scratch_off, u64 count,
    u32 tag, char* my_arena, u32 dst_rank, float addv) {
  const int lane = int(threadIdx.x) & 63;
  for (u64 i = lane; i < count; i += 64)
    scratch_arena_buf[i] = in[i] + addv;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (lane == 0) {
    CallDesc d{};
    d.scenario = u32(Op::stream_put);
    d.count_lo = u32(count & 0xFFFFFFFFu);
    d.count_hi = u32(count >> 32);
    d.comm_id = 0;
    d.root_src_dst = dst_rank;
    d.tag = tag;
    d.arith = u32(DataType::float32) | (u32(DataType::float32) << 8);
    d.addr0 = scratch_off;
    d.flags = F_SRC_ARENA;
    u64 tok = device_api::device_call(my_arena, d);
    u32 e = device_api::device_call_wait(my_arena, tok);
    (void)e;
  }
}

void launch_vadd_devicecall(const void* in, void* scratch, u64 scratch_off,
                            u64 count, u32 tag, void* my_arena, u32 dst_rank,
                            float addv, void* stream) {
  hipLaunchKernelGGL(vadd_devicecall_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const float*)in, (float*)scratch,
                     scratch_off, count, tag, (char*)my_arena, dst_rank, addv);
}

void launch_vadd_put(const void* in, u64 count, u32 tag, void* my_arena,
                     void* peer_arena, u32 me, u32 peer, u32 seg_bytes,
                     float addv, void* stream) {
  hipLaunchKernelGGL(vadd_put_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const float*)in, count, tag,
                     (char*)my_arena, (char*)peer_arena, me, peer, seg_bytes,
                     addv);
}

void launch_stream_drain(void* out, u64 max_elems, u32 nseg, void* my_arena,
                         void* peer_arena, u32 me, u32 peer, u64 start_seq,
                         void* stream) {
  hipLaunchKernelGGL(stream_drain_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (float*)out, max_elems, nseg,
                     (char*)my_arena, (char*)peer_arena, me, peer, start_seq);
}

}  // namespace accl
