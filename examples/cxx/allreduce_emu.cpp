// Standalone C++ usage of the accl::ACCL facade (no Python): single-rank
// allreduce + combine on the CPU emulator backend. The same code drives the
// GPU backend by constructing GpuDevice instead (see csrc/gpu/gpudevice.hpp).
//
// Build (from the repo root, objects built by `python -m accl_amd.build`;
// the host runtime is plain C++17 — no HIP headers needed off the GPU path):
//   g++ -std=c++17 examples/cxx/allreduce_emu.cpp \
//     accl_amd/.build/core_util.cpp.o accl_amd/.build/core_accl.cpp.o \
//     accl_amd/.build/emu_emudevice.cpp.o -o allreduce_emu -lpthread
#include <cassert>
#include <cstdio>
#include <memory>
#include <vector>

#include "../../accl_amd/csrc/core/accl.hpp"
#include "../../accl_amd/csrc/emu/emudevice.hpp"

int main() {
  using namespace accl;
  auto dev = std::make_unique<EmuDevice>(1, 0, "cxx_example");
  ACCL a(std::move(dev));
  auto blob = a.local_blob();
  a.connect({blob});

  const u64 n = 1 << 16;
  auto src = a.create_buffer(n, DataType::float32);
  auto dst = a.create_buffer(n, DataType::float32);
  float* s = (float*)src->host_ptr();
  for (u64 i = 0; i < n; ++i) s[i] = float(i % 100);

  a.allreduce(*src, *dst, n, ReduceFunction::SUM);  // P=1: identity
  float* d = (float*)dst->host_ptr();
  dst->sync_from_device();
  for (u64 i = 0; i < n; ++i) assert(d[i] == s[i]);

  Request* r = a.combine(n, ReduceFunction::MAX, *src, *dst, *dst);
  assert(r->retcode() == 0 && r->duration_us() >= 0.0);
  std::printf("cxx example OK (engine: %s)\n",
              a.dump_engine_status().c_str());
  return 0;
}
