// Sanitizer harness: 2 forked ranks over shm run the protocol surface
// (allreduce, out-of-order tags, stream_put, barrier) under ASan/UBSan.
// Built and run by scripts/sanitize.sh.
#include <sys/wait.h>
#include <unistd.h>
#include <cassert>
#include <cstdio>
#include <cstring>
#include <memory>
#include <string>
#include <vector>

#include "../../accl_amd/csrc/core/accl.hpp"
#include "../../accl_amd/csrc/emu/emudevice.hpp"

using namespace accl;

static void rank_main(u32 rank, const std::string& job) {
  ProtoConfig cfg = default_proto_config(2, rank);
  cfg.n_slots = 4;
  cfg.slot_bytes = 4096;
  cfg.timeout_us = 20u * 1000 * 1000;
  auto dev = std::make_unique<EmuDevice>(2, rank, job, &cfg, 64u << 20);
  ACCL a(std::move(dev));
  std::vector<char> mine = a.local_blob();
  std::vector<std::vector<char>> blobs(2);
  for (u32 r = 0; r < 2; ++r) {
    std::string nm = "/" + job + "_r" + std::to_string(r);
    blobs[r] = std::vector<char>(nm.begin(), nm.end());
  }
  blobs[rank] = mine;
  a.connect(blobs);

  const u64 n = 5000;
  auto s = a.create_buffer(n, DataType::float32);
  auto d = a.create_buffer(n, DataType::float32);
  float* sp = (float*)s->host_ptr();
  for (u64 i = 0; i < n; ++i) sp[i] = float((i * 7 + rank * 13) % 61) - 30;
  a.allreduce(*s, *d, n, ReduceFunction::SUM);
  float* dp = (float*)d->host_ptr();
  for (u64 i = 0; i < n; ++i) {
    float e = (float((i * 7) % 61) - 30) + (float((i * 7 + 13) % 61) - 30);
    assert(dp[i] == e);
  }

  // out-of-order tags across segmentation
  if (rank == 0) {
    for (u32 tag : {5u, 6u}) {
      for (u64 i = 0; i < n; ++i) sp[i] = float(tag * 1000 + i % 97);
      s->sync_to_device();
      a.send(*s, n, 1, tag);
    }
  } else {
    for (u32 tag : {6u, 5u}) {
      a.recv(*d, n, 0, tag);
      for (u64 i = 0; i < n; ++i) assert(dp[i] == float(tag * 1000 + i % 97));
    }
  }
  a.barrier();

  // stream_put -> pop_stream
  if (rank == 0) {
    a.stream_put(*s, 1024, 1, 3);
  } else {
    std::vector<float> out(1024);
    u32 tag = 0;
    u64 nb = a.pop_stream(0, out.data(), out.size() * 4, &tag);
    assert(nb == 1024 * 4 && tag == 3);
  }
  a.barrier();

  // round-2 surface: direct (rendezvous) collectives + windowed n-ary
  // fan-in under the sanitizers (forced by the tiny max_eager above)
  const u64 m = 6000;  // 24 KB > max_eager
  auto s2 = a.create_buffer(m * 2, DataType::float32);
  auto d2 = a.create_buffer(m, DataType::float32);
  float* s2p = (float*)s2->host_ptr();
  for (u64 i = 0; i < m * 2; ++i)
    s2p[i] = float((i * 3 + rank * 11) % 53) - 26;
  a.reduce_scatter(*s2, *d2, m, ReduceFunction::SUM);
  float* d2p = (float*)d2->host_ptr();
  for (u64 i = 0; i < m; ++i) {
    u64 gi = rank * m + i;
    float e = (float((gi * 3) % 53) - 26) + (float((gi * 3 + 11) % 53) - 26);
    assert(d2p[i] == e);
  }
  a.barrier();

  // parked-call interleaving: rank 1 posts the recv seconds "early" (rank 0
  // delays its send behind other traffic), exercising park/retry + resume
  if (rank == 1) {
    Request* rq = a.recv(*d, n, 0, 77, 0, false, DataType::none, true);
    for (int it = 0; it < 5; ++it) a.allreduce(*s, *d2, 100, ReduceFunction::SUM);
    // the engine must have served the allreduces while the recv was parked
    a.barrier();           // release rank 0's send
    assert(rq->wait() == 0);
    a.free_request(rq);
  } else {
    for (int it = 0; it < 5; ++it) a.allreduce(*s, *d2, 100, ReduceFunction::SUM);
    a.barrier();
    for (u64 i = 0; i < n; ++i) sp[i] = float(i % 31);
    s->sync_to_device();
    a.send(*s, n, 1, 77);
  }
  a.barrier();
  std::printf("rank %u OK\n", rank);
}

int main() {
  std::string job = "asan" + std::to_string(getpid());
  pid_t pid = fork();
  if (pid == 0) {
    rank_main(1, job);
    _exit(0);
  }
  rank_main(0, job);
  int st = 0;
  waitpid(pid, &st, 0);
  assert(WIFEXITED(st) && WEXITSTATUS(st) == 0);
  std::puts("asan multirank OK");
  return 0;
}
