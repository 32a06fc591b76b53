// accl_amd core types — shared by the host runtime, the CPU emulator engine and
// the GPU persistent-engine kernel.
//
// Modeled on the semantics of the reference ACCL driver/firmware
// (reference: driver/xrt/include/accl/constants.hpp:191-384,
//  kernels/cclo/fw/sw_apps/ccl_offload_control/src/ccl_offload_control.c) but
// re-designed for an MI355X-native engine: the 15-word MicroBlaze call
// descriptor becomes a 64-byte cache-line descriptor consumed by a persistent
// HIP kernel; addresses are raw device pointers (peer windows are IPC-mapped).
//
// This header must compile as plain C++17 *and* as HIP device code.
#pragma once

#include <cstdint>
#include <cstddef>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define ACCL_HD __host__ __device__
#else
#define ACCL_HD
#endif

namespace accl {

using u8 = std::uint8_t;
using u16 = std::uint16_t;
using u32 = std::uint32_t;
using u64 = std::uint64_t;
using i32 = std::int32_t;
using i64 = std::int64_t;

// ---------------------------------------------------------------- operations
// Scenario IDs (reference: constants.hpp:191-210 operation enum).
enum class Op : u32 {
  config = 0,
  copy = 1,
  combine = 2,
  send = 3,
  recv = 4,
  bcast = 5,
  scatter = 6,
  gather = 7,
  reduce = 8,
  allgather = 9,
  allreduce = 10,
  reduce_scatter = 11,
  alltoall = 12,
  barrier = 13,
  stream_put = 14,  // send into the peer's stream ring (reference:
                    // stream_put scenario, accl.hpp:204-238; remote side is
                    // consumed by the application, not a posted recv)
  halt = 254,  // engine shutdown (no reference analogue; replaces kernel exit)
  nop = 255,
};

// config sub-functions (reference: constants.hpp cfgFunc)
enum class CfgFunc : u32 {
  reset = 0,
  enable_pkt = 1,
  set_timeout = 2,
  set_max_eager_size = 3,
  set_max_rendezvous_size = 4,
  set_tuning = 5,
  dump_state = 6,   // write a parked/pending/spill summary into the dbg
                    // region (observability; no reference analogue beyond
                    // the exchange-memory dumps, accl.cpp:964-1048)
};

enum class ReduceFunction : u32 { SUM = 0, MAX = 1 };

// ------------------------------------------------------------------- dtypes
// (reference: constants.hpp:256-264 dataType; bf16 added — MI355X native.)
enum class DataType : u32 {
  none = 0,
  float16 = 1,
  float32 = 2,
  float64 = 3,
  int32 = 4,
  int64 = 5,
  bfloat16 = 6,
  int8 = 7,
};

ACCL_HD constexpr u32 dtype_size(DataType d) {
  switch (d) {
    case DataType::float16: return 2;
    case DataType::bfloat16: return 2;
    case DataType::float32: return 4;
    case DataType::int32: return 4;
    case DataType::float64: return 8;
    case DataType::int64: return 8;
    case DataType::int8: return 1;
    default: return 0;
  }
}

// -------------------------------------------------------------------- flags
// Per-call flags (reference: constants.hpp:279-326 stream/host/compression).
enum CallFlags : u32 {
  F_NONE = 0,
  F_COMPRESS_OP0 = 1u << 0,   // convert op0 dtype->wire dtype before send
  F_COMPRESS_OP1 = 1u << 1,
  F_COMPRESS_RES = 1u << 2,
  F_ASYNC = 1u << 3,
  F_SRC_ARENA = 1u << 4,      // addresses are arena offsets (rendezvous-capable)
  F_DST_ARENA = 1u << 5,
  F_OP1_ARENA = 1u << 6,
  F_SRC_STREAM = 1u << 7,     // addr0 = stream lane id: the engine consumes
                              // segments from its own stream ring (reference:
                              // OP0_STREAM send/reduce-from-krnl-stream,
                              // dma_mover.cpp:497)
  F_DST_PEER = 1u << 8,       // copy: addr2 is an arena OFFSET in rank
                              // root_src_dst's arena — one-sided put over
                              // xGMI (reference: copy into a p2p buffer,
                              // test_copy_p2p; user synchronizes, e.g.
                              // barrier)
};

// -------------------------------------------------------------------- errors
// Error bitmask accumulated through the collective (reference:
// constants.hpp:355-384, 27 error codes; ours is condensed but same idea:
// every failure is a bit, bits OR together, 0 == success).
enum ErrorCode : u32 {
  E_OK = 0,
  E_TIMEOUT = 1u << 0,            // spin bound exceeded waiting on a peer
  E_MATCH = 1u << 1,              // rx matching inconsistency
  E_SEGMENT = 1u << 2,            // segmentation/size error
  E_COMPRESSION = 1u << 3,        // unsupported dtype conversion
  E_ARITH = 1u << 4,              // unsupported reduce function/dtype
  E_INVALID_OP = 1u << 5,         // unknown scenario
  E_ENGINE_DOWN = 1u << 6,        // engine halted / not initialized
  E_RNDZV = 1u << 7,              // rendezvous protocol failure
  E_CREDIT = 1u << 8,             // eager flow-control failure
  E_TRANSPORT = 1u << 9,          // socket/IPC failure
  E_INVALID_ARG = 1u << 10,       // bad count/root/addr
  E_INFLIGHT_OVERFLOW = 1u << 11, // too many outstanding ops
  E_COMM = 1u << 12,              // bad communicator id / membership
  E_NOT_READY = 1u << 31,         // INTERNAL: op parked before any progress
                                  // (the reference's NOT_READY_ERROR retry
                                  // path, ccl_offload_control.c:2460-2478);
                                  // never surfaces to the host
};

// --------------------------------------------------------------- descriptor
// The call descriptor: one 64-B cache line, the analogue of the reference's
// 15-word CMD_CALL bundle (ccl_offload_control.c:2317-2336). Host writes it
// into a ring; engine (CPU thread or GPU control workgroup) pops and runs it.
struct alignas(64) CallDesc {
  u32 scenario;    // Op
  u32 count_lo;    // element count (low 32)
  u32 count_hi;    //               (high 32)
  u32 comm_id;     // index into the engine's communicator table
  u32 root_src_dst;// root rank (collectives) or peer (send/recv), local index
  u32 function;    // ReduceFunction / CfgFunc
  u32 tag;         // user tag (send/recv) or internal tag
  u32 arith;       // packed dtypes: [7:0]=op dtype, [15:8]=wire dtype
  u64 addr0;       // src operand (raw pointer / arena offset per flags)
  u64 addr1;       // second operand (combine) / scratch
  u64 addr2;       // destination
  u32 flags;       // CallFlags
  u32 seq;         // call sequence number (matches RetEntry.seq)
};
static_assert(sizeof(CallDesc) == 64, "CallDesc must be one cache line");

ACCL_HD inline u64 desc_count(const CallDesc& d) {
  return (u64(d.count_hi) << 32) | d.count_lo;
}
ACCL_HD inline DataType desc_dtype(const CallDesc& d) {
  return DataType(d.arith & 0xFF);
}
ACCL_HD inline DataType desc_wire_dtype(const CallDesc& d) {
  return DataType((d.arith >> 8) & 0xFF);
}

// Completion record (analogue of RETVAL_OFFSET + perf counter readback,
// ccl_offload_control.c:2291-2306).
struct alignas(32) RetEntry {
  u32 seq;        // published LAST (release) — nonzero means valid for seq
  u32 errcode;    // ErrorCode bitmask
  u64 t_start;    // engine clock ticks at call start
  u64 t_end;      // engine clock ticks at call end
  u64 _pad;
};
static_assert(sizeof(RetEntry) == 32, "");

// ------------------------------------------------------------- communicator
// Flat communicator record shared host/device (reference: rank table in
// exchange memory, communicator.cpp:25-52). Members are *global* rank ids
// (indices into the engine's channel table).
constexpr int MAX_RANKS = 64;
constexpr int MAX_COMMS = 16;

struct CommView {
  u32 id;
  u32 rank;              // my local index within this communicator
  u32 size;
  u32 _pad;
  u32 members[MAX_RANKS];// local index -> global rank
  ACCL_HD u32 global(u32 local) const { return members[local]; }
};

// ------------------------------------------------------------ engine limits
constexpr int MAX_INFLIGHT = 32;    // outstanding nonblocking ops per engine
constexpr u32 TAG_ANY = 0xFFFFFFFFu;

// Internal tag namespace for collective-internal messages: collectives stamp
// tag = TAG_COLL | (comm_id << 20) | (opseq & 0xFFFFF). User tags must be
// < 0x40000000 (enforced at the API).
constexpr u32 TAG_COLL = 0x40000000u;
constexpr u32 MAX_USER_TAG = 0x3FFFFFFFu;

}  // namespace accl
