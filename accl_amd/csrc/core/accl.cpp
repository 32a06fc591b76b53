#include "accl.hpp"
#include <algorithm>
#include <cstdlib>
#include <cstdio>
#include <cstring>
#include <string>

namespace accl {

// ------------------------------------------------------------- BaseBuffer
BaseBuffer::BaseBuffer(ACCL* owner, u64 arena_off, u64 count, DataType dt,
                       bool own_host, void* host_ptr, u64 root_off)
    : owner_(owner), off_(arena_off), count_(count), dt_(dt),
      own_host_(own_host), host_(host_ptr), root_off_(root_off),
      own_arena_(root_off != 0) {}

BaseBuffer::~BaseBuffer() {
  if (own_arena_ && owner_) owner_->backend()->free_block(root_off_);
  if (own_host_ && host_) std::free(host_);
}

void BaseBuffer::sync_to_device() {
  if (host_) owner_->backend()->write_arena(off_, host_, bytes());
}
void BaseBuffer::sync_from_device() {
  if (host_) owner_->backend()->read_arena(off_, host_, bytes());
}
void* BaseBuffer::device_ptr() const {
  return owner_->backend()->arena_local() + off_;
}
std::unique_ptr<BaseBuffer> BaseBuffer::slice(u64 start, u64 end) {
  if (end > count_ || start > end) throw accl_error("buffer slice out of range");
  u64 esz = dtype_size(dt_);
  void* h = host_ ? (char*)host_ + start * esz : nullptr;
  return std::unique_ptr<BaseBuffer>(new BaseBuffer(
      owner_, off_ + start * esz, end - start, dt_, false, h, 0));
}

// ---------------------------------------------------------------- Request
u32 Request::wait(u64 timeout_ms) {
  if (!done_) {
    be_->wait(seq_, &ret_, timeout_ms);
    done_ = true;
  }
  return ret_.errcode;
}
bool Request::test() {
  if (done_) return true;
  done_ = be_->test(seq_, &ret_);
  return done_;
}
u32 Request::retcode() { return done_ ? ret_.errcode : 0; }
double Request::duration_us() {
  wait();
  return double(ret_.t_end - ret_.t_start) / TICKS_PER_US;
}

// ------------------------------------------------------------------- ACCL
ACCL::ACCL(std::unique_ptr<Backend> backend) : be_(std::move(backend)) {
  comm_sizes_.push_back(be_->cfg().nranks);
  comm_ranks_.push_back(be_->cfg().rank);
}

ACCL::~ACCL() {
  for (Request* r : reqs_) delete r;
  deinit();
}

void ACCL::deinit() {
  if (be_) be_->shutdown();
}

std::unique_ptr<BaseBuffer> ACCL::create_buffer(u64 count, DataType dt) {
  u64 bytes = count * dtype_size(dt);
  u64 off = be_->alloc(bytes);
  void* host = std::malloc(bytes ? bytes : 1);
  return std::unique_ptr<BaseBuffer>(
      new BaseBuffer(this, off, count, dt, true, host, off));
}
std::unique_ptr<BaseBuffer> ACCL::create_buffer(void* host, u64 count, DataType dt) {
  u64 off = be_->alloc(count * dtype_size(dt));
  return std::unique_ptr<BaseBuffer>(
      new BaseBuffer(this, off, count, dt, false, host, off));
}
std::unique_ptr<BaseBuffer> ACCL::create_buffer_device(u64 count, DataType dt) {
  u64 off = be_->alloc(count * dtype_size(dt));
  return std::unique_ptr<BaseBuffer>(
      new BaseBuffer(this, off, count, dt, false, nullptr, off));
}

u32 ACCL::create_communicator(const std::vector<u32>& global_ranks, u32 my_local) {
  u32 id = be_->add_comm(global_ranks, my_local);
  comm_sizes_.resize(id + 1);
  comm_ranks_.resize(id + 1);
  comm_sizes_[id] = u32(global_ranks.size());
  comm_ranks_[id] = my_local;
  return id;
}
u32 ACCL::split_communicator(const std::vector<u32>& global_ranks) {
  auto it = std::find(global_ranks.begin(), global_ranks.end(), rank());
  if (it == global_ranks.end())
    throw accl_error("split_communicator: caller not in group");
  return create_communicator(global_ranks, u32(it - global_ranks.begin()));
}

CallDesc ACCL::make_desc(Op op, u64 count, DataType dt, DataType wire) {
  CallDesc d{};
  d.scenario = u32(op);
  d.count_lo = u32(count & 0xFFFFFFFFu);
  d.count_hi = u32(count >> 32);
  if (wire == DataType::none) wire = dt;
  d.arith = u32(dt) | (u32(wire) << 8);
  return d;
}

static const char* op_name(u32 sc) {
  switch (Op(sc)) {
    case Op::copy: return "copy";
    case Op::combine: return "combine";
    case Op::send: return "send";
    case Op::recv: return "recv";
    case Op::bcast: return "bcast";
    case Op::scatter: return "scatter";
    case Op::gather: return "gather";
    case Op::allgather: return "allgather";
    case Op::reduce: return "reduce";
    case Op::allreduce: return "allreduce";
    case Op::reduce_scatter: return "reduce_scatter";
    case Op::alltoall: return "alltoall";
    case Op::barrier: return "barrier";
    case Op::stream_put: return "stream_put";
    case Op::config: return "config";
    default: return "op";
  }
}

Request* ACCL::finish(CallDesc d, bool run_async, BaseBuffer* sync_out,
                      u64 out_count, BaseBuffer* sync_in0, u64 in0_count,
                      BaseBuffer* sync_in1, u64 in1_count) {
  if (sync_in0 && sync_in0->host_ptr())
    be_->write_arena(sync_in0->arena_offset(), sync_in0->host_ptr(),
                     in0_count * dtype_size(sync_in0->dtype()));
  if (sync_in1 && sync_in1->host_ptr())
    be_->write_arena(sync_in1->arena_offset(), sync_in1->host_ptr(),
                     in1_count * dtype_size(sync_in1->dtype()));
  u64 seq = be_->submit(d);
  if (debug_enabled())
    debug_log(std::string(op_name(d.scenario)) + " seq=" +
              std::to_string(seq) + " count=" +
              std::to_string((u64(d.count_hi) << 32) | d.count_lo) +
              " tag=" + std::to_string(d.tag) +
              (run_async ? " async" : ""));
  Request* r = new Request(be_.get(), seq);
  reqs_.push_back(r);
  if (!run_async) {
    u32 e = r->wait();
    if (e) {
      debug_log(std::string(op_name(d.scenario)) + " seq=" +
                std::to_string(seq) + " FAILED: " + error_to_string(e));
      std::string msg = "accl op failed: " + error_to_string(e);
      if (e & E_TIMEOUT) msg += "\n" + be_->timeout_dump_str();
      throw accl_error(msg, e);
    }
    if (sync_out && sync_out->host_ptr())
      be_->read_arena(sync_out->arena_offset(), sync_out->host_ptr(),
                      out_count * dtype_size(sync_out->dtype()));
  }
  return r;
}

void ACCL::free_request(Request* r) {
  auto it = std::find(reqs_.begin(), reqs_.end(), r);
  if (it != reqs_.end()) {
    reqs_.erase(it);
    delete r;
  }
}

// --------------------------------------------------------------- op calls
Request* ACCL::copy(BaseBuffer& src, BaseBuffer& dst, u64 count,
                    bool from_device, bool to_device, bool run_async) {
  CallDesc d = make_desc(Op::copy, count, src.dtype(), src.dtype());
  d.addr0 = src.arena_offset();
  d.addr2 = dst.arena_offset();
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  return finish(d, run_async, to_device ? nullptr : &dst, count,
                from_device ? nullptr : &src, count);
}

Request* ACCL::put(BaseBuffer& src, u64 count, u32 dst_rank,
                   u64 peer_arena_offset, bool from_device, bool run_async) {
  // one-sided xGMI write into a peer-resident buffer (reference:
  // test_copy_p2p — CCLO copy into a p2p bo); synchronize with barrier()
  CallDesc d = make_desc(Op::copy, count, src.dtype(), src.dtype());
  d.addr0 = src.arena_offset();
  d.addr2 = peer_arena_offset;
  d.root_src_dst = dst_rank;
  d.flags = F_SRC_ARENA | F_DST_PEER;
  return finish(d, run_async, nullptr, 0, from_device ? nullptr : &src, count);
}

Request* ACCL::combine(u64 count, ReduceFunction f, BaseBuffer& op0,
                       BaseBuffer& op1, BaseBuffer& res, bool from_device,
                       bool to_device, bool run_async) {
  CallDesc d = make_desc(Op::combine, count, op0.dtype(), op0.dtype());
  d.addr0 = op0.arena_offset();
  d.addr1 = op1.arena_offset();
  d.addr2 = res.arena_offset();
  d.function = u32(f);
  d.flags = F_SRC_ARENA | F_DST_ARENA | F_OP1_ARENA;
  return finish(d, run_async, to_device ? nullptr : &res, count,
                from_device ? nullptr : &op0, count,
                from_device ? nullptr : &op1, count);
}

Request* ACCL::send(BaseBuffer& src, u64 count, u32 dst, u32 tag, u32 comm,
                    bool from_device, DataType compress, bool run_async) {
  if (tag != TAG_ANY && tag > MAX_USER_TAG)
    throw accl_error("send: tag out of range");
  CallDesc d = make_desc(Op::send, count, src.dtype(), compress);
  d.addr0 = src.arena_offset();
  d.root_src_dst = dst;
  d.tag = tag;
  d.comm_id = comm;
  d.flags = F_SRC_ARENA;
  return finish(d, run_async, nullptr, 0, from_device ? nullptr : &src, count);
}

Request* ACCL::stream_put(BaseBuffer& src, u64 count, u32 dst, u32 tag,
                          u32 comm, bool from_device, DataType compress,
                          bool run_async) {
  if (tag > MAX_USER_TAG) throw accl_error("stream_put: tag out of range");
  CallDesc d = make_desc(Op::stream_put, count, src.dtype(), compress);
  d.addr0 = src.arena_offset();
  d.root_src_dst = dst;
  d.tag = tag;
  d.comm_id = comm;
  d.flags = F_SRC_ARENA;
  return finish(d, run_async, nullptr, 0, from_device ? nullptr : &src, count);
}

// ---- host-side stream consumption (the application end of the ring) ----
namespace {
struct StreamLane {
  u64 ctl_off, hdr_off, payload_off;
};
StreamLane stream_lane(const ProtoConfig& c, u32 src_lane) {
  ArenaLayout L = arena_layout(c);
  u64 lane_bytes = sizeof(EagerChanCtl) +
                   u64(c.n_stream) * sizeof(SlotHdr) +
                   u64(c.n_stream) * c.stream_bytes;
  StreamLane s{};
  s.ctl_off = L.stream_off + u64(src_lane) * lane_bytes;
  s.hdr_off = s.ctl_off + sizeof(EagerChanCtl);
  s.payload_off = s.hdr_off + u64(c.n_stream) * sizeof(SlotHdr);
  return s;
}
}  // namespace

u64 ACCL::push_stream(u32 dst, const void* data, u64 bytes, u32 tag,
                      u64 timeout_ms) {
  // Host-side twin of device_api::stream_push — the CCLO_BFM analogue
  // (reference test/model/bfm/cclo_bfm.cpp:28-178): lets user "kernel"
  // logic co-simulate its stream-producer role against the emulator
  // without a GPU, interoperating with engine ops and real device kernels
  // through the shared tx_ctr allocator and credit words.
  const ProtoConfig& c = be_->cfg();
  if (dst >= c.nranks) throw accl_error("push_stream: bad dst");
  if (be_->is_gpu())
    throw accl_error(
        "push_stream is the emulator-side BFM producer; on the GPU use "
        "device_api::stream_push from a kernel (or ACCL::stream_put)");
  char* my = be_->arena_local();
  // channel (me -> dst): ctl (credit + tx_ctr) lives in MY arena lane [dst]
  StreamLane mine = stream_lane(c, dst);
  auto* ctl = (EagerChanCtl*)(my + mine.ctl_off);
  if (bytes > c.stream_bytes)
    throw accl_error("push_stream: segment larger than stream slot");
  u64 seq = __atomic_fetch_add(&ctl->tx_ctr, 1ull, __ATOMIC_RELAXED) + 1;
  u64 t0 = wallclock_host_ns();
  while (__atomic_load_n(&ctl->credit, __ATOMIC_ACQUIRE) + c.n_stream < seq) {
    if (wallclock_host_ns() - t0 > timeout_ms * 1000000ull)
      throw accl_error("push_stream: no stream credit (consumer stalled)");
    cpu_pause();
  }
  // payload + header into DST's arena lane [me]
  StreamLane theirs = stream_lane(c, c.rank);
  u32 slot = u32((seq - 1) % c.n_stream);
  be_->write_peer(dst, theirs.payload_off + u64(slot) * c.stream_bytes, data,
                  bytes);
  SlotHdr h{};
  h.tag = tag;
  h.bytes = u32(bytes);
  h.msg_count = bytes;
  h.arith = 0;
  h.flags = SEG_FIRST | SEG_LAST;
  h.seq = 0;  // published separately, last
  u64 hdr_off = theirs.hdr_off + u64(slot) * sizeof(SlotHdr);
  be_->write_peer(dst, hdr_off, &h, sizeof(h));
  __atomic_thread_fence(__ATOMIC_RELEASE);
  be_->write_peer(dst, hdr_off + offsetof(SlotHdr, seq), &seq, sizeof(seq));
  return seq;
}

bool ACCL::stream_ready(u32 src) {
  const ProtoConfig& c = be_->cfg();
  StreamLane L = stream_lane(c, src);
  u64 seq = stream_rx_seq_[src] + 1;
  u32 slot = u32((seq - 1) % c.n_stream);
  SlotHdr h{};
  be_->read_arena(L.hdr_off + slot * sizeof(SlotHdr), &h, sizeof(h));
  return h.seq == seq;
}

u64 ACCL::pop_stream(u32 src, void* out, u64 max_bytes, u32* tag,
                     u64 timeout_ms) {
  const ProtoConfig& c = be_->cfg();
  if (src >= c.nranks) throw accl_error("pop_stream: bad src");
  StreamLane L = stream_lane(c, src);
  u64 seq = stream_rx_seq_[src] + 1;
  u32 slot = u32((seq - 1) % c.n_stream);
  SlotHdr h{};
  u64 t0 = wallclock_host_ns();
  for (;;) {
    be_->read_arena(L.hdr_off + slot * sizeof(SlotHdr), &h, sizeof(h));
    if (h.seq == seq) break;
    if (wallclock_host_ns() - t0 > timeout_ms * 1000000ull) return 0;
    cpu_pause();
  }
  u64 n = h.bytes;
  if (n > max_bytes)
    throw accl_error("pop_stream: segment larger than out buffer");
  be_->read_arena(L.payload_off + u64(slot) * c.stream_bytes, out, n);
  if (tag) *tag = h.tag;
  stream_rx_seq_[src] = seq;
  // return credit: the word lives in the SENDER's arena, lane [me]
  StreamLane mine = stream_lane(c, c.rank);
  be_->write_peer(src, mine.ctl_off + offsetof(EagerChanCtl, credit), &seq,
                  sizeof(seq));
  return n;
}

void ACCL::set_timeout_ms(u64 ms) {
  CallDesc d = make_desc(Op::config, ms, DataType::none, DataType::none);
  d.function = u32(CfgFunc::set_timeout);
  be_->call(d);
}
void ACCL::set_max_eager_size(u64 bytes) {
  // protocol-selection config: a PARKED call that chose its eager/rndzv
  // path under the old threshold must not be re-interpreted under the new
  // one (its ParkState encoding is path-specific) — drain the queue first
  be_->quiesce();
  CallDesc d = make_desc(Op::config, bytes, DataType::none, DataType::none);
  d.function = u32(CfgFunc::set_max_eager_size);
  be_->call(d);
}
void ACCL::set_tuning(u32 knob, u64 value) {
  CallDesc d = make_desc(Op::config, value, DataType::none, DataType::none);
  d.function = u32(CfgFunc::set_tuning);
  d.root_src_dst = knob;
  be_->call(d);
}
void ACCL::set_max_rendezvous_size(u64 bytes) {
  be_->quiesce();  // same rule as set_max_eager_size (window sizing)
  CallDesc d = make_desc(Op::config, bytes, DataType::none, DataType::none);
  d.function = u32(CfgFunc::set_max_rendezvous_size);
  be_->call(d);
}

Request* ACCL::copy_from_stream(u32 lane, BaseBuffer& dst, u64 count,
                                bool to_device, bool run_async) {
  CallDesc d = make_desc(Op::copy, count, dst.dtype(), DataType::none);
  d.addr0 = lane;
  d.addr2 = dst.arena_offset();
  d.flags = F_SRC_STREAM | F_DST_ARENA;
  return finish(d, run_async, to_device ? nullptr : &dst, count);
}

Request* ACCL::send_from_stream(u32 lane, u64 count, u32 dst, u32 tag,
                                u32 comm, DataType compress, bool run_async) {
  CallDesc d = make_desc(Op::send, count, DataType::float32, compress);
  d.addr0 = lane;
  d.root_src_dst = dst;
  d.tag = tag;
  d.comm_id = comm;
  d.flags = F_SRC_STREAM;
  return finish(d, run_async, nullptr, 0);
}

void ACCL::soft_reset() {
  CallDesc d = make_desc(Op::config, 0, DataType::none, DataType::none);
  d.function = u32(CfgFunc::reset);
  be_->call(d);
}

bool ACCL::alive() {
  auto v = be_->ctrl_view();
  return v.engine_up != 0;
}

// ---------------- debug dumps ----------------
// reference: dump_communicator (accl.cpp:1429-1439), dump_rx_buffers
// (accl.cpp:964-1048: status/occupancy/tag/seqn per rx buffer), here read
// from the live arena (slot headers + credit words are the ground truth).
std::string ACCL::dump_communicator(u32 comm) {
  const CommView& c = be_->comm_view(comm);
  std::string out = "communicator " + std::to_string(comm) + ": rank " +
                    std::to_string(c.rank) + " of " + std::to_string(c.size) +
                    ", members [";
  for (u32 i = 0; i < c.size; ++i)
    out += (i ? " " : "") + std::to_string(c.members[i]);
  out += "]\n";
  return out;
}

std::string ACCL::dump_engine_status() {
  auto v = be_->ctrl_view();
  char buf[256];
  std::snprintf(buf, sizeof(buf),
                "engine: up=%llu doorbell=%llu submitted=%llu retired=%llu "
                "heartbeat=%llu ncomms=%llu\n",
                (unsigned long long)v.engine_up,
                (unsigned long long)v.doorbell,
                (unsigned long long)v.submitted,
                (unsigned long long)v.retired,
                (unsigned long long)v.heartbeat,
                (unsigned long long)v.ncomms);
  std::string out = buf;
  // live engine-internal state (parked/pending/spill): ask the engine to
  // snapshot into the dbg region, then decode
  if (v.engine_up) {
    // bounded: a WEDGED engine (the main diagnostic case) must not make
    // the dump itself block — fall back to the basic line after 2 s
    CallDesc d = make_desc(Op::config, 0, DataType::none, DataType::none);
    d.function = u32(CfgFunc::dump_state);
    try {
      u64 seq = be_->submit(d);
      be_->wait(seq, nullptr, 2000);
    } catch (const accl_error&) {
      return out + "(engine busy/wedged: internal-state snapshot skipped)\n";
    }
    ArenaLayout L = arena_layout(be_->cfg());
    std::vector<u64> w(68);
    be_->read_arena(L.dbg_off + 4096, w.data(), w.size() * sizeof(u64));
    std::snprintf(buf, sizeof(buf),
                  "parked=%llu pending_addr=%llu pending_done=%llu "
                  "unexpected=%llu spill_busy=0x%llx\n",
                  (unsigned long long)w[1],
                  (unsigned long long)(w[66] & 0xFFFF),
                  (unsigned long long)((w[66] >> 16) & 0xFFFF),
                  (unsigned long long)(w[66] >> 32),
                  (unsigned long long)w[67]);
    out += buf;
    for (u32 k = 2; k < 2 + 2 * 32 && w[1]; k += 2) {
      if (!w[k]) break;
      std::snprintf(buf, sizeof(buf),
                    "  parked[%u]: op=%llu peer=%llu tag=0x%llx step=%llu "
                    "ring_idx=%llu progress=%llu\n",
                    (k - 2) / 2, (unsigned long long)(w[k] & 0xFF),
                    (unsigned long long)((w[k] >> 8) & 0xFF),
                    (unsigned long long)((w[k] >> 16) & 0xFFFFFFFF),
                    (unsigned long long)(w[k] >> 48),
                    (unsigned long long)(w[k + 1] & 0xFFFFFFFF),
                    (unsigned long long)(w[k + 1] >> 32));
      out += buf;
    }
  }
  return out;
}

std::string ACCL::dump_rendezvous() {
  // Host-readable snapshot of MY arena's rendezvous surface: inbound addr
  // records (windows receivers posted to me as a sender), inbound done
  // records, and the flow-control credits my own posts run against.
  // (reference analogue: exchange-memory debug dumps, accl.cpp:964-1048)
  const ProtoConfig& c = be_->cfg();
  ArenaLayout L = arena_layout(c);
  std::string out = "rendezvous rings (" + std::to_string(c.n_rndzv) +
                    " records/pair, prog pool " + std::to_string(N_PROG) +
                    "):\n";
  char b[200];
  for (u32 p = 0; p < c.nranks; ++p) {
    u64 a_new = 0, d_new = 0;
    std::string recs;
    for (u32 i = 0; i < c.n_rndzv; ++i) {
      RndzvRec r{};
      be_->read_arena(L.rndzv_addr_off +
                          (u64(p) * c.n_rndzv + i) * sizeof(RndzvRec),
                      &r, sizeof(r));
      if (r.seq > a_new) a_new = r.seq;
      if (r.seq) {
        std::snprintf(b, sizeof(b),
                      "    addr[%u]: seq=%llu tag=0x%x count=%llu "
                      "prog_idx=%llu\n",
                      i, (unsigned long long)r.seq, r.tag,
                      (unsigned long long)r.count,
                      (unsigned long long)r.prog_idx);
        recs += b;
      }
      RndzvRec d{};
      be_->read_arena(L.rndzv_done_off +
                          (u64(p) * c.n_rndzv + i) * sizeof(RndzvRec),
                      &d, sizeof(d));
      if (d.seq > d_new) d_new = d.seq;
    }
    EagerChanCtl ctl{};
    u64 lane_bytes = sizeof(EagerChanCtl) + u64(c.n_slots) * sizeof(SlotHdr);
    be_->read_arena(L.eager_off + u64(p) * lane_bytes, &ctl, sizeof(ctl));
    std::snprintf(b, sizeof(b),
                  "  pair %u: newest_inbound_addr_seq=%llu "
                  "newest_inbound_done_seq=%llu my_posts_consumed: "
                  "addr=%llu done=%llu\n",
                  p, (unsigned long long)a_new, (unsigned long long)d_new,
                  (unsigned long long)ctl.addr_ret,
                  (unsigned long long)ctl.done_ret);
    out += b;
    out += recs;
  }
  return out;
}

std::string ACCL::dump_eager_rx_buffers(bool verbose) {
  const ProtoConfig& c = be_->cfg();
  ArenaLayout L = arena_layout(c);
  std::string out = "eager rx channels (" + std::to_string(c.n_slots) +
                    " slots x " + std::to_string(c.slot_bytes) + " B):\n";
  u64 lane_bytes = sizeof(EagerChanCtl) + u64(c.n_slots) * sizeof(SlotHdr);
  for (u32 s = 0; s < c.nranks; ++s) {
    u64 ctl_off = L.eager_off + u64(s) * lane_bytes;
    EagerChanCtl ctl{};
    be_->read_arena(ctl_off, &ctl, sizeof(ctl));
    u64 newest = 0, occupied = 0;
    std::string slots;
    for (u32 i = 0; i < c.n_slots; ++i) {
      SlotHdr h{};
      be_->read_arena(ctl_off + sizeof(EagerChanCtl) + i * sizeof(SlotHdr),
                      &h, sizeof(h));
      if (h.seq > newest) newest = h.seq;
      if (h.seq) occupied++;
      if (verbose) {
        char b[128];
        std::snprintf(b, sizeof(b), "    slot %u: seq=%llu tag=%u bytes=%u\n",
                      i, (unsigned long long)h.seq, h.tag, h.bytes);
        slots += b;
      }
    }
    char b[160];
    std::snprintf(b, sizeof(b),
                  "  from rank %u: newest_seq=%llu written_slots=%llu "
                  "credit_to_peer=%llu\n",
                  s, (unsigned long long)newest, (unsigned long long)occupied,
                  (unsigned long long)ctl.credit);
    out += b;
    out += slots;
  }
  return out;
}

std::string ACCL::dump_streams() {
  const ProtoConfig& c = be_->cfg();
  std::string out = "stream rings (" + std::to_string(c.n_stream) +
                    " slots x " + std::to_string(c.stream_bytes) + " B):\n";
  for (u32 s = 0; s < c.nranks; ++s) {
    StreamLane L = stream_lane(c, s);
    EagerChanCtl ctl{};
    be_->read_arena(L.ctl_off, &ctl, sizeof(ctl));
    u64 newest = 0;
    for (u32 i = 0; i < c.n_stream; ++i) {
      SlotHdr h{};
      be_->read_arena(L.hdr_off + i * sizeof(SlotHdr), &h, sizeof(h));
      if (h.seq > newest) newest = h.seq;
    }
    char b[200];
    std::snprintf(b, sizeof(b),
                  "  lane %u: newest_rx_seq=%llu consumed=%llu "
                  "tx_ctr(as sender to %u)=%llu credit=%llu\n",
                  s, (unsigned long long)newest,
                  (unsigned long long)stream_rx_seq_[s], s,
                  (unsigned long long)ctl.tx_ctr,
                  (unsigned long long)ctl.credit);
    out += b;
  }
  return out;
}

Request* ACCL::recv(BaseBuffer& dst, u64 count, u32 src, u32 tag, u32 comm,
                    bool to_device, DataType compress, bool run_async) {
  CallDesc d = make_desc(Op::recv, count, dst.dtype(), compress);
  d.addr2 = dst.arena_offset();
  d.root_src_dst = src;
  d.tag = tag;
  d.comm_id = comm;
  d.flags = F_DST_ARENA;
  return finish(d, run_async, to_device ? nullptr : &dst, count);
}

Request* ACCL::bcast(BaseBuffer& buf, u64 count, u32 root, u32 comm,
                     bool from_device, bool to_device, DataType compress,
                     bool run_async) {
  CallDesc d = make_desc(Op::bcast, count, buf.dtype(), compress);
  d.addr0 = buf.arena_offset();
  d.addr2 = buf.arena_offset();
  d.root_src_dst = root;
  d.comm_id = comm;
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  bool is_root = comm_rank(comm) == root;
  return finish(d, run_async, (!to_device && !is_root) ? &buf : nullptr, count,
                (!from_device && is_root) ? &buf : nullptr, count);
}

Request* ACCL::scatter(BaseBuffer& src, BaseBuffer& dst, u64 count, u32 root,
                       u32 comm, bool from_device, bool to_device,
                       DataType compress, bool run_async) {
  CallDesc d = make_desc(Op::scatter, count, src.dtype(), compress);
  d.addr0 = src.arena_offset();
  d.addr2 = dst.arena_offset();
  d.root_src_dst = root;
  d.comm_id = comm;
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  bool is_root = comm_rank(comm) == root;
  return finish(d, run_async, to_device ? nullptr : &dst, count,
                (!from_device && is_root) ? &src : nullptr,
                count * comm_size(comm));
}

Request* ACCL::gather(BaseBuffer& src, BaseBuffer& dst, u64 count, u32 root,
                      u32 comm, bool from_device, bool to_device,
                      DataType compress, bool run_async) {
  CallDesc d = make_desc(Op::gather, count, src.dtype(), compress);
  d.addr0 = src.arena_offset();
  d.addr2 = dst.arena_offset();
  d.root_src_dst = root;
  d.comm_id = comm;
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  bool is_root = comm_rank(comm) == root;
  return finish(d, run_async,
                (!to_device && is_root) ? &dst : nullptr,
                count * comm_size(comm), from_device ? nullptr : &src, count);
}

Request* ACCL::allgather(BaseBuffer& src, BaseBuffer& dst, u64 count, u32 comm,
                         bool from_device, bool to_device, DataType compress,
                         bool run_async) {
  CallDesc d = make_desc(Op::allgather, count, src.dtype(), compress);
  d.addr0 = src.arena_offset();
  d.addr2 = dst.arena_offset();
  d.comm_id = comm;
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  return finish(d, run_async, to_device ? nullptr : &dst,
                count * comm_size(comm), from_device ? nullptr : &src, count);
}

Request* ACCL::reduce(BaseBuffer& src, BaseBuffer& dst, u64 count, u32 root,
                      ReduceFunction f, u32 comm, bool from_device,
                      bool to_device, DataType compress, bool run_async) {
  CallDesc d = make_desc(Op::reduce, count, src.dtype(), compress);
  d.addr0 = src.arena_offset();
  d.addr2 = dst.arena_offset();
  d.root_src_dst = root;
  d.function = u32(f);
  d.comm_id = comm;
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  bool is_root = comm_rank(comm) == root;
  return finish(d, run_async, (!to_device && is_root) ? &dst : nullptr, count,
                from_device ? nullptr : &src, count);
}

Request* ACCL::allreduce(BaseBuffer& src, BaseBuffer& dst, u64 count,
                         ReduceFunction f, u32 comm, bool from_device,
                         bool to_device, DataType compress, bool run_async) {
  CallDesc d = make_desc(Op::allreduce, count, src.dtype(), compress);
  d.addr0 = src.arena_offset();
  d.addr2 = dst.arena_offset();
  d.function = u32(f);
  d.comm_id = comm;
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  return finish(d, run_async, to_device ? nullptr : &dst, count,
                from_device ? nullptr : &src, count);
}

Request* ACCL::reduce_scatter(BaseBuffer& src, BaseBuffer& dst, u64 count,
                              ReduceFunction f, u32 comm, bool from_device,
                              bool to_device, DataType compress,
                              bool run_async) {
  CallDesc d = make_desc(Op::reduce_scatter, count, src.dtype(), compress);
  d.addr0 = src.arena_offset();
  d.addr2 = dst.arena_offset();
  d.function = u32(f);
  d.comm_id = comm;
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  return finish(d, run_async, to_device ? nullptr : &dst, count,
                from_device ? nullptr : &src, count * comm_size(comm));
}

Request* ACCL::alltoall(BaseBuffer& src, BaseBuffer& dst, u64 count, u32 comm,
                        bool from_device, bool to_device, bool run_async) {
  CallDesc d = make_desc(Op::alltoall, count, src.dtype(), src.dtype());
  d.addr0 = src.arena_offset();
  d.addr2 = dst.arena_offset();
  d.comm_id = comm;
  d.flags = F_SRC_ARENA | F_DST_ARENA;
  return finish(d, run_async, to_device ? nullptr : &dst,
                count * comm_size(comm), from_device ? nullptr : &src,
                count * comm_size(comm));
}

Request* ACCL::barrier(u32 comm, bool run_async) {
  CallDesc d = make_desc(Op::barrier, 0, DataType::float32, DataType::float32);
  d.comm_id = comm;
  return finish(d, run_async, nullptr, 0);
}

Request* ACCL::nop(bool run_async) {
  CallDesc d = make_desc(Op::nop, 0, DataType::float32, DataType::float32);
  return finish(d, run_async, nullptr, 0);
}

}  // namespace accl
