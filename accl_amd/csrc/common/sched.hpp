// accl_amd collective scheduler — single source for the GPU persistent engine
// and the CPU emulator engine.
//
// This is the rebuild of the reference's collective microcode
// (reference: kernels/cclo/fw/sw_apps/ccl_offload_control/src/
// ccl_offload_control.c — run() dispatch :2375-2459, send :573-649,
// recv :653-710, bcast :796-988, scatter :992-1123, gather :1128-1294,
// allgather :1297-1503, reduce :1507-1744, reduce_scatter :1748-1852,
// allreduce :1855-2075, barrier :2078-2120, all_to_all :2123-2218), built
// MI355X-first: instead of emitting AXI DataMover instructions it runs a set
// of *flows* (segmented, credit-throttled, gate-chained data streams) over
// xGMI peer memory, and instead of ring-only schedules it uses fullmesh
// schedules that drive all 7 xGMI links of a GPU at once where P <= the
// flow budget.
//
// Template parameter `Mover` is the data plane:
//   u32  submit(const MoveDesc&)  — async; returns token (monotonic)
//   bool poll(u32 token)          — completed? (acquire semantics on true)
// GPU: movers = all engine workgroups (gpu/engine.hip). CPU: synchronous
// execution (emu/emudevice.cpp).
#pragma once
#include "types.hpp"
#include "proto.hpp"
#include "move.hpp"

namespace accl {

constexpr u32 MAX_FLOWS = 72;        // >= 4*(P-1)+1 for fullmesh at P<=16
constexpr u32 FLOW_INFLIGHT = 4;     // outstanding segments per flow
                                     // (reference keeps <=3 eager moves in
                                     // flight, ccl_offload_control.c:626-648)

enum FlowKind : u8 {
  FLOW_IDLE = 0,
  FLOW_LOCAL,      // local move src(+red) -> dst
  FLOW_TX,         // eager send to gpeer via rx slots
  FLOW_RX,         // eager recv from gpeer (copy or fused reduce)
  FLOW_TX_DIRECT,  // direct write into peer arena (rendezvous data phase)
  FLOW_RX_DIRECT,  // wait on peer's direct-write progress word
};

struct PendSeg {
  u32 token;
  u32 slot;       // eager: slot index; direct: unused
  u64 elems;
  u64 seq;        // eager: segment seq to publish / credit value to return
  u32 first_last; // SEG_FIRST/SEG_LAST for hdr publish
  u32 spill;      // FLOW_RX: spare_slot+1 when the segment was consumed
                  // from the unexpected-spill pool (0 = live ring; credit
                  // was already returned at spill time)
};

struct Flow {
  u8 kind;
  u8 to_stream;   // FLOW_TX: target the peer's STREAM ring, not eager slots
  u8 func;        // ReduceFunction+1; 0 = plain copy
  u8 sdt, wdt, ddt, bdt;  // src, wire, dst, reduce-operand dtypes
  u32 gpeer;
  u32 tag;        // expected tag (rx) / stamped tag (tx); TAG_ANY allowed on rx
  u32 matched_tag;
  const char* src;
  char* dst;
  const char* red;        // rx fused-reduce second operand (may alias dst)
  u64 count;              // total elements
  u64 submitted;          // elements handed to the mover
  u64 done;               // elements retired IN ORDER (gates dependents)
  const u64* gate;        // if set: submitted may not pass *gate
  u64 claimed;            // stream tx: seq claimed from the shared allocator
                          // but not yet submitted (0 = none)
  // direct mode:
  u64 peer_off;           // destination offset in peer arena (tx_direct)
  u64 prog_addr;          // window progress word: tx_direct = in PEER's
                          // arena (we store cumulative window bytes),
                          // rx_direct = in OWN arena (we poll)
  // pending segment fifo:
  PendSeg pend[FLOW_INFLIGHT];
  u32 ph, pt;             // head/tail (pt-ph = in flight)
  u32 win_slot;           // rx_direct: progress-word pool index (freed when
                          // the window is fully consumed)
  u64 msg_count_hdr;      // value for SlotHdr.msg_count (tx)
};

// ---------------------------------------------------------------- engine
template <class Mover>
struct Cclo {
  TransportView tv;
  ProtoConfig cfg;
  PairSeq sq;
  Mover* mv;
  CommView comms[MAX_COMMS];
  u32 ncomms;
  u64 timeout_ticks;
  u64 max_eager_bytes;     // above this, arena<->arena transfers go direct
  u64 max_rndzv_bytes;     // window cap for a single posted rendezvous
                           // window (0 = unlimited); reference:
                           // set_max_rendezvous_size (accl.hpp:103-104)
  u32 tune_oneshot_max;    // one-shot fan-in cutoff bytes (0 = 64 KB)
  u32 tune_fullmesh_max;   // allreduce fullmesh->ring cutoff (runtime
                           // tuning register; reference flat-tree caps,
                           // driver/xrt/src/accl.cpp:1198-1208); 0 = default
  u32 err;                 // error bits of the current call
  u32 nflows_;             // size of the flow set run_flows is executing
                           // (sibling-flow scan for cross-call spills)
  Flow flows[MAX_FLOWS];

  // ---- unexpected-message queue (the rxbuf_seek match-engine analogue:
  // reference kernels/cclo/hls/rxbuf_offload/rxbuf_seek.cpp:53-72 searches
  // pending rx buffers by (tag, src, seqn) in any order). Segments whose tag
  // does not match the posted recv are spilled to the spare region's upper
  // half so later recvs can match them out of order.
  static constexpr u32 UQ_DEPTH = 8;
  struct Unexpected {
    u32 tag; u32 arith; u64 bytes; u64 msg_count; u32 spare_slot; u32 flags;
  };

  // ---- rendezvous pending sets (the RNDZV_PENDING spill queue analogue,
  // reference ccl_offload_control.c:154-408): in-seq records whose tag does
  // not match the current wait are parked here for later out-of-order match.
  static constexpr u32 RNDZV_PEND = 8;
  struct PendRndzv { u64 seq; u64 offset; u64 count; u32 tag; u32 arith;
                     u32 valid; u32 prog; };

  // ---- multi-call interleaving (the CMD_CALL_RETRY requeue analogue,
  // reference ccl_offload_control.c:2460-2478 + current_step resume
  // :671-675): a send/recv that would block before making ANY progress is
  // parked and the engine serves later descriptors; parked calls are
  // re-probed each loop. ParkState carries the resume cursors (posted
  // rendezvous windows) so re-entry never re-posts.
  struct ParkState { u64 w[6]; u32 step; u32 _pad; };
  struct ParkedCall {
    CallDesc d; u64 ring_idx; u64 deadline; u64 t_start; ParkState ps;
    u32 used; u32 _pad;
  };

  // Cold match/park tables live OUTSIDE the Cclo (GPU: device-global
  // via GpuEngineState; emulator: heap) — the GPU scheduler stages Cclo
  // in LDS and the per-workgroup LDS budget is 64 KB; these tables are
  // slow-path state that does not need LDS residency.
  struct ColdState {
    Unexpected uq[MAX_RANKS][UQ_DEPTH];
    u32 uq_h[MAX_RANKS], uq_t[MAX_RANKS];
    PendRndzv pa[MAX_RANKS][RNDZV_PEND];  // addr records
    PendRndzv pd[MAX_RANKS][RNDZV_PEND];  // done records
    ParkedCall parked[MAX_INFLIGHT];
    u64 prog_busy[MAX_RANKS][2];  // window progress-word pool (N_PROG bits)
  };
  ColdState* cold;
  u64 spill_busy;          // bitmap over spill slots (<= 64)
  u32 nparked;
  // retry_parked() completion report:
  u64 done_ring_idx; u64 done_t0; u32 done_err;
  // probe plumbing (set around run_call_inner)
  u32 probe_;
  ParkState* ps_;
  u32 in_drain_;  // re-entrancy guard for drain_for_parked()
  // parked index currently being retried, +1 (0 = none): the drain must
  // not touch an entry whose own probe is running above it — both would
  // consume the same windows/segments through the same ParkState
  u32 active_parked1_;
  // last non-flow waitpoint (dump_timeout diagnosis): what a flows=0
  // timeout was actually spinning on
  u64 wp_kind_;   // 0 none, 1 wait_addr, 2 wait_done
  u64 wp_info_;   // peer | tag<<32
  u64 wp_seq_;    // awaited ring seq | observed head seq<<32
  // pairs whose live channel head is HELD (unreleased SlotRef from a
  // one-shot collect): the drain must not spill/consume that pair's ring
  // or the held payload gets overwritten once its credit returns early
  u64 drain_hold_;

  ACCL_HD u32 me() const { return cfg.rank; }
  // wildcard (TAG_ANY) matching must never capture engine-internal
  // collective traffic: TAG_COLL-tagged segments/records belong to the
  // collective state machines, not to user receives
  ACCL_HD static bool any_ok(u32 tag) { return !(tag & TAG_COLL); }

  // device-only micro-timeline (GPU: GpuMover::dbg; emulator mover: no-op)
  template <class M>
  ACCL_HD auto stamp_impl(M* m, int i, int) -> decltype((void)m->dbg) {
#if defined(__HIP_DEVICE_COMPILE__)
    m->dbg[i] = wallclock();
#else
    (void)m; (void)i;
#endif
  }
  template <class M>
  ACCL_HD void stamp_impl(M*, int, long) {}
  ACCL_HD void stamp(int i) { stamp_impl(mv, i, 0); }

  // ---------------- tiny helpers ----------------
  ACCL_HD char* local_ptr(u64 addr, bool arena) {
    return arena ? tv.arena[me()] + addr : (char*)addr;
  }
  ACCL_HD static u64 min64(u64 a, u64 b) { return a < b ? a : b; }

  ACCL_HD bool wait_pred_tick(u64& deadline) {
#if defined(__HIP_DEVICE_COMPILE__)
    // Drop stale L1/XCD-L2 lines before the next re-read. Peer-written
    // control words (another PROCESS over IPC, another GPU over xGMI) do
    // NOT invalidate this XCD's L2 copies on coarse-grained memory, and
    // sc0/sc1 polls are L2-served — a line cached by our own earlier polls
    // would read stale forever (round-2 fresh-box wedge: peer's published
    // slot header invisible for the whole 10 s deadline).
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
#endif
    // Flow-free progress for PARKED calls while this op spins: without it,
    // two engines deadlock when a committed multi-segment send waits for
    // credit that only a parked recv (stuck behind this very wait on the
    // peer) would return — the reference's rxbuf-offload engines drain the
    // wire independently of the consumer for exactly this reason
    // (kernels/cclo/hls/rxbuf_offload/*).
    drain_for_parked();
    cpu_pause();
    if (wallclock() > deadline) { err |= E_TIMEOUT; return false; }
    return true;
  }
  ACCL_HD u64 deadline_now() { return wallclock() + timeout_ticks; }

  // ------------- 64-byte record publish / consume (one lane) -------------
  // Publication rule per MI355X_MICROARCH §Workgroup dispatch: plain field
  // stores, release fence (system), then the seq word.
  ACCL_HD void publish_rec(volatile u64* rec8, const u64* val8) {
    for (int i = 1; i < 8; ++i) rec8[i] = val8[i];
    fence_release_sys();
    st_sys(&rec8[0], val8[0]);
  }

  // ---------------- eager slot helpers ----------------
  // Channel (s -> r): headers+payload in r's arena lane [s]; credit word in
  // s's arena (chan_ctl(s, r)->credit), advanced by r.
  ACCL_HD u64 tx_credit(u32 peer) {  // slots consumed by peer (cumulative)
    return ld_sys(&tv.chan_ctl(me(), peer)->credit);
  }
  // Return credit MONOTONICALLY: consumers can finish out of order (a
  // flow's pending segment copy retires after the drain already consumed
  // and credited newer segments), and a stale write would shrink the
  // sender's window below its outstanding count — a permanent stall.
  ACCL_HD void ret_credit(u32 peer, u64 seq) {
    if (seq > sq.credit_ret[peer]) {
      sq.credit_ret[peer] = seq;
      st_sys(&tv.chan_ctl(peer, me())->credit, seq);
    }
  }

  // Flow-free single-segment eager send (direct mover submit + header
  // publish): the parked-send resume/drain path — safe to run from INSIDE
  // another op's wait (never touches the flow table). Returns elements
  // sent (0 = no credit / would block).
  ACCL_HD u64 eager_send_segment(u32 peer, const char* src, DataType sdt,
                                 DataType wdt, u64 off, u64 remaining,
                                 u64 total, u32 tag) {
    u64 next = sq.eager_tx[peer];
    if (next - tx_credit(peer) >= cfg.n_slots) return 0;  // no credit
    const u64 seg_cap = u64(cfg.slot_bytes) / dtype_size(wdt);
    u64 n = min64(remaining, seg_cap);
    u32 slot = u32(next % cfg.n_slots);
    MoveDesc m{};
    m.dst = (u64)tv.slot_payload(peer, me(), slot);
    m.dst_dt = u8(wdt);
    m.src[0] = (u64)(src + off * dtype_size(sdt));
    m.src_dt[0] = u8(sdt);
    m.nsrc = 1;
    m.count = n;
    u32 tok = mv->submit(m);
    u64 deadline = deadline_now();
    while (!mv->poll(tok))
      if (!wait_pred_tick_nodrain(deadline)) return 0;
    SlotHdr* h = tv.slot_hdr(peer, me(), slot);
    h->tag = tag;
    h->bytes = u32(n * dtype_size(wdt));
    h->msg_count = total;
    h->arith = u32(wdt);
    h->flags = (off == 0 ? SEG_FIRST : 0) |
               (off + n >= total ? SEG_LAST : 0);
    fence_release_sys();
    st_sys(&h->seq, next + 1);
    sq.eager_tx[peer] = next + 1;
    return n;
  }

  // flow-free synchronous mover copy (safe from inside another op's wait:
  // never touches the flow table)
  ACCL_HD bool copy_free(const void* src, DataType sdt, char* dstp,
                         DataType ddt, u64 count) {
    MoveDesc m{};
    m.dst = (u64)dstp;
    m.dst_dt = u8(ddt);
    m.src[0] = (u64)src;
    m.src_dt[0] = u8(sdt);
    m.nsrc = 1;
    m.count = count;
    u32 tok = mv->submit(m);
    u64 deadline = deadline_now();
    while (!mv->poll(tok))
      if (!wait_pred_tick_nodrain(deadline)) return false;
    return true;
  }

  // wait tick WITHOUT the parked-drain hook (used inside the drain itself
  // and other non-reentrant spots)
  ACCL_HD bool wait_pred_tick_nodrain(u64& deadline) {
#if defined(__HIP_DEVICE_COMPILE__)
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
#endif
    cpu_pause();
    if (wallclock() > deadline) { err |= E_TIMEOUT; return false; }
    return true;
  }

  // Bounded flow-free progress for parked calls (called from in-op waits):
  //  - parked eager recv: spill ONE arrived head segment to the unexpected
  //    pool (returns the sender's credit — the deadlock breaker)
  //  - parked eager send: push ONE segment when credit allows (progress
  //    recorded in its ParkState; the retry resumes from there)
  ACCL_HD void drain_for_parked() {
    if (!nparked || in_drain_) return;
    in_drain_ = 1;
    for (u32 i = 0; i < MAX_INFLIGHT; ++i) {
      ParkedCall& p = cold->parked[i];
      if (!p.used) continue;
      if (i + 1 == active_parked1_) continue;  // its own probe is running
      // per-(pair,tag) FIFO: skip if an older parked call shares the key
      bool blocked = false;
      for (u32 j = 0; j < MAX_INFLIGHT; ++j) {
        const ParkedCall& q = cold->parked[j];
        if (!q.used || j == i) continue;
        if (q.ring_idx < p.ring_idx && q.d.scenario == p.d.scenario &&
            q.d.root_src_dst == p.d.root_src_dst &&
            q.d.comm_id == p.d.comm_id && q.d.tag == p.d.tag) {
          blocked = true;
          break;
        }
      }
      if (blocked) continue;
      if (p.d.comm_id >= ncomms) continue;
      const CommView& c = comms[p.d.comm_id];
      u32 peer = c.global(p.d.root_src_dst);
      if (peer == me()) continue;
      // a one-shot collect holds this pair's live head unreleased — any
      // spill/consume here returns its credit early and the sender
      // overwrites the held payload
      if ((drain_hold_ >> (peer & 63)) & 1) continue;
      Op op = Op(p.d.scenario);
      DataType dt = desc_dtype(p.d), wdt = desc_wire_dtype(p.d);
      u64 n = desc_count(p.d);
      if (op == Op::recv && !use_rndzv(n, dt, wdt)) {
        // deliver matching segments straight into the parked recv's dst
        // (progress in its ParkState — op_recv_eager's step-2 encoding)
        // and spill mismatched heads. Direct delivery matters: the spill
        // pool is BOUNDED (UQ_DEPTH), so a recv that only spills livelocks
        // once the pool fills with its own message.
        char* dst = local_ptr(p.d.addr2, p.d.flags & F_DST_ARENA);
        u64 got = (p.ps.step == 2) ? p.ps.w[0] : 0;
        u32 mtag = (p.ps.step == 2) ? u32(p.ps.w[1]) : p.d.tag;
        const u32 wsz = dtype_size(wdt), dsz = dtype_size(dt);
        bool prog = true;
        while (prog && got < n) {
          prog = false;
          for (u32 qi = cold->uq_h[peer]; qi != cold->uq_t[peer]; ++qi) {
            Unexpected& u = cold->uq[peer][qi % UQ_DEPTH];
            if (u.bytes == 0) continue;
            if (mtag == TAG_ANY ? !any_ok(u.tag) : u.tag != mtag) continue;
            if (got > 0 && u.tag != mtag) continue;
            // arith/segment violations are the retry's to report
            // (E_COMPRESSION/E_SEGMENT); stop pool delivery here
            if (u.arith != u32(wdt)) break;
            u64 nseg = u.bytes / wsz;
            if (nseg > n - got) break;
            if (got == 0) mtag = u.tag;
            if (!copy_free(spill_ptr(u.spare_slot), wdt, dst + got * dsz,
                           dt, nseg))
              break;
            spill_busy &= ~(1ull << u.spare_slot);
            u.bytes = 0;
            while (cold->uq_h[peer] != cold->uq_t[peer] &&
                   cold->uq[peer][cold->uq_h[peer] % UQ_DEPTH].bytes == 0)
              cold->uq_h[peer]++;
            got += nseg;
            prog = true;
            break;
          }
          if (prog) continue;
          u64 seq = sq.eager_rx[peer] + 1;
          u32 sl = u32((seq - 1) % cfg.n_slots);
          SlotHdr* h = tv.slot_hdr(me(), peer, sl);
          if (ld_sys(&h->seq) != seq) break;
          fence_acquire_sys();
          bool match = mtag == TAG_ANY ? any_ok(h->tag) : (h->tag == mtag);
          u64 nseg = u64(h->bytes) / wsz;
          if (match && h->arith == u32(wdt) && nseg <= n - got) {
            if (got == 0) mtag = h->tag;
            if (!copy_free(tv.slot_payload(me(), peer, sl), wdt,
                           dst + got * dsz, dt, nseg))
              break;
            sq.eager_rx[peer] = seq;
            ret_credit(peer, seq);
            got += nseg;
            prog = true;
          } else if (!match) {
            prog = spill_head(peer, h, sl, seq);
          }  // mismatched arith / overlong segment: leave for the retry
        }
        p.ps.w[0] = got;
        p.ps.w[1] = mtag;
        p.ps.step = 2;
      } else if (op == Op::recv && (p.d.flags & F_DST_ARENA)) {
        // windowed rendezvous recv: consume completed windows and rotate
        // new ones in the background (data lands directly in dst — the
        // sender writes the arena window; "consumption" is observing the
        // progress word and freeing its pool slot)
        ParkState& ps = p.ps;
        if (ps.step == 3) continue;  // only the done record outstanding
        const u32 esz = dtype_size(dt);
        u64 wmax = max_rndzv_bytes ? max_rndzv_bytes / esz : n;
        if (!wmax) wmax = 1;
        u64 posted = 0, got = 0;
        u32 wslot[2] = {0, 0};
        u64 wcnt[2] = {0, 0};
        u32 wi = 0, nw = 0;
        if (ps.step == 1) {
          posted = ps.w[0]; got = ps.w[1];
          wcnt[0] = ps.w[2]; wcnt[1] = ps.w[3];
          wslot[0] = u32(ps.w[4]); wslot[1] = u32(ps.w[4] >> 32);
          wi = u32(ps.w[5]); nw = u32(ps.w[5] >> 32);
        }
        bool prog = true;
        while (prog && got < n) {
          prog = false;
          while (posted < n && nw < 2 && addr_room(peer)) {
            u64 w = min64(n - posted, wmax);
            wslot[(wi + nw) % 2] =
                post_addr(peer, p.d.addr2 + posted * esz, w, p.d.tag, u32(dt));
            wcnt[(wi + nw) % 2] = w;
            posted += w; nw++; prog = true;
          }
          if (nw &&
              ld_sys(tv.direct_word(me(), peer, wslot[wi])) >=
                  wcnt[wi] * u64(esz)) {
            fence_acquire_sys();
            cold->prog_busy[peer & 63][(wslot[wi] >> 6) & 1] &=
                ~(1ull << (wslot[wi] & 63));
            got += wcnt[wi];
            wi ^= 1; nw--; prog = true;
          }
        }
        ps.w[0] = posted; ps.w[1] = got;
        ps.w[2] = wcnt[0]; ps.w[3] = wcnt[1];
        ps.w[4] = u64(wslot[0]) | (u64(wslot[1]) << 32);
        ps.w[5] = u64(wi) | (u64(nw) << 32);
        ps.step = (got >= n) ? 3 : 1;
      } else if (op == Op::send && !use_rndzv(n, dt, wdt)) {
        u64 sent = (p.ps.step == 1) ? p.ps.w[0] : 0;
        if (sent < n) {
          const char* src =
              local_ptr(p.d.addr0, p.d.flags & F_SRC_ARENA);
          u64 did = eager_send_segment(peer, src, dt, wdt, sent, n - sent, n,
                                       p.d.tag);
          if (did) {
            p.ps.w[0] = sent + did;
            p.ps.step = 1;
          }
        }
      } else if (op == Op::send) {
        // parked rendezvous send: push whatever windows the receiver has
        // posted (the receiver may be committed-blocking on them while OUR
        // engine is blocked in an unrelated op — same cycle as eager)
        (void)rndzv_send_push(peer, p.d, p.ps);
      }
    }
    in_drain_ = 0;
  }

  // ---------------- flow stepping ----------------
  ACCL_HD bool flow_done(const Flow& f) {
    return f.kind == FLOW_IDLE || (f.done >= f.count && f.ph == f.pt);
  }

  ACCL_HD u64 gate_limit(const Flow& f) {
    if (!f.gate) return f.count;
    return min64(f.count, ld_sys((const volatile u64*)f.gate));
  }

  // retire completed segments in order; apply post actions
  ACCL_HD bool flow_retire(Flow& f) {
    bool any = false;
    while (f.ph != f.pt) {
      PendSeg& p = f.pend[f.ph % FLOW_INFLIGHT];
      if (!mv->poll(p.token)) break;
      switch (f.kind) {
        case FLOW_TX: {
          // payload complete -> publish the slot header (release chain:
          // mover released on completion, poll() acquired).
          SlotHdr* h = f.to_stream ? tv.stream_hdr(f.gpeer, me(), p.slot)
                                   : tv.slot_hdr(f.gpeer, me(), p.slot);
          h->tag = f.tag;
          h->bytes = u32(p.elems * dtype_size(DataType(f.wdt)));
          h->msg_count = f.msg_count_hdr;
          h->arith = u32(f.wdt);
          h->flags = p.first_last;
          fence_release_sys();
          st_sys(&h->seq, p.seq);
          break;
        }
        case FLOW_RX: {
          if (p.spill) {
            // segment came from the spill pool (credit returned when it
            // was spilled); free the spare slot now that the move retired
            spill_busy &= ~(1ull << (p.spill - 1));
            break;
          }
          // payload consumed -> return credit to the sender (cumulative).
          ret_credit(f.gpeer, p.seq);
          break;
        }
        case FLOW_TX_DIRECT: {
          // advance the WINDOW's progress word in the peer's arena
          // (cumulative bytes within this posted window)
          fence_release_sys();
          st_sys((volatile u64*)f.prog_addr,
                 (f.done + p.elems) * dtype_size(DataType(f.ddt)));
          break;
        }
        default: break;
      }
      f.done += p.elems;
      f.ph++;
      any = true;
    }
    return any;
  }

  ACCL_HD bool flow_submit(Flow& f) {
    bool any = false;
    while (f.submitted < f.count && (f.pt - f.ph) < FLOW_INFLIGHT) {
      u64 avail = gate_limit(f);
      if (f.submitted >= avail) break;
      MoveDesc m{};
      switch (f.kind) {
        case FLOW_LOCAL: {
          // Local moves need no wire segmentation: one move exposes every
          // tile to the whole mover fleet at once (max data-plane
          // parallelism). Gated flows still chase the gate in segments.
          const u64 seg_cap = f.gate
              ? (8u << 20) / dtype_size(DataType(f.ddt))
              : f.count;
          u64 n = min64(avail - f.submitted, seg_cap);
          m.dst = (u64)(f.dst + f.submitted * dtype_size(DataType(f.ddt)));
          m.dst_dt = f.ddt;
          m.src[0] = (u64)(f.src + f.submitted * dtype_size(DataType(f.sdt)));
          m.src_dt[0] = f.sdt;
          m.nsrc = 1;
          if (f.func) {
            m.src[1] = (u64)(f.red + f.submitted * dtype_size(DataType(f.bdt)));
            m.src_dt[1] = f.bdt;
            m.nsrc = 2;
            m.func = f.func - 1;
          }
          m.count = n;
          u32 tok = mv->submit(m);
          f.pend[f.pt % FLOW_INFLIGHT] = {tok, 0, n, 0, 0};
          f.pt++; f.submitted += n; any = true;
          break;
        }
        case FLOW_TX: {
          u64 next;
          const u32 n_slots = f.to_stream ? cfg.n_stream : cfg.n_slots;
          if (f.to_stream) {
            // seq from the channel's shared allocator (engine + host + device
            // producers interoperate); claim once, hold across credit stalls
            EagerChanCtl* ctl = tv.stream_ctl(me(), f.gpeer);
            if (!f.claimed) f.claimed = afadd_sys(&ctl->tx_ctr, 1) + 1;
            next = f.claimed - 1;
            u64 credit = ld_sys(&ctl->credit);
            if (credit + n_slots < f.claimed) return any;  // slot not yet free
          } else {
            next = sq.eager_tx[f.gpeer];            // segments sent so far
            if (next - tx_credit(f.gpeer) >= n_slots) return any;  // no credit
          }
          const u32 seg_bytes = f.to_stream ? cfg.stream_bytes : cfg.slot_bytes;
          const u64 seg_cap = u64(seg_bytes) / dtype_size(DataType(f.wdt));
          u64 n = min64(avail - f.submitted, seg_cap);
          u32 slot = u32(next % n_slots);
          m.dst = f.to_stream ? (u64)tv.stream_payload(f.gpeer, me(), slot)
                              : (u64)tv.slot_payload(f.gpeer, me(), slot);
          m.dst_dt = f.wdt;
          m.src[0] = (u64)(f.src + f.submitted * dtype_size(DataType(f.sdt)));
          m.src_dt[0] = f.sdt;
          m.nsrc = 1;
          m.count = n;
          u32 tok = mv->submit(m);
          u32 fl = (f.submitted == 0 ? SEG_FIRST : 0) |
                   (f.submitted + n >= f.count ? SEG_LAST : 0);
          f.pend[f.pt % FLOW_INFLIGHT] = {tok, slot, n, next + 1, fl};
          f.pt++; f.submitted += n;
          if (f.to_stream) f.claimed = 0;
          else sq.eager_tx[f.gpeer] = next + 1;
          any = true;
          break;
        }
        case FLOW_RX: {
          // 0) spill pool first: a parked recv's probe may have spilled
          // this flow's segment out of the live ring (spilled entries are
          // strictly older than the current head, so consuming them first
          // preserves per-tag send order)
          {
            bool consumed_spill = false;
            for (u32 qi = cold->uq_h[f.gpeer]; qi != cold->uq_t[f.gpeer];
                 ++qi) {
              Unexpected& u = cold->uq[f.gpeer][qi % UQ_DEPTH];
              if (u.bytes == 0) continue;
              u32 want = (f.submitted == 0) ? f.tag : f.matched_tag;
              if (want == TAG_ANY ? !any_ok(u.tag) : u.tag != want) continue;
              if (u.arith != u32(f.wdt)) { err |= E_COMPRESSION; return any; }
              u32 wsz0 = dtype_size(DataType(f.wdt));
              u64 n0 = u.bytes / wsz0;
              if (n0 > f.count - f.submitted) { err |= E_SEGMENT; return any; }
              if (f.gate && f.submitted + n0 > gate_limit(f)) return any;
              if (f.submitted == 0) f.matched_tag = u.tag;
              m.dst = (u64)(f.dst + f.submitted * dtype_size(DataType(f.ddt)));
              m.dst_dt = f.ddt;
              m.src[0] = (u64)spill_ptr(u.spare_slot);
              m.src_dt[0] = f.wdt;
              m.nsrc = 1;
              if (f.func) {
                m.src[1] =
                    (u64)(f.red + f.submitted * dtype_size(DataType(f.bdt)));
                m.src_dt[1] = f.bdt;
                m.nsrc = 2;
                m.func = f.func - 1;
              }
              m.count = n0;
              u32 tok0 = mv->submit(m);
              f.pend[f.pt % FLOW_INFLIGHT] =
                  PendSeg{tok0, 0, n0, 0, 0, u.spare_slot + 1};
              f.pt++;
              f.submitted += n0;
              u.bytes = 0;  // entry claimed; spill_busy freed at retire
              while (cold->uq_h[f.gpeer] != cold->uq_t[f.gpeer] &&
                     cold->uq[f.gpeer][cold->uq_h[f.gpeer] % UQ_DEPTH].bytes ==
                         0)
                cold->uq_h[f.gpeer]++;
              any = true;
              consumed_spill = true;
              break;
            }
            if (consumed_spill) continue;
          }
          u64 next = sq.eager_rx[f.gpeer];          // segments consumed
          u32 slot = u32(next % cfg.n_slots);
          SlotHdr* h = tv.slot_hdr(me(), f.gpeer, slot);
          if (ld_sys(&h->seq) != next + 1) return any;   // not arrived
          fence_acquire_sys();
          // Per-segment tag demultiplex: several rx flows may drain one
          // pair channel concurrently (e.g. allreduce phase 1 + phase 2);
          // a head segment that belongs to a SIBLING flow in this set is
          // left in place, but a CROSS-CALL segment (a user send whose
          // matching recv comes later in program order) must be spilled to
          // the unexpected pool or it head-of-line-blocks this collective
          // forever. (reference analogue: rxbuf_seek matching by
          // (tag, src, seqn), rxbuf_seek.cpp:53-72)
          u32 want = (f.submitted == 0) ? f.tag : f.matched_tag;
          bool wmatch = want == TAG_ANY ? any_ok(h->tag) : (h->tag == want);
          if (!wmatch) {
            bool sibling = false;
            for (u32 j = 0; j < nflows_; ++j) {
              const Flow& g = flows[j];
              if (g.kind != FLOW_RX || flow_done(g) || g.gpeer != f.gpeer)
                continue;
              u32 gw = (g.submitted == 0) ? g.tag : g.matched_tag;
              if ((gw == TAG_ANY && any_ok(h->tag)) || gw == h->tag) {
                sibling = true;
                break;
              }
            }
            if (!sibling) {
              if (spill_head(f.gpeer, h, slot, next + 1)) { any = true; continue; }
              if (err) return any;
            }
            return any;
          }
          if (f.submitted == 0) f.matched_tag = h->tag;
          u32 wsz = dtype_size(DataType(f.wdt));
          if (h->arith != u32(f.wdt)) { err |= E_COMPRESSION; return any; }
          u64 n = u64(h->bytes) / wsz;
          if (n > f.count - f.submitted) { err |= E_SEGMENT; return any; }
          // a segment is consumed whole: a gated flow (fused reduce chained
          // on its predecessor's done counter) must wait until the gate
          // covers the ENTIRE segment, or the reduce would read dst elements
          // the predecessor has not produced yet (race on the GPU's async
          // movers; the synchronous emulator mover never exposes it)
          if (f.gate && f.submitted + n > avail) return any;
          m.dst = (u64)(f.dst + f.submitted * dtype_size(DataType(f.ddt)));
          m.dst_dt = f.ddt;
          m.src[0] = (u64)tv.slot_payload(me(), f.gpeer, slot);
          m.src_dt[0] = f.wdt;
          m.nsrc = 1;
          if (f.func) {
            m.src[1] = (u64)(f.red + f.submitted * dtype_size(DataType(f.bdt)));
            m.src_dt[1] = f.bdt;
            m.nsrc = 2;
            m.func = f.func - 1;
          }
          m.count = n;
          u32 tok = mv->submit(m);
          f.pend[f.pt % FLOW_INFLIGHT] = {tok, slot, n, next + 1, 0};
          f.pt++; f.submitted += n;
          sq.eager_rx[f.gpeer] = next + 1;
          any = true;
          break;
        }
        case FLOW_TX_DIRECT: {
          const u64 seg_cap = (8u << 20) / dtype_size(DataType(f.ddt));
          u64 n = min64(avail - f.submitted, seg_cap);
          u32 dsz = dtype_size(DataType(f.ddt));
          m.dst = (u64)(tv.arena[f.gpeer] + f.peer_off + f.submitted * dsz);
          m.dst_dt = f.ddt;
          m.src[0] = (u64)(f.src + f.submitted * dtype_size(DataType(f.sdt)));
          m.src_dt[0] = f.sdt;
          m.nsrc = 1;
          m.count = n;
          u32 tok = mv->submit(m);
          f.pend[f.pt % FLOW_INFLIGHT] = {tok, 0, n, 0, 0};
          f.pt++; f.submitted += n; any = true;
          break;
        }
        case FLOW_RX_DIRECT: {
          // pure wait: done advances with the window's progress word
          u64 w = ld_sys((const volatile u64*)f.prog_addr);
          u64 avail_elems = min64(f.count, w / dtype_size(DataType(f.ddt)));
          if (avail_elems > f.done) {
            fence_acquire_sys();
            f.done = avail_elems;
            f.submitted = avail_elems;
            if (f.done >= f.count)  // window consumed: free its progress word
              cold->prog_busy[f.gpeer & 63][(f.win_slot >> 6) & 1] &=
                  ~(1ull << (f.win_slot & 63));
            any = true;
          }
          return any;
        }
        default: return any;
      }
    }
    return any;
  }

  // run a set of flows to completion (the engine inner loop)
  ACCL_HD u32 run_flows(u32 n) {
    nflows_ = n;
    stamp(10);
    u64 deadline = deadline_now();
    for (;;) {
      bool any = false, alldone = true;
      for (u32 i = 0; i < n; ++i) {
        Flow& f = flows[i];
        if (flow_done(f)) continue;
        alldone = false;
        any |= flow_retire(f);
        if (err) return err;
        any |= flow_submit(f);
        if (err) return err;
      }
      if (alldone) { stamp(14); return E_OK; }
      if (any) deadline = deadline_now();
      else if (!wait_pred_tick(deadline)) return err;
    }
  }

  // ---------------- flow constructors ----------------
  ACCL_HD Flow& fl(u32 i) { Flow& f = flows[i]; f = Flow{}; return f; }

  ACCL_HD void mk_local(u32 i, const char* src, DataType sdt, char* dst,
                        DataType ddt, u64 count, const char* red = nullptr,
                        DataType bdt = DataType::none, int func = -1,
                        const u64* gate = nullptr) {
    Flow& f = fl(i);
    f.kind = FLOW_LOCAL; f.src = src; f.dst = dst; f.red = red;
    f.sdt = u8(sdt); f.ddt = u8(ddt); f.bdt = u8(bdt);
    f.func = u8(func + 1); f.count = count; f.gate = gate;
  }
  ACCL_HD void mk_tx(u32 i, u32 gpeer, const char* src, DataType sdt,
                     DataType wdt, u64 count, u32 tag,
                     const u64* gate = nullptr, bool to_stream = false) {
    Flow& f = fl(i);
    f.kind = FLOW_TX; f.gpeer = gpeer; f.src = src;
    f.sdt = u8(sdt); f.wdt = u8(wdt); f.count = count; f.tag = tag;
    f.gate = gate; f.msg_count_hdr = count; f.to_stream = to_stream ? 1 : 0;
  }
  ACCL_HD void mk_rx(u32 i, u32 gpeer, char* dst, DataType ddt, DataType wdt,
                     u64 count, u32 tag, const char* red = nullptr,
                     DataType bdt = DataType::none, int func = -1,
                     const u64* gate = nullptr) {
    Flow& f = fl(i);
    f.kind = FLOW_RX; f.gpeer = gpeer; f.dst = dst; f.red = red;
    f.ddt = u8(ddt); f.wdt = u8(wdt); f.bdt = u8(bdt);
    f.func = u8(func + 1); f.count = count; f.tag = tag; f.gate = gate;
  }
  // slot = the addr-ring slot of the matched/posted window (determines the
  // progress word both sides use)
  ACCL_HD void mk_tx_direct(u32 i, u32 gpeer, const char* src, DataType sdt,
                            DataType ddt, u64 count, u64 peer_off, u32 slot,
                            const u64* gate = nullptr) {
    Flow& f = fl(i);
    f.kind = FLOW_TX_DIRECT; f.gpeer = gpeer; f.src = src;
    f.sdt = u8(sdt); f.ddt = u8(ddt); f.count = count;
    f.peer_off = peer_off; f.gate = gate;
    f.prog_addr = (u64)tv.direct_word(gpeer, me(), slot);
  }
  ACCL_HD void mk_rx_direct(u32 i, u32 gpeer, u64 count, DataType ddt,
                            u32 slot) {
    Flow& f = fl(i);
    f.kind = FLOW_RX_DIRECT; f.gpeer = gpeer; f.ddt = u8(ddt); f.count = count;
    f.prog_addr = (u64)tv.direct_word(me(), gpeer, slot);
    f.win_slot = slot;
  }

  // ---------------- rendezvous record rings ----------------
  // post {offset,count,tag} into PEER's addr ring (peer = the sender that
  // will write to us); zero the window's progress word first. Returns the
  // ring slot. reference: rendezvous_send_addr
  // (ccl_offload_control.c:142-150).
  // ring flow control: cumulative consumed counts, published by the
  // consumer into the POSTER's arena (same placement rule as eager credit)
  ACCL_HD bool addr_room(u32 gpeer) {
    return sq.rndzv_addr_tx[gpeer] -
               ld_sys(&tv.chan_ctl(me(), gpeer)->addr_ret) < cfg.n_rndzv;
  }
  ACCL_HD bool done_room(u32 gpeer) {
    return sq.rndzv_done_tx[gpeer] -
               ld_sys(&tv.chan_ctl(me(), gpeer)->done_ret) < cfg.n_rndzv;
  }
  // one consumption step of MY incoming addr/done ring from gpeer: spill
  // the head into the pending set. Run while blocked in post_addr/post_done
  // so two ranks posting to each other symmetrically always drain.
  ACCL_HD void consume_addr_tick(u32 gpeer) {
    u64 seq = sq.rndzv_addr_rx[gpeer] + 1;
    RndzvRec* r = tv.rndzv_addr(me(), gpeer, u32((seq - 1) % cfg.n_rndzv));
    if (ld_sys(&r->seq) != seq) return;
    fence_acquire_sys();
    u32 k = 0;
    while (k < RNDZV_PEND && cold->pa[gpeer][k].valid) ++k;
    if (k >= RNDZV_PEND) return;
    cold->pa[gpeer][k] = PendRndzv{seq, r->offset, r->count, r->tag, r->arith,
                                   1, u32(r->prog_idx)};
    sq.rndzv_addr_rx[gpeer] = seq;
    st_sys(&tv.chan_ctl(gpeer, me())->addr_ret, seq);
  }
  ACCL_HD void consume_done_tick(u32 gpeer) {
    u64 seq = sq.rndzv_done_rx[gpeer] + 1;
    RndzvRec* r = tv.rndzv_done(me(), gpeer, u32((seq - 1) % cfg.n_rndzv));
    if (ld_sys(&r->seq) != seq) return;
    fence_acquire_sys();
    u32 k = 0;
    while (k < RNDZV_PEND && cold->pd[gpeer][k].valid) ++k;
    if (k >= RNDZV_PEND) return;
    cold->pd[gpeer][k] = PendRndzv{seq, 0, 0, r->tag, 0, 1};
    sq.rndzv_done_rx[gpeer] = seq;
    st_sys(&tv.chan_ctl(gpeer, me())->done_ret, seq);
  }

  ACCL_HD u32 post_addr(u32 gpeer, u64 offset, u64 count, u32 tag, u32 arith) {
    if (!addr_room(gpeer)) {
      // ring full: records outlive ops (parked recvs), so wait for the
      // consumer — draining our own inbound rings meanwhile so symmetric
      // posters can't block each other. On deadline: post anyway (err is
      // set; the op fails loud downstream instead of silently losing a
      // record).
      u64 deadline = deadline_now();
      while (!addr_room(gpeer)) {
        consume_addr_tick(gpeer);
        consume_done_tick(gpeer);
        if (!wait_pred_tick(deadline)) break;
      }
    }
    // allocate a progress word (the pool is sized to never exhaust
    // structurally; the deadline wait is a belt for error-path leaks and
    // fails loud via E_TIMEOUT)
    u64* pb = cold->prog_busy[gpeer & 63];
    u32 idx = 0;
    u64 deadline = 0;
    for (;;) {
      idx = 0;
      while (idx < N_PROG && ((pb[idx >> 6] >> (idx & 63)) & 1)) ++idx;
      if (idx < N_PROG) break;
      if (!deadline) deadline = deadline_now();
      consume_addr_tick(gpeer);
      consume_done_tick(gpeer);
      if (!wait_pred_tick(deadline)) { idx = 0; break; }
    }
    pb[idx >> 6] |= 1ull << (idx & 63);
    u64 seq = ++sq.rndzv_addr_tx[gpeer];
    u32 slot = u32((seq - 1) % cfg.n_rndzv);
    st_sys(tv.direct_word(me(), gpeer, idx), 0);  // reset window progress
    RndzvRec* r = tv.rndzv_addr(gpeer, me(), slot);
    u64 val[8] = {seq, (u64(arith) << 32) | tag, offset, count, idx, 0, 0, 0};
    publish_rec((volatile u64*)r, val);
    return idx;
  }
  // progress-word pool index carried in the record (decoupled from the
  // addr-ring slot, which recycles while windows are pending)
  ACCL_HD static u32 rec_slot(const RndzvRec& rec) {
    return u32(rec.prog_idx);
  }

  // Wait for an addr record from gpeer MATCHING want_tag; non-matching
  // in-seq records spill to a pending set so overlapping rendezvous ops
  // match out of order (reference: the RNDZV_PENDING spill queue,
  // ccl_offload_control.c:154-212 + rxbuf_seek-style any-order matching).
  ACCL_HD bool wait_addr(u32 gpeer, u32 want_tag, RndzvRec& out) {
    u64 deadline = deadline_now();
    for (;;) {
      int best = -1;
      for (u32 k = 0; k < RNDZV_PEND; ++k) {
        PendRndzv& p = cold->pa[gpeer][k];
        if (!p.valid) continue;
        bool m = want_tag == TAG_ANY ? any_ok(p.tag)
                 : (p.tag == want_tag ||
                    (p.tag == TAG_ANY && any_ok(want_tag)));
        if (!m) continue;
        if (best < 0 || p.seq < cold->pa[gpeer][best].seq) best = int(k);
      }
      if (best >= 0) {
        PendRndzv& p = cold->pa[gpeer][best];
        out.seq = p.seq; out.tag = p.tag; out.arith = p.arith;
        out.offset = p.offset; out.count = p.count;
        out.prog_idx = p.prog;
        p.valid = 0;
        return true;
      }
      u64 seq = sq.rndzv_addr_rx[gpeer] + 1;
      RndzvRec* r = tv.rndzv_addr(me(), gpeer, u32((seq - 1) % cfg.n_rndzv));
      if (ld_sys(&r->seq) == seq) {
        fence_acquire_sys();
        // copy the record BEFORE publishing consumption — the poster may
        // reuse the slot the instant addr_ret advances
        u32 rtag = r->tag, rarith = r->arith;
        u64 roff = r->offset, rcnt = r->count, rprog = r->prog_idx;
        sq.rndzv_addr_rx[gpeer] = seq;
        st_sys(&tv.chan_ctl(gpeer, me())->addr_ret, seq);
        bool m = want_tag == TAG_ANY ? any_ok(rtag)
                 : (rtag == want_tag || (rtag == TAG_ANY && any_ok(want_tag)));
        if (m) {
          out.seq = seq; out.tag = rtag; out.arith = rarith;
          out.offset = roff; out.count = rcnt;
          out.prog_idx = rprog;
          return true;
        }
        u32 k = 0;
        while (k < RNDZV_PEND && cold->pa[gpeer][k].valid) ++k;
        if (k >= RNDZV_PEND) { err |= E_RNDZV; return false; }
        cold->pa[gpeer][k] = PendRndzv{seq, roff, rcnt, rtag, rarith, 1,
                                       u32(rprog)};
        continue;
      }
      wp_kind_ = 1;
      wp_info_ = u64(gpeer) | (u64(want_tag) << 32);
      wp_seq_ = (seq & 0xFFFFFFFFull) | (ld_sys(&r->seq) << 32);
      if (!wait_pred_tick(deadline)) return false;
    }
  }
  // probe: is a TAG-MATCHING addr record available right now? Foreign-tag
  // head records are SPILLED into the pending set while probing (their
  // owners match them from there), so a record queued BEHIND a foreign
  // head is still discoverable — without this, a parked send whose sync
  // caller blocks the next submission can deadlock (its record second in
  // the ring, nothing ever consuming the first). A matching head must NOT
  // make the probe commit blindly either: committing on a foreign head
  // would block the engine inside wait_addr.
  ACCL_HD bool addr_ready(u32 gpeer, u32 want_tag) {
    for (;;) {
      for (u32 k = 0; k < RNDZV_PEND; ++k) {
        PendRndzv& p = cold->pa[gpeer][k];
        if (p.valid && (want_tag == TAG_ANY
                            ? any_ok(p.tag)
                            : (p.tag == want_tag ||
                               (p.tag == TAG_ANY && any_ok(want_tag)))))
          return true;
      }
      u64 seq = sq.rndzv_addr_rx[gpeer] + 1;
      RndzvRec* r = tv.rndzv_addr(me(), gpeer, u32((seq - 1) % cfg.n_rndzv));
      if (ld_sys(&r->seq) != seq) return false;
      fence_acquire_sys();
      u32 t = r->tag;
      if (want_tag == TAG_ANY ? any_ok(t)
                              : (t == want_tag ||
                                 (t == TAG_ANY && any_ok(want_tag))))
        return true;
      // spill the foreign head to pending and keep looking
      u32 k = 0;
      while (k < RNDZV_PEND && cold->pa[gpeer][k].valid) ++k;
      if (k >= RNDZV_PEND) return false;  // pending full: try again later
      cold->pa[gpeer][k] = PendRndzv{seq, r->offset, r->count, r->tag,
                                     r->arith, 1, u32(r->prog_idx)};
      sq.rndzv_addr_rx[gpeer] = seq;
      st_sys(&tv.chan_ctl(gpeer, me())->addr_ret, seq);
    }
  }
  // probe: is a tag-matching DONE record available right now? (mirror of
  // addr_ready — foreign heads spill to the pending set). Needed because a
  // probe-mode recv must never block in wait_done: the sender may be a
  // PARKED call on the peer whose post_done only happens at ITS retry,
  // and that retry needs the peer engine free (mutual-block otherwise).
  ACCL_HD bool done_ready(u32 gpeer, u32 want_tag) {
    for (;;) {
      for (u32 k = 0; k < RNDZV_PEND; ++k) {
        PendRndzv& p = cold->pd[gpeer][k];
        if (p.valid && (want_tag == TAG_ANY
                            ? any_ok(p.tag)
                            : (p.tag == want_tag ||
                               (p.tag == TAG_ANY && any_ok(want_tag)))))
          return true;
      }
      u64 seq = sq.rndzv_done_rx[gpeer] + 1;
      RndzvRec* r = tv.rndzv_done(me(), gpeer, u32((seq - 1) % cfg.n_rndzv));
      if (ld_sys(&r->seq) != seq) return false;
      fence_acquire_sys();
      u32 t = r->tag;
      if (want_tag == TAG_ANY ? any_ok(t)
                              : (t == want_tag ||
                                 (t == TAG_ANY && any_ok(want_tag))))
        return true;
      u32 k = 0;
      while (k < RNDZV_PEND && cold->pd[gpeer][k].valid) ++k;
      if (k >= RNDZV_PEND) return false;  // pending full: try again later
      cold->pd[gpeer][k] = PendRndzv{seq, 0, 0, r->tag, 0, 1};
      sq.rndzv_done_rx[gpeer] = seq;
      st_sys(&tv.chan_ctl(gpeer, me())->done_ret, seq);
    }
  }

  ACCL_HD void post_done(u32 gpeer, u32 tag) {
    if (!done_room(gpeer)) {
      u64 deadline = deadline_now();
      while (!done_room(gpeer)) {
        consume_addr_tick(gpeer);
        consume_done_tick(gpeer);
        if (!wait_pred_tick(deadline)) break;
      }
    }
    u64 seq = ++sq.rndzv_done_tx[gpeer];
    RndzvRec* r = tv.rndzv_done(gpeer, me(), u32((seq - 1) % cfg.n_rndzv));
    u64 val[8] = {seq, tag, 0, 0, 0, 0, 0, 0};
    publish_rec((volatile u64*)r, val);
  }
  ACCL_HD bool wait_done(u32 gpeer, u32 want_tag) {
    u64 deadline = deadline_now();
    for (;;) {
      int best = -1;
      for (u32 k = 0; k < RNDZV_PEND; ++k) {
        PendRndzv& p = cold->pd[gpeer][k];
        if (!p.valid) continue;
        bool m = want_tag == TAG_ANY ? any_ok(p.tag)
                 : (p.tag == want_tag ||
                    (p.tag == TAG_ANY && any_ok(want_tag)));
        if (!m) continue;
        if (best < 0 || p.seq < cold->pd[gpeer][best].seq) best = int(k);
      }
      if (best >= 0) { cold->pd[gpeer][best].valid = 0; return true; }
      u64 seq = sq.rndzv_done_rx[gpeer] + 1;
      RndzvRec* r = tv.rndzv_done(me(), gpeer, u32((seq - 1) % cfg.n_rndzv));
      if (ld_sys(&r->seq) == seq) {
        fence_acquire_sys();
        u32 rtag = r->tag;  // copy before publishing consumption
        sq.rndzv_done_rx[gpeer] = seq;
        st_sys(&tv.chan_ctl(gpeer, me())->done_ret, seq);
        if (want_tag == TAG_ANY ? any_ok(rtag)
                                : (rtag == want_tag ||
                                   (rtag == TAG_ANY && any_ok(want_tag))))
          return true;
        u32 k = 0;
        while (k < RNDZV_PEND && cold->pd[gpeer][k].valid) ++k;
        if (k >= RNDZV_PEND) { err |= E_RNDZV; return false; }
        cold->pd[gpeer][k] = PendRndzv{seq, 0, 0, rtag, 0, 1};
        continue;
      }
      wp_kind_ = 2;
      wp_info_ = u64(gpeer) | (u64(want_tag) << 32);
      wp_seq_ = (seq & 0xFFFFFFFFull) | (ld_sys(&r->seq) << 32);
      if (!wait_pred_tick(deadline)) return false;
    }
  }

  // ================= collectives =================
  // All take global rank ids resolved through CommView. `d` fields follow
  // CallDesc. Returns error bits (0 = ok).

  ACCL_HD u32 op_copy(const CallDesc& d) {
    stamp(8);
    u64 n = desc_count(d);
    if (d.flags & F_SRC_STREAM)  // stream2mem: drain ring lane addr0
      return stream_fed(u32(d.addr0), local_ptr(d.addr2, d.flags & F_DST_ARENA),
                        desc_dtype(d), n, -1, desc_dtype(d), 0);
    if (d.flags & F_DST_PEER) {
      // one-sided put: write straight into the peer's mapped arena over
      // xGMI (reference copy_p2p semantics); user-level synchronization
      u32 peer = d.root_src_dst < cfg.nranks ? d.root_src_dst : cfg.rank;
      mk_local(0, local_ptr(d.addr0, d.flags & F_SRC_ARENA), desc_dtype(d),
               tv.arena[peer] + d.addr2, desc_dtype(d), n);
      return run_flows(1);
    }
    mk_local(0, local_ptr(d.addr0, d.flags & F_SRC_ARENA), desc_dtype(d),
             local_ptr(d.addr2, d.flags & F_DST_ARENA), desc_dtype(d), n);
    stamp(9);
    u32 e = run_flows(1);
    stamp(11);
    return e;
  }

  ACCL_HD u32 op_combine(const CallDesc& d) {
    // reference: combine (ccl_offload_control.c:551-569)
    u64 n = desc_count(d);
    mk_local(0, local_ptr(d.addr0, d.flags & F_SRC_ARENA), desc_dtype(d),
             local_ptr(d.addr2, d.flags & F_DST_ARENA), desc_dtype(d), n,
             local_ptr(d.addr1, d.flags & F_OP1_ARENA), desc_dtype(d),
             int(d.function));
    return run_flows(1);
  }

  // protocol selection shared by both sides (reference: eager if bytes <=
  // max_eager or compressed, ccl_offload_control.c:587-610)
  ACCL_HD bool use_rndzv(u64 count, DataType dt, DataType wdt) {
    return dt == wdt && count * dtype_size(dt) > max_eager_bytes;
  }

  ACCL_HD u32 op_send(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    u32 peer = c.global(d.root_src_dst);
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    if (d.flags & F_SRC_STREAM)  // send-from-stream (OP0_STREAM analogue)
      return stream_fed(u32(d.addr0), nullptr, dt, n, i64(peer), wdt, d.tag);
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    // self-send uses the generic eager path (loopback slots): it is
    // probe-aware, so a large self-send parks on credit instead of
    // committing the engine to a wait only its own matching recv (queued
    // BEHIND it) could satisfy. Rendezvous is peer-only — op_recv matches
    // self traffic on the eager channel.
    if (use_rndzv(n, dt, wdt) && peer != me()) {
      // follow the receiver's posted windows (tag-matched, out of order
      // w.r.t. other rendezvous ops on this pair). Flow-free resumable
      // pushes in BOTH modes: a probe parks when no window is posted
      // (committing to an in-op wait risks the cross-rank deadlock); the
      // blocking fallback (park table full) resumes from the same
      // ParkState — it must never resend windows a probe already pushed.
      if (probe_) {
        if (!rndzv_send_push(peer, d, *ps_)) return E_NOT_READY;
        if (!done_room(peer)) return E_NOT_READY;  // never block in probe
        post_done(peer, d.tag);
        return E_OK;
      }
      ParkState local{};
      ParkState& ps = ps_ ? *ps_ : local;
      u64 deadline = deadline_now();
      while (!rndzv_send_push(peer, d, ps)) {
        if (err) return err;
        if (!wait_pred_tick(deadline)) return err;
      }
      post_done(peer, d.tag);
      return E_OK;
    }
    // eager: flow-free per-segment pushes (<= max_eager, so at most a
    // couple of slot-sized segments on production geometry); probe mode
    // parks on no-credit with progress saved, blocking mode waits
    u64 sent = (ps_ && ps_->step == 1) ? ps_->w[0] : 0;
    u64 deadline = deadline_now();
    while (sent < n) {
      u64 did = eager_send_segment(peer, src, dt, wdt, sent, n - sent, n,
                                   d.tag);
      if (err) return err;
      if (!did) {
        if (probe_) {
          if (ps_) { ps_->w[0] = sent; ps_->step = 1; }
          return E_NOT_READY;
        }
        if (!wait_pred_tick(deadline)) return err;
        continue;
      }
      sent += did;
      if (ps_) { ps_->w[0] = sent; ps_->step = 1; }
      deadline = deadline_now();
    }
    return E_OK;
  }

  // flow-free rendezvous window pushes (parked-send resume/drain): true
  // when the whole message has been written
  ACCL_HD bool rndzv_send_push(u32 peer, const CallDesc& d, ParkState& ps) {
    u64 n = desc_count(d);
    DataType dt = desc_dtype(d);
    const u32 esz = dtype_size(dt);
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    u64 sent = (ps.step == 1) ? ps.w[0] : 0;
    while (sent < n) {
      if (!addr_ready(peer, d.tag)) {
        ps.w[0] = sent;
        ps.step = 1;
        return false;
      }
      RndzvRec rec{};
      if (!wait_addr(peer, d.tag, rec)) return false;  // immediate (ready)
      u64 w = min64(n - sent, rec.count);
      MoveDesc m{};
      m.dst = (u64)(tv.arena[peer] + rec.offset);
      m.dst_dt = u8(rec.arith ? rec.arith : u32(dt));
      m.src[0] = (u64)(src + sent * esz);
      m.src_dt[0] = u8(dt);
      m.nsrc = 1;
      m.count = w;
      u32 tok = mv->submit(m);
      u64 deadline = deadline_now();
      while (!mv->poll(tok))
        if (!wait_pred_tick_nodrain(deadline)) return false;
      fence_release_sys();
      st_sys(tv.direct_word(peer, me(), rec_slot(rec)),
             w * u64(dtype_size(DataType(m.dst_dt))));
      sent += w;
      ps.w[0] = sent;
      ps.step = 1;
    }
    return true;
  }

  ACCL_HD u32 spill_slot_count() {
    u64 ns = (tv.hdr(me())->spare_bytes / 2) / cfg.slot_bytes;
    return ns > 64 ? 64 : u32(ns);
  }
  ACCL_HD char* spill_ptr(u32 sp) {
    const ArenaHdr* h = tv.hdr(me());
    return tv.arena[me()] + h->spare_off + h->spare_bytes / 2 +
           u64(sp) * cfg.slot_bytes;
  }

  // move the head segment of channel (peer -> me) into the spill pool and
  // release the rx slot; false when queue/pool is full (caller keeps
  // waiting). NON-REENTRANT w.r.t. the flow table: the copy goes straight
  // to the mover (no flows), so it is safe to call from INSIDE
  // flow_submit when a flow meets a foreign-tag head segment.
  ACCL_HD bool spill_head(u32 peer, const SlotHdr* hd, u32 sl, u64 seq) {
    if (cold->uq_t[peer] - cold->uq_h[peer] >= UQ_DEPTH) return false;
    u32 ns = spill_slot_count();
    u32 sp = 0;
    while (sp < ns && ((spill_busy >> sp) & 1)) ++sp;
    if (sp >= ns || hd->bytes > cfg.slot_bytes) return false;
    MoveDesc m{};
    m.dst = (u64)spill_ptr(sp);
    m.dst_dt = u8(DataType::int8);
    m.src[0] = (u64)tv.slot_payload(me(), peer, sl);
    m.src_dt[0] = u8(DataType::int8);
    m.nsrc = 1;
    m.count = hd->bytes;
    u32 tok = mv->submit(m);
    u64 deadline = deadline_now();
    while (!mv->poll(tok))
      if (!wait_pred_tick(deadline)) return false;
    Unexpected& u = cold->uq[peer][cold->uq_t[peer] % UQ_DEPTH];
    u.tag = hd->tag; u.arith = hd->arith; u.bytes = hd->bytes;
    u.msg_count = hd->msg_count; u.flags = hd->flags; u.spare_slot = sp;
    spill_busy |= 1ull << sp;
    cold->uq_t[peer]++;
    sq.eager_rx[peer] = seq;
    ret_credit(peer, seq);
    return true;
  }

  // eager receive with out-of-order tag matching: drain matching spilled
  // segments first, then the live channel; mismatched head segments are
  // spilled so a different-tag recv can run ahead (MPI matching semantics,
  // reference rxbuf_seek + pending queue).
  ACCL_HD u32 op_recv_eager(u32 peer, char* dst, DataType ddt, DataType wdt,
                            u64 n, u32 want_tag) {
    u64 got = 0;
    u32 mtag = want_tag;
    // resumable in probe mode: partial progress (and the matched tag, if
    // TAG_ANY) survives a park — the retry continues where it left off
    if (probe_ && ps_ && ps_->step == 2) {
      got = ps_->w[0];
      mtag = u32(ps_->w[1]);
    }
    const u32 wsz = dtype_size(wdt), dsz = dtype_size(ddt);
    u64 deadline = deadline_now();
    while (got < n) {
      bool progressed = false;
      // 1) spill queue, in arrival order (per-tag FIFO preserved)
      for (u32 qi = cold->uq_h[peer]; qi != cold->uq_t[peer]; ++qi) {
        Unexpected& u = cold->uq[peer][qi % UQ_DEPTH];
        if (u.bytes == 0) continue;  // consumed hole
        if (mtag == TAG_ANY ? !any_ok(u.tag) : u.tag != mtag) continue;
        if (u.arith != u32(wdt)) { err |= E_COMPRESSION; return err; }
        if (got == 0) mtag = u.tag;
        else if (u.tag != mtag) continue;
        u64 nseg = u.bytes / wsz;
        if (nseg > n - got) { err |= E_SEGMENT; return err; }
        mk_local(0, spill_ptr(u.spare_slot), wdt, dst + got * dsz, ddt, nseg);
        u32 e = run_flows(1);
        if (e) return e;
        spill_busy &= ~(1ull << u.spare_slot);
        u.bytes = 0;
        while (cold->uq_h[peer] != cold->uq_t[peer] &&
               cold->uq[peer][cold->uq_h[peer] % UQ_DEPTH].bytes == 0)
          cold->uq_h[peer]++;
        got += nseg;
        progressed = true;
        deadline = deadline_now();
        break;
      }
      if (progressed) continue;
      if (got >= n) break;
      // 2) live channel head
      u64 seq = sq.eager_rx[peer] + 1;
      u32 sl = u32((seq - 1) % cfg.n_slots);
      SlotHdr* hd = tv.slot_hdr(me(), peer, sl);
      if (ld_sys(&hd->seq) == seq) {
        fence_acquire_sys();
        bool match = mtag == TAG_ANY ? any_ok(hd->tag) : (hd->tag == mtag);
        if (match) {
          if (hd->arith != u32(wdt)) { err |= E_COMPRESSION; return err; }
          if (got == 0) mtag = hd->tag;
          u64 nseg = u64(hd->bytes) / wsz;
          if (nseg > n - got) { err |= E_SEGMENT; return err; }
          mk_local(0, tv.slot_payload(me(), peer, sl), wdt, dst + got * dsz,
                   ddt, nseg);
          u32 e = run_flows(1);
          if (e) return e;
          sq.eager_rx[peer] = seq;
          ret_credit(peer, seq);
          got += nseg;
          deadline = deadline_now();
          continue;
        }
        if (spill_head(peer, hd, sl, seq)) { deadline = deadline_now(); continue; }
        if (err) return err;
      }
      // probe mode NEVER spins: park (with progress) whenever the channel
      // has nothing for us right now — committing to an in-op wait here is
      // the classic cross-rank deadlock (peer blocked behind its own op)
      if (probe_) {
        if (ps_) { ps_->w[0] = got; ps_->w[1] = mtag; ps_->step = 2; }
        return E_NOT_READY;
      }
      if (!wait_pred_tick(deadline)) return err;
    }
    return E_OK;
  }

  ACCL_HD u32 op_recv(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    u32 peer = c.global(d.root_src_dst);
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    if (use_rndzv(n, dt, wdt) && peer != me()) {
      const u32 esz = dtype_size(dt);
      // resumed park in the final phase: all data landed, only the done
      // record outstanding (probe-mode ops must never block in wait_done —
      // the sender may be a parked call whose post_done happens at ITS
      // retry, which needs the peer engine free)
      if (probe_ && ps_ && ps_->step == 3) {
        if (!done_ready(peer, d.tag)) return E_NOT_READY;
        return wait_done(peer, d.tag) ? E_OK : err;
      }
      if (d.flags & F_DST_ARENA) {
        // window the posting by max_rndzv_bytes (reference:
        // set_max_rendezvous_size caps a single rendezvous transfer);
        // sender follows each posted window (op_send's wait_addr loop).
        // Resumable: a parked re-entry restores the window cursors from
        // ParkState instead of re-posting (the current_step analogue).
        u64 wmax = max_rndzv_bytes ? max_rndzv_bytes / esz : n;
        if (!wmax) wmax = 1;
        u64 posted = 0, got = 0;
        u32 wslot[2] = {0, 0}; u64 wcnt[2] = {0, 0}; u32 wi = 0, nw = 0;
        if (ps_ && ps_->step == 1) {
          posted = ps_->w[0]; got = ps_->w[1];
          wcnt[0] = ps_->w[2]; wcnt[1] = ps_->w[3];
          wslot[0] = u32(ps_->w[4]); wslot[1] = u32(ps_->w[4] >> 32);
          wi = u32(ps_->w[5]); nw = u32(ps_->w[5] >> 32);
        }
        while (got < n) {
          while (posted < n && nw < 2) {
            if (probe_ && !addr_room(peer)) break;  // never block in probe
            u64 w = min64(n - posted, wmax);
            wslot[(wi + nw) % 2] =
                post_addr(peer, d.addr2 + posted * esz, w, d.tag, u32(dt));
            wcnt[(wi + nw) % 2] = w;
            posted += w; nw++;
          }
          if (probe_ &&
              (nw == 0 || (got == 0 && ld_sys(tv.direct_word(
                                           me(), peer, wslot[wi])) == 0))) {
            if (ps_) {
              ps_->w[0] = posted; ps_->w[1] = got;
              ps_->w[2] = wcnt[0]; ps_->w[3] = wcnt[1];
              ps_->w[4] = u64(wslot[0]) | (u64(wslot[1]) << 32);
              ps_->w[5] = u64(wi) | (u64(nw) << 32);
              ps_->step = 1;
            }
            return E_NOT_READY;
          }
          mk_rx_direct(0, peer, wcnt[wi], dt, wslot[wi]);
          u32 e = run_flows(1);
          if (e) return e;
          got += wcnt[wi];
          wi ^= 1; nw--;
        }
        if (probe_ && !done_ready(peer, d.tag)) {
          if (ps_) ps_->step = 3;
          return E_NOT_READY;
        }
        return wait_done(peer, d.tag) ? E_OK : err;
      }
      // stage through the spare region's LOWER half in windows (double-
      // buffered); the upper half is the unexpected-message spill pool
      ArenaHdr* h = tv.hdr(me());
      u64 bank = (h->spare_bytes / 4) / esz;  // elems per staging bank
      if (max_rndzv_bytes) bank = min64(bank, max_rndzv_bytes / esz);
      if (!bank) { err |= E_INVALID_ARG; return err; }
      u64 posted = 0, got = 0;
      u32 wslot[2]; u64 wcnt[2], woff[2]; u32 wi = 0, nw = 0; int cur = 0;
      while (got < n) {
        while (posted < n && nw < 2) {
          u64 w = min64(n - posted, bank);
          u64 off = h->spare_off + u64(cur) * (h->spare_bytes / 4);
          wslot[(wi + nw) % 2] = post_addr(peer, off, w, d.tag, u32(dt));
          wcnt[(wi + nw) % 2] = w;
          woff[(wi + nw) % 2] = off;
          posted += w; cur ^= 1; nw++;
        }
        mk_rx_direct(0, peer, wcnt[wi], dt, wslot[wi]);
        u32 e = run_flows(1);
        if (e) return e;
        mk_local(0, tv.arena[me()] + woff[wi], dt, dst + got * esz, dt,
                 wcnt[wi]);
        if ((e = run_flows(1))) return e;
        got += wcnt[wi];
        wi ^= 1; nw--;
      }
      if (probe_ && !done_ready(peer, d.tag)) {
        if (ps_) ps_->step = 3;
        return E_NOT_READY;
      }
      return wait_done(peer, d.tag) ? E_OK : err;
    }
    return op_recv_eager(peer, dst, dt, wdt, n, d.tag);
  }

  // fullmesh bcast: root pushes to every peer; direct when dst offsets can
  // be exchanged (arena), else eager. reference: broadcast
  // (ccl_offload_control.c:796-988 — binary/flat tree; xGMI is all-to-all
  // so a flat push uses P-1 links at once and needs no relay).
  ACCL_HD u32 op_bcast(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    u32 root = d.root_src_dst;
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    bool rndzv = use_rndzv(n, dt, wdt) && (d.flags & F_DST_ARENA);
    u32 tag = TAG_COLL | (u32(Op::bcast) << 16) | d.comm_id;
    if (c.size == 1) return E_OK;
    if (c.rank == root) {
      const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
      u32 nf = 0;
      for (u32 p = 0; p < c.size; ++p) {
        if (p == root) continue;
        if (rndzv) {
          RndzvRec rec{};
          if (!wait_addr(c.global(p), tag, rec)) return err;
          mk_tx_direct(nf++, c.global(p), src, dt, dt, n, rec.offset,
                       rec_slot(rec));
        } else {
          mk_tx(nf++, c.global(p), src, dt, wdt, n, tag);
        }
      }
      return run_flows(nf);
    }
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    if (rndzv) {
      u32 s = post_addr(c.global(root), d.addr2, n, tag, u32(dt));
      mk_rx_direct(0, c.global(root), n, dt, s);
      return run_flows(1);
    }
    mk_rx(0, c.global(root), dst, dt, wdt, n, tag);
    return run_flows(1);
  }

  // both src and dst arena-resident, large, same wire dtype -> the
  // address-exchange direct path is legal on every rank (flags must agree
  // across ranks, same contract as the existing bcast/allgather direct)
  ACCL_HD bool coll_direct_ok(const CallDesc& d, u64 n, DataType dt,
                              DataType wdt) const {
    return dt == wdt && n * dtype_size(dt) > max_eager_bytes &&
           (d.flags & F_SRC_ARENA) && (d.flags & F_DST_ARENA);
  }

  // reference: scatter (ccl_offload_control.c:992-1123 — the rendezvous
  // variant there is root self-copy + out-of-order addr-driven writes,
  // :1010-1070; ours: every leaf posts its dst window, root direct-writes
  // all P-1 links concurrently)
  ACCL_HD u32 op_scatter(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);  // per-rank count
    u32 root = d.root_src_dst;
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::scatter) << 16) | d.comm_id;
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    bool direct = coll_direct_ok(d, n, dt, wdt) && c.size <= MAX_FLOWS;
    if (c.rank == root) {
      const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
      u32 nf = 0;
      mk_local(nf++, src + u64(root) * n * dtype_size(dt), dt, dst, dt, n);
      for (u32 p = 0; p < c.size; ++p) {
        if (p == root) continue;
        const char* s = src + u64(p) * n * dtype_size(dt);
        if (direct) {
          RndzvRec rec{};
          if (!wait_addr(c.global(p), tag, rec)) return err;
          mk_tx_direct(nf++, c.global(p), s, dt, dt, n, rec.offset,
                       rec_slot(rec));
        } else {
          mk_tx(nf++, c.global(p), s, dt, wdt, n, tag);
        }
      }
      return run_flows(nf);
    }
    if (direct) {
      u32 s = post_addr(c.global(root), d.addr2, n, tag, u32(dt));
      mk_rx_direct(0, c.global(root), n, dt, s);
      return run_flows(1);
    }
    mk_rx(0, c.global(root), dst, dt, wdt, n, tag);
    return run_flows(1);
  }

  // reference: gather (ccl_offload_control.c:1128-1294; rendezvous flat
  // tree :1146-1184 — leaves write into the root's posted dst slots; ours:
  // root posts per-leaf windows, every inbound link lands concurrently)
  ACCL_HD u32 op_gather(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    u32 root = d.root_src_dst;
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::gather) << 16) | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    bool direct = coll_direct_ok(d, n, dt, wdt) && c.size <= MAX_FLOWS;
    if (c.rank == root) {
      char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
      u32 nf = 0;
      mk_local(nf++, src, dt, dst + u64(root) * n * dtype_size(dt), dt, n);
      if (direct) {
        for (u32 p = 0; p < c.size; ++p) {
          if (p == root) continue;
          u32 s = post_addr(c.global(p), d.addr2 + u64(p) * n * dtype_size(dt),
                            n, tag, u32(dt));
          mk_rx_direct(nf++, c.global(p), n, dt, s);
        }
        return run_flows(nf);
      }
      for (u32 p = 0; p < c.size; ++p) {
        if (p == root) continue;
        mk_rx(nf++, c.global(p), dst + u64(p) * n * dtype_size(dt), dt, wdt, n,
              tag);
      }
      return run_flows(nf);
    }
    if (direct) {
      RndzvRec rec{};
      if (!wait_addr(c.global(root), tag, rec)) return err;
      mk_tx_direct(0, c.global(root), src, dt, dt, n, rec.offset,
                   rec_slot(rec));
      return run_flows(1);
    }
    mk_tx(0, c.global(root), src, dt, wdt, n, tag);
    return run_flows(1);
  }

  // fullmesh allgather (reference: ring store-and-forward,
  // ccl_offload_control.c:1297-1503; fullmesh pushes each chunk once over
  // its own link instead of P-1 relay hops)
  ACCL_HD u32 op_allgather(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::allgather) << 16) | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    u32 r = c.rank;
    if (2 * u64(c.size) - 1 > MAX_FLOWS) return allgather_batched(d, c);
    u32 nf = 0;
    mk_local(nf++, src, dt, dst + u64(r) * n * dtype_size(dt), dt, n);
    bool direct = use_rndzv(n, dt, wdt) && (d.flags & F_DST_ARENA) &&
                  (d.flags & F_SRC_ARENA) && 2 * u64(c.size) - 1 <= MAX_FLOWS;
    if (direct) {
      // to peer p I post the slot for P'S DATA in MY dst (addr2 + p*n) —
      // the record's offset is where the CONSUMING sender writes in MY
      // arena (bug history: posting my OWN block index here sent every
      // peer's vector to the same region; it survived all small tests
      // because an opts-plumbing bug kept max_eager at the 4 MiB default,
      // so the direct branch never actually ran below 4 MiB)
      u32 myslot[MAX_RANKS];
      for (u32 p = 0; p < c.size; ++p)
        if (p != r)
          myslot[p] = post_addr(c.global(p),
                                d.addr2 + u64(p) * n * dtype_size(dt), n, tag,
                                u32(dt));
      for (u32 p = 0; p < c.size; ++p) {
        if (p == r) continue;
        RndzvRec rec{};
        if (!wait_addr(c.global(p), tag, rec)) return err;
        mk_tx_direct(nf++, c.global(p), src, dt, dt, n, rec.offset,
                     rec_slot(rec));
        mk_rx_direct(nf++, c.global(p), n, dt, myslot[p]);
      }
      return run_flows(nf);
    }
    for (u32 p = 0; p < c.size; ++p) {
      if (p == r) continue;
      mk_tx(nf++, c.global(p), src, dt, wdt, n, tag);
      mk_rx(nf++, c.global(p), dst + u64(p) * n * dtype_size(dt), dt, wdt, n, tag);
    }
    return run_flows(nf);
  }

  // large-P fallback: (tx,rx) pairs in batches that fit the flow table
  // (P > MAX_FLOWS/2 would overflow flows[] — LDS corruption otherwise)
  ACCL_HD u32 allgather_batched(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::allgather) << 16) | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    u32 r = c.rank;
    mk_local(0, src, dt, dst + u64(r) * n * dtype_size(dt), dt, n);
    u32 e = run_flows(1);
    if (e) return e;
    const u32 B = (MAX_FLOWS / 2) - 1;
    for (u32 base = 0; base < c.size; base += B) {
      u32 nf = 0;
      for (u32 j = base; j < c.size && j < base + B; ++j) {
        if (j == r) continue;
        mk_tx(nf++, c.global(j), src, dt, wdt, n, tag);
        mk_rx(nf++, c.global(j), dst + u64(j) * n * dtype_size(dt), dt, wdt,
              n, tag);
      }
      if (nf && (e = run_flows(nf))) return e;
    }
    return E_OK;
  }

  // ---- windowed n-ary direct fan-in (the reduce family's large-message
  // path; reference: flat-tree reduce with spare-buffer scratchpads,
  // ccl_offload_control.c:1531-1602, redesigned MI355X-first: P-1 peers
  // direct-write window stages over their own xGMI links concurrently,
  // then ONE n-ary mover pass folds stage slots + own chunk into dst —
  // root HBM traffic is O(P*read + write) per window, depth O(1)) ----
  struct FanGeom { u64 bank_off[2]; u64 slot_bytes; u64 W; };
  ACCL_HD bool fan_geom(u32 P, u32 esz, FanGeom& g) {
    const ArenaHdr* h = tv.hdr(me());
    u64 bank_bytes = (h->spare_bytes / 4) & ~4095ull;
    g.slot_bytes = (bank_bytes / (P - 1)) & ~255ull;
    g.W = g.slot_bytes / esz;
    if (max_rndzv_bytes) g.W = min64(g.W, max_rndzv_bytes / esz);
    g.bank_off[0] = h->spare_off;
    g.bank_off[1] = h->spare_off + bank_bytes;
    return g.W != 0;
  }

  // dst[0..count) = f(first, extra[0], ..., extra[nx-1]), chained in groups
  // of MOVE_MAX_SRC sources (one mover pass per group; later groups
  // accumulate into dst)
  ACCL_HD u32 nary_reduce(char* dst, const char* first, DataType dt, u64 count,
                          const char* const* extra, u32 nx, int func) {
    u32 k = 0;
    const char* head = first;
    for (;;) {
      MoveDesc m{};
      m.dst = (u64)dst;
      m.dst_dt = u8(dt);
      m.count = count;
      m.func = u32(func);
      m.src[0] = (u64)head;
      m.src_dt[0] = u8(dt);
      m.nsrc = 1;
      while (m.nsrc < MOVE_MAX_SRC && k < nx) {
        m.src[m.nsrc] = (u64)extra[k];
        m.src_dt[m.nsrc] = u8(dt);
        m.nsrc++;
        k++;
      }
      u32 tok = mv->submit(m);
      u64 deadline = deadline_now();
      while (!mv->poll(tok))
        if (!wait_pred_tick(deadline)) return err;
      if (k >= nx) return E_OK;
      head = dst;
    }
  }

  // non-root side of the direct reduce family: follow the root's posted
  // stage windows (tag-matched)
  ACCL_HD u32 reduce_direct_leaf(const CommView& c, u32 root, const char* src,
                                 DataType dt, u64 n, u32 tag) {
    const u32 esz = dtype_size(dt);
    u64 sent = 0;
    while (sent < n) {
      RndzvRec rec{};
      if (!wait_addr(c.global(root), tag, rec)) return err;
      u64 w = min64(n - sent, rec.count);
      mk_tx_direct(0, c.global(root), src + sent * esz, dt, dt, w, rec.offset,
                   rec_slot(rec));
      u32 e = run_flows(1);
      if (e) return e;
      sent += w;
    }
    return E_OK;
  }

  ACCL_HD u32 reduce_direct_root(const CallDesc& d, const CommView& c,
                                 u32 tag) {
    u64 n = desc_count(d);
    DataType dt = desc_dtype(d);
    const u32 esz = dtype_size(dt);
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    const u32 P = c.size, r = c.rank;
    FanGeom g{};
    if (!fan_geom(P, esz, g)) { err |= E_INVALID_ARG; return err; }
    u64 nwin = (n + g.W - 1) / g.W;
    u32 sp[2][MAX_RANKS];
    u64 w0 = min64(n, g.W);
    for (u32 p = 0; p < P; ++p) {
      if (p == r) continue;
      u32 idx = p < r ? p : p - 1;
      sp[0][p] = post_addr(c.global(p), g.bank_off[0] + idx * g.slot_bytes,
                           w0, tag, u32(dt));
    }
    for (u64 w = 0; w < nwin; ++w) {
      int bank = int(w & 1);
      u64 off = w * g.W;
      u64 wc = min64(g.W, n - off);
      // post window w+1 into the other bank FIRST: senders stream it over
      // xGMI while we still collect/reduce window w
      if (w + 1 < nwin) {
        u64 nc = min64(g.W, n - (w + 1) * g.W);
        for (u32 p = 0; p < P; ++p) {
          if (p == r) continue;
          u32 idx = p < r ? p : p - 1;
          sp[bank ^ 1][p] =
              post_addr(c.global(p), g.bank_off[bank ^ 1] + idx * g.slot_bytes,
                        nc, tag, u32(dt));
        }
      }
      u32 nf = 0;
      for (u32 p = 0; p < P; ++p)
        if (p != r) mk_rx_direct(nf++, c.global(p), wc, dt, sp[bank][p]);
      u32 e = run_flows(nf);
      if (e) return e;
      const char* extra[MAX_RANKS];
      u32 nx = 0;
      for (u32 p = 0; p < P; ++p) {
        if (p == r) continue;
        u32 idx = p < r ? p : p - 1;
        extra[nx++] = tv.arena[me()] + g.bank_off[bank] + idx * g.slot_bytes;
      }
      e = nary_reduce(dst + off * esz, src + off * esz, dt, wc, extra, nx,
                      int(d.function));
      if (e) return e;
    }
    return E_OK;
  }

  // every rank runs root-and-leaf at once: direct-write window stages for
  // chunk p to rank p while collecting own chunk's stages from all peers
  // (reference ring reduce_scatter ccl_offload_control.c:1748-1852; ours
  // drives all xGMI links simultaneously)
  ACCL_HD u32 reduce_scatter_direct(const CallDesc& d, const CommView& c,
                                    u32 tag) {
    u64 n = desc_count(d);  // per-rank chunk
    DataType dt = desc_dtype(d);
    const u32 esz = dtype_size(dt);
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    const u32 P = c.size, r = c.rank;
    FanGeom g{};
    if (!fan_geom(P, esz, g)) { err |= E_INVALID_ARG; return err; }
    u64 nwin = (n + g.W - 1) / g.W;
    u32 sp[2][MAX_RANKS];
    u64 w0 = min64(n, g.W);
    for (u32 p = 0; p < P; ++p) {
      if (p == r) continue;
      u32 idx = p < r ? p : p - 1;
      sp[0][p] = post_addr(c.global(p), g.bank_off[0] + idx * g.slot_bytes,
                           w0, tag, u32(dt));
    }
    for (u64 w = 0; w < nwin; ++w) {
      int bank = int(w & 1);
      u64 off = w * g.W;
      u64 wc = min64(g.W, n - off);
      if (w + 1 < nwin) {
        u64 nc = min64(g.W, n - (w + 1) * g.W);
        for (u32 p = 0; p < P; ++p) {
          if (p == r) continue;
          u32 idx = p < r ? p : p - 1;
          sp[bank ^ 1][p] =
              post_addr(c.global(p), g.bank_off[bank ^ 1] + idx * g.slot_bytes,
                        nc, tag, u32(dt));
        }
      }
      u32 nf = 0;
      for (u32 p = 0; p < P; ++p) {
        if (p == r) continue;
        RndzvRec rec{};
        if (!wait_addr(c.global(p), tag, rec)) return err;
        mk_tx_direct(nf++, c.global(p), src + (u64(p) * n + off) * esz, dt,
                     dt, min64(wc, rec.count), rec.offset,
                     rec_slot(rec));
      }
      for (u32 p = 0; p < P; ++p)
        if (p != r) mk_rx_direct(nf++, c.global(p), wc, dt, sp[bank][p]);
      u32 e = run_flows(nf);
      if (e) return e;
      const char* extra[MAX_RANKS];
      u32 nx = 0;
      for (u32 p = 0; p < P; ++p) {
        if (p == r) continue;
        u32 idx = p < r ? p : p - 1;
        extra[nx++] = tv.arena[me()] + g.bank_off[bank] + idx * g.slot_bytes;
      }
      e = nary_reduce(dst + off * esz, src + (u64(r) * n + off) * esz, dt, wc,
                      extra, nx, int(d.function));
      if (e) return e;
    }
    return E_OK;
  }

  // ---- one-shot small-message fan-in (reference: flat-tree reduce for
  // small messages, ccl_offload_control.c:1531-1602). Each contributor's
  // whole message fits ONE eager slot; the consumer waits for all P-1 slot
  // headers (pool-aware) and folds slot payloads + own src into dst with a
  // single n-ary mover pass — one network round, no dst round-trips.
  // Returns E_OK and fills extras/meta, or E_NOT_READY-free blocking wait.
  struct SlotRef { const char* pay; u64 seq; u32 from_pool; u32 pool_qi; };
  ACCL_HD bool collect_one_slot(u32 gpeer, u32 tag, DataType wdt, u64 n,
                                SlotRef& out) {
    const u32 wsz = dtype_size(wdt);
    u64 deadline = deadline_now();
    for (;;) {
      // pool first (a parked recv's probe may have spilled it)
      for (u32 qi = cold->uq_h[gpeer]; qi != cold->uq_t[gpeer]; ++qi) {
        Unexpected& u = cold->uq[gpeer][qi % UQ_DEPTH];
        if (u.bytes == 0 || u.tag != tag) continue;
        if (u.arith != u32(wdt) || u.bytes / wsz != n) {
          err |= E_SEGMENT;
          return false;
        }
        out = SlotRef{spill_ptr(u.spare_slot), 0, 1, qi};
        return true;
      }
      u64 seq = sq.eager_rx[gpeer] + 1;
      u32 sl = u32((seq - 1) % cfg.n_slots);
      SlotHdr* h = tv.slot_hdr(me(), gpeer, sl);
      if (ld_sys(&h->seq) == seq) {
        fence_acquire_sys();
        if (h->tag == tag) {
          if (h->arith != u32(wdt) || u64(h->bytes) / wsz != n) {
            err |= E_SEGMENT;
            return false;
          }
          out = SlotRef{tv.slot_payload(me(), gpeer, sl), seq, 0, 0};
          drain_hold_ |= 1ull << (gpeer & 63);
          return true;
        }
        if (spill_head(gpeer, h, sl, seq)) { deadline = deadline_now(); continue; }
        if (err) return false;
      }
      if (!wait_pred_tick(deadline)) return false;
    }
  }
  ACCL_HD void release_slot(u32 gpeer, const SlotRef& s) {
    if (s.from_pool) {
      Unexpected& u = cold->uq[gpeer][s.pool_qi % UQ_DEPTH];
      spill_busy &= ~(1ull << u.spare_slot);
      u.bytes = 0;
      while (cold->uq_h[gpeer] != cold->uq_t[gpeer] &&
             cold->uq[gpeer][cold->uq_h[gpeer] % UQ_DEPTH].bytes == 0)
        cold->uq_h[gpeer]++;
    } else {
      sq.eager_rx[gpeer] = s.seq;
      ret_credit(gpeer, s.seq);
      drain_hold_ &= ~(1ull << (gpeer & 63));
    }
  }

  // small allreduce/reduce eligibility: whole message in one slot, no
  // compression, fan-in within the n-ary move budget
  ACCL_HD bool one_shot_ok(u64 n, DataType dt, DataType wdt, u32 P) const {
    u64 cap = tune_oneshot_max ? tune_oneshot_max : (64u << 10);
    return dt == wdt && P >= 2 && P <= MOVE_MAX_SRC &&
           n * dtype_size(dt) <= cap && n * dtype_size(dt) <= cfg.slot_bytes;
  }

  // one-shot allreduce: everyone broadcasts its vector; every rank folds
  // P-1 slot payloads + own src into dst with one n-ary pass
  ACCL_HD u32 allreduce_one_shot(const CallDesc& d, const CommView& c,
                                 u32 tag) {
    u64 n = desc_count(d);
    DataType dt = desc_dtype(d);
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    const u32 P = c.size, r = c.rank;
    u32 nf = 0;
    for (u32 p = 0; p < P; ++p)
      if (p != r) mk_tx(nf++, c.global(p), src, dt, dt, n, tag);
    u32 e = run_flows(nf);
    if (e) return e;
    // fold in LOCAL-RANK order with own src in position r: every rank
    // computes the identical fp reduction (replicas must stay bitwise in
    // sync — DDP/c10d semantics)
    SlotRef refs[MOVE_MAX_SRC];
    const char* contrib[MOVE_MAX_SRC];
    u32 nx = 0;
    for (u32 p = 0; p < P; ++p) {
      if (p == r) {
        contrib[p] = src;
        continue;
      }
      if (!collect_one_slot(c.global(p), tag, dt, n, refs[nx])) return err;
      contrib[p] = refs[nx].pay;
      nx++;
    }
    e = nary_reduce(dst, contrib[0], dt, n, &contrib[1], P - 1,
                    int(d.function));
    u32 k = 0;
    for (u32 p = 0; p < P; ++p)
      if (p != r) release_slot(c.global(p), refs[k++]);
    return e;
  }

  // one-shot reduce at root: leaves send one slot each; root folds them
  ACCL_HD u32 reduce_one_shot(const CallDesc& d, const CommView& c, u32 root,
                              u32 tag) {
    u64 n = desc_count(d);
    DataType dt = desc_dtype(d);
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    const u32 P = c.size;
    if (c.rank != root) {
      mk_tx(0, c.global(root), src, dt, dt, n, tag);
      return run_flows(1);
    }
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    // local-rank fold order (matches allreduce_one_shot for reproducible
    // numerics across the reduce family)
    SlotRef refs[MOVE_MAX_SRC];
    const char* contrib[MOVE_MAX_SRC];
    u32 nx = 0;
    for (u32 p = 0; p < P; ++p) {
      if (p == root) {
        contrib[p] = src;
        continue;
      }
      if (!collect_one_slot(c.global(p), tag, dt, n, refs[nx])) return err;
      contrib[p] = refs[nx].pay;
      nx++;
    }
    u32 e = nary_reduce(dst, contrib[0], dt, n, &contrib[1], P - 1,
                        int(d.function));
    u32 k = 0;
    for (u32 p = 0; p < P; ++p)
      if (p != root) release_slot(c.global(p), refs[k++]);
    return e;
  }

  // reduce at root: fan-in with a serialized reduce chain per segment
  // (reference: reduce, ccl_offload_control.c:1507-1744)
  ACCL_HD u32 op_reduce(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    u32 root = d.root_src_dst;
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::reduce) << 16) | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    if (c.size == 1) {
      mk_local(0, src, dt, local_ptr(d.addr2, d.flags & F_DST_ARENA), dt, n);
      return run_flows(1);
    }
    if (one_shot_ok(n, dt, wdt, c.size))
      return reduce_one_shot(d, c, root, tag);
    if (coll_direct_ok(d, n, dt, wdt)) {
      if (c.rank != root) return reduce_direct_leaf(c, root, src, dt, n, tag);
      return reduce_direct_root(d, c, tag);
    }
    if (c.rank != root) {
      mk_tx(0, c.global(root), src, dt, wdt, n, tag);
      return run_flows(1);
    }
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    u32 nf = 0;
    mk_local(nf, src, dt, dst, dt, n);
    const u64* gate = &flows[nf].done;
    nf++;
    for (u32 p = 0; p < c.size; ++p) {
      if (p == root) continue;
      mk_rx(nf, c.global(p), dst, dt, wdt, n, tag, dst, dt, int(d.function), gate);
      gate = &flows[nf].done;
      nf++;
    }
    return run_flows(nf);
  }

  // fullmesh reduce_scatter (reference ring: ccl_offload_control.c:1748-1852)
  // Every rank owns chunk r: peers push their chunk r, owner chains reduces.
  ACCL_HD u32 op_reduce_scatter(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);  // per-rank result count
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::reduce_scatter) << 16) | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    u32 r = c.rank;
    if (c.size == 1) {
      mk_local(0, src, dt, dst, dt, n);
      return run_flows(1);
    }
    if (coll_direct_ok(d, n, dt, wdt) && 2 * u64(c.size - 1) <= MAX_FLOWS)
      return reduce_scatter_direct(d, c, tag);
    if (2 * u64(c.size) - 1 > MAX_FLOWS) return reduce_scatter_batched(d, c);
    u32 nf = 0;
    // outbound: my chunk p -> rank p, all links at once
    for (u32 p = 0; p < c.size; ++p)
      if (p != r)
        mk_tx(nf++, c.global(p), src + u64(p) * n * dtype_size(dt), dt, wdt, n, tag);
    // inbound chain into dst
    mk_local(nf, src + u64(r) * n * dtype_size(dt), dt, dst, dt, n);
    const u64* gate = &flows[nf].done;
    nf++;
    for (u32 p = 0; p < c.size; ++p) {
      if (p == r) continue;
      mk_rx(nf, c.global(p), dst, dt, wdt, n, tag, dst, dt, int(d.function), gate);
      gate = &flows[nf].done;
      nf++;
    }
    return run_flows(nf);
  }

  // large-P fallback: (tx, fused recv-reduce) PAIRS batched to the flow
  // table, reduce chain serialized by gates. Running tx and rx in the same
  // flow set is what keeps credit flowing: an all-tx phase would stall once
  // the per-peer message exceeds the eager window, with every rank waiting
  // for credit only a posted recv returns (deadlock).
  ACCL_HD u32 reduce_scatter_batched(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::reduce_scatter) << 16) | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    u32 r = c.rank;
    u32 e;
    mk_local(0, src + u64(r) * n * dtype_size(dt), dt, dst, dt, n);
    if ((e = run_flows(1))) return e;
    const u32 B = MAX_FLOWS / 2;
    for (u32 base = 0; base < c.size; base += B) {
      u32 nf = 0;
      const u64* gate = nullptr;  // predecessor finished in an earlier batch
      for (u32 p = base; p < c.size && p < base + B; ++p) {
        if (p == r) continue;
        mk_tx(nf++, c.global(p), src + u64(p) * n * dtype_size(dt), dt, wdt,
              n, tag);
        mk_rx(nf, c.global(p), dst, dt, wdt, n, tag, dst, dt, int(d.function),
              gate);
        gate = &flows[nf].done;
        nf++;
      }
      if (nf && (e = run_flows(nf))) return e;
    }
    return E_OK;
  }

  // allreduce = fullmesh reduce_scatter + fullmesh allgather, all phases as
  // one flow set so phase 2 streams out as phase 1 chunks retire.
  // (reference: segmented ring RS+AG, ccl_offload_control.c:1855-2075.)
  ACCL_HD u32 op_allreduce(const CallDesc& d, const CommView& c) {
    u64 total = desc_count(d);
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::allreduce) << 16) | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    const u32 P = c.size, r = c.rank;
    if (P == 1) {
      mk_local(0, src, dt, dst, dt, total);
      return run_flows(1);
    }
    // Small-message one-shot path: single round + single n-ary fold —
    // halves latency vs RS+AG on the latency-bound end of the curve
    if (one_shot_ok(total, dt, wdt, P))
      return allreduce_one_shot(d, c, tag);
    // Large-message direct path: compose direct reduce_scatter (windowed
    // n-ary fan-in over xGMI stages) + direct allgather (single peer
    // writes) — no eager slot staging on either phase. (reference shape:
    // rendezvous allreduce = reduce+bcast composition,
    // ccl_offload_control.c:1878-1887; ours keeps the RS+AG structure.)
    if (total % P == 0 && coll_direct_ok(d, total / P, dt, wdt) &&
        2 * u64(P - 1) <= MAX_FLOWS) {
      u64 chunk = total / P;
      CallDesc rs = d;
      rs.scenario = u32(Op::reduce_scatter);
      rs.count_lo = u32(chunk);
      rs.count_hi = u32(chunk >> 32);
      rs.addr2 = d.addr2 + u64(r) * chunk * dtype_size(dt);
      u32 e = op_reduce_scatter(rs, c);
      if (e) return e;
      CallDesc ag = d;
      ag.scenario = u32(Op::allgather);
      ag.count_lo = u32(chunk);
      ag.count_hi = u32(chunk >> 32);
      ag.addr0 = d.addr2 + u64(r) * chunk * dtype_size(dt);
      ag.addr2 = d.addr2;
      return op_allgather(ag, c);
    }
    u32 fm_max = tune_fullmesh_max ? tune_fullmesh_max : 9;
    if (fm_max > 17) fm_max = 17;  // 4(P-1)+1 flows must fit MAX_FLOWS
    if (P > fm_max) return ring_allreduce(d, c);
    const u32 dsz = dtype_size(dt);
    // chunk partition: chunk i = [off(i), off(i+1)), balanced
    u64 base = total / P, rem = total % P;
    u64 off[MAX_RANKS + 1];
    off[0] = 0;
    for (u32 i = 0; i < P; ++i) off[i + 1] = off[i] + base + (i < rem ? 1 : 0);
    u64 myn = off[r + 1] - off[r];
    u32 nf = 0;
    // phase 1 outbound: src chunk p -> rank p
    for (u32 p = 0; p < P; ++p)
      if (p != r)
        mk_tx(nf++, c.global(p), src + off[p] * dsz, dt, wdt,
              off[p + 1] - off[p], tag);
    // phase 1 inbound chain into dst chunk r
    mk_local(nf, src + off[r] * dsz, dt, dst + off[r] * dsz, dt, myn);
    const u64* gate = &flows[nf].done;
    nf++;
    for (u32 p = 0; p < P; ++p) {
      if (p == r) continue;
      mk_rx(nf, c.global(p), dst + off[r] * dsz, dt, wdt, myn, tag,
            dst + off[r] * dsz, dt, int(d.function), gate);
      gate = &flows[nf].done;
      nf++;
    }
    const u64* phase1_done = gate;
    // phase 2: broadcast my reduced chunk, receive everyone else's
    u32 tag2 = tag + 1;
    for (u32 p = 0; p < P; ++p) {
      if (p == r) continue;
      mk_tx(nf++, c.global(p), dst + off[r] * dsz, dt, wdt, myn, tag2, phase1_done);
      mk_rx(nf++, c.global(p), dst + off[p] * dsz, dt, wdt,
            off[p + 1] - off[p], tag2);
    }
    return run_flows(nf);
  }

  // ring allreduce for P > fullmesh budget (reference schedule shape:
  // ccl_offload_control.c:1888-2071). Works on dst as the accumulator.
  ACCL_HD u32 ring_allreduce(const CallDesc& d, const CommView& c) {
    u64 total = desc_count(d);
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::allreduce) << 16) | 0x8000 | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    const u32 P = c.size, r = c.rank;
    const u32 dsz = dtype_size(dt);
    u64 base = total / P, rem = total % P;
    u64 off[MAX_RANKS + 1];
    off[0] = 0;
    for (u32 i = 0; i < P; ++i) off[i + 1] = off[i] + base + (i < rem ? 1 : 0);
    u32 next = c.global((r + 1) % P), prev = c.global((r + P - 1) % P);
    // local init: dst = src
    mk_local(0, src, dt, dst, dt, total);
    u32 e = run_flows(1);
    if (e) return e;
    // P-1 reduce-scatter steps: send dst chunk (r-s), recv+reduce chunk (r-s-1)
    for (u32 s = 0; s < P - 1; ++s) {
      u32 ct = (r + P - s) % P;        // chunk to send
      u32 cr = (r + P - s - 1) % P;    // chunk to receive+reduce
      mk_tx(0, next, dst + off[ct] * dsz, dt, wdt, off[ct + 1] - off[ct], tag);
      mk_rx(1, prev, dst + off[cr] * dsz, dt, wdt, off[cr + 1] - off[cr], tag,
            dst + off[cr] * dsz, dt, int(d.function));
      if ((e = run_flows(2))) return e;
    }
    // P-1 allgather steps
    for (u32 s = 0; s < P - 1; ++s) {
      u32 ct = (r + 1 + P - s) % P;
      u32 cr = (r + P - s) % P;
      mk_tx(0, next, dst + off[ct] * dsz, dt, wdt, off[ct + 1] - off[ct], tag);
      mk_rx(1, prev, dst + off[cr] * dsz, dt, wdt, off[cr + 1] - off[cr], tag);
      if ((e = run_flows(2))) return e;
    }
    return E_OK;
  }

  // alltoall: P-1 pairwise exchanges + local copy (reference: fused flat
  // broadcasts, ccl_offload_control.c:2123-2218)
  ACCL_HD u32 op_alltoall(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);  // per-pair count
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    u32 tag = TAG_COLL | (u32(Op::alltoall) << 16) | d.comm_id;
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    char* dst = local_ptr(d.addr2, d.flags & F_DST_ARENA);
    u32 r = c.rank;
    const u32 esz = dtype_size(dt);
    if (coll_direct_ok(d, n, dt, wdt) && 2 * u64(c.size) - 1 <= MAX_FLOWS) {
      // pairwise address exchange + direct peer writes, every link at once
      // (reference rendezvous all_to_all: publish P-1 addrs, write-on-addr
      // out of order, ccl_offload_control.c:2123-2218)
      u32 sp[MAX_RANKS];
      for (u32 p = 0; p < c.size; ++p)
        if (p != r)
          sp[p] = post_addr(c.global(p), d.addr2 + u64(p) * n * esz, n, tag,
                            u32(dt));
      u32 nf = 0;
      mk_local(nf++, src + u64(r) * n * esz, dt, dst + u64(r) * n * esz, dt, n);
      for (u32 p = 0; p < c.size; ++p) {
        if (p == r) continue;
        RndzvRec rec{};
        if (!wait_addr(c.global(p), tag, rec)) return err;
        mk_tx_direct(nf++, c.global(p), src + u64(p) * n * esz, dt, dt, n,
                     rec.offset, rec_slot(rec));
        mk_rx_direct(nf++, c.global(p), n, dt, sp[p]);
      }
      return run_flows(nf);
    }
    u32 nf = 0;
    mk_local(nf++, src + u64(r) * n * dtype_size(dt), dt,
             dst + u64(r) * n * dtype_size(dt), dt, n);
    if (2 * u64(c.size) - 1 > MAX_FLOWS) {
      u32 e = run_flows(1);  // self copy first
      if (e) return e;
      const u32 B = (MAX_FLOWS / 2) - 1;
      for (u32 base = 0; base < c.size; base += B) {
        nf = 0;
        for (u32 j = base; j < c.size && j < base + B; ++j) {
          if (j == r) continue;
          mk_tx(nf++, c.global(j), src + u64(j) * n * dtype_size(dt), dt, wdt,
                n, tag);
          mk_rx(nf++, c.global(j), dst + u64(j) * n * dtype_size(dt), dt, wdt,
                n, tag);
        }
        if (nf && (e = run_flows(nf))) return e;
      }
      return E_OK;
    }
    for (u32 p = 0; p < c.size; ++p) {
      if (p == r) continue;
      mk_tx(nf++, c.global(p), src + u64(p) * n * dtype_size(dt), dt, wdt, n, tag);
      mk_rx(nf++, c.global(p), dst + u64(p) * n * dtype_size(dt), dt, wdt, n, tag);
    }
    return run_flows(nf);
  }

  // dissemination-free fullmesh barrier: bump my token in every peer's
  // arena, wait for all peers' tokens locally.
  // (reference: gather+scatter notification barrier,
  // ccl_offload_control.c:2078-2120)
  ACCL_HD u32 op_barrier(const CallDesc&, const CommView& c) {
    // epochs are per PAIR: two ranks' common-barrier counts always agree,
    // while a global per-rank epoch would not across subgroup communicators.
    fence_release_sys();
    for (u32 p = 0; p < c.size; ++p) {
      if (p == c.rank) continue;
      u32 g = c.global(p);
      st_sys(tv.barrier_word(g, me()), ++sq.barrier_epoch[g]);
    }
    u64 deadline = deadline_now();
    for (u32 p = 0; p < c.size; ++p) {
      if (p == c.rank) continue;
      u32 g = c.global(p);
      volatile u64* w = tv.barrier_word(me(), g);
      while (ld_sys(w) < sq.barrier_epoch[g])
        if (!wait_pred_tick(deadline)) return err;
    }
    fence_acquire_sys();
    return E_OK;
  }

  // Drain `count` elements from my stream ring lane [lane] into either a
  // local buffer (stream2mem; dst != null) or a forward-to-peer eager tx
  // (send-from-stream; fwd_peer >= 0). One segment at a time: wait for the
  // slot, move its payload, return credit. reference: OP0_STREAM operand
  // routing (dma_mover.cpp:497, router :92-98) and the stream2mem /
  // mem2stream reduce+send test matrix (test.cpp).
  ACCL_HD u32 stream_fed(u32 lane, char* dst, DataType ddt, u64 count,
                         i64 fwd_peer, DataType wdt, u32 tag) {
    u64 drained = 0;
    const ArenaHdr* h = tv.hdr(me());
    while (drained < count) {
      u64 seq = sq.stream_fed_rx[lane] + 1;
      u32 slot = u32((seq - 1) % h->n_stream);
      SlotHdr* sh = tv.stream_hdr(me(), lane, slot);
      u64 deadline = deadline_now();
      while (ld_sys(&sh->seq) != seq)
        if (!wait_pred_tick(deadline)) return err;
      fence_acquire_sys();
      u32 esz = dtype_size(ddt);
      u64 n = sh->bytes / esz;
      if (n > count - drained) { err |= E_SEGMENT; return err; }
      const char* pay = tv.stream_payload(me(), lane, slot);
      if (dst) {
        mk_local(0, pay, ddt, dst + drained * esz, ddt, n);
      } else {
        mk_tx(0, u32(fwd_peer), pay, ddt, wdt, n, tag);
      }
      u32 e = run_flows(1);
      if (e) return e;
      sq.stream_fed_rx[lane] = seq;
      // credit back to the producer's arena so its ring keeps flowing
      fence_release_sys();
      st_sys(&tv.stream_ctl(lane, me())->credit, seq);
      drained += n;
    }
    return E_OK;
  }

  // stream_put: push a buffer into the peer's STREAM ring — consumed by the
  // application at the peer (host pop_stream or a device kernel), never by a
  // posted recv. reference: stream_put (accl.hpp:204-238), depacketizer
  // strm-TDEST bypass (udp_depacketizer.cpp:135-148).
  ACCL_HD u32 op_stream_put(const CallDesc& d, const CommView& c) {
    u64 n = desc_count(d);
    u32 peer = c.global(d.root_src_dst);
    DataType dt = desc_dtype(d), wdt = desc_wire_dtype(d);
    const char* src = local_ptr(d.addr0, d.flags & F_SRC_ARENA);
    mk_tx(0, peer, src, dt, wdt, n, d.tag, nullptr, /*to_stream=*/true);
    return run_flows(1);
  }

  // ---- timeout forensics ----
  // On E_TIMEOUT, snapshot every live flow (plus what it is waiting on:
  // awaited slot seq vs observed header, credit, direct progress, mover poll
  // state of the head pending segment) into the arena's dbg region. The
  // host appends the formatted dump to the thrown error
  // (Backend::timeout_dump_str), so a wedged GPU run is root-causeable from
  // the pytest log alone.
  ACCL_HD void dump_timeout(u32 scen) {
    volatile u64* w = (volatile u64*)(tv.arena[me()] + tv.hdr(me())->dbg_off);
    u32 nd = 0;
    for (u32 i = 0; i < MAX_FLOWS && nd < 24; ++i) {
      Flow& f = flows[i];
      if (f.kind == FLOW_IDLE || flow_done(f)) continue;
      volatile u64* fw = w + 8 + u64(nd) * 16;
      for (int k = 0; k < 16; ++k) fw[k] = 0;
      fw[0] = u64(f.kind) | (u64(f.to_stream) << 8) | (u64(f.func) << 16) |
              (u64(f.gpeer) << 32);
      fw[1] = u64(f.tag) | (u64(f.matched_tag) << 32);
      fw[2] = f.count; fw[3] = f.submitted; fw[4] = f.done;
      fw[5] = u64(f.ph) | (u64(f.pt) << 32);
      fw[6] = f.gate ? ld_sys((const volatile u64*)f.gate) : ~0ull;
      if (f.kind == FLOW_RX) {
        u64 seq = sq.eager_rx[f.gpeer] + 1;
        u32 sl = u32((seq - 1) % cfg.n_slots);
        SlotHdr* h = tv.slot_hdr(me(), f.gpeer, sl);
        fw[7] = seq;
        fw[8] = ld_sys(&h->seq);
        fw[9] = u64(h->tag) | (u64(h->bytes) << 32);
      } else if (f.kind == FLOW_TX) {
        fw[7] = sq.eager_tx[f.gpeer];
        fw[8] = tx_credit(f.gpeer);
      } else if (f.kind == FLOW_RX_DIRECT) {
        fw[7] = f.count * dtype_size(DataType(f.ddt));  // window bytes wanted
        fw[8] = ld_sys((const volatile u64*)f.prog_addr);
      }
      if (f.ph != f.pt) {
        PendSeg& p = f.pend[f.ph % FLOW_INFLIGHT];
        fw[11] = u64(p.token) | (u64(mv->poll(p.token) ? 1 : 0) << 32);
        fw[12] = p.elems;
        fw[13] = p.seq;
      }
      nd++;
    }
    w[1] = u64(scen) | (u64(err) << 32);
    w[2] = nd;
    w[3] = wallclock();
    w[4] = u64(cfg.rank) | (u64(cfg.nranks) << 32);
    w[5] = wp_kind_;
    w[6] = wp_info_;
    w[7] = wp_seq_;
    if (Op(scen) == Op::barrier && nd == 0) {
      // barrier diagnosis: observed peer token vs expected epoch per pair
      for (u32 g = 0; g < cfg.nranks && g < 24; ++g) {
        w[8 + 2 * g] = g == cfg.rank ? ~0ull : ld_sys(tv.barrier_word(me(), g));
        w[8 + 2 * g + 1] = sq.barrier_epoch[g];
      }
      w[2] = 0;
    }
    fence_release_sys();
    st_sys((volatile u64*)&w[0], ld_sys((const volatile u64*)&w[0]) + 1);
  }

  ACCL_HD bool device_call_pending(u64 consumed) const {
    const DevCallSlot* s = tv.devcall_slot(cfg.rank, u32(consumed % DEVCALL_RING));
    return ld_sys(&s->seq) == consumed + 1;
  }

  // Consume pending device-initiated calls (client_arbiter analogue):
  // returns number executed. Called from both engine main loops between
  // host-ring batches.
  ACCL_HD u32 poll_device_calls(u64& consumed) {
    DevCallRing* ring = tv.devcall_ring(me());
    u32 did = 0;
    for (;;) {
      u32 i = u32(consumed % DEVCALL_RING);
      DevCallSlot* s = tv.devcall_slot(me(), i);
      if (ld_sys(&s->seq) != consumed + 1) break;
      fence_acquire_sys();
      DevCallRet* r = tv.devcall_ret(me(), i);
      r->t_start = wallclock();
      u32 e = run_call(s->d);
      r->t_end = wallclock();
      r->errcode = e;
      fence_release_sys();
      st_sys(&r->seq, consumed + 1);
      consumed++;
      did++;
    }
    (void)ring;
    return did;
  }

  // ---------------- dispatch ----------------
  // reference: run() scenario switch (ccl_offload_control.c:2375-2459)
  ACCL_HD u32 run_call(const CallDesc& d) {
    ParkState ps{};
    return run_call_probe(d, false, ps);
  }

  ACCL_HD u32 run_call_probe(const CallDesc& d, bool probe, ParkState& ps) {
    probe_ = probe ? 1 : 0;
    ps_ = &ps;
    u32 e = run_call_inner(d);
    probe_ = 0;
    ps_ = nullptr;
    if (e & E_NOT_READY) return e;
    if (e & E_TIMEOUT) dump_timeout(d.scenario);
    return e;
  }

  // would running `d` now violate per-(pair, tag) FIFO with a parked call?
  ACCL_HD bool parked_key_match(const CallDesc& d) const {
    if (!nparked) return false;
    for (u32 i = 0; i < MAX_INFLIGHT; ++i) {
      const ParkedCall& p = cold->parked[i];
      if (p.used && p.d.scenario == d.scenario &&
          p.d.root_src_dst == d.root_src_dst && p.d.comm_id == d.comm_id &&
          p.d.tag == d.tag)
        return true;
    }
    return false;
  }

  ACCL_HD bool park(const CallDesc& d, u64 ring_idx, const ParkState& ps) {
    for (u32 i = 0; i < MAX_INFLIGHT; ++i) {
      if (cold->parked[i].used) continue;
      cold->parked[i].d = d;
      cold->parked[i].ring_idx = ring_idx;
      cold->parked[i].deadline = wallclock() + timeout_ticks;
      cold->parked[i].t_start = wallclock();
      cold->parked[i].ps = ps;
      cold->parked[i].used = 1;
      nparked++;
      return true;
    }
    return false;
  }

  // One retry round over the parked set (FIFO per channel key). Returns a
  // parked-entry index that COMPLETED (engine publishes its ret using
  // done_ring_idx/done_err/done_t0), or -1 if none finished this round.
  ACCL_HD int retry_parked() {
    if (!nparked) return -1;
    for (u32 i = 0; i < MAX_INFLIGHT; ++i) {
      ParkedCall& p = cold->parked[i];
      if (!p.used) continue;
      bool blocked = false;
      for (u32 j = 0; j < MAX_INFLIGHT; ++j) {
        const ParkedCall& q = cold->parked[j];
        if (!q.used || j == i) continue;
        if (q.ring_idx < p.ring_idx && q.d.scenario == p.d.scenario &&
            q.d.root_src_dst == p.d.root_src_dst &&
            q.d.comm_id == p.d.comm_id && q.d.tag == p.d.tag) {
          blocked = true;
          break;
        }
      }
      if (blocked) continue;
      active_parked1_ = i + 1;
      u32 e = run_call_probe(p.d, true, p.ps);
      active_parked1_ = 0;
      if (e & E_NOT_READY) {
        if (wallclock() <= p.deadline) continue;
        err = E_TIMEOUT;
        dump_timeout(p.d.scenario);
        e = E_TIMEOUT;
      }
      done_ring_idx = p.ring_idx;
      done_err = e & ~E_NOT_READY;
      done_t0 = p.t_start;
      p.used = 0;
      nparked--;
      return int(i);
    }
    return -1;
  }

  // Serve one fresh descriptor with parking. Bit E_NOT_READY in the return
  // means "parked — publish nothing"; otherwise the value is the errcode.
  // Parkable ops ALWAYS probe: a not-ready send/recv parks immediately
  // (even with an empty queue — later submissions must not block behind
  // it), and the engine loop re-probes at its retry cadence.
  ACCL_HD u32 serve_desc(const CallDesc& d, u64 ring_idx, bool) {
    Op op = Op(d.scenario);
    bool parkable = (op == Op::send || op == Op::recv) &&
                    !(d.flags & F_SRC_STREAM);
    if (parkable && parked_key_match(d)) {
      if (park(d, ring_idx, ParkState{})) return E_NOT_READY;
    }
    ParkState ps{};
    u32 e = run_call_probe(d, parkable, ps);
    if (e & E_NOT_READY) {
      if (park(d, ring_idx, ps)) return E_NOT_READY;
      e = run_call_probe(d, false, ps);  // park table full: run blocking
    }
    return e & ~E_NOT_READY;
  }

  // engine shutdown with calls still parked: fail them (engine going away)
  ACCL_HD int fail_parked() {
    for (u32 i = 0; i < MAX_INFLIGHT; ++i) {
      ParkedCall& p = cold->parked[i];
      if (!p.used) continue;
      done_ring_idx = p.ring_idx;
      done_err = E_ENGINE_DOWN;
      done_t0 = p.t_start;
      p.used = 0;
      nparked--;
      return int(i);
    }
    return -1;
  }

  ACCL_HD u32 run_call_inner(const CallDesc& d) {
    err = 0;
    drain_hold_ = 0;  // holds never span calls (belt for error paths)
    if (d.comm_id >= ncomms && Op(d.scenario) != Op::copy &&
        Op(d.scenario) != Op::combine && Op(d.scenario) != Op::config &&
        Op(d.scenario) != Op::nop)
      return E_COMM;
    const CommView& c = comms[d.comm_id < ncomms ? d.comm_id : 0];
    switch (Op(d.scenario)) {
      case Op::nop: return E_OK;
      case Op::copy: return op_copy(d);
      case Op::combine: return op_combine(d);
      case Op::send: return op_send(d, c);
      case Op::recv: return op_recv(d, c);
      case Op::bcast: return op_bcast(d, c);
      case Op::scatter: return op_scatter(d, c);
      case Op::gather: return op_gather(d, c);
      case Op::allgather: return op_allgather(d, c);
      case Op::reduce: return op_reduce(d, c);
      case Op::allreduce: return op_allreduce(d, c);
      case Op::reduce_scatter: return op_reduce_scatter(d, c);
      case Op::alltoall: return op_alltoall(d, c);
      case Op::barrier: return op_barrier(d, c);
      case Op::stream_put: return op_stream_put(d, c);
      case Op::config: return run_config(d);
      default: return E_INVALID_OP;
    }
  }

  ACCL_HD u32 run_config(const CallDesc& d) {
    switch (CfgFunc(d.function)) {
      case CfgFunc::set_timeout:
        timeout_ticks = desc_count(d) * 1000 * TICKS_PER_US;  // ms -> ticks
        return E_OK;
      case CfgFunc::set_max_eager_size:
        max_eager_bytes = desc_count(d);
        return E_OK;
      case CfgFunc::set_max_rendezvous_size:
        max_rndzv_bytes = desc_count(d);
        return E_OK;
      case CfgFunc::dump_state: {
        // summary at dbg_off+4096 (flow dump uses [0,~3.2K), handshake
        // probes use the last 2 KB)
        volatile u64* w =
            (volatile u64*)(tv.arena[me()] + tv.hdr(me())->dbg_off + 4096);
        w[1] = nparked;
        u32 k = 2;
        for (u32 i = 0; i < MAX_INFLIGHT && k < 2 + 2 * MAX_INFLIGHT; ++i) {
          const ParkedCall& p = cold->parked[i];
          if (!p.used) continue;
          w[k++] = u64(p.d.scenario) | (u64(p.d.root_src_dst) << 8) |
                   (u64(p.d.tag) << 16) | (u64(p.ps.step) << 48);
          // progress cursor (w[0] of the ParkState: elements sent/got or
          // windows posted) packed with the ring index for the host decode
          w[k++] = (p.ring_idx & 0xFFFFFFFFull) | (p.ps.w[0] << 32);
        }
        u64 pa_n = 0, pd_n = 0, uq_n = 0;
        for (u32 r = 0; r < cfg.nranks; ++r) {
          for (u32 q = 0; q < RNDZV_PEND; ++q) {
            if (cold->pa[r][q].valid) pa_n++;
            if (cold->pd[r][q].valid) pd_n++;
          }
          uq_n += cold->uq_t[r] - cold->uq_h[r];
        }
        w[66] = pa_n | (pd_n << 16) | (uq_n << 32);
        w[67] = spill_busy;
        fence_release_sys();
        st_sys((volatile u64*)&w[0], ld_sys((const volatile u64*)&w[0]) + 1);
        return E_OK;
      }
      case CfgFunc::set_tuning:
        // tuning registers (reference configure_tuning_parameters):
        // knob id in root_src_dst, value in count
        if (d.root_src_dst == 0) tune_fullmesh_max = u32(desc_count(d));
        if (d.root_src_dst == 1) tune_oneshot_max = u32(desc_count(d));
        return E_OK;
      case CfgFunc::reset: {
        // soft reset (reference: encore_soft_reset drains retry queue +
        // resets peripherals, ccl_offload_control.c:2249-2261). Local-only:
        // clears the flow table, unexpected-message queue and spill pool;
        // pair sequence counters are PROTOCOL state shared with peers and
        // survive (desynced pairs need a reset on both ends).
        for (u32 i = 0; i < MAX_FLOWS; ++i) flows[i] = Flow{};
        for (u32 r = 0; r < MAX_RANKS; ++r) { cold->uq_h[r] = cold->uq_t[r] = 0; }
        for (u32 r = 0; r < MAX_RANKS; ++r)
          cold->prog_busy[r][0] = cold->prog_busy[r][1] = 0;
        spill_busy = 0;
        err = 0;
        return E_OK;
      }
      case CfgFunc::enable_pkt:
        return E_OK;
      default: return E_OK;
    }
  }
};

}  // namespace accl
