#include "p2p_transport.hpp"

#include <algorithm>
#include <cstdlib>
#include <cstring>
#include <mutex>

#include "../core/config.hpp"
#include "../core/log.hpp"
#include "../hip/kernels.hpp"
#include "bootstrap.hpp"
#include "context.hpp"
#include "group.hpp"
#include "request.hpp"

namespace mlsl {

#define HIP_CHECKP(cmd)                                                       \
    do {                                                                      \
        hipError_t e_ = (cmd);                                                \
        if (e_ != hipSuccess)                                                 \
            MLSL_THROW(std::string("HIP error (p2p): ") +                     \
                       hipGetErrorString(e_));                                \
    } while (0)

namespace {
// Flag mailboxes are 128-byte-strided so flags written by different peers
// never share a cache line (remote stores would otherwise ping-pong lines
// between GPUs over xGMI).
constexpr size_t kFlagStride = 128;
// Sub-messages at or below this ride the fully-fused bounded-grid kernels
// (one launch per side); larger ones use wide kernels with separate 1-wg
// waits so the payload kernels never spin. Tunable: MLSL_P2P_FUSED_MAX_KB.
size_t FusedMaxBytes() {
    static const size_t v = [] {
        if (const char* e = std::getenv("MLSL_P2P_FUSED_MAX_KB")) {
            long long kb = std::atoll(e);
            if (kb >= 0) return static_cast<size_t>(kb) * 1024;
        }
        return static_cast<size_t>(1u << 20);
    }();
    return v;
}
std::mutex g_issue_mu;  // enqueue-order == counter-order per group

struct WireHandle {
    hipIpcMemHandle_t h;
    int valid;
};
}  // namespace

std::unique_ptr<P2pGroup> P2pGroup::Create(ProcessGroup* g, size_t nlanes,
                                           size_t nslots, size_t slot_bytes) {
    Context& ctx = Context::Get();
    auto pg = std::unique_ptr<P2pGroup>(new P2pGroup());
    pg->gsize_ = g->Size();
    pg->my_idx_ = g->MyIdx();
    pg->nlanes_ = nlanes;
    pg->nslots_ = nslots;
    pg->slot_bytes_ = slot_bytes;
    const size_t N = static_cast<size_t>(pg->gsize_);
    pg->flags_bytes_ = N * nlanes * 2 * kFlagStride;
    pg->win_bytes_ = pg->flags_bytes_ + N * nlanes * nslots * slot_bytes;

    WireHandle mine{};
    std::memset(&mine, 0, sizeof(mine));
    const bool member = g->IsMember() && g->Size() > 1;
    if (member) {
        HIP_CHECKP(hipMalloc(reinterpret_cast<void**>(&pg->my_base_),
                             pg->win_bytes_));
        // Zero ONLY the flag region: flags/acks are accessed exclusively
        // with system-scope atomics (coherent). The slot payload area is
        // deliberately left untouched — a local memset would leave this
        // process's caches holding zero lines that can shadow the peer's
        // later payload stores (cold-box page-scrub lines shadowed arrivals
        // the same way; slots are fully overwritten before every consume,
        // so they never need initialization).
        HIP_CHECKP(hipMemset(pg->my_base_, 0, pg->flags_bytes_));
        HIP_CHECKP(hipIpcGetMemHandle(&mine.h, pg->my_base_));
        mine.valid = 1;
        HIP_CHECKP(hipHostMalloc(reinterpret_cast<void**>(&pg->abort_host_),
                                 2 * sizeof(uint32_t)));
        pg->status_host_ = pg->abort_host_ + 1;
        pg->abort_host_[0] = 0;
        pg->status_host_[0] = 0;
        const int tmo = GlobalConfig().timeout_sec;
        pg->max_ticks_ = static_cast<uint64_t>(tmo > 0 ? tmo : 300) * 100000000ull;
    }

    // Same exchange pattern as the RCCL unique-id path: world-wide
    // bootstrap allgather, indexed by world rank.
    std::vector<WireHandle> all(static_cast<size_t>(ctx.Boot()->Size()));
    ctx.Boot()->Allgather(&mine, sizeof(WireHandle), all.data());
    if (member) {
        pg->peer_base_.assign(N, nullptr);
        pg->sent_.assign(N * nlanes, 0);
        pg->rcvd_.assign(N * nlanes, 0);
        pg->fused_sent_.assign(N * nlanes, 0);
        pg->fused_rcvd_.assign(N * nlanes, 0);
        pg->fanin_adds_.assign(nlanes, 0);
        const size_t nctr = 2 * N * nlanes + nlanes;
        HIP_CHECKP(hipMalloc(reinterpret_cast<void**>(&pg->ctr_dev_),
                             nctr * sizeof(uint64_t)));
        HIP_CHECKP(hipMemset(pg->ctr_dev_, 0, nctr * sizeof(uint64_t)));
        for (int i = 0; i < pg->gsize_; ++i) {
            if (i == pg->my_idx_) {
                pg->peer_base_[i] = pg->my_base_;
                continue;
            }
            const WireHandle& wh = all[static_cast<size_t>(g->WorldRank(i))];
            MLSL_CHECK(wh.valid, "p2p window handle missing for a group peer");
            void* p = nullptr;
            HIP_CHECKP(hipIpcOpenMemHandle(&p, wh.h,
                                           hipIpcMemLazyEnablePeerAccess));
            pg->peer_base_[i] = static_cast<uint8_t*>(p);
        }
        // hipStreamWriteValue64 publishes measured SLOWER than the 1-wg
        // SetFlag kernel in a same-box A/B (4 KiB: 45 vs 39 us; 256 MiB
        // ring: 310 vs 353 GB/s — the write packet stalls the queue), so
        // the HW path is opt-in: MLSL_P2P_HW_WRITE=1.
        const char* hw_env = std::getenv("MLSL_P2P_HW_WRITE");
        const bool hw_allowed = hw_env && std::atoi(hw_env) != 0;
        // Probe hipStreamWriteValue64 on window memory: if the runtime
        // accepts it, publishes/acks ride the queue as packets (no kernel
        // launch). The value written here is 0 == the initial state.
        if (hw_allowed) {
            hipStream_t ps;
            if (hipStreamCreateWithFlags(&ps, hipStreamNonBlocking) == hipSuccess) {
                hipError_t we = hipStreamWriteValue64(
                    ps, pg->MyInFlag(pg->my_idx_, 0), 0ull, 0);
                if (we == hipSuccess && hipStreamSynchronize(ps) == hipSuccess)
                    pg->hw_write_ = true;
                else
                    (void)hipGetLastError();
                (void)hipStreamDestroy(ps);
            }
        }
        MLSL_LOG(DEBUG,
                 "p2p group ready: size=%d lanes=%zu slots=%zu slot_bytes=%zu "
                 "window=%zu MiB hw_write=%d",
                 pg->gsize_, nlanes, nslots, slot_bytes,
                 pg->win_bytes_ >> 20, (int)pg->hw_write_);
    }
    return pg;
}

P2pGroup::~P2pGroup() {
    if (!my_base_) return;
    // Unblock any in-flight wait kernels, let the device drain, then unmap.
    if (abort_host_) abort_host_[0] = 1;
    (void)hipDeviceSynchronize();
    for (int i = 0; i < gsize_; ++i)
        if (peer_base_[i] && i != my_idx_)
            (void)hipIpcCloseMemHandle(peer_base_[i]);
    (void)hipFree(my_base_);
    if (ctr_dev_) (void)hipFree(ctr_dev_);
    if (abort_host_) (void)hipHostFree(abort_host_);
}

bool P2pGroup::Healthy() const {
    return !status_host_ ||
           __atomic_load_n(status_host_, __ATOMIC_ACQUIRE) == 0;
}

void P2pGroup::Abort() {
    if (abort_host_) __atomic_store_n(abort_host_, 1u, __ATOMIC_RELEASE);
}

// --- window geometry -------------------------------------------------------
// Window of owner o: [flags][data]. Flag pair for (src s, lane l) at
// (s*nlanes+l)*2*kFlagStride: in_flag then ack_flag. in_flag = number of
// slot-messages s has pushed to o (written by s). ack_flag = number of
// slot-messages s has CONSUMED of those o pushed to s (also written by s).
// Data slot (s, l, k): payload staging for s -> o traffic.

uint8_t* P2pGroup::MySlot(int src, size_t lane, size_t slot) const {
    return my_base_ + flags_bytes_ +
           (((static_cast<size_t>(src) * nlanes_ + lane) * nslots_) + slot) *
               slot_bytes_;
}
uint8_t* P2pGroup::PeerSlot(int peer, size_t lane, size_t slot) const {
    return peer_base_[peer] + flags_bytes_ +
           (((static_cast<size_t>(my_idx_) * nlanes_ + lane) * nslots_) + slot) *
               slot_bytes_;
}
void* P2pGroup::MyInFlag(int src, size_t lane) const {
    return my_base_ + (static_cast<size_t>(src) * nlanes_ + lane) * 2 * kFlagStride;
}
void* P2pGroup::MyAckFlag(int peer, size_t lane) const {
    return my_base_ +
           (static_cast<size_t>(peer) * nlanes_ + lane) * 2 * kFlagStride +
           kFlagStride;
}
void* P2pGroup::PeerInFlag(int peer, size_t lane) const {
    return peer_base_[peer] +
           (static_cast<size_t>(my_idx_) * nlanes_ + lane) * 2 * kFlagStride;
}
void* P2pGroup::PeerAckFlag(int peer, size_t lane) const {
    return peer_base_[peer] +
           (static_cast<size_t>(my_idx_) * nlanes_ + lane) * 2 * kFlagStride +
           kFlagStride;
}

void P2pGroup::PublishFlag(void* mbox, uint64_t val, hipStream_t s) {
    if (hw_write_) {
        if (hipStreamWriteValue64(s, mbox, val, 0) == hipSuccess) return;
        (void)hipGetLastError();
        hw_write_ = false;  // runtime changed its mind: kernels from now on
    }
    LaunchSetFlag(mbox, val, s);
}

// --- transport ops ---------------------------------------------------------

void P2pGroup::IssueSchedule(CommRequest* req, ChunkExec& ce, size_t lane,
                             hipStream_t s, const uint8_t* sbase,
                             uint8_t* rbase, uint8_t* tmp) {
    std::lock_guard<std::mutex> lk(g_issue_mu);
    const size_t es = DtypeSize(req->Dtype());
    const bool quant = ce.sch.quant_block > 0;
    const size_t unit = quant ? req->QParams().WireBlockBytes() : es;
    const size_t blk = ce.sch.quant_block;
    MLSL_CHECK(lane < nlanes_, "p2p lane out of range");

    auto ptr = [&](const BufRef& b) -> uint8_t* {
        switch (b.space) {
            case Space::SEND: return const_cast<uint8_t*>(sbase) + b.off;
            case Space::RECV: return rbase + b.off;
            case Space::TMP: return tmp + b.off;
        }
        return nullptr;
    };
    auto same_ref = [](const BufRef& a, const BufRef& b) {
        return a.space == b.space && a.off == b.off && a.bytes == b.bytes;
    };

    const size_t msg_max = (slot_bytes_ / unit) * unit;
    MLSL_CHECK(msg_max > 0, "p2p slot smaller than one element/block");
    // Adaptive sub-message size: ~8 sub-messages per transfer (pipelines
    // DMA against the consumer's reduce) but never below 4 MiB (each
    // wide-path sub-message costs ~5 kernel launches) and never above the
    // slot.
    // Sender and receiver derive the identical size from the transfer
    // length, so the slot partition always agrees.
    auto sub_size = [&](size_t bytes) -> size_t {
        size_t v = std::max<size_t>((bytes + 7) / 8,
                                    std::min<size_t>(4u << 20, msg_max));
        v = std::min(v, msg_max);
        v = (v / unit) * unit;
        if (v < unit) v = unit;
        return std::min(v, msg_max);
    };
    for (int phase = 0; phase < ce.sch.num_phases; ++phase) {
        // One-shot (direct) allreduce fast path: the whole exchange phase
        // becomes TWO kernels — a fan-out pushing the payload to every
        // peer and a fan-in waiting all arrivals and reducing them in a
        // single pass over dst. Falls back to the generic loop for large
        // payloads, >8 peers or uncovered dtypes.
        if (ce.sch.one_shot && phase == 1 && !quant) {
            std::vector<const Step*> fan;
            for (const auto& st : ce.sch.steps)
                if (st.phase == 1) fan.push_back(&st);
            const size_t B = fan.empty() ? 0 : fan[0]->send.bytes;
            const bool covered = req->Dtype() == DataType::F32 ||
                                 req->Dtype() == DataType::BF16;
            if (!fan.empty() && fan.size() <= 8 && B > 0 && B <= msg_max &&
                B <= FusedMaxBytes() && covered) {
                XferPoll ab{};
                ab.abort_word = abort_host_;
                ab.status = status_host_;
                ab.max_ticks = max_ticks_;
                FanPeer sp[8], rp[8];
                const int np = static_cast<int>(fan.size());
                for (int i = 0; i < np; ++i) {
                    const Step* st = fan[i];
                    const int peer = st->send_peer;
                    const size_t e = static_cast<size_t>(peer) * nlanes_ + lane;
                    const uint64_t sseq = ++sent_[e];
                    sp[i].slot = PeerSlot(peer, lane, (sseq - 1) % nslots_);
                    sp[i].flag = PeerInFlag(peer, lane);
                    sp[i].flag_val = sseq;
                    sp[i].wait_mbox =
                        sseq > nslots_ ? MyAckFlag(peer, lane) : nullptr;
                    sp[i].wait_target = sseq - nslots_;
                    sp[i].ctr = ctr_dev_ + e;
                    sp[i].ctr_target =
                        fused_sent_[e] += static_cast<uint64_t>(
                            FanOutWgsPerPeer(static_cast<int>(fan.size())));
                    const uint64_t rseq = ++rcvd_[e];
                    rp[i].slot = MySlot(peer, lane, (rseq - 1) % nslots_);
                    rp[i].flag = PeerAckFlag(peer, lane);
                    rp[i].flag_val = rseq;
                    rp[i].wait_mbox = MyInFlag(peer, lane);
                    rp[i].wait_target = rseq;
                    rp[i].ctr = nullptr;
                    rp[i].ctr_target = 0;
                }
                LaunchFanOutSend(ptr(fan[0]->send), B, sp, np, &ab, s);
                uint64_t* fctr =
                    ctr_dev_ + 2 * static_cast<size_t>(gsize_) * nlanes_ + lane;
                const bool ok = LaunchFanInReduce(
                    ptr(fan[0]->local_dst), B / es, req->Dtype(), ce.sch.rop,
                    rp, np, fctr, fanin_adds_[lane] += kXferFusedGrid, &ab, s);
                MLSL_CHECK(ok, "fan-in dtype dispatch failed");
                continue;
            }
        }
        // Collect this phase's send and recv jobs, then interleave their
        // sub-messages round-robin. The interleave is what makes the
        // bounded slot ring deadlock-free: a sender blocked on backpressure
        // (ack >= seq-kSlots) is guaranteed that its OWN next consume —
        // which is what its upstream neighbor is waiting on — was already
        // enqueued at the previous sub-message index. Issuing all sends of
        // a phase before any recv would cycle-deadlock the ring whenever a
        // segment needs more sub-messages than there are slots.
        struct RecvJob {
            const Step* st;
            bool fuse_into, fuse_out;
        };
        std::vector<const Step*> sends;
        std::vector<RecvJob> recvs;
        size_t max_msgs = 0;
        for (const auto& st : ce.sch.steps) {
            if (st.phase != phase) continue;
            if (st.send_peer >= 0 && st.send.bytes > 0) {
                MLSL_CHECK(st.send_peer != my_idx_, "p2p self-send in schedule");
                const size_t sub = sub_size(st.send.bytes);
                sends.push_back(&st);
                max_msgs = std::max(max_msgs, (st.send.bytes + sub - 1) / sub);
            }
            if (st.recv_peer >= 0 && st.recv.bytes > 0) {
                MLSL_CHECK(st.recv_peer != my_idx_, "p2p self-recv in schedule");
                // Fusion cases: the received data IS the local op's source
                // (ring/RHD reduce), or the local op targets the recv range
                // (reduce-scatter's out-of-place accumulate).
                const bool fuse_into = st.local == Step::LocalOp::REDUCE &&
                                       same_ref(st.local_src, st.recv);
                const bool fuse_out = !fuse_into &&
                                      st.local == Step::LocalOp::REDUCE &&
                                      same_ref(st.local_dst, st.recv) &&
                                      st.local_src.bytes == st.recv.bytes;
                const size_t sub = sub_size(st.recv.bytes);
                recvs.push_back({&st, fuse_into, fuse_out});
                max_msgs = std::max(max_msgs, (st.recv.bytes + sub - 1) / sub);
            }
        }
        for (size_t k = 0; k < max_msgs; ++k) {
            for (const Step* st : sends) {
                const size_t sub = sub_size(st->send.bytes);
                const size_t off = k * sub;
                if (off >= st->send.bytes) continue;
                const size_t n = std::min(sub, st->send.bytes - off);
                const int peer = st->send_peer;
                uint64_t& sent = sent_[static_cast<size_t>(peer) * nlanes_ + lane];
                const uint64_t seq = ++sent;
                const size_t slot = (seq - 1) % nslots_;
                if (n <= FusedMaxBytes()) {
                    // Small sub-message: ONE bounded-grid kernel does the
                    // backpressure poll, the slot copy and the publish
                    // (32 workgroups can never starve the peer — the
                    // full-device fused variant deadlocked; see below).
                    const size_t e = static_cast<size_t>(peer) * nlanes_ + lane;
                    XferPoll bp{};
                    bp.mbox = seq > nslots_ ? MyAckFlag(peer, lane) : nullptr;
                    bp.target = seq - nslots_;
                    bp.abort_word = abort_host_;
                    bp.status = status_host_;
                    bp.max_ticks = max_ticks_;
                    LaunchXferSendFused(PeerSlot(peer, lane, slot),
                                        ptr(st->send) + off, n,
                                        bp.mbox ? &bp : nullptr, ctr_dev_ + e,
                                        fused_sent_[e] += kXferFusedGrid,
                                        PeerInFlag(peer, lane), seq, s);
                    continue;
                }
                // Backpressure (slot reuse) as a SEPARATE 1-wg wait kernel:
                // a poll fused into a FULL-DEVICE copy kernel deadlocked two
                // same-device ranks — each rank's spinner starved the
                // peer's consumer kernel of CUs (measured at 256 MiB
                // world-2). The 1-wg wait always co-schedules.
                if (seq > nslots_)
                    LaunchWaitFlag(MyAckFlag(peer, lane), seq - nslots_,
                                   abort_host_, status_host_, max_ticks_, s);
                LaunchXferCopy(PeerSlot(peer, lane, slot), ptr(st->send) + off,
                               n, nullptr, s);
                PublishFlag(PeerInFlag(peer, lane), seq, s);
            }
            for (const RecvJob& rj : recvs) {
                const Step& st = *rj.st;
                const size_t sub = sub_size(st.recv.bytes);
                const size_t off = k * sub;
                if (off >= st.recv.bytes) continue;
                const size_t n = std::min(sub, st.recv.bytes - off);
                const int peer = st.recv_peer;
                uint64_t& rcvd = rcvd_[static_cast<size_t>(peer) * nlanes_ + lane];
                const uint64_t seq = ++rcvd;
                const size_t slot = (seq - 1) % nslots_;
                const uint8_t* sl = MySlot(peer, lane, slot);
                XferPoll wp{};
                wp.mbox = MyInFlag(peer, lane);
                wp.target = seq;
                wp.abort_word = abort_host_;
                wp.status = status_host_;
                wp.max_ticks = max_ticks_;
                if (n <= FusedMaxBytes()) {
                    const size_t e = static_cast<size_t>(peer) * nlanes_ + lane;
                    uint64_t* rctr = ctr_dev_ +
                                     static_cast<size_t>(gsize_) * nlanes_ + e;
                    if (quant && rj.fuse_into) {
                        // compressed-domain accumulate with the wait fused in
                        LaunchXferRecvQuantAccum(
                            ptr(st.local_dst) + off, sl, (n / unit) * blk, blk,
                            &wp, rctr, fused_rcvd_[e] += kXferFusedGrid,
                            PeerAckFlag(peer, lane), seq, s);
                        continue;
                    }
                    if (!quant || (!rj.fuse_into && !rj.fuse_out)) {
                        // mode 0 (byte copy) is dtype-agnostic, so quant AG
                        // forwards ride it too
                        const int mode = rj.fuse_into ? 1 : (rj.fuse_out ? 2 : 0);
                        uint8_t* d = mode == 1 ? ptr(st.local_dst) + off
                                               : ptr(st.recv) + off;
                        const uint8_t* o =
                            mode == 2 ? ptr(st.local_src) + off : nullptr;
                        if (LaunchXferRecvFused(d, sl, o,
                                                mode == 0 ? n : n / es,
                                                req->Dtype(), ce.sch.rop, mode,
                                                &wp, rctr,
                                                fused_rcvd_[e] += kXferFusedGrid,
                                                PeerAckFlag(peer, lane), seq, s))
                            continue;
                        fused_rcvd_[e] -= kXferFusedGrid;  // dtype not covered
                    }
                }
                // Arrival wait as a 1-wg kernel (same deadlock avoidance as
                // the sender backpressure), then the wide consume kernel.
                LaunchWaitFlag(MyInFlag(peer, lane), seq, abort_host_,
                               status_host_, max_ticks_, s);
                if (rj.fuse_into && !quant &&
                    LaunchXferReduce(ptr(st.local_dst) + off, sl, nullptr,
                                     n / es, req->Dtype(), ce.sch.rop, nullptr,
                                     s)) {
                } else if (rj.fuse_out && !quant &&
                           LaunchXferReduce(ptr(st.recv) + off, sl,
                                            ptr(st.local_src) + off, n / es,
                                            req->Dtype(), ce.sch.rop, nullptr,
                                            s)) {
                } else if (!rj.fuse_into && !rj.fuse_out) {
                    LaunchXferCopy(ptr(st.recv) + off, sl, n, nullptr, s);
                } else {
                    if (rj.fuse_into) {
                        if (quant)
                            LaunchQuantAccum(ptr(st.local_dst) + off, sl,
                                             (n / unit) * blk, blk, s);
                        else
                            LaunchReduce(ptr(st.local_dst) + off, sl, n / es,
                                         req->Dtype(), ce.sch.rop, s);
                    } else {  // fuse_out
                        if (quant) {
                            // acc = slot, then acc += local_src slice
                            LaunchCopyVariant(ptr(st.recv) + off, sl, n, false, s);
                            LaunchQuantAccum(ptr(st.recv) + off,
                                             ptr(st.local_src) + off,
                                             (n / unit) * blk, blk, s);
                        } else {
                            LaunchReduceOut(ptr(st.recv) + off, sl,
                                            ptr(st.local_src) + off, n / es,
                                            req->Dtype(), ce.sch.rop, s);
                        }
                    }
                }
                PublishFlag(PeerAckFlag(peer, lane), seq, s);
            }
        }
        // Unfused local ops (pure copy steps, or reduce with unrelated
        // source/dest geometry) run after the phase's traffic, step order.
        for (const auto& st : ce.sch.steps) {
            if (st.phase != phase || st.local == Step::LocalOp::NONE) continue;
            if (st.local_dst.bytes == 0) continue;  // zero-length segment
            const bool was_recv_step = st.recv_peer >= 0 && st.recv.bytes > 0;
            if (was_recv_step) {
                const bool fuse_into = st.local == Step::LocalOp::REDUCE &&
                                       same_ref(st.local_src, st.recv);
                const bool fuse_out = !fuse_into &&
                                      st.local == Step::LocalOp::REDUCE &&
                                      same_ref(st.local_dst, st.recv) &&
                                      st.local_src.bytes == st.recv.bytes;
                if (fuse_into || fuse_out) continue;  // handled per sub-msg
            }
            uint8_t* d = ptr(st.local_dst);
            uint8_t* src = ptr(st.local_src);
            if (st.local == Step::LocalOp::COPY) {
                if (d != src) LaunchCopyVariant(d, src, st.local_src.bytes, false, s);
            } else if (quant) {
                LaunchQuantAccum(d, src, (st.local_dst.bytes / unit) * blk,
                                 blk, s);
            } else {
                LaunchReduce(d, src, st.local_dst.bytes / es, req->Dtype(),
                             ce.sch.rop, s);
            }
        }
    }
}

}  // namespace mlsl
