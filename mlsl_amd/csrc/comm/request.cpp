#include "request.hpp"

#include <algorithm>
#include <cstring>

#include "../core/config.hpp"
#include "../core/log.hpp"
#include "context.hpp"
#include "device_comm.hpp"
#include "device_state.hpp"
#include "engine.hpp"
#include "group.hpp"
#include "mesh.hpp"
#include "quant.hpp"

namespace mlsl {

const char* CollOpName(CollOp op) {
    switch (op) {
        case CollOp::ALLREDUCE: return "allreduce";
        case CollOp::REDUCE: return "reduce";
        case CollOp::REDUCE_SCATTER: return "reduce_scatter";
        case CollOp::ALLGATHER: return "allgather";
        case CollOp::ALLGATHERV: return "allgatherv";
        case CollOp::BCAST: return "bcast";
        case CollOp::GATHER: return "gather";
        case CollOp::SCATTER: return "scatter";
        case CollOp::ALLTOALL: return "alltoall";
        case CollOp::ALLTOALLV: return "alltoallv";
        case CollOp::BARRIER: return "barrier";
        case CollOp::SRLIST: return "srlist";
    }
    return "?";
}

void ChunkExec::Reset() {
    cur_phase = sch.steps.empty() ? sch.num_phases : 0;
    finished = sch.num_phases == 0;
    prologue_done = false;
    state.assign(sch.steps.size(), StepState{});
}

struct CommRequest::DeviceState : DeviceReqState {};

CommRequest::CommRequest(ProcessGroup* group, DataType dt, CompType ctype)
    : group_(group), dtype_(dt), ctype_(ctype) {}

CommRequest::~CommRequest() = default;

#define MLSL_ADD_OP()                                                        \
    MLSL_CHECK(!has_op_, "request already describes an op");                 \
    has_op_ = true;

void CommRequest::AddAllReduce(size_t count, ReduceOp op) {
    MLSL_ADD_OP();
    spec_.op = CollOp::ALLREDUCE;
    spec_.count = count;
    spec_.rop = op;
}

void CommRequest::AddReduce(size_t count, ReduceOp op, int root) {
    MLSL_ADD_OP();
    spec_.op = CollOp::REDUCE;
    spec_.count = count;
    spec_.rop = op;
    spec_.root = root;
}

void CommRequest::AddReduceScatter(size_t recv_count, ReduceOp op) {
    MLSL_ADD_OP();
    spec_.op = CollOp::REDUCE_SCATTER;
    spec_.count = recv_count;
    spec_.rop = op;
}

void CommRequest::AddAllGather(size_t send_count) {
    MLSL_ADD_OP();
    spec_.op = CollOp::ALLGATHER;
    spec_.count = send_count;
}

void CommRequest::AddAllGatherv(size_t send_count, const std::vector<size_t>& recv_counts) {
    MLSL_ADD_OP();
    spec_.op = CollOp::ALLGATHERV;
    spec_.count = send_count;
    spec_.recv_counts = recv_counts;
}

void CommRequest::AddBcast(size_t count, int root) {
    MLSL_ADD_OP();
    spec_.op = CollOp::BCAST;
    spec_.count = count;
    spec_.root = root;
}

void CommRequest::AddGather(size_t send_count, int root) {
    MLSL_ADD_OP();
    spec_.op = CollOp::GATHER;
    spec_.count = send_count;
    spec_.root = root;
}

void CommRequest::AddScatter(size_t recv_count, int root) {
    MLSL_ADD_OP();
    spec_.op = CollOp::SCATTER;
    spec_.count = recv_count;
    spec_.root = root;
}

void CommRequest::AddAlltoAll(size_t send_count) {
    MLSL_ADD_OP();
    spec_.op = CollOp::ALLTOALL;
    spec_.count = send_count;
}

void CommRequest::AddAlltoAllv(const std::vector<size_t>& scnt, const std::vector<size_t>& soff,
                               const std::vector<size_t>& rcnt, const std::vector<size_t>& roff) {
    MLSL_ADD_OP();
    spec_.op = CollOp::ALLTOALLV;
    spec_.send_counts = scnt;
    spec_.send_offs = soff;
    spec_.recv_counts = rcnt;
    spec_.recv_offs = roff;
}

void CommRequest::AddBarrier() {
    MLSL_ADD_OP();
    spec_.op = CollOp::BARRIER;
}

void CommRequest::AddSendRecvList(const std::vector<SRPair>& pairs) {
    MLSL_ADD_OP();
    spec_.op = CollOp::SRLIST;
    spec_.pairs = pairs;
}

void CommRequest::SetCompression(Compression c, const QuantParams& qp) {
    MLSL_CHECK(!setup_done_, "SetCompression after Setup");
    comp_ = c;
    qparams_ = qp;
    plugin_ = LoadQuantPlugin(qparams_);
}

bool CommRequest::Compressed() const {
    return comp_ == Compression::QUANT_INT8 && spec_.op == CollOp::ALLREDUCE &&
           (dtype_ == DataType::F32 || dtype_ == DataType::BF16) &&
           spec_.rop == ReduceOp::SUM;
}

bool CommRequest::UsesDeviceSchedule() const {
    const Config& cfg = GlobalConfig();
    // IPC window transport: EVERY op walks its schedule (there are no
    // fused RCCL collectives to fall back on).
    Context& ctx = Context::Get();
    if (ctx.DeviceMode() && ctx.Device() && ctx.Device()->UsesP2p() &&
        group_->Size() > 1)
        return !Compressed();
    return (cfg.allreduce_algo == AllReduceAlgo::RING ||
            cfg.allreduce_algo == AllReduceAlgo::RHD ||
            cfg.allreduce_algo == AllReduceAlgo::DIRECT) &&
           spec_.op == CollOp::ALLREDUCE && group_->Size() > 1 && !Compressed();
}

size_t CommRequest::WireBytesFor(const ChunkExec& ce) const {
    // compressed chunks: Schedule::result is already in wire bytes
    return ce.sch.result.bytes;
}

size_t CommRequest::MessageBytes() const {
    const size_t es = DtypeSize(dtype_);
    switch (spec_.op) {
        case CollOp::ALLREDUCE:
        case CollOp::REDUCE:
        case CollOp::BCAST:
            return spec_.count * es;
        case CollOp::REDUCE_SCATTER:
        case CollOp::ALLGATHER:
        case CollOp::GATHER:
        case CollOp::SCATTER:
        case CollOp::ALLTOALL:
            return spec_.count * es * static_cast<size_t>(group_->Size());
        case CollOp::ALLGATHERV: {
            size_t t = 0;
            for (size_t c : spec_.recv_counts) t += c;
            return t * es;
        }
        case CollOp::ALLTOALLV: {
            size_t t = 0;
            for (size_t c : spec_.send_counts) t += c;
            return t * es;
        }
        default:
            return 0;
    }
}

void CommRequest::BuildChunks() {
    const Config& cfg = GlobalConfig();
    const int gr = group_->MyIdx(), gs = group_->Size();
    MLSL_CHECK(group_->IsMember(), "request on a group this rank is not in");
    const size_t es = DtypeSize(dtype_);

    // Chunk fan-out. Element-splittable ops slice the message; the ring/
    // pairwise ops (RS, AG(v), AlltoAll(v)) use dedicated chunked builders
    // that slice every rank-block (reference endpoint split of those ops,
    // src/comm_ep.cpp:598-736 + MLSL_ALLTOALL_SPLIT).
    size_t n_chunks = 1;
    const bool splittable = spec_.op == CollOp::ALLREDUCE || spec_.op == CollOp::BCAST ||
                            spec_.op == CollOp::REDUCE;
    const bool block_splittable =
        spec_.op == CollOp::REDUCE_SCATTER || spec_.op == CollOp::ALLGATHER ||
        spec_.op == CollOp::ALLGATHERV || spec_.op == CollOp::ALLTOALL ||
        spec_.op == CollOp::ALLTOALLV;
    // Splitting only pays when chunks land on DIFFERENT channel streams
    // (concurrent RCCL comms over distinct link schedules). On a single
    // channel the chunks serialize on one stream, so splitting is pure
    // launch overhead — unlike the reference, whose endpoint servers gave
    // every chunk its own progress process (comm_ep.cpp GET_EP_PAYLOAD).
    // Block-splitting applies only where the executor WALKS the chunk
    // schedules (host path, p2p transport): the fused RCCL path has no
    // strided sub-collective to map a block-chunk onto.
    const bool walks_schedules =
        !Context::Get().DeviceMode() || UsesDeviceSchedule();
    if ((splittable || (block_splittable && walks_schedules)) && gs > 1 &&
        !Compressed() && cfg.num_channels > 1) {
        n_chunks = cfg.num_channels;
        if (splittable && MessageBytes() >= cfg.large_msg_mb * (1024 * 1024) &&
            cfg.large_msg_chunks > 1)
            n_chunks *= cfg.large_msg_chunks;
        // Keep chunks >= 4 KB so per-message overhead stays amortized.
        const size_t min_chunk_elems = std::max<size_t>(1, 4096 / es);
        const size_t per_chunk_basis =
            splittable ? spec_.count
                       : std::max<size_t>(1, MessageBytes() / es /
                                                 static_cast<size_t>(gs));
        n_chunks = std::max<size_t>(
            1, std::min(n_chunks,
                        per_chunk_basis / std::max<size_t>(1, min_chunk_elems)));
        n_chunks = std::min<size_t>(n_chunks, 4096);
        if (n_chunks == 0) n_chunks = 1;
    }

    AllReduceAlgo algo = cfg.allreduce_algo;
    if (algo == AllReduceAlgo::AUTO) {
        // Measured crossovers on MI355X (profiles/multirank_transport_r2.md
        // + sweep_mr_w4_rhd_prefix_r2.jsonl): at world 2 the ring's single
        // exchange matches RHD at every size; at pow2 worlds >= 4 RHD's
        // 2*log2(N) phases beat the ring's 2(N-1) until a few MiB, after
        // which the bandwidth-optimal ring wins.
        const bool pow2 = (gs & (gs - 1)) == 0;
        // One-shot direct exchange for genuinely small messages (single
        // round over all N-1 links; (N-1)x wire cost), then RHD for pow2
        // worlds >= 4 up to a few MiB, then the bandwidth-optimal ring.
        if (MessageBytes() <= (1u << 20) && gs <= 8)
            algo = AllReduceAlgo::DIRECT;
        else if (pow2 && gs >= 4 && MessageBytes() <= (4u << 20))
            algo = AllReduceAlgo::RHD;
        else
            algo = AllReduceAlgo::RING;
    }

    chunks_.clear();
    chunks_.resize(n_chunks);
    total_tmp_bytes_ = 0;
    for (size_t c = 0; c < n_chunks; ++c) {
        ChunkExec& ce = chunks_[c];
        ce.chunk_idx = c;
        size_t cnt = spec_.count, off = 0;
        if (n_chunks > 1) {
            off = SegOffset(spec_.count, n_chunks, c);
            cnt = SegCount(spec_.count, n_chunks, c);
        }
        ce.elem_off = off;
        switch (spec_.op) {
            case CollOp::ALLREDUCE:
                if (Compressed()) {
                    // Quantized allreduce: ring over int8 wire blocks with
                    // compressed-domain accumulation (quant/quant.c path
                    // fused into the collective — reference cqueue.c:1977,
                    // 2283). Single chunk; the wire lives in TMP.
                    const size_t blk = qparams_.block_elems;
                    const size_t nblocks = (cnt + blk - 1) / blk;
                    ce.sch = BuildAllReduceRingUnits(gr, gs, nblocks,
                                                     qparams_.WireBlockBytes(), blk);
                } else if (algo == AllReduceAlgo::DIRECT) {
                    ce.sch = BuildAllReduceDirect(gr, gs, cnt, dtype_, spec_.rop);
                } else if (algo == AllReduceAlgo::RHD && (gs & (gs - 1)) == 0) {
                    ce.sch = BuildAllReduceRHD(gr, gs, cnt, dtype_, spec_.rop);
                } else {
                    // Channel c rides a stride-rotated ring: disjoint xGMI
                    // links per channel (multi-endpoint fan-out analog).
                    const int stride = RingStrideForChannel(
                        c % std::max<size_t>(1, cfg.num_channels), gs);
                    ce.sch = BuildAllReduceRing(gr, gs, cnt, dtype_, spec_.rop,
                                                stride);
                }
                break;
            case CollOp::REDUCE:
                ce.sch = BuildReduce(gr, gs, cnt, dtype_, spec_.rop, spec_.root);
                break;
            case CollOp::BCAST:
                ce.sch = BuildBcast(gr, gs, cnt, dtype_, spec_.root);
                break;
            case CollOp::REDUCE_SCATTER:
                if (n_chunks > 1) {
                    ce.elem_off = 0;  // chunk offsets are absolute
                    ce.sch = BuildReduceScatterChunk(
                        gr, gs, spec_.count, SegOffset(spec_.count, n_chunks, c),
                        SegCount(spec_.count, n_chunks, c), dtype_, spec_.rop);
                } else {
                    ce.sch = BuildReduceScatter(gr, gs, spec_.count, dtype_, spec_.rop);
                }
                break;
            case CollOp::ALLGATHER:
                if (n_chunks > 1) {
                    ce.elem_off = 0;
                    std::vector<size_t> counts(static_cast<size_t>(gs), spec_.count);
                    ce.sch = BuildAllGathervChunk(gr, gs, counts, c, n_chunks, dtype_);
                } else {
                    ce.sch = BuildAllGather(gr, gs, spec_.count, dtype_);
                }
                break;
            case CollOp::ALLGATHERV:
                if (n_chunks > 1) {
                    ce.elem_off = 0;
                    ce.sch = BuildAllGathervChunk(gr, gs, spec_.recv_counts, c,
                                                  n_chunks, dtype_);
                } else {
                    ce.sch = BuildAllGatherv(gr, gs, spec_.recv_counts, dtype_);
                }
                break;
            case CollOp::GATHER:
                ce.sch = BuildGather(gr, gs, spec_.count, dtype_, spec_.root);
                break;
            case CollOp::SCATTER:
                ce.sch = BuildScatter(gr, gs, spec_.count, dtype_, spec_.root);
                break;
            case CollOp::ALLTOALL:
                if (n_chunks > 1) {
                    ce.elem_off = 0;
                    std::vector<size_t> cnt(static_cast<size_t>(gs), spec_.count);
                    std::vector<size_t> off(static_cast<size_t>(gs));
                    for (int i = 0; i < gs; ++i)
                        off[static_cast<size_t>(i)] = static_cast<size_t>(i) * spec_.count;
                    ce.sch = BuildAlltoAllvChunk(gr, gs, cnt, off, cnt, off, c,
                                                 n_chunks, dtype_);
                } else {
                    ce.sch = BuildAlltoAll(gr, gs, spec_.count, dtype_);
                }
                break;
            case CollOp::ALLTOALLV:
                if (n_chunks > 1) {
                    ce.elem_off = 0;
                    ce.sch = BuildAlltoAllvChunk(gr, gs, spec_.send_counts,
                                                 spec_.send_offs, spec_.recv_counts,
                                                 spec_.recv_offs, c, n_chunks, dtype_);
                } else {
                    ce.sch = BuildAlltoAllv(gr, gs, spec_.send_counts, spec_.send_offs,
                                            spec_.recv_counts, spec_.recv_offs, dtype_);
                }
                break;
            case CollOp::BARRIER:
                ce.sch = BuildBarrier(gr, gs);
                break;
            case CollOp::SRLIST:
                ce.sch = BuildSendRecvList(gr, gs, spec_.pairs, dtype_);
                break;
        }
        total_tmp_bytes_ += ce.sch.tmp_bytes;
        // Phase byte 0xFF is reserved for edge-sequenced p2p tags (PairTag);
        // a schedule reaching phase 255 (ring allreduce at 129+ ranks,
        // alltoall at 256+) would collapse the p2p/collective tag spaces.
        MLSL_CHECK(ce.sch.num_phases < 255,
                   "schedule phase count exceeds the 8-bit tag field "
                   "(group too large for this algorithm)");
    }
}

void CommRequest::Setup() {
    MLSL_CHECK(has_op_, "Setup() before describing the op");
    BuildChunks();
    Context& ctx = Context::Get();
    if (ctx.DeviceMode()) {
        dev_ = std::make_unique<DeviceState>();
        DeviceSetupRequest(this, *dev_);
    } else {
        for (auto& ce : chunks_) {
            size_t t = ce.sch.tmp_bytes;
            if (Compressed()) {
                // [wire][schedule scratch][error-feedback residual]
                const size_t wire = ce.sch.result.bytes;
                const size_t err = spec_.count * DtypeSize(dtype_);
                t = wire + ce.sch.tmp_bytes + err;
            }
            ce.tmp.assign(t, 0);
        }
    }
    setup_done_ = true;
}

uint64_t CommRequest::PairTag(uint32_t seq) const {
    // phase byte 0xFF marks an edge-sequenced p2p tag (schedule phases
    // never reach 255); seq wraps at 2^24 in-flight-window safety.
    return (static_cast<uint64_t>(group_->Uid() & 0xFFFFF) << 44) |
           (static_cast<uint64_t>(seq & 0xFFFFFF) << 8) | 0xFFull;
}

uint64_t CommRequest::MakeTag(size_t chunk, int phase) const {
    return (static_cast<uint64_t>(group_->Uid() & 0xFFFFF) << 44) |
           (static_cast<uint64_t>(flow_ & 0xFFFFFF) << 20) |
           (static_cast<uint64_t>(chunk & 0xFFF) << 8) |
           (static_cast<uint64_t>(phase) & 0xFF);
}

void CommRequest::Start(const void* sbuf, void* rbuf) {
    MLSL_CHECK(setup_done_, "Start() before Setup()");
    ReqState st = state_.load(std::memory_order_acquire);
    MLSL_CHECK(st == ReqState::IDLE || st == ReqState::DONE,
               "Start() while request in flight");
    sbuf_ = static_cast<const uint8_t*>(sbuf);
    rbuf_ = static_cast<uint8_t*>(rbuf);
    // AlltoAll(v)'s pairwise schedule reads send-block j while writing
    // recv-block i of the same iteration — aliased buffers would corrupt
    // silently (the reference splits these ops out of in-place too,
    // src/comm_ep.cpp:623-736). Fail loudly instead.
    MLSL_CHECK(!((spec_.op == CollOp::ALLTOALL ||
                  spec_.op == CollOp::ALLTOALLV) &&
                 sbuf_ && sbuf_ == rbuf_),
               "alltoall(v) does not support in-place (sbuf == rbuf)");
    dev_sbuf_ = nullptr;
    dev_rbuf_ = nullptr;
    flow_ = spec_.op == CollOp::SRLIST ? 0 : group_->NextFlow();
    for (auto& ce : chunks_) ce.Reset();
    if (dev_) dev_->issued = false;
    error_.clear();

    Context& ctx = Context::Get();
    if (GlobalConfig().check_pointers) {
        // Pointer checker (reference src/pointer_checker.*): every buffer
        // handed to a collective must be a known registered allocation.
        const size_t bytes = MessageBytes();
        MLSL_CHECK(ctx.CheckBuffer(sbuf_, bytes) && ctx.CheckBuffer(rbuf_, 0),
                   "collective buffer not from Environment::Alloc "
                   "(MLSL_CHECK_POINTERS=1)");
    }
    MLSL_LOG(TRACE, "start %s count=%zu dtype=%s group=%d flow=%u",
             CollOpName(spec_.op), spec_.count, DtypeName(dtype_), group_->Uid(), flow_);
    if (ctx.DeviceMode()) {
        // Device path: issue inline — RCCL/kernel enqueues are non-blocking
        // and the STREAMS are the progress engine; Wait/Test poll the
        // completion events directly. This keeps small-message latency at
        // enqueue cost (no ring/thread/condvar hop).
        state_.store(ReqState::ACTIVE, std::memory_order_release);
        start_seqno_ = ctx.GetEngine()->NextSeqno();
        if (AdvanceDevice()) MarkDone();
        return;
    }
    MLSL_LOG(TRACE, "start %s count=%zu dtype=%s group=%d flow=%u sbuf=%p rbuf=%p",
             CollOpName(spec_.op), spec_.count, DtypeName(dtype_), group_->Uid(),
             flow_, static_cast<const void*>(sbuf_), static_cast<void*>(rbuf_));
    state_.store(ReqState::QUEUED, std::memory_order_release);
    ctx.GetEngine()->Submit(this);
}

void* CommRequest::Wait() {
    Context& ctx = Context::Get();
    if (ctx.DeviceMode()) {
        while (state_.load(std::memory_order_acquire) == ReqState::ACTIVE) {
            try {
                if (AdvanceDevice()) MarkDone();
            } catch (const std::exception& e) {
                MarkFailed(e.what());
            }
        }
    } else {
        ctx.GetEngine()->WaitFor(this);
    }
    if (state_.load(std::memory_order_acquire) == ReqState::FAILED)
        MLSL_THROW("request failed: " + error_);
    state_.store(ReqState::IDLE, std::memory_order_release);
    if (chunks_.empty()) return rbuf_;
    if (Compressed()) return rbuf_;  // dequantized into the user recv buffer
    const BufRef& res = chunks_[0].sch.result;
    switch (res.space) {
        case Space::RECV: return rbuf_ + res.off;
        case Space::SEND: return const_cast<uint8_t*>(sbuf_) + res.off;
        case Space::TMP: return nullptr;
    }
    return rbuf_;
}

bool CommRequest::Test() {
    ReqState st = state_.load(std::memory_order_acquire);
    if (st == ReqState::FAILED) MLSL_THROW("request failed: " + error_);
    if (st == ReqState::DONE || st == ReqState::IDLE) return true;
    Context& ctx = Context::Get();
    if (ctx.DeviceMode()) {
        try {
            if (AdvanceDevice()) {
                MarkDone();
                return true;
            }
        } catch (const std::exception& e) {
            MarkFailed(e.what());
            MLSL_THROW("request failed: " + error_);
        }
        return false;
    }
    return ctx.GetEngine()->TestFor(this);
}

void CommRequest::MarkDone() {
    state_.store(ReqState::DONE, std::memory_order_release);
}

void CommRequest::MarkFailed(const std::string& what) {
    error_ = what;
    state_.store(ReqState::FAILED, std::memory_order_release);
}

// ---------------------------------------------------------------------------
// Host executor: advance every chunk's phase program against the mesh.

bool CommRequest::AdvanceHost(Mesh* mesh) {
    const size_t es = DtypeSize(dtype_);
    bool all_done = true;

    const bool compressed = Compressed();
    for (auto& ce : chunks_) {
        if (ce.finished) continue;
        const uint8_t* sbase = sbuf_ + ce.elem_off * es;
        uint8_t* rbase = rbuf_ + ce.elem_off * es;
        uint8_t* wire = ce.tmp.data();                         // compressed only
        const size_t wire_bytes = compressed ? ce.sch.result.bytes : 0;
        auto ptr = [&](const BufRef& b) -> uint8_t* {
            if (compressed) {
                switch (b.space) {
                    case Space::SEND: return wire + b.off;
                    case Space::RECV: return wire + b.off;
                    case Space::TMP: return ce.tmp.data() + wire_bytes + b.off;
                }
            }
            switch (b.space) {
                case Space::SEND: return const_cast<uint8_t*>(sbase) + b.off;
                case Space::RECV: return rbase + b.off;
                case Space::TMP: return ce.tmp.data() + b.off;
            }
            return nullptr;
        };
        if (compressed && !ce.prologue_done) {
            // quantize user send (+ residual) into the wire
            uint8_t* err = ce.tmp.data() + wire_bytes + ce.sch.tmp_bytes;
            if (plugin_) {
                // reference ABI: dtype enum {INT8=0,F16=1,F32=2,F64=3},
                // comp_ratio = input-elem-bytes (int8 wire), method DFP=1
                const int sdt = dtype_ == DataType::F32 ? 2
                                : dtype_ == DataType::F64 ? 3 : 1;
                int rc = plugin_->quant(const_cast<uint8_t*>(sbase), wire,
                                        spec_.count, err, sdt,
                                        DtypeSize(dtype_), 1);
                MLSL_CHECK(rc == 0, "quant plugin compress failed");
            } else {
                HostQuantize(sbase, err, wire, spec_.count, qparams_.block_elems,
                             dtype_, true);
            }
            ce.prologue_done = true;
        }

        while (!ce.finished) {
            bool phase_done = true;
            for (size_t i = 0; i < ce.sch.steps.size(); ++i) {
                const Step& st = ce.sch.steps[i];
                if (st.phase != ce.cur_phase) continue;
                auto& ss = ce.state[i];
                // Post receive first (so early sends always land).
                // SRLIST tags come from per-directed-edge sequences (p2p
                // matching like NCCL — the sequence is drawn once per step
                // and survives retries); collective tags from the group
                // flow, which SRLIST requests deliberately do not consume.
                const bool p2p = spec_.op == CollOp::SRLIST;
                if (st.recv_peer >= 0 && st.recv.bytes > 0 && !ss.recv_posted) {
                    const uint64_t tag =
                        p2p ? PairTag(group_->NextRecvSeq(st.recv_peer))
                            : MakeTag(ce.chunk_idx, st.phase);
                    mesh->PostRecv(group_->WorldRank(st.recv_peer), tag,
                                   ptr(st.recv), st.recv.bytes, &ss.recv_done);
                    ss.recv_posted = true;
                }
                if (st.send_peer >= 0 && st.send.bytes > 0 && !ss.send_started) {
                    if (!ss.tag_drawn) {
                        ss.tag = p2p ? PairTag(group_->NextSendSeq(st.send_peer))
                                     : MakeTag(ce.chunk_idx, st.phase);
                        ss.tag_drawn = true;
                    }
                    if (mesh->StartSend(group_->WorldRank(st.send_peer), ss.tag,
                                        ptr(st.send), st.send.bytes, &ss.send_done))
                        ss.send_started = true;
                }
                const bool send_ok =
                    st.send_peer < 0 || st.send.bytes == 0 || (ss.send_started && ss.send_done);
                const bool recv_ok = st.recv_peer < 0 || st.recv.bytes == 0 || ss.recv_done;
                if (send_ok && recv_ok) {
                    if (st.local != Step::LocalOp::NONE && !ss.local_done) {
                        uint8_t* d = ptr(st.local_dst);
                        uint8_t* s = ptr(st.local_src);
                        if (st.local == Step::LocalOp::COPY) {
                            if (d != s) std::memmove(d, s, st.local_src.bytes);
                        } else if (ce.sch.quant_block > 0) {
                            const size_t blk = ce.sch.quant_block;
                            const size_t units =
                                st.local_dst.bytes / qparams_.WireBlockBytes();
                            if (plugin_) {
                                int rc = plugin_->reduce_sum(s, d, units);
                                MLSL_CHECK(rc == 0, "quant plugin reduce failed");
                            } else {
                                HostQuantAccum(d, s, units * blk, blk);
                            }
                        } else {
                            HostReduce(d, s, st.local_dst.bytes / es, dtype_, ce.sch.rop);
                        }
                        ss.local_done = true;
                    }
                } else {
                    phase_done = false;
                }
            }
            if (!phase_done) break;
            ce.cur_phase++;
            if (ce.cur_phase >= ce.sch.num_phases) {
                ce.finished = true;
                if (compressed) {
                    if (plugin_) {
                        int rc = plugin_->dequant(wire, rbase, spec_.count);
                        MLSL_CHECK(rc == 0, "quant plugin decompress failed");
                    } else {
                        HostDequantize(wire, rbase, spec_.count,
                                       qparams_.block_elems, dtype_);
                    }
                }
            }
        }
        if (!ce.finished) all_done = false;
    }
    return all_done;
}

double CommRequest::LastDeviceCommMs() const {
    return dev_ ? DeviceRequestCommMs(*dev_) : -1.0;
}

bool CommRequest::AdvanceDevice() {
    MLSL_CHECK(dev_ != nullptr, "device state missing");
    return DeviceAdvanceRequest(this, *dev_);
}

}  // namespace mlsl
