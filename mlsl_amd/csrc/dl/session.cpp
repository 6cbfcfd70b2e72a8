// Planner implementation. Collective-selection parity with reference
// src/mlsl_impl.cpp:139-241 (the 5 peer-connection cases) and block
// geometry :243-347; distributed-update (ZeRO-1 ancestor) owned-kernel math
// :388-444. Fresh code on the MI355X comm core.
#include "session.hpp"

#include <x86intrin.h>

#include <algorithm>
#include <cstdio>

#include "../comm/context.hpp"
#include "../comm/group.hpp"
#include "../core/log.hpp"

namespace mlsl {

// ---------------------------------------------------------------------------
// OperationRegInfo

size_t OperationRegInfo::AddInput(size_t fm_count, size_t fm_size, DataType dt) {
    inputs_.push_back(RegEntry{fm_count, fm_size, dt});
    return inputs_.size() - 1;
}

size_t OperationRegInfo::AddOutput(size_t fm_count, size_t fm_size, DataType dt) {
    outputs_.push_back(RegEntry{fm_count, fm_size, dt});
    return outputs_.size() - 1;
}

size_t OperationRegInfo::AddParameterSet(size_t kernel_count, size_t kernel_size,
                                         DataType dt, bool distributed_update,
                                         Compression comp) {
    params_.push_back(RegEntry{kernel_count, kernel_size, dt, distributed_update, comp});
    return params_.size() - 1;
}

void OperationRegInfo::Validate(Distribution* d) {
    if (!d) return;
    const size_t mp = d->GetModelParts();
    for (const auto& e : inputs_)
        MLSL_CHECK(e.count % mp == 0, "input fm count not divisible by model parts");
    for (const auto& e : params_)
        MLSL_CHECK(e.count % mp == 0, "kernel count not divisible by model parts");
}

// ---------------------------------------------------------------------------
// Activation

// Construction parity: reference mlsl_impl.cpp:36-66 — CC outputs keep the
// full fm range locally (partial sums, needReduce); everything else is
// sharded over the model group.
Activation::Activation(Operation* op, const RegEntry& re, bool is_input, size_t idx)
    : op_(op), act_idx_(idx), is_input_(is_input) {
    global_fm_count_ = re.count;
    fm_size_ = re.size;
    dt_ = re.dt;
    Distribution* dist = op->GetDistribution();
    const size_t mp = dist->GetModelParts();
    if (!is_input_ && op->Kind() == OpKind::CC) {
        local_fm_count_ = global_fm_count_;
        global_fm_offset_ = 0;
        need_reduce_ = mp > 1;
    } else {
        MLSL_CHECK(global_fm_count_ % mp == 0, "fm count not divisible by model parts");
        local_fm_count_ = global_fm_count_ / mp;
        global_fm_offset_ = local_fm_count_ * dist->GetProcessIdx(GroupKind::MODEL);
        need_reduce_ = false;
    }
}

Activation::~Activation() {
    if (owned_comm_buf_ && Context::Initialized())
        Context::Get().Free(owned_comm_buf_);
}

void* Activation::GetCommBuf() {
    if (!comm_buf_bytes_) return nullptr;
    if (!owned_comm_buf_) owned_comm_buf_ = Context::Get().Alloc(comm_buf_bytes_, 64);
    return owned_comm_buf_;
}

void Activation::SetPeer(Activation* act) {
    if (act == nullptr) {
        peer_ = nullptr;
        peer_set_ = true;
        need_comm_ = false;
        return;
    }
    MLSL_CHECK(act->global_fm_count_ * act->fm_size_ == global_fm_count_ * fm_size_,
               "peer activation sizes must match");
    MLSL_CHECK(is_input_ != act->is_input_, "peer must pair an input with an output");
    MLSL_CHECK(dt_ == act->dt_, "peer dtypes must match");
    MLSL_CHECK(peer_ == nullptr || peer_ == act, "peer can be set only once");
    peer_ = act;
    act->peer_ = this;
    peer_set_ = true;
    act->peer_set_ = true;
}

// Case 1 (same-dist model parallel): fwd ReduceScatter of partial sums +
// bwd AllGather (reference :159-175).
void Activation::BuildCase1(Activation* in, Activation* out) {
    Distribution* dist = in->op_->GetDistribution();
    ProcessGroup* mg = dist->Group(GroupKind::MODEL);
    const size_t mp = static_cast<size_t>(mg->Size());
    const size_t midx = static_cast<size_t>(mg->MyIdx());
    const size_t local_mb = in->op_->GetLocalMinibatchSize();
    const size_t seg_elems = in->local_fm_count_ * local_mb * in->fm_size_;
    const size_t es = DtypeSize(in->dt_);

    out->req_ = std::make_unique<CommRequest>(mg, out->dt_, CompType::FPROP);
    out->req_->AddReduceScatter(seg_elems, ReduceOp::SUM);
    out->req_->Setup();
    out->send_off_bytes_ = 0;
    out->recv_off_bytes_ = 0;
    out->comm_buf_bytes_ = mp * seg_elems * es;
    // pack: one block per model peer, fm-sliced (BIPackReduceScatter :243).
    const size_t fm_per_seg = out->local_fm_count_ / mp;
    for (size_t i = 0; i < mp; ++i)
        out->pack_blocks_.emplace_back(0, local_mb, i * fm_per_seg, fm_per_seg,
                                       out->fm_size_, out->dt_, i * seg_elems);
    in->unpack_blocks_.emplace_back(0, local_mb, 0, in->local_fm_count_, in->fm_size_,
                                    in->dt_, 0);

    in->req_ = std::make_unique<CommRequest>(mg, in->dt_, CompType::BPROP);
    in->req_->AddAllGather(seg_elems);
    in->req_->Setup();
    in->send_off_bytes_ = midx * seg_elems * es;  // slot packing
    in->recv_off_bytes_ = 0;
    in->comm_buf_bytes_ = mp * seg_elems * es;
    in->pack_blocks_.emplace_back(0, local_mb, 0, in->local_fm_count_, in->fm_size_,
                                  in->dt_, midx * seg_elems);
    for (size_t i = 0; i < mp; ++i)
        out->unpack_blocks_.emplace_back(0, local_mb, i * fm_per_seg, fm_per_seg,
                                         out->fm_size_, out->dt_, i * seg_elems);
}

// Case 3 (MP -> wider DP): fwd ReduceScatter over the producer's model group
// with minibatch repartition, bwd AllGather (reference :187-202).
void Activation::BuildCase3(Activation* in, Activation* out) {
    Distribution* odist = out->op_->GetDistribution();
    ProcessGroup* mg = odist->Group(GroupKind::MODEL);
    const size_t mp = static_cast<size_t>(mg->Size());
    const size_t midx = static_cast<size_t>(mg->MyIdx());
    const size_t in_local_mb = in->op_->GetLocalMinibatchSize();
    const size_t seg_elems = in->local_fm_count_ * in_local_mb * in->fm_size_;
    const size_t es = DtypeSize(in->dt_);

    out->req_ = std::make_unique<CommRequest>(mg, out->dt_, CompType::FPROP);
    out->req_->AddReduceScatter(seg_elems, ReduceOp::SUM);
    out->req_->Setup();
    out->comm_buf_bytes_ = mp * seg_elems * es;
    // pack: mb-sliced blocks (BIPackReduceScatter2 :259).
    for (size_t i = 0; i < mp; ++i)
        out->pack_blocks_.emplace_back(i * in_local_mb, in_local_mb, 0,
                                       out->local_fm_count_, out->fm_size_, out->dt_,
                                       i * seg_elems);
    in->unpack_blocks_.emplace_back(0, in_local_mb, 0, in->local_fm_count_, in->fm_size_,
                                    in->dt_, 0);

    in->req_ = std::make_unique<CommRequest>(mg, in->dt_, CompType::BPROP);
    in->req_->AddAllGather(seg_elems);
    in->req_->Setup();
    in->send_off_bytes_ = midx * seg_elems * es;
    in->comm_buf_bytes_ = mp * seg_elems * es;
    in->pack_blocks_.emplace_back(0, in_local_mb, 0, in->local_fm_count_, in->fm_size_,
                                  in->dt_, midx * seg_elems);
    for (size_t i = 0; i < mp; ++i)
        out->unpack_blocks_.emplace_back(i * in_local_mb, in_local_mb, 0,
                                         out->local_fm_count_, out->fm_size_, out->dt_,
                                         i * seg_elems);
}

// Cases 4/5 (DP <-> MP layout transition): AlltoAll both directions with the
// reference's block geometry (BIBuildAlltoAll :330-347). `out`/`in` follow
// the data-flow direction; `pack_act` gets pack blocks, `unpack_act` unpack.
void Activation::BuildAlltoAllBlocks(Activation* pack_act, Activation* unpack_act) {
    ProcessGroup* mg = pack_act->op_->GetDistribution()->GetModelParts() == 1
                           ? unpack_act->op_->GetDistribution()->Group(GroupKind::MODEL)
                           : pack_act->op_->GetDistribution()->Group(GroupKind::MODEL);
    const size_t gsz = static_cast<size_t>(mg->Size());
    const size_t local_mb = std::min(pack_act->op_->GetLocalMinibatchSize(),
                                     unpack_act->op_->GetLocalMinibatchSize());
    const size_t fm_elems = std::min(pack_act->local_fm_count_ * pack_act->fm_size_,
                                     unpack_act->local_fm_count_ * unpack_act->fm_size_);
    const size_t pack_fm = fm_elems / pack_act->fm_size_;
    const size_t unpack_fm = fm_elems / unpack_act->fm_size_;

    size_t blk = 0;
    for (size_t i = 0; i < pack_act->op_->GetLocalMinibatchSize(); i += local_mb)
        for (size_t j = 0; j < pack_act->local_fm_count_; j += pack_fm) {
            pack_act->pack_blocks_.emplace_back(i, local_mb, j, pack_fm,
                                                pack_act->fm_size_, pack_act->dt_,
                                                blk * local_mb * fm_elems);
            ++blk;
        }
    MLSL_CHECK(blk == gsz, "alltoall pack block count != group size");
    blk = 0;
    for (size_t i = 0; i < unpack_act->op_->GetLocalMinibatchSize(); i += local_mb)
        for (size_t j = 0; j < unpack_act->local_fm_count_; j += unpack_fm) {
            unpack_act->unpack_blocks_.emplace_back(i, local_mb, j, unpack_fm,
                                                    unpack_act->fm_size_, unpack_act->dt_,
                                                    blk * local_mb * fm_elems);
            ++blk;
        }
    MLSL_CHECK(blk == gsz, "alltoall unpack block count != group size");
}

// Collective selection — the 5 cases (reference mlsl_impl.cpp:139-241).
void Activation::InitPeerConnection() {
    if (peer_ == nullptr) return;
    Activation* out = is_input_ ? peer_ : this;
    Activation* in = is_input_ ? this : peer_;
    Distribution* odist = out->op_->GetDistribution();
    Distribution* idist = in->op_->GetDistribution();
    Context& ctx = Context::Get();

    if (ctx.Size() > 1 && (out->need_reduce_ || odist != idist)) {
        out->need_comm_ = true;
        in->need_comm_ = true;
    }
    if (!out->need_comm_) return;

    const size_t out_mp = odist->GetModelParts();
    const size_t in_mp = idist->GetModelParts();
    const size_t out_dp = odist->GetDataParts();
    const size_t in_dp = idist->GetDataParts();
    const size_t es = DtypeSize(out->dt_);

    if (out->need_reduce_ && odist == idist) {
        BuildCase1(in, out);
    } else if (out->need_reduce_ && in_mp == 1 && out_dp == in_dp) {
        // Case 2: fwd AllReduce of partial sums over the model group; no bwd
        // comm (reference :176-186).
        ProcessGroup* mg = odist->Group(GroupKind::MODEL);
        const size_t elems =
            out->local_fm_count_ * out->op_->GetLocalMinibatchSize() * out->fm_size_;
        out->req_ = std::make_unique<CommRequest>(mg, out->dt_, CompType::FPROP);
        out->req_->AddAllReduce(elems, ReduceOp::SUM);
        out->req_->Setup();
        out->comm_buf_bytes_ = elems * es;
        out->pack_blocks_.emplace_back(0, out->op_->GetLocalMinibatchSize(), 0,
                                       out->local_fm_count_, out->fm_size_, out->dt_, 0);
        in->unpack_blocks_.emplace_back(0, in->op_->GetLocalMinibatchSize(), 0,
                                        in->local_fm_count_, in->fm_size_, in->dt_, 0);
        in->comm_buf_bytes_ = elems * es;
        // in keeps need_comm_=true with no request of its own: its WaitComm
        // waits the producer's AllReduce; bwd has no exchange (ref :176-186).
    } else if (out->need_reduce_ && in_mp == 1 && in_dp % out_dp == 0 &&
               in_dp == out_mp * out_dp) {
        BuildCase3(in, out);
    } else if (!out->need_reduce_ && (out_mp == 1 || in_mp == 1)) {
        // Cases 4/5: layout transition via AlltoAll (reference :203-226).
        ProcessGroup* mg = (out_mp == 1 ? idist : odist)->Group(GroupKind::MODEL);
        const size_t gsz = static_cast<size_t>(mg->Size());
        const size_t local_mb = std::min(out->op_->GetLocalMinibatchSize(),
                                         in->op_->GetLocalMinibatchSize());
        const size_t fm_elems = std::min(out->local_fm_count_ * out->fm_size_,
                                         in->local_fm_count_ * in->fm_size_);
        const size_t pair_elems = local_mb * fm_elems;
        const size_t payload = gsz * pair_elems * es;

        out->req_ = std::make_unique<CommRequest>(mg, out->dt_, CompType::FPROP);
        out->req_->AddAlltoAll(pair_elems);
        out->req_->Setup();
        out->send_off_bytes_ = 0;
        out->recv_off_bytes_ = payload;  // out-of-place: recv region after send
        out->comm_buf_bytes_ = 2 * payload;

        in->req_ = std::make_unique<CommRequest>(mg, in->dt_, CompType::BPROP);
        in->req_->AddAlltoAll(pair_elems);
        in->req_->Setup();
        in->send_off_bytes_ = 0;
        in->recv_off_bytes_ = payload;
        in->comm_buf_bytes_ = 2 * payload;

        BuildAlltoAllBlocks(out, in);  // fwd: out packs, in unpacks
        BuildAlltoAllBlocks(in, out);  // bwd: in packs grads, out unpacks
    } else {
        MLSL_THROW("unsupported activation distribution transition");
    }
}

void Activation::StartComm(void* buf) {
    Statistics* stats = op_->GetSession()->GetStats();
    Statistics::Event ev{op_->OpIndex(), act_idx_, true, false, is_input_,
                         Statistics::Event::Action::Start};
    stats->Update(ev);
    if (need_comm_ && req_) {
        uint8_t* b = static_cast<uint8_t*>(buf);
        req_->Start(b + send_off_bytes_, b + recv_off_bytes_);
    }
    ev.is_compute = false;
    stats->Update(ev, need_comm_ && req_ ? req_->MessageBytes() : 0);
}

void* Activation::WaitComm() {
    Statistics* stats = op_->GetSession()->GetStats();
    Statistics::Event ev{op_->OpIndex(), act_idx_, true, false, is_input_,
                         Statistics::Event::Action::Wait};
    stats->Update(ev);
    void* ptr = nullptr;
    // Wait the PEER's request: the producer started the exchange
    // (reference mlsl_impl.cpp:379-380). A null peer request (case 2
    // backward) means no exchange in this direction.
    if (need_comm_ && peer_ && peer_->req_) ptr = peer_->req_->Wait();
    ev.is_compute = false;
    stats->Update(ev, 0,
                  need_comm_ && peer_ && peer_->req_
                      ? peer_->req_->LastDeviceCommMs()
                      : -1.0);
    return ptr;
}

// ---------------------------------------------------------------------------
// ParameterSet (reference mlsl_impl.cpp:388-539)

ParameterSet::ParameterSet(Operation* op, const RegEntry& re, size_t idx)
    : op_(op), param_idx_(idx), distributed_update_(re.dist_update) {
    Distribution* dist = op->GetDistribution();
    ProcessGroup* dg = dist->Group(GroupKind::DATA);
    const size_t mp = dist->GetModelParts();
    const size_t dp = static_cast<size_t>(dg->Size());

    global_kernel_count_ = re.count;
    MLSL_CHECK(global_kernel_count_ % mp == 0, "kernel count not divisible by model parts");
    local_kernel_count_ = global_kernel_count_ / mp;
    global_kernel_offset_ = local_kernel_count_ * dist->GetProcessIdx(GroupKind::MODEL);
    kernel_size_ = re.size;
    dt_ = re.dt;
    comp_ = re.comp;
    need_comm_ = dp > 1;

    if (distributed_update_) {
        // ZeRO-1 ancestor: each data-rank owns ceil(local/dp) kernels and
        // only runs the optimizer there (reference :401-411).
        owned_kernel_count_ = (local_kernel_count_ + dp - 1) / dp;
        local_kernel_count_ = owned_kernel_count_ * dp;  // padded
        owned_kernel_offset_ = owned_kernel_count_ * static_cast<size_t>(dg->MyIdx());
    } else {
        owned_kernel_count_ = local_kernel_count_;
        owned_kernel_offset_ = 0;
    }

    if (need_comm_) {
        grad_req_ = std::make_unique<CommRequest>(dg, dt_, CompType::PARAM_GRAD);
        if (comp_ == Compression::QUANT_INT8 && !distributed_update_) {
            // int8 gradient compression with error feedback, fused into the
            // allreduce (reference quant path, cqueue.c:1977-1994).
            grad_req_->SetCompression(comp_,
                                      Environment::GetEnv().GetQuantizationParams());
        }
        if (distributed_update_) {
            grad_req_->AddReduceScatter(owned_kernel_count_ * kernel_size_, ReduceOp::SUM);
        } else {
            grad_req_->AddAllReduce(owned_kernel_count_ * kernel_size_, ReduceOp::SUM);
        }
        grad_req_->Setup();
        if (distributed_update_) {
            inc_req_ = std::make_unique<CommRequest>(dg, dt_, CompType::PARAM_INC);
            inc_req_->AddAllGather(owned_kernel_count_ * kernel_size_);
            inc_req_->Setup();
        }
    }
}

ParameterSet::~ParameterSet() = default;

void ParameterSet::StartGradientComm(void* buf) {
    Statistics* stats = op_->GetSession()->GetStats();
    Statistics::Event ev{op_->OpIndex(), param_idx_, true, true, false,
                         Statistics::Event::Action::Start};
    stats->Update(ev);
    last_grad_buf_ = buf;
    if (need_comm_) {
        uint8_t* b = static_cast<uint8_t*>(buf);
        const size_t es = DtypeSize(dt_);
        if (distributed_update_) {
            // In-place shard: the reduced segment lands at the owned offset.
            grad_req_->Start(b, b + owned_kernel_offset_ * kernel_size_ * es);
        } else {
            grad_req_->Start(b, b);
        }
    }
    ev.is_compute = false;
    stats->Update(ev, need_comm_ ? grad_req_->MessageBytes() : 0);
}

void* ParameterSet::WaitGradientComm() {
    Statistics* stats = op_->GetSession()->GetStats();
    Statistics::Event ev{op_->OpIndex(), param_idx_, true, true, false,
                         Statistics::Event::Action::Wait};
    stats->Update(ev);
    void* r = nullptr;
    if (need_comm_) r = grad_req_->Wait();
    else r = last_grad_buf_;
    ev.is_compute = false;
    stats->Update(ev, 0, need_comm_ ? grad_req_->LastDeviceCommMs() : -1.0);
    return r;
}

bool ParameterSet::TestGradientComm(void** result) {
    if (!need_comm_) {
        if (result) *result = last_grad_buf_;
        return true;
    }
    Statistics* stats = op_->GetSession()->GetStats();
    Statistics::Event ev{op_->OpIndex(), param_idx_, true, true, false,
                         Statistics::Event::Action::Test};
    stats->Update(ev);
    bool done = grad_req_->Test();
    if (done && result) *result = grad_req_->Wait();
    ev.is_compute = false;
    stats->Update(ev);
    return done;
}

void ParameterSet::StartIncrementComm(void* buf) {
    Statistics* stats = op_->GetSession()->GetStats();
    Statistics::Event ev{op_->OpIndex(), param_idx_, true, true, true,
                         Statistics::Event::Action::Start};
    stats->Update(ev);
    if (need_comm_ && inc_req_) {
        uint8_t* b = static_cast<uint8_t*>(buf);
        const size_t es = DtypeSize(dt_);
        inc_req_->Start(b + owned_kernel_offset_ * kernel_size_ * es, b);
    }
    ev.is_compute = false;
    stats->Update(ev, need_comm_ && inc_req_ ? inc_req_->MessageBytes() : 0);
}

void* ParameterSet::WaitIncrementComm() {
    Statistics* stats = op_->GetSession()->GetStats();
    Statistics::Event ev{op_->OpIndex(), param_idx_, true, true, true,
                         Statistics::Event::Action::Wait};
    stats->Update(ev);
    void* r = need_comm_ && inc_req_ ? inc_req_->Wait() : nullptr;
    ev.is_compute = false;
    stats->Update(ev, 0,
                  need_comm_ && inc_req_ ? inc_req_->LastDeviceCommMs() : -1.0);
    return r;
}

// ---------------------------------------------------------------------------
// Operation

Operation::Operation(Session* s, const OperationRegInfo& info, Distribution* dist,
                     size_t op_idx)
    : session_(s), dist_(dist), kind_(info.Kind()), name_(info.Name()), op_idx_(op_idx) {
    MLSL_CHECK(dist_ != nullptr, "operation needs a distribution");
    MLSL_CHECK(s->GetGlobalMinibatchSize() > 0, "set global minibatch size first");
    MLSL_CHECK(s->GetGlobalMinibatchSize() % dist->GetDataParts() == 0,
               "global minibatch not divisible by data parts");
    for (size_t i = 0; i < info.inputs_.size(); ++i)
        inputs_.push_back(std::make_unique<Activation>(this, info.inputs_[i], true, i));
    for (size_t i = 0; i < info.outputs_.size(); ++i)
        outputs_.push_back(std::make_unique<Activation>(this, info.outputs_[i], false, i));
    for (size_t i = 0; i < info.params_.size(); ++i)
        params_.push_back(std::make_unique<ParameterSet>(this, info.params_[i], i));
}

Operation::~Operation() = default;

size_t Operation::GetGlobalMinibatchSize() const {
    return session_->GetGlobalMinibatchSize();
}

size_t Operation::GetLocalMinibatchSize() const {
    return GetGlobalMinibatchSize() / dist_->GetDataParts();
}

size_t Operation::GetGlobalMinibatchOffset() const {
    return GetLocalMinibatchSize() * dist_->GetProcessIdx(GroupKind::DATA);
}

void Operation::SetPrev(Operation* prev, size_t in_idx, size_t prev_out_idx) {
    Activation* mine = GetInput(in_idx);
    if (!prev) {
        mine->SetPeer(nullptr);
        return;
    }
    MLSL_CHECK(prev->GetSession() == session_, "operations in different sessions");
    prev->GetOutput(prev_out_idx)->SetPeer(mine);
}

void Operation::SetNext(Operation* next, size_t out_idx, size_t next_in_idx) {
    Activation* mine = GetOutput(out_idx);
    if (!next) {
        mine->SetPeer(nullptr);
        return;
    }
    MLSL_CHECK(next->GetSession() == session_, "operations in different sessions");
    mine->SetPeer(next->GetInput(next_in_idx));
}

void Operation::Commit() {
    for (auto& out : outputs_) out->InitPeerConnection();
}

// ---------------------------------------------------------------------------
// Statistics

static inline unsigned long long Now() { return __rdtsc(); }

Statistics::Statistics(Session* s) : session_(s) {
    env_enabled_ = GlobalConfig().stats;
    started_ = env_enabled_;
    last_ts_ = Now();
}

void Statistics::EnsureSize(size_t n) {
    if (per_op_.size() < n) per_op_.resize(n);
}

void Statistics::Update(const Event& ev, size_t bytes, double device_ms) {
    if (!env_enabled_ || !started_) return;
    EnsureSize(ev.op_idx + 1);
    const unsigned long long now = Now();
    const unsigned long long delta = now - last_ts_;
    last_ts_ = now;
    OpStats& os = per_op_[ev.op_idx];
    const int cls = ev.is_param ? (ev.is_input_or_inc ? INC : GRAD)
                                : (ev.is_input_or_inc ? IA : OA);
    if (ev.is_compute) {
        os.compute_cycles += delta;
    } else {
        os.comm_cycles += delta;
        os.ent[cls].cycles += delta;
    }
    os.comm_bytes += bytes;
    os.ent[cls].bytes += bytes;
    if (device_ms >= 0.0) {
        const unsigned long long ns =
            static_cast<unsigned long long>(device_ms * 1e6);
        os.comm_device_ns += ns;
        os.ent[cls].device_ns += ns;
    }
}

void Statistics::CollectIsolation() {
    if (!env_enabled_ || !started_) return;
    // Reference protocol: 10 iterations, skip the first 4
    // (src/mlsl_impl_stats.cpp:48-49). Each entity's request runs standalone
    // against scratch buffers; ranks iterate in identical order so the
    // collectives match up.
    constexpr int kIters = 10, kSkip = 4;
    Context& ctx = Context::Get();
    const size_t nops = session_->GetOperationCount();
    EnsureSize(nops);
    for (size_t oi = 0; oi < nops; ++oi) {
        Operation* op = session_->GetOperation(oi);
        auto bench = [&](CommRequest* req, size_t buf_bytes, size_t send_off,
                         size_t recv_off) {
            if (!req || buf_bytes == 0) return;
            uint8_t* buf = static_cast<uint8_t*>(ctx.Alloc(buf_bytes + 64, 64));
            unsigned long long acc = 0;
            for (int it = 0; it < kIters; ++it) {
                const unsigned long long t0 = Now();
                req->Start(buf + send_off, buf + recv_off);
                req->Wait();
                const unsigned long long t1 = Now();
                if (it >= kSkip) acc += t1 - t0;
            }
            ctx.Free(buf);
            per_op_[oi].isolation_cycles += acc / (kIters - kSkip);
        };
        for (size_t i = 0; i < op->GetOutputCount(); ++i) {
            Activation* a = op->GetOutput(i);
            if (a->NeedComm() && a->req_)
                bench(a->req_.get(), a->comm_buf_bytes_, a->send_off_bytes_,
                      a->recv_off_bytes_);
        }
        for (size_t i = 0; i < op->GetInputCount(); ++i) {
            Activation* a = op->GetInput(i);
            if (a->NeedComm() && a->req_)
                bench(a->req_.get(), a->comm_buf_bytes_, a->send_off_bytes_,
                      a->recv_off_bytes_);
        }
        for (size_t i = 0; i < op->GetParameterSetCount(); ++i) {
            ParameterSet* p = op->GetParameterSet(i);
            if (!p->need_comm_) continue;
            const size_t es = DtypeSize(p->GetDataType());
            const size_t bytes = p->GetLocalKernelCount() * p->GetKernelSize() * es;
            if (p->grad_req_) {
                const size_t roff = p->IsDistributedUpdate()
                                        ? p->GetOwnedKernelOffset() * p->GetKernelSize() * es
                                        : 0;
                bench(p->grad_req_.get(), bytes, 0, roff);
            }
            if (p->inc_req_)
                bench(p->inc_req_.get(), bytes,
                      p->GetOwnedKernelOffset() * p->GetKernelSize() * es, 0);
        }
    }
}

void Statistics::Reset() {
    per_op_.clear();
    last_ts_ = Now();
}

void Statistics::Print() {
    // Rank-local table to mlsl_stats.log (reference file name,
    // src/mlsl_impl.hpp:40).
    FILE* f = std::fopen("mlsl_stats.log", "a");
    if (!f) return;
    std::fprintf(f, "# op | compute_cyc | comm_cyc | comm_KB | isolation_cyc |"
                    " dev_comm_us |"
                    " IA KB/cyc/devus | OA KB/cyc/devus | GRAD KB/cyc/devus |"
                    " INC KB/cyc/devus\n");
    for (size_t i = 0; i < per_op_.size(); ++i) {
        const OpStats& os = per_op_[i];
        const char* name = i < session_->GetOperationCount()
                               ? session_->GetOperation(i)->GetName()
                               : "?";
        std::fprintf(f, "%zu(%s) | %llu | %llu | %zu | %llu | %llu", i, name,
                     os.compute_cycles, os.comm_cycles, os.comm_bytes / 1024,
                     os.isolation_cycles, os.comm_device_ns / 1000);
        for (int c = 0; c < 4; ++c)
            std::fprintf(f, " | %zu/%llu/%llu", os.ent[c].bytes / 1024,
                         os.ent[c].cycles, os.ent[c].device_ns / 1000);
        std::fprintf(f, "\n");
    }
    std::fclose(f);
}

unsigned long long Statistics::GetIsolationCommCycles(size_t op) const {
    return op < per_op_.size() ? per_op_[op].isolation_cycles : 0;
}
unsigned long long Statistics::GetCommCycles(size_t op) const {
    return op < per_op_.size() ? per_op_[op].comm_cycles : 0;
}
unsigned long long Statistics::GetComputeCycles(size_t op) const {
    return op < per_op_.size() ? per_op_[op].compute_cycles : 0;
}
size_t Statistics::GetCommSize(size_t op) const {
    return op < per_op_.size() ? per_op_[op].comm_bytes : 0;
}
unsigned long long Statistics::GetTotalIsolationCommCycles() const {
    unsigned long long t = 0;
    for (auto& o : per_op_) t += o.isolation_cycles;
    return t;
}
unsigned long long Statistics::GetTotalCommCycles() const {
    unsigned long long t = 0;
    for (auto& o : per_op_) t += o.comm_cycles;
    return t;
}
unsigned long long Statistics::GetTotalComputeCycles() const {
    unsigned long long t = 0;
    for (auto& o : per_op_) t += o.compute_cycles;
    return t;
}
unsigned long long Statistics::GetCommDeviceNs(size_t op) const {
    return op < per_op_.size() ? per_op_[op].comm_device_ns : 0;
}
unsigned long long Statistics::GetTotalCommDeviceNs() const {
    unsigned long long t = 0;
    for (const auto& o : per_op_) t += o.comm_device_ns;
    return t;
}
size_t Statistics::GetTotalCommSize() const {
    size_t t = 0;
    for (auto& o : per_op_) t += o.comm_bytes;
    return t;
}

// ---------------------------------------------------------------------------
// Session

Session::Session(PhaseKind phase) : phase_(phase) {
    stats_ = std::make_unique<Statistics>(this);
}

Session::~Session() {
    // MLSL_STATS runs always leave the table behind (the reference dumps
    // mlsl_stats.log without an explicit call, mlsl_impl_stats.cpp:97).
    if (stats_ && stats_->IsEnabled() && GetOperationCount() > 0) stats_->Print();
}

void Session::SetGlobalMinibatchSize(size_t mb) {
    MLSL_CHECK(mb > 0, "global minibatch must be positive");
    global_mb_ = mb;
}

OperationRegInfo* Session::CreateOperationRegInfo(OpKind kind) {
    reg_infos_.push_back(std::make_unique<OperationRegInfo>(kind));
    return reg_infos_.back().get();
}

void Session::DeleteOperationRegInfo(OperationRegInfo* i) {
    for (auto it = reg_infos_.begin(); it != reg_infos_.end(); ++it)
        if (it->get() == i) {
            reg_infos_.erase(it);
            return;
        }
}

size_t Session::AddOperation(OperationRegInfo* info, Distribution* dist) {
    MLSL_CHECK(info != nullptr, "null reg info");
    ops_.push_back(std::make_unique<Operation>(this, *info, dist, ops_.size()));
    return ops_.size() - 1;
}

void Session::RemoveOperations() { ops_.clear(); }

void Session::Commit() {
    for (auto& op : ops_) op->Commit();
    stats_->CollectIsolation();
}

// Environment hooks that live here to keep session.hpp self-contained.
Session* Environment::CreateSession(PhaseKind phase) { return new Session(phase); }
void Environment::DeleteSession(Session* s) { delete s; }

}  // namespace mlsl
