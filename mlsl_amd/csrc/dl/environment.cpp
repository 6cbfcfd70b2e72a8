// Environment + Distribution implementation (reference src/mlsl.cpp dispatch
// + src/mlsl_impl.hpp:174-305 grid construction, rebuilt on the MI355X comm
// core: ProcessGroup = RCCL communicator set / TCP mesh subset).
#include "../include/mlsl/mlsl.hpp"

#include <hip/hip_runtime.h>

#include <cstring>
#include <functional>
#include <memory>
#include <mutex>
#include <unordered_map>

#include "../comm/context.hpp"
#include "../comm/group.hpp"
#include "../comm/request.hpp"
#include "../core/log.hpp"

namespace mlsl {

namespace {

// RequestStorage analog (reference src/mlsl_impl.hpp:60-94): owns the
// one-shot requests handed out by Distribution collectives so
// Environment::Wait can free them.
class RequestStorage {
  public:
    static RequestStorage& Get() {
        static RequestStorage s;
        return s;
    }
    CommRequest* Register(std::unique_ptr<CommRequest> r) {
        std::lock_guard<std::mutex> lk(mu_);
        CommRequest* p = r.get();
        reqs_.emplace(p, std::move(r));
        return p;
    }
    void Release(CommRequest* r) {
        std::lock_guard<std::mutex> lk(mu_);
        reqs_.erase(r);
    }
    bool Owns(CommRequest* r) {
        std::lock_guard<std::mutex> lk(mu_);
        return reqs_.count(r) > 0;
    }
    void Clear() {
        std::lock_guard<std::mutex> lk(mu_);
        reqs_.clear();
    }

  private:
    std::mutex mu_;
    std::unordered_map<CommRequest*, std::unique_ptr<CommRequest>> reqs_;
};

QuantParams g_quant_params;

CommRequest* OneShot(ProcessGroup* g, DataType dt,
                     const std::function<void(CommRequest&)>& describe,
                     const void* sbuf, void* rbuf) {
    auto req = std::make_unique<CommRequest>(g, dt, CompType::GENERIC);
    describe(*req);
    req->Setup();
    CommRequest* p = RequestStorage::Get().Register(std::move(req));
    p->Start(sbuf, rbuf);
    return p;
}

}  // namespace

// ---------------------------------------------------------------------------
// Distribution

// Grid construction (behavioral parity with reference
// src/mlsl_impl.hpp:212-278): rank -> (replica iR, data iM, model iF) by
// lId = rank % (d*m); iM = lId / modelParts; iF = lId % modelParts;
// modelColor = iR*lSize + iM (ranks sharing a model group differ in iF),
// dataColor = iR*lSize + iF.
Distribution::Distribution(size_t data_parts, size_t model_parts) {
    MLSL_CHECK(data_parts >= 1 && model_parts >= 1, "grid parts must be >= 1");
    Context& ctx = Context::Get();
    data_parts_ = data_parts;
    model_parts_ = model_parts;

    const size_t world = static_cast<size_t>(ctx.Size());
    const size_t rank = static_cast<size_t>(ctx.Rank());
    const size_t lsize = data_parts * model_parts;
    MLSL_CHECK(lsize <= world, "grid larger than world");
    MLSL_CHECK(world % lsize == 0, "world size not divisible by grid size");
    const size_t replicas = world / lsize;
    const size_t lid = rank % lsize;
    const size_t ir = rank / lsize;
    const size_t im = lid / model_parts;   // data index
    const size_t ifm = lid % model_parts;  // model index

    const int model_color = static_cast<int>(ir * lsize + im);
    const int data_color = static_cast<int>(ir * lsize + ifm);
    const int replica_color = static_cast<int>(lid);

    if (model_parts == 1) model_group_ = ctx.Self();
    else if (model_parts == world) model_group_ = ctx.World();
    else model_group_ = ctx.CreateGroup(model_color);

    if (data_parts == 1) data_group_ = ctx.Self();
    else if (data_parts == world) data_group_ = ctx.World();
    else data_group_ = ctx.CreateGroup(data_color);

    if (replicas == 1) replica_group_ = ctx.Self();
    else if (replicas == world) replica_group_ = ctx.World();
    else replica_group_ = ctx.CreateGroup(replica_color);
}

Distribution::Distribution(int data_color, int model_color) {
    Context& ctx = Context::Get();
    model_group_ = ctx.CreateGroup(model_color);
    data_group_ = ctx.CreateGroup(data_color);
    replica_group_ = ctx.Self();
    data_parts_ = static_cast<size_t>(data_group_->Size());
    model_parts_ = static_cast<size_t>(model_group_->Size());
}

Distribution::~Distribution() = default;

ProcessGroup* Distribution::Group(GroupKind g) const {
    switch (g) {
        case GroupKind::DATA: return data_group_;
        case GroupKind::MODEL: return model_group_;
        case GroupKind::GLOBAL: return Context::Get().World();
    }
    return nullptr;
}

size_t Distribution::GetProcessIdx(GroupKind g) const {
    return static_cast<size_t>(Group(g)->MyIdx());
}

size_t Distribution::GetProcessCount(GroupKind g) const {
    return static_cast<size_t>(Group(g)->Size());
}

CommRequest* Distribution::Bcast(void* buf, size_t count, DataType dt, size_t root,
                                 GroupKind g) {
    return OneShot(Group(g), dt,
                   [&](CommRequest& r) { r.AddBcast(count, static_cast<int>(root)); },
                   buf, buf);
}

CommRequest* Distribution::Reduce(const void* sbuf, void* rbuf, size_t count,
                                  DataType dt, ReduceOp op, size_t root, GroupKind g) {
    return OneShot(Group(g), dt,
                   [&](CommRequest& r) { r.AddReduce(count, op, static_cast<int>(root)); },
                   sbuf, rbuf);
}

CommRequest* Distribution::AllReduce(const void* sbuf, void* rbuf, size_t count,
                                     DataType dt, ReduceOp op, GroupKind g) {
    return OneShot(Group(g), dt, [&](CommRequest& r) { r.AddAllReduce(count, op); },
                   sbuf, rbuf);
}

CommRequest* Distribution::AlltoAll(const void* sbuf, size_t send_count, void* rbuf,
                                    DataType dt, GroupKind g) {
    return OneShot(Group(g), dt, [&](CommRequest& r) { r.AddAlltoAll(send_count); },
                   sbuf, rbuf);
}

CommRequest* Distribution::AlltoAllv(const void* sbuf, const size_t* scnt,
                                     const size_t* soff, void* rbuf, const size_t* rcnt,
                                     const size_t* roff, DataType dt, GroupKind g) {
    ProcessGroup* grp = Group(g);
    const size_t n = static_cast<size_t>(grp->Size());
    return OneShot(grp, dt,
                   [&](CommRequest& r) {
                       r.AddAlltoAllv({scnt, scnt + n}, {soff, soff + n},
                                      {rcnt, rcnt + n}, {roff, roff + n});
                   },
                   sbuf, rbuf);
}

CommRequest* Distribution::Gather(const void* sbuf, size_t send_count, void* rbuf,
                                  DataType dt, size_t root, GroupKind g) {
    return OneShot(Group(g), dt,
                   [&](CommRequest& r) { r.AddGather(send_count, static_cast<int>(root)); },
                   sbuf, rbuf);
}

CommRequest* Distribution::AllGather(const void* sbuf, size_t send_count, void* rbuf,
                                     DataType dt, GroupKind g) {
    return OneShot(Group(g), dt, [&](CommRequest& r) { r.AddAllGather(send_count); },
                   sbuf, rbuf);
}

CommRequest* Distribution::AllGatherv(const void* sbuf, size_t send_count, void* rbuf,
                                      const size_t* rcnt, DataType dt, GroupKind g) {
    ProcessGroup* grp = Group(g);
    const size_t n = static_cast<size_t>(grp->Size());
    (void)send_count;
    return OneShot(grp, dt,
                   [&](CommRequest& r) { r.AddAllGatherv(send_count, {rcnt, rcnt + n}); },
                   sbuf, rbuf);
}

CommRequest* Distribution::Scatter(const void* sbuf, void* rbuf, size_t recv_count,
                                   DataType dt, size_t root, GroupKind g) {
    return OneShot(Group(g), dt,
                   [&](CommRequest& r) { r.AddScatter(recv_count, static_cast<int>(root)); },
                   sbuf, rbuf);
}

CommRequest* Distribution::ReduceScatter(const void* sbuf, void* rbuf, size_t recv_count,
                                         DataType dt, ReduceOp op, GroupKind g) {
    return OneShot(Group(g), dt,
                   [&](CommRequest& r) { r.AddReduceScatter(recv_count, op); },
                   sbuf, rbuf);
}

CommRequest* Distribution::SendRecvList(const void* sbuf, void* rbuf,
                                        const std::vector<SRPair>& pairs, DataType dt,
                                        GroupKind g) {
    return OneShot(Group(g), dt, [&](CommRequest& r) { r.AddSendRecvList(pairs); },
                   sbuf, rbuf);
}

void Distribution::Barrier(GroupKind g) {
    CommRequest* r = OneShot(Group(g), DataType::U8,
                             [&](CommRequest& rq) { rq.AddBarrier(); }, nullptr, nullptr);
    Environment::GetEnv().Wait(r);
}

// ---------------------------------------------------------------------------
// Environment

Environment& Environment::GetEnv() {
    static Environment env;
    return env;
}

int Environment::GetVersion() { return (1 << 16) | 0; }

void Environment::Init(int* argc, char** argv[]) {
    (void)argc;
    (void)argv;
    Context::Get().Init(-1, -1);
}

void Environment::Init(int rank, int size) { Context::Get().Init(rank, size); }

void Environment::Configure(const char* config) {
    if (!config) return;
    const char* p = std::strstr(config, "color=");
    MLSL_CHECK(p != nullptr, "Configure expects \"color=N\"");
    Context::Get().Configure(std::atoi(p + 6));
}

void Environment::Finalize() {
    RequestStorage::Get().Clear();
    Context::Get().Finalize();
}

bool Environment::IsInitialized() const { return Context::Initialized(); }

size_t Environment::GetProcessIdx() const {
    return static_cast<size_t>(Context::Get().Rank());
}

size_t Environment::GetProcessCount() const {
    return static_cast<size_t>(Context::Get().Size());
}

Distribution* Environment::CreateDistribution(size_t d, size_t m) {
    return new Distribution(d, m);
}

Distribution* Environment::CreateDistributionWithColors(int dc, int mc) {
    return new Distribution(dc, mc);
}

void Environment::DeleteDistribution(Distribution* d) { delete d; }

void* Environment::Wait(CommRequest* req) {
    void* res = req->Wait();
    if (RequestStorage::Get().Owns(req)) RequestStorage::Get().Release(req);
    return res;
}

bool Environment::Test(CommRequest* req, void** result) {
    if (!req->Test()) return false;
    void* res = req->Wait();  // completed: harvest result + reset state
    if (result) *result = res;
    if (RequestStorage::Get().Owns(req)) RequestStorage::Get().Release(req);
    return true;
}

void* Environment::Alloc(size_t size, size_t alignment) {
    return Context::Get().Alloc(size, alignment);
}

void Environment::Free(void* ptr) { Context::Get().Free(ptr); }

void Environment::Memcpy(void* dst, const void* src, size_t bytes) {
    Context& ctx = Context::Get();
    if (ctx.DeviceMode()) {
        MLSL_CHECK(hipMemcpy(dst, src, bytes, hipMemcpyDefault) == hipSuccess,
                   "Environment::Memcpy failed");
    } else {
        std::memcpy(dst, src, bytes);
    }
}

void Environment::SetQuantizationParams(const QuantParams& p) { g_quant_params = p; }

const QuantParams& Environment::GetQuantizationParams() const { return g_quant_params; }

}  // namespace mlsl
