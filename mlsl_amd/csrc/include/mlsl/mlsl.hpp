// mlsl_amd public C++ API.
//
// Capability-parity surface with intel/MLSL's include/mlsl.hpp (Environment,
// Distribution, Session, Operation(+RegInfo), Activation, ParameterSet,
// Statistics, CommBlockInfo — reference lines 177-906), re-designed
// MI355X-native: RCCL over xGMI underneath, HIP-stream progress, no MPI.
// This is a fresh implementation; reference file:line cites in comments are
// behavioral parity markers, not provenance.
#pragma once

#include <cstddef>
#include <vector>

#include "../../core/types.hpp"
#include "../../comm/schedule.hpp"

namespace mlsl {

class CommRequest;
class ProcessGroup;
class Distribution;
class Session;
class Operation;
class OperationRegInfo;
class Activation;
class ParameterSet;
class Statistics;
class CommBlockInfo;

// ---------------------------------------------------------------------------
// Distribution: the data x model process grid (reference mlsl.hpp:350-502).
class Distribution {
  public:
    Distribution(size_t data_parts, size_t model_parts);
    Distribution(int data_color, int model_color);  // WithColors variant
    ~Distribution();

    size_t GetProcessIdx(GroupKind g) const;
    size_t GetProcessCount(GroupKind g) const;
    size_t GetDataParts() const { return data_parts_; }
    size_t GetModelParts() const { return model_parts_; }
    ProcessGroup* Group(GroupKind g) const;

    // Nonblocking generic collectives; complete with Environment::Wait/Test.
    CommRequest* Bcast(void* buf, size_t count, DataType dt, size_t root, GroupKind g);
    CommRequest* Reduce(const void* sbuf, void* rbuf, size_t count, DataType dt,
                        ReduceOp op, size_t root, GroupKind g);
    CommRequest* AllReduce(const void* sbuf, void* rbuf, size_t count, DataType dt,
                           ReduceOp op, GroupKind g);
    CommRequest* AlltoAll(const void* sbuf, size_t send_count, void* rbuf,
                          DataType dt, GroupKind g);
    CommRequest* AlltoAllv(const void* sbuf, const size_t* scnt, const size_t* soff,
                           void* rbuf, const size_t* rcnt, const size_t* roff,
                           DataType dt, GroupKind g);
    CommRequest* Gather(const void* sbuf, size_t send_count, void* rbuf, DataType dt,
                        size_t root, GroupKind g);
    CommRequest* AllGather(const void* sbuf, size_t send_count, void* rbuf,
                           DataType dt, GroupKind g);
    CommRequest* AllGatherv(const void* sbuf, size_t send_count, void* rbuf,
                            const size_t* rcnt, DataType dt, GroupKind g);
    CommRequest* Scatter(const void* sbuf, void* rbuf, size_t recv_count, DataType dt,
                         size_t root, GroupKind g);
    CommRequest* ReduceScatter(const void* sbuf, void* rbuf, size_t recv_count,
                               DataType dt, ReduceOp op, GroupKind g);
    // Neighbor exchange (reference CommOpSRList; unimplemented there, live here).
    CommRequest* SendRecvList(const void* sbuf, void* rbuf,
                              const std::vector<SRPair>& pairs, DataType dt, GroupKind g);
    void Barrier(GroupKind g);

  private:
    size_t data_parts_ = 1, model_parts_ = 1;
    ProcessGroup* data_group_ = nullptr;
    ProcessGroup* model_group_ = nullptr;
    ProcessGroup* replica_group_ = nullptr;
    friend class Environment;
};

// ---------------------------------------------------------------------------
// Environment: library lifecycle + allocation + request completion
// (reference mlsl.hpp:799-906).
class Environment {
  public:
    static Environment& GetEnv();
    static int GetVersion();  // (major<<16)|minor

    void Init(int* argc = nullptr, char** argv[] = nullptr);
    void Init(int rank, int size);
    // Multi-tenant world re-split: Configure("color=N") makes ranks with
    // the same N their own world (reference mlsl.hpp Configure semantics).
    void Configure(const char* config = nullptr);
    void Finalize();
    bool IsInitialized() const;

    size_t GetProcessIdx() const;
    size_t GetProcessCount() const;

    Session* CreateSession(PhaseKind phase = PhaseKind::TRAIN);
    void DeleteSession(Session* s);
    Distribution* CreateDistribution(size_t data_parts, size_t model_parts);
    Distribution* CreateDistributionWithColors(int data_color, int model_color);
    void DeleteDistribution(Distribution* d);

    // Completes a generic-collective request and releases it.
    void* Wait(CommRequest* req);
    bool Test(CommRequest* req, void** result);

    void* Alloc(size_t size, size_t alignment = 64);
    void Free(void* ptr);
    // Coherent copy between host memory and Alloc'd (HBM) buffers — the
    // sanctioned way to fill/read comm buffers from the host. Direct CPU
    // stores to device memory over BAR are NOT coherent with the GPU's
    // per-XCD L2s (stale lines, e.g. the driver's page-scrub zeros, can
    // shadow them); hipMemcpy goes through the coherent path.
    void Memcpy(void* dst, const void* src, size_t bytes);

    void SetQuantizationParams(const QuantParams& p);
    const QuantParams& GetQuantizationParams() const;

  private:
    Environment() = default;
};

}  // namespace mlsl
