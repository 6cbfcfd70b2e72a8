// Persistent communication requests: the CommDesc/CommRequest analog of the
// reference (src/comm.hpp:250-409). A request is described once
// (Add{AllReduce,...}), Setup() compiles it into chunked schedules and sizes
// the persistent scratch, then Start/Wait/Test re-run it every iteration —
// the reference's "persistent request" hot-loop contract
// (src/mlsl_impl.cpp:349-364, 446-539).
//
// Chunking over channels is the xGMI analog of the reference's
// message-splitting across endpoint servers (GET_EP_PAYLOAD,
// src/comm_ep.cpp:99-115): a large message is split into K chunks that
// progress independently (on TCP: interleaved flows; on RCCL: separate
// comms/streams over distinct xGMI rings).
#pragma once

#include <atomic>
#include <condition_variable>
#include <memory>
#include <mutex>
#include <vector>

#include "../core/types.hpp"
#include "schedule.hpp"

namespace mlsl {

class Mesh;
class ProcessGroup;
class Engine;
class DeviceComm;

enum class CollOp : int {
    ALLREDUCE = 0,
    REDUCE,
    REDUCE_SCATTER,
    ALLGATHER,
    ALLGATHERV,
    BCAST,
    GATHER,
    SCATTER,
    ALLTOALL,
    ALLTOALLV,
    BARRIER,
    SRLIST,
};

const char* CollOpName(CollOp op);

// What a request does (CommDesc analog). One collective per request, over
// one group. CompType tags the request for statistics attribution
// (reference CommDesc::CompType, src/comm.hpp:252-258).
enum class CompType : int { FPROP = 0, BPROP = 1, PARAM_GRAD = 2, PARAM_INC = 3, GENERIC = 4 };

struct OpSpec {
    CollOp op = CollOp::BARRIER;
    size_t count = 0;            // elements (per-rank semantics per op)
    DataType dtype = DataType::F32;
    ReduceOp rop = ReduceOp::SUM;
    int root = 0;
    std::vector<size_t> send_counts, send_offs, recv_counts, recv_offs;
    std::vector<SRPair> pairs;
};

// Per-chunk execution state for the host (TCP) path: a resumable schedule
// instance (phase cursor + per-step flags) — the allreduce_pr state machine
// generalized (eplib/allreduce_pr.c:69-343).
struct ChunkExec {
    Schedule sch;
    size_t elem_off = 0;      // element offset of this chunk in the message
    size_t chunk_idx = 0;
    bool prologue_done = false;   // compressed path: quantize ran
    int cur_phase = -1;
    struct StepState {
        bool send_started = false, send_done = false;
        bool recv_posted = false, recv_done = false;
        bool local_done = false;
        bool tag_drawn = false;   // p2p: edge sequence drawn exactly once
        uint64_t tag = 0;
    };
    std::vector<StepState> state;
    std::vector<uint8_t> tmp;  // host scratch
    bool finished = false;

    void Reset();
};

enum class ReqState : int { IDLE = 0, QUEUED, ACTIVE, DONE, FAILED };

class CommRequest {
  public:
    CommRequest(ProcessGroup* group, DataType dt, CompType ctype);
    ~CommRequest();

    // ---- description (CommDesc analog) ----
    void AddAllReduce(size_t count, ReduceOp op);
    void AddReduce(size_t count, ReduceOp op, int root);
    void AddReduceScatter(size_t recv_count, ReduceOp op);
    void AddAllGather(size_t send_count);
    void AddAllGatherv(size_t send_count, const std::vector<size_t>& recv_counts);
    void AddBcast(size_t count, int root);
    void AddGather(size_t send_count, int root);
    void AddScatter(size_t recv_count, int root);
    void AddAlltoAll(size_t send_count);
    void AddAlltoAllv(const std::vector<size_t>& scnt, const std::vector<size_t>& soff,
                      const std::vector<size_t>& rcnt, const std::vector<size_t>& roff);
    void AddBarrier();
    void AddSendRecvList(const std::vector<SRPair>& pairs);
    // Gradient compression (reference quant/quant.c contract): int8 block
    // quantization with error feedback, fused into the allreduce. Call
    // before Setup(); honored for ALLREDUCE on f32/bf16.
    void SetCompression(Compression c, const QuantParams& qp);
    Compression GetCompression() const { return comp_; }
    const QuantParams& QParams() const { return qparams_; }
    const struct QuantPluginApi* Plugin() const { return plugin_; }
    bool Compressed() const;
    // True when the device executor will walk the schedule (custom ring/RHD
    // over ncclSend/Recv) instead of the fused RCCL op.
    bool UsesDeviceSchedule() const;
    // Compressed-path layout inside each chunk's tmp:
    // [wire W][schedule scratch][error-feedback residual].
    size_t WireBytesFor(const ChunkExec& ce) const;

    // ---- lifecycle ----
    void Setup();                         // compile schedules, size scratch
    void Start(const void* sbuf, void* rbuf);
    void* Wait();                         // returns result pointer
    bool Test();                          // true when complete
    size_t GetTmpBytes() const { return total_tmp_bytes_; }

    ProcessGroup* Group() const { return group_; }
    const OpSpec& Spec() const { return spec_; }
    CompType GetCompType() const { return ctype_; }
    DataType Dtype() const { return dtype_; }
    size_t MessageBytes() const;          // payload bytes (for stats/priority)

    // ---- engine interface ----
    // Advance host-path execution; returns true when all chunks finished.
    bool AdvanceHost(Mesh* mesh);
    // Device-path: issue all chunks onto streams (once) and poll completion.
    bool AdvanceDevice();
    void MarkDone();
    void MarkFailed(const std::string& what);

    ReqState State() const { return state_.load(std::memory_order_acquire); }
    uint64_t StartSeqno() const { return start_seqno_; }
    // hipEvent-measured device comm time of the last completed Start (ms);
    // -1 on the host path or with MLSL_STATS off.
    double LastDeviceCommMs() const;

    // Device-executor access (device_comm.cpp). SendBuf/RecvBuf return the
    // staging override when host buffers were staged into HBM
    // (ReplaceIn/Out analog); User*Buf always return the caller's pointers.
    std::vector<ChunkExec>& Chunks() { return chunks_; }
    const uint8_t* SendBuf() const { return dev_sbuf_ ? dev_sbuf_ : sbuf_; }
    uint8_t* RecvBuf() const { return dev_rbuf_ ? dev_rbuf_ : rbuf_; }
    const uint8_t* UserSendBuf() const { return sbuf_; }
    uint8_t* UserRecvBuf() const { return rbuf_; }
    void SetDeviceBuffers(const uint8_t* s, uint8_t* r) {
        dev_sbuf_ = s;
        dev_rbuf_ = r;
    }

  private:
    friend class Engine;
    void BuildChunks();
    uint64_t MakeTag(size_t chunk, int phase) const;
    uint64_t PairTag(uint32_t seq) const;

    ProcessGroup* group_;
    DataType dtype_;
    CompType ctype_;
    OpSpec spec_;
    bool has_op_ = false;
    bool setup_done_ = false;
    Compression comp_ = Compression::NONE;
    QuantParams qparams_;
    const struct QuantPluginApi* plugin_ = nullptr;  // dlopen'd compression

    std::vector<ChunkExec> chunks_;
    size_t total_tmp_bytes_ = 0;

    // per-Start state
    const uint8_t* sbuf_ = nullptr;
    uint8_t* rbuf_ = nullptr;
    const uint8_t* dev_sbuf_ = nullptr;  // staging overrides (device mode)
    uint8_t* dev_rbuf_ = nullptr;
    uint32_t flow_ = 0;            // group-consistent sequence at Start
    uint64_t start_seqno_ = 0;     // engine-local, for priority ordering
    uint64_t start_ns_ = 0;        // steady-clock ns at Submit (watchdog)
    std::atomic<ReqState> state_{ReqState::IDLE};
    std::string error_;

    // Device-path state (opaque to keep HIP out of this header).
    struct DeviceState;
    std::unique_ptr<DeviceState> dev_;

    std::mutex mu_;
    std::condition_variable cv_;
};

}  // namespace mlsl
