// DL-semantic planner: Session / Operation / Activation / ParameterSet /
// Statistics / CommBlockInfo.
//
// Behavioral parity with the reference planner (src/mlsl_impl.{hpp,cpp}):
// given layer shapes and a Distribution, derive which collective each tensor
// exchange needs (the 5 peer-connection cases, mlsl_impl.cpp:139-241),
// pre-create persistent requests at Commit, compute pack/unpack block
// geometry (mlsl_impl.cpp:243-347), and expose Start/Wait/Test pairs so
// compute and communication overlap. Fresh MI355X-native implementation:
// requests are RCCL/stream plans, scratch lives in HBM, timing is
// rdtsc/hipEvent.
//
// Buffer contract (documented difference from the reference): a request's
// scratch is library-owned (sized at Setup into HBM); the user comm buffer
// passed to StartComm/StartGradientComm only carries payload
// (GetCommBufSize()). WaitComm returns the result pointer; unpack-block
// buf_offsets index into that result region.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "../comm/request.hpp"
#include "../core/types.hpp"
#include "../include/mlsl/mlsl.hpp"

namespace mlsl {

class Session;
class Operation;
class Statistics;

class CommBlockInfo {
  public:
    CommBlockInfo(size_t mb_off, size_t mb_cnt, size_t fm_off, size_t fm_cnt,
                  size_t fm_size, DataType dt, size_t buf_off)
        : mb_off_(mb_off), mb_cnt_(mb_cnt), fm_off_(fm_off), fm_cnt_(fm_cnt),
          fm_size_(fm_size), dt_(dt), buf_off_(buf_off) {}
    size_t GetMbOffset() const { return mb_off_; }
    size_t GetMbCount() const { return mb_cnt_; }
    size_t GetFmOffset() const { return fm_off_; }
    size_t GetFmCount() const { return fm_cnt_; }
    size_t GetFmSize() const { return fm_size_; }
    DataType GetDataType() const { return dt_; }
    size_t GetBufOffset() const { return buf_off_; }

  private:
    size_t mb_off_, mb_cnt_, fm_off_, fm_cnt_, fm_size_;
    DataType dt_;
    size_t buf_off_;
};

// Registration info (reference OperationRegInfo, mlsl.hpp:510-556).
struct RegEntry {
    size_t count;        // feature maps / kernels
    size_t size;         // elements per fm / kernel
    DataType dt;
    bool dist_update = false;
    Compression comp = Compression::NONE;
};

class OperationRegInfo {
  public:
    explicit OperationRegInfo(OpKind kind) : kind_(kind) {}
    void SetName(const char* n) { name_ = n ? n : ""; }
    size_t AddInput(size_t fm_count, size_t fm_size, DataType dt);
    size_t AddOutput(size_t fm_count, size_t fm_size, DataType dt);
    size_t AddParameterSet(size_t kernel_count, size_t kernel_size, DataType dt,
                           bool distributed_update = false,
                           Compression comp = Compression::NONE);
    void Validate(Distribution* d = nullptr);

    OpKind Kind() const { return kind_; }
    const std::string& Name() const { return name_; }

  private:
    friend class Operation;
    OpKind kind_;
    std::string name_;
    std::vector<RegEntry> inputs_, outputs_, params_;
};

class Activation {
  public:
    Activation(Operation* op, const RegEntry& re, bool is_input, size_t idx);
    ~Activation();

    size_t GetGlobalFmCount() const { return global_fm_count_; }
    size_t GetGlobalFmOffset() const { return global_fm_offset_; }
    size_t GetLocalFmCount() const { return local_fm_count_; }
    size_t GetFmSize() const { return fm_size_; }
    DataType GetDataType() const { return dt_; }
    size_t GetPackBlockCount() const { return pack_blocks_.size(); }
    size_t GetUnpackBlockCount() const { return unpack_blocks_.size(); }
    const CommBlockInfo* GetPackBlock(size_t i) const { return &pack_blocks_[i]; }
    const CommBlockInfo* GetUnpackBlock(size_t i) const { return &unpack_blocks_[i]; }
    size_t GetCommBufSize() const { return comm_buf_bytes_; }
    // Library-owned comm buffer (reference Activation::GetCommBuf,
    // include/mlsl.hpp:210-275): lazily allocated at GetCommBufSize() bytes
    // from Environment::Alloc (HBM in device mode), freed with the
    // activation. Users may still pass their own buffer to StartComm.
    void* GetCommBuf();

    void StartComm(void* buf);
    void* WaitComm();

    // wiring
    void SetPeer(Activation* peer);
    bool NeedComm() const { return need_comm_; }
    bool NeedReduce() const { return need_reduce_; }
    Operation* Op() const { return op_; }

  private:
    friend class Operation;
    friend class Session;
    friend class Statistics;
    void InitPeerConnection();  // the 5-case collective selection
    void BuildCase1(Activation* in, Activation* out);
    void BuildCase3(Activation* in, Activation* out);
    void BuildAlltoAllBlocks(Activation* out, Activation* in);

    Operation* op_;
    size_t act_idx_;
    bool is_input_;
    size_t global_fm_count_ = 0, global_fm_offset_ = 0, local_fm_count_ = 0;
    size_t fm_size_ = 0;
    DataType dt_ = DataType::F32;
    bool need_reduce_ = false, need_comm_ = false, peer_set_ = false;
    Activation* peer_ = nullptr;
    std::unique_ptr<CommRequest> req_;
    size_t send_off_bytes_ = 0;   // added to the user buffer at Start
    size_t recv_off_bytes_ = 0;   // recv region offset within the user buffer
    size_t comm_buf_bytes_ = 0;
    void* owned_comm_buf_ = nullptr;
    std::vector<CommBlockInfo> pack_blocks_, unpack_blocks_;
};

class ParameterSet {
  public:
    ParameterSet(Operation* op, const RegEntry& re, size_t idx);
    ~ParameterSet();

    size_t GetGlobalKernelCount() const { return global_kernel_count_; }
    size_t GetGlobalKernelOffset() const { return global_kernel_offset_; }
    size_t GetLocalKernelCount() const { return local_kernel_count_; }
    size_t GetOwnedKernelCount() const { return owned_kernel_count_; }
    size_t GetOwnedKernelOffset() const { return owned_kernel_offset_; }
    size_t GetKernelSize() const { return kernel_size_; }
    DataType GetDataType() const { return dt_; }
    bool IsDistributedUpdate() const { return distributed_update_; }

    void StartGradientComm(void* buf);
    void* WaitGradientComm();
    bool TestGradientComm(void** result);
    void StartIncrementComm(void* buf);
    void* WaitIncrementComm();

    Operation* Op() const { return op_; }

  private:
    friend class Operation;
    friend class Session;
    friend class Statistics;
    Operation* op_;
    size_t param_idx_;
    bool distributed_update_;
    size_t global_kernel_count_ = 0, global_kernel_offset_ = 0;
    size_t local_kernel_count_ = 0, owned_kernel_count_ = 0, owned_kernel_offset_ = 0;
    size_t kernel_size_ = 0;
    DataType dt_ = DataType::F32;
    Compression comp_ = Compression::NONE;
    bool need_comm_ = false;
    std::unique_ptr<CommRequest> grad_req_, inc_req_;
    void* last_grad_buf_ = nullptr;
};

class Operation {
  public:
    Operation(Session* s, const OperationRegInfo& info, Distribution* dist, size_t op_idx);
    ~Operation();

    void SetDistribution(Distribution* d) { dist_ = d; }
    Distribution* GetDistribution() const { return dist_; }
    const char* GetName() const { return name_.c_str(); }
    OpKind Kind() const { return kind_; }
    OpKind GetOpType() const { return kind_; }            // reference name
    bool HasParameterSets() const { return !params_.empty(); }
    size_t OpIndex() const { return op_idx_; }
    Session* GetSession() const { return session_; }

    void SetPrev(Operation* prev, size_t in_idx, size_t prev_out_idx);
    void SetNext(Operation* next, size_t out_idx, size_t next_in_idx);

    size_t GetGlobalMinibatchSize() const;
    size_t GetLocalMinibatchSize() const;
    size_t GetGlobalMinibatchOffset() const;

    size_t GetInputCount() const { return inputs_.size(); }
    Activation* GetInput(size_t i) const { return inputs_[i].get(); }
    size_t GetOutputCount() const { return outputs_.size(); }
    Activation* GetOutput(size_t i) const { return outputs_[i].get(); }
    size_t GetParameterSetCount() const { return params_.size(); }
    ParameterSet* GetParameterSet(size_t i) const { return params_[i].get(); }

    void Commit();  // finalize peer connections + setup requests

  private:
    friend class Session;
    Session* session_;
    Distribution* dist_;
    OpKind kind_;
    std::string name_;
    size_t op_idx_;
    std::vector<std::unique_ptr<Activation>> inputs_, outputs_;
    std::vector<std::unique_ptr<ParameterSet>> params_;
};

// Cycle-accounting statistics (reference src/mlsl_impl_stats.cpp; rdtsc on
// host, per (op, entity) comm/compute attribution, isolation microbenchmark
// at Commit, MLSL_STATS gate).
class Statistics {
  public:
    explicit Statistics(Session* s);

    struct Event {
        size_t op_idx, ent_idx;
        bool is_compute, is_param, is_input_or_inc;
        enum class Action { Start, Wait, Test } action;
    };

    // `device_ms` >= 0 attributes hipEvent-measured GPU comm time of the
    // just-completed request (device mode; Wait-side emitters pass it).
    void Update(const Event& ev, size_t bytes = 0, double device_ms = -1.0);
    void CollectIsolation();   // 10 iterations, skip 4 (ref :48-49)

    // IsEnabled = MLSL_STATS env gate; IsStarted = currently collecting
    // (reference Statistics::IsEnabled/IsStarted, include/mlsl.hpp:655-660).
    void Start() { started_ = true; }
    void Stop() { started_ = false; }
    bool IsEnabled() const { return env_enabled_; }
    bool IsStarted() const { return started_; }
    void Reset();
    void Print();

    unsigned long long GetIsolationCommCycles(size_t op) const;
    unsigned long long GetCommCycles(size_t op) const;
    unsigned long long GetComputeCycles(size_t op) const;
    size_t GetCommSize(size_t op) const;
    unsigned long long GetTotalIsolationCommCycles() const;
    unsigned long long GetTotalCommCycles() const;
    unsigned long long GetTotalComputeCycles() const;
    size_t GetTotalCommSize() const;
    // hipEvent GPU comm time (ns); 0 in host mode / MLSL_STATS off.
    unsigned long long GetCommDeviceNs(size_t op) const;
    unsigned long long GetTotalCommDeviceNs() const;

    // per-entity classes in the printed table (reference per-op table of
    // [KB, Kcycles] per IA/OA/GRAD/INC, mlsl_impl_stats.cpp:97-363)
    enum EntClass { IA = 0, OA = 1, GRAD = 2, INC = 3 };

  private:
    struct OpStats {
        unsigned long long comm_cycles = 0, compute_cycles = 0, isolation_cycles = 0;
        size_t comm_bytes = 0;
        // hipEvent-measured GPU comm nanoseconds (device mode): unlike the
        // host rdtsc deltas, overlapped async comm is attributed to the
        // request that ran it, not to whoever happened to call Wait.
        unsigned long long comm_device_ns = 0;
        struct Ent {
            unsigned long long cycles = 0;
            size_t bytes = 0;
            unsigned long long device_ns = 0;
        } ent[4];
    };
    void EnsureSize(size_t n);

    Session* session_;
    bool env_enabled_ = false;
    bool started_ = false;
    unsigned long long last_ts_ = 0;
    std::vector<OpStats> per_op_;
};

class Session {
  public:
    explicit Session(PhaseKind phase);
    ~Session();

    void SetGlobalMinibatchSize(size_t mb);
    size_t GetGlobalMinibatchSize() const { return global_mb_; }
    PhaseKind Phase() const { return phase_; }
    PhaseKind GetPhaseType() const { return phase_; }     // reference name

    OperationRegInfo* CreateOperationRegInfo(OpKind kind);
    void DeleteOperationRegInfo(OperationRegInfo* i);
    size_t AddOperation(OperationRegInfo* info, Distribution* dist);
    void RemoveOperations();
    size_t GetOperationCount() const { return ops_.size(); }
    Operation* GetOperation(size_t i) const { return ops_[i].get(); }
    void Commit();
    Statistics* GetStats() { return stats_.get(); }

  private:
    PhaseKind phase_;
    size_t global_mb_ = 0;
    std::vector<std::unique_ptr<Operation>> ops_;
    std::vector<std::unique_ptr<OperationRegInfo>> reg_infos_;
    std::unique_ptr<Statistics> stats_;
};

}  // namespace mlsl
