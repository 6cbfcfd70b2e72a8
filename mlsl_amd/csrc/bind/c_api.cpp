// Flat C API over the C++ core (reference analog: src/c_bind.cpp with its
// TRY_CATCH_RETURN convention; ~95 functions). Every call catches C++
// exceptions and returns MLSL_FAILURE with the message retrievable via
// mlsl_last_error().
#include "../include/mlsl/c_api.h"

#include <cstring>
#include <string>

#include "../comm/group.hpp"
#include "../core/log.hpp"
#include "../dl/rma.hpp"
#include "../dl/session.hpp"
#include "../include/mlsl/mlsl.hpp"

using namespace mlsl;

namespace {
thread_local std::string g_last_error;
}

extern "C" {

const char* mlsl_last_error(void) { return g_last_error.c_str(); }

/* used by ops_api.cpp so all bindings share one error slot */
void mlsl_set_last_error_impl(const char* msg) { g_last_error = msg ? msg : ""; }

#define C_TRY try {
#define C_CATCH                                                                \
    return MLSL_SUCCESS;                                                       \
    } catch (const std::exception& e) {                                        \
        g_last_error = e.what();                                               \
        return MLSL_FAILURE;                                                   \
    } catch (...) {                                                            \
        g_last_error = "unknown error";                                        \
        return MLSL_FAILURE;                                                   \
    }

/* ---- environment ---- */

int mlsl_init(int rank, int size) {
    C_TRY Environment::GetEnv().Init(rank, size);
    C_CATCH
}

int mlsl_finalize(void) {
    C_TRY Environment::GetEnv().Finalize();
    C_CATCH
}

int mlsl_configure(const char* config) {
    C_TRY Environment::GetEnv().Configure(config);
    C_CATCH
}

int mlsl_initialized(int* out) {
    C_TRY* out = Environment::GetEnv().IsInitialized() ? 1 : 0;
    C_CATCH
}

int mlsl_get_version(int* out) {
    C_TRY* out = Environment::GetVersion();
    C_CATCH
}

int mlsl_rank(size_t* out) {
    C_TRY* out = Environment::GetEnv().GetProcessIdx();
    C_CATCH
}

int mlsl_world_size(size_t* out) {
    C_TRY* out = Environment::GetEnv().GetProcessCount();
    C_CATCH
}

int mlsl_alloc(size_t sz, size_t align, void** out) {
    C_TRY* out = Environment::GetEnv().Alloc(sz, align);
    C_CATCH
}

int mlsl_dealloc(void* ptr) {
    C_TRY Environment::GetEnv().Free(ptr);
    C_CATCH
}

int mlsl_wait(mlsl_request req, void** result) {
    C_TRY void* r = Environment::GetEnv().Wait(static_cast<CommRequest*>(req));
    if (result) *result = r;
    C_CATCH
}

int mlsl_test(mlsl_request req, int* done, void** result) {
    C_TRY void* r = nullptr;
    bool d = Environment::GetEnv().Test(static_cast<CommRequest*>(req), &r);
    if (done) *done = d ? 1 : 0;
    if (result) *result = r;
    C_CATCH
}

int mlsl_set_quant_params(size_t block_elems) {
    C_TRY QuantParams p;
    p.block_elems = block_elems;
    Environment::GetEnv().SetQuantizationParams(p);
    C_CATCH
}

/* ---- distribution ---- */

int mlsl_distribution_create(size_t d, size_t m, mlsl_distribution* out) {
    C_TRY* out = Environment::GetEnv().CreateDistribution(d, m);
    C_CATCH
}

int mlsl_distribution_create_with_colors(int dc, int mc, mlsl_distribution* out) {
    C_TRY* out = Environment::GetEnv().CreateDistributionWithColors(dc, mc);
    C_CATCH
}

int mlsl_distribution_free(mlsl_distribution d) {
    C_TRY Environment::GetEnv().DeleteDistribution(static_cast<Distribution*>(d));
    C_CATCH
}

int mlsl_distribution_process_idx(mlsl_distribution d, mlsl_group g, size_t* out) {
    C_TRY* out = static_cast<Distribution*>(d)->GetProcessIdx(static_cast<GroupKind>(g));
    C_CATCH
}

int mlsl_distribution_process_count(mlsl_distribution d, mlsl_group g, size_t* out) {
    C_TRY* out = static_cast<Distribution*>(d)->GetProcessCount(static_cast<GroupKind>(g));
    C_CATCH
}

int mlsl_distribution_barrier(mlsl_distribution d, mlsl_group g) {
    C_TRY static_cast<Distribution*>(d)->Barrier(static_cast<GroupKind>(g));
    C_CATCH
}

#define DIST(d) static_cast<Distribution*>(d)
#define DT(x) static_cast<DataType>(x)
#define ROP(x) static_cast<ReduceOp>(x)
#define GK(x) static_cast<GroupKind>(x)

int mlsl_distribution_bcast(mlsl_distribution d, void* buf, size_t count, mlsl_data_type dt,
                            size_t root, mlsl_group g, mlsl_request* out) {
    C_TRY* out = DIST(d)->Bcast(buf, count, DT(dt), root, GK(g));
    C_CATCH
}

int mlsl_distribution_reduce(mlsl_distribution d, const void* sbuf, void* rbuf, size_t count,
                             mlsl_data_type dt, mlsl_reduction op, size_t root, mlsl_group g,
                             mlsl_request* out) {
    C_TRY* out = DIST(d)->Reduce(sbuf, rbuf, count, DT(dt), ROP(op), root, GK(g));
    C_CATCH
}

int mlsl_distribution_all_reduce(mlsl_distribution d, const void* sbuf, void* rbuf,
                                 size_t count, mlsl_data_type dt, mlsl_reduction op,
                                 mlsl_group g, mlsl_request* out) {
    C_TRY* out = DIST(d)->AllReduce(sbuf, rbuf, count, DT(dt), ROP(op), GK(g));
    C_CATCH
}

int mlsl_distribution_all_to_all(mlsl_distribution d, const void* sbuf, size_t send_count,
                                 void* rbuf, mlsl_data_type dt, mlsl_group g,
                                 mlsl_request* out) {
    C_TRY* out = DIST(d)->AlltoAll(sbuf, send_count, rbuf, DT(dt), GK(g));
    C_CATCH
}

int mlsl_distribution_all_to_allv(mlsl_distribution d, const void* sbuf, const size_t* scnt,
                                  const size_t* soff, void* rbuf, const size_t* rcnt,
                                  const size_t* roff, mlsl_data_type dt, mlsl_group g,
                                  mlsl_request* out) {
    C_TRY* out = DIST(d)->AlltoAllv(sbuf, scnt, soff, rbuf, rcnt, roff, DT(dt), GK(g));
    C_CATCH
}

int mlsl_distribution_gather(mlsl_distribution d, const void* sbuf, size_t send_count,
                             void* rbuf, mlsl_data_type dt, size_t root, mlsl_group g,
                             mlsl_request* out) {
    C_TRY* out = DIST(d)->Gather(sbuf, send_count, rbuf, DT(dt), root, GK(g));
    C_CATCH
}

int mlsl_distribution_all_gather(mlsl_distribution d, const void* sbuf, size_t send_count,
                                 void* rbuf, mlsl_data_type dt, mlsl_group g,
                                 mlsl_request* out) {
    C_TRY* out = DIST(d)->AllGather(sbuf, send_count, rbuf, DT(dt), GK(g));
    C_CATCH
}

int mlsl_distribution_all_gatherv(mlsl_distribution d, const void* sbuf, size_t send_count,
                                  void* rbuf, const size_t* rcnt, mlsl_data_type dt,
                                  mlsl_group g, mlsl_request* out) {
    C_TRY* out = DIST(d)->AllGatherv(sbuf, send_count, rbuf, rcnt, DT(dt), GK(g));
    C_CATCH
}

int mlsl_distribution_scatter(mlsl_distribution d, const void* sbuf, void* rbuf,
                              size_t recv_count, mlsl_data_type dt, size_t root, mlsl_group g,
                              mlsl_request* out) {
    C_TRY* out = DIST(d)->Scatter(sbuf, rbuf, recv_count, DT(dt), root, GK(g));
    C_CATCH
}

int mlsl_distribution_reduce_scatter(mlsl_distribution d, const void* sbuf, void* rbuf,
                                     size_t recv_count, mlsl_data_type dt, mlsl_reduction op,
                                     mlsl_group g, mlsl_request* out) {
    C_TRY* out = DIST(d)->ReduceScatter(sbuf, rbuf, recv_count, DT(dt), ROP(op), GK(g));
    C_CATCH
}

int mlsl_distribution_send_recv_list(mlsl_distribution d, const void* sbuf, void* rbuf,
                                     const size_t* peers, const size_t* soffs,
                                     const size_t* scnts, const size_t* roffs,
                                     const size_t* rcnts, size_t npairs,
                                     mlsl_data_type dt, mlsl_group g, mlsl_request* out) {
    C_TRY std::vector<SRPair> pairs(npairs);
    for (size_t i = 0; i < npairs; ++i)
        pairs[i] = SRPair{static_cast<int>(peers[i]), soffs[i], scnts[i], roffs[i],
                          rcnts[i]};
    *out = DIST(d)->SendRecvList(sbuf, rbuf, pairs, DT(dt), GK(g));
    C_CATCH
}

/* ---- persistent requests ---- */

static CommRequest* NewPersistent(mlsl_distribution d, mlsl_group g, mlsl_data_type dt) {
    Distribution* dist = DIST(d);
    return new CommRequest(dist->Group(GK(g)), DT(dt), CompType::GENERIC);
}

int mlsl_persistent_all_reduce(mlsl_distribution d, size_t count, mlsl_data_type dt,
                               mlsl_reduction op, mlsl_group g, int quantized,
                               mlsl_request* out) {
    C_TRY CommRequest* r = NewPersistent(d, g, dt);
    r->AddAllReduce(count, ROP(op));
    if (quantized)
        r->SetCompression(Compression::QUANT_INT8,
                          Environment::GetEnv().GetQuantizationParams());
    r->Setup();
    *out = r;
    C_CATCH
}

int mlsl_persistent_reduce_scatter(mlsl_distribution d, size_t recv_count,
                                   mlsl_data_type dt, mlsl_reduction op, mlsl_group g,
                                   mlsl_request* out) {
    C_TRY CommRequest* r = NewPersistent(d, g, dt);
    r->AddReduceScatter(recv_count, ROP(op));
    r->Setup();
    *out = r;
    C_CATCH
}

int mlsl_persistent_all_gather(mlsl_distribution d, size_t send_count, mlsl_data_type dt,
                               mlsl_group g, mlsl_request* out) {
    C_TRY CommRequest* r = NewPersistent(d, g, dt);
    r->AddAllGather(send_count);
    r->Setup();
    *out = r;
    C_CATCH
}

int mlsl_persistent_all_to_all(mlsl_distribution d, size_t send_count, mlsl_data_type dt,
                               mlsl_group g, mlsl_request* out) {
    C_TRY CommRequest* r = NewPersistent(d, g, dt);
    r->AddAlltoAll(send_count);
    r->Setup();
    *out = r;
    C_CATCH
}

int mlsl_request_start(mlsl_request req, const void* sbuf, void* rbuf) {
    C_TRY static_cast<CommRequest*>(req)->Start(sbuf, rbuf);
    C_CATCH
}

int mlsl_request_wait(mlsl_request req, void** result) {
    C_TRY void* r = static_cast<CommRequest*>(req)->Wait();
    if (result) *result = r;
    C_CATCH
}

int mlsl_request_test(mlsl_request req, int* done) {
    C_TRY* done = static_cast<CommRequest*>(req)->Test() ? 1 : 0;
    C_CATCH
}

int mlsl_request_destroy(mlsl_request req) {
    C_TRY delete static_cast<CommRequest*>(req);
    C_CATCH
}

/* ---- session / planner ---- */

#define SES(s) static_cast<Session*>(s)
#define OP(o) static_cast<Operation*>(o)
#define ACT(a) static_cast<Activation*>(a)
#define PS(p) static_cast<ParameterSet*>(p)
#define REG(i) static_cast<OperationRegInfo*>(i)
#define STATS(x) static_cast<Statistics*>(x)
#define CBI(b) static_cast<CommBlockInfo*>(b)

int mlsl_session_create(mlsl_phase phase, mlsl_session* out) {
    C_TRY* out = Environment::GetEnv().CreateSession(static_cast<PhaseKind>(phase));
    C_CATCH
}

int mlsl_session_free(mlsl_session s) {
    C_TRY Environment::GetEnv().DeleteSession(SES(s));
    C_CATCH
}

int mlsl_session_set_global_minibatch_size(mlsl_session s, size_t mb) {
    C_TRY SES(s)->SetGlobalMinibatchSize(mb);
    C_CATCH
}

int mlsl_session_get_global_minibatch_size(mlsl_session s, size_t* out) {
    C_TRY* out = SES(s)->GetGlobalMinibatchSize();
    C_CATCH
}

int mlsl_session_create_op_reg_info(mlsl_session s, mlsl_op_type ot, mlsl_op_reg_info* out) {
    C_TRY* out = SES(s)->CreateOperationRegInfo(static_cast<OpKind>(ot));
    C_CATCH
}

int mlsl_session_delete_op_reg_info(mlsl_session s, mlsl_op_reg_info info) {
    C_TRY SES(s)->DeleteOperationRegInfo(REG(info));
    C_CATCH
}

int mlsl_session_add_operation(mlsl_session s, mlsl_op_reg_info info, mlsl_distribution d,
                               size_t* out_idx) {
    C_TRY* out_idx = SES(s)->AddOperation(REG(info), DIST(d));
    C_CATCH
}

int mlsl_session_remove_operations(mlsl_session s) {
    C_TRY SES(s)->RemoveOperations();
    C_CATCH
}

int mlsl_session_get_operation_count(mlsl_session s, size_t* out) {
    C_TRY* out = SES(s)->GetOperationCount();
    C_CATCH
}

int mlsl_session_get_operation(mlsl_session s, size_t idx, mlsl_operation* out) {
    C_TRY* out = SES(s)->GetOperation(idx);
    C_CATCH
}

int mlsl_session_commit(mlsl_session s) {
    C_TRY SES(s)->Commit();
    C_CATCH
}

int mlsl_session_get_stats(mlsl_session s, mlsl_statistics* out) {
    C_TRY* out = SES(s)->GetStats();
    C_CATCH
}

/* ---- op reg info ---- */

int mlsl_op_reg_info_set_name(mlsl_op_reg_info i, const char* name) {
    C_TRY REG(i)->SetName(name);
    C_CATCH
}

int mlsl_op_reg_info_add_input(mlsl_op_reg_info i, size_t c, size_t s, mlsl_data_type dt,
                               size_t* out_idx) {
    C_TRY size_t r = REG(i)->AddInput(c, s, DT(dt));
    if (out_idx) *out_idx = r;
    C_CATCH
}

int mlsl_op_reg_info_add_output(mlsl_op_reg_info i, size_t c, size_t s, mlsl_data_type dt,
                                size_t* out_idx) {
    C_TRY size_t r = REG(i)->AddOutput(c, s, DT(dt));
    if (out_idx) *out_idx = r;
    C_CATCH
}

int mlsl_op_reg_info_add_parameter_set(mlsl_op_reg_info i, size_t kc, size_t ks,
                                       mlsl_data_type dt, int dist_update,
                                       mlsl_compression comp, size_t* out_idx) {
    C_TRY size_t r = REG(i)->AddParameterSet(kc, ks, DT(dt), dist_update != 0,
                                             static_cast<Compression>(comp));
    if (out_idx) *out_idx = r;
    C_CATCH
}

int mlsl_op_reg_info_validate(mlsl_op_reg_info i, mlsl_distribution d) {
    C_TRY REG(i)->Validate(DIST(d));
    C_CATCH
}

/* ---- operation ---- */

int mlsl_operation_set_distribution(mlsl_operation o, mlsl_distribution d) {
    C_TRY OP(o)->SetDistribution(DIST(d));
    C_CATCH
}

int mlsl_operation_get_distribution(mlsl_operation o, mlsl_distribution* out) {
    C_TRY* out = OP(o)->GetDistribution();
    C_CATCH
}

int mlsl_operation_set_prev(mlsl_operation o, mlsl_operation prev, size_t a, size_t b) {
    C_TRY OP(o)->SetPrev(OP(prev), a, b);
    C_CATCH
}

int mlsl_operation_set_next(mlsl_operation o, mlsl_operation next, size_t a, size_t b) {
    C_TRY OP(o)->SetNext(OP(next), a, b);
    C_CATCH
}

int mlsl_operation_get_name(mlsl_operation o, const char** out) {
    C_TRY* out = OP(o)->GetName();
    C_CATCH
}

int mlsl_operation_get_global_minibatch_size(mlsl_operation o, size_t* out) {
    C_TRY* out = OP(o)->GetGlobalMinibatchSize();
    C_CATCH
}

int mlsl_operation_get_local_minibatch_size(mlsl_operation o, size_t* out) {
    C_TRY* out = OP(o)->GetLocalMinibatchSize();
    C_CATCH
}

int mlsl_operation_get_global_minibatch_offset(mlsl_operation o, size_t* out) {
    C_TRY* out = OP(o)->GetGlobalMinibatchOffset();
    C_CATCH
}

int mlsl_operation_get_input_count(mlsl_operation o, size_t* out) {
    C_TRY* out = OP(o)->GetInputCount();
    C_CATCH
}

int mlsl_operation_get_input(mlsl_operation o, size_t idx, mlsl_activation* out) {
    C_TRY* out = OP(o)->GetInput(idx);
    C_CATCH
}

int mlsl_operation_get_output_count(mlsl_operation o, size_t* out) {
    C_TRY* out = OP(o)->GetOutputCount();
    C_CATCH
}

int mlsl_operation_get_output(mlsl_operation o, size_t idx, mlsl_activation* out) {
    C_TRY* out = OP(o)->GetOutput(idx);
    C_CATCH
}

int mlsl_operation_get_parameter_set_count(mlsl_operation o, size_t* out) {
    C_TRY* out = OP(o)->GetParameterSetCount();
    C_CATCH
}

int mlsl_operation_get_parameter_set(mlsl_operation o, size_t idx, mlsl_parameter_set* out) {
    C_TRY* out = OP(o)->GetParameterSet(idx);
    C_CATCH
}

/* ---- activation ---- */

int mlsl_activation_get_global_fm_count(mlsl_activation a, size_t* out) {
    C_TRY* out = ACT(a)->GetGlobalFmCount();
    C_CATCH
}

int mlsl_activation_get_global_fm_offset(mlsl_activation a, size_t* out) {
    C_TRY* out = ACT(a)->GetGlobalFmOffset();
    C_CATCH
}

int mlsl_activation_get_local_fm_count(mlsl_activation a, size_t* out) {
    C_TRY* out = ACT(a)->GetLocalFmCount();
    C_CATCH
}

int mlsl_activation_get_fm_size(mlsl_activation a, size_t* out) {
    C_TRY* out = ACT(a)->GetFmSize();
    C_CATCH
}

int mlsl_activation_get_data_type(mlsl_activation a, mlsl_data_type* out) {
    C_TRY* out = static_cast<mlsl_data_type>(ACT(a)->GetDataType());
    C_CATCH
}

int mlsl_activation_get_comm_buf_size(mlsl_activation a, size_t* out) {
    C_TRY* out = ACT(a)->GetCommBufSize();
    C_CATCH
}

int mlsl_activation_get_pack_block_count(mlsl_activation a, size_t* out) {
    C_TRY* out = ACT(a)->GetPackBlockCount();
    C_CATCH
}

int mlsl_activation_get_unpack_block_count(mlsl_activation a, size_t* out) {
    C_TRY* out = ACT(a)->GetUnpackBlockCount();
    C_CATCH
}

int mlsl_activation_get_pack_block(mlsl_activation a, size_t idx, mlsl_comm_block_info* out) {
    C_TRY* out = const_cast<CommBlockInfo*>(ACT(a)->GetPackBlock(idx));
    C_CATCH
}

int mlsl_activation_get_unpack_block(mlsl_activation a, size_t idx, mlsl_comm_block_info* out) {
    C_TRY* out = const_cast<CommBlockInfo*>(ACT(a)->GetUnpackBlock(idx));
    C_CATCH
}

int mlsl_activation_start_comm(mlsl_activation a, void* buf) {
    C_TRY ACT(a)->StartComm(buf);
    C_CATCH
}

int mlsl_activation_wait_comm(mlsl_activation a, void** out) {
    C_TRY void* r = ACT(a)->WaitComm();
    if (out) *out = r;
    C_CATCH
}

/* ---- comm block info ---- */

int mlsl_comm_block_info_get_mb_offset(mlsl_comm_block_info b, size_t* out) {
    C_TRY* out = CBI(b)->GetMbOffset();
    C_CATCH
}
int mlsl_comm_block_info_get_mb_count(mlsl_comm_block_info b, size_t* out) {
    C_TRY* out = CBI(b)->GetMbCount();
    C_CATCH
}
int mlsl_comm_block_info_get_fm_offset(mlsl_comm_block_info b, size_t* out) {
    C_TRY* out = CBI(b)->GetFmOffset();
    C_CATCH
}
int mlsl_comm_block_info_get_fm_count(mlsl_comm_block_info b, size_t* out) {
    C_TRY* out = CBI(b)->GetFmCount();
    C_CATCH
}
int mlsl_comm_block_info_get_fm_size(mlsl_comm_block_info b, size_t* out) {
    C_TRY* out = CBI(b)->GetFmSize();
    C_CATCH
}
int mlsl_comm_block_info_get_data_type(mlsl_comm_block_info b, mlsl_data_type* out) {
    C_TRY* out = static_cast<mlsl_data_type>(CBI(b)->GetDataType());
    C_CATCH
}
int mlsl_comm_block_info_get_buf_offset(mlsl_comm_block_info b, size_t* out) {
    C_TRY* out = CBI(b)->GetBufOffset();
    C_CATCH
}

/* ---- parameter set ---- */

int mlsl_parameter_set_get_global_kernel_count(mlsl_parameter_set p, size_t* out) {
    C_TRY* out = PS(p)->GetGlobalKernelCount();
    C_CATCH
}
int mlsl_parameter_set_get_global_kernel_offset(mlsl_parameter_set p, size_t* out) {
    C_TRY* out = PS(p)->GetGlobalKernelOffset();
    C_CATCH
}
int mlsl_parameter_set_get_local_kernel_count(mlsl_parameter_set p, size_t* out) {
    C_TRY* out = PS(p)->GetLocalKernelCount();
    C_CATCH
}
int mlsl_parameter_set_get_owned_kernel_count(mlsl_parameter_set p, size_t* out) {
    C_TRY* out = PS(p)->GetOwnedKernelCount();
    C_CATCH
}
int mlsl_parameter_set_get_owned_kernel_offset(mlsl_parameter_set p, size_t* out) {
    C_TRY* out = PS(p)->GetOwnedKernelOffset();
    C_CATCH
}
int mlsl_parameter_set_get_kernel_size(mlsl_parameter_set p, size_t* out) {
    C_TRY* out = PS(p)->GetKernelSize();
    C_CATCH
}
int mlsl_parameter_set_get_data_type(mlsl_parameter_set p, mlsl_data_type* out) {
    C_TRY* out = static_cast<mlsl_data_type>(PS(p)->GetDataType());
    C_CATCH
}
int mlsl_parameter_set_is_distributed_update(mlsl_parameter_set p, int* out) {
    C_TRY* out = PS(p)->IsDistributedUpdate() ? 1 : 0;
    C_CATCH
}
int mlsl_parameter_set_start_gradient_comm(mlsl_parameter_set p, void* buf) {
    C_TRY PS(p)->StartGradientComm(buf);
    C_CATCH
}
int mlsl_parameter_set_wait_gradient_comm(mlsl_parameter_set p, void** out) {
    C_TRY void* r = PS(p)->WaitGradientComm();
    if (out) *out = r;
    C_CATCH
}
int mlsl_parameter_set_test_gradient_comm(mlsl_parameter_set p, int* done, void** out) {
    C_TRY void* r = nullptr;
    bool d = PS(p)->TestGradientComm(&r);
    if (done) *done = d ? 1 : 0;
    if (out) *out = r;
    C_CATCH
}
int mlsl_parameter_set_start_increment_comm(mlsl_parameter_set p, void* buf) {
    C_TRY PS(p)->StartIncrementComm(buf);
    C_CATCH
}
int mlsl_parameter_set_wait_increment_comm(mlsl_parameter_set p, void** out) {
    C_TRY void* r = PS(p)->WaitIncrementComm();
    if (out) *out = r;
    C_CATCH
}

/* ---- statistics ---- */

int mlsl_statistics_start(mlsl_statistics st) {
    C_TRY STATS(st)->Start();
    C_CATCH
}
int mlsl_statistics_stop(mlsl_statistics st) {
    C_TRY STATS(st)->Stop();
    C_CATCH
}
int mlsl_statistics_reset(mlsl_statistics st) {
    C_TRY STATS(st)->Reset();
    C_CATCH
}
int mlsl_statistics_is_enabled(mlsl_statistics st, int* out) {
    C_TRY* out = STATS(st)->IsEnabled() ? 1 : 0;
    C_CATCH
}
int mlsl_statistics_print(mlsl_statistics st) {
    C_TRY STATS(st)->Print();
    C_CATCH
}
int mlsl_statistics_get_isolation_comm_cycles(mlsl_statistics st, size_t op,
                                              unsigned long long* out) {
    C_TRY* out = STATS(st)->GetIsolationCommCycles(op);
    C_CATCH
}
int mlsl_statistics_get_comm_size(mlsl_statistics st, size_t op, size_t* out) {
    C_TRY* out = STATS(st)->GetCommSize(op);
    C_CATCH
}
int mlsl_statistics_get_comm_cycles(mlsl_statistics st, size_t op, unsigned long long* out) {
    C_TRY* out = STATS(st)->GetCommCycles(op);
    C_CATCH
}
int mlsl_statistics_get_compute_cycles(mlsl_statistics st, size_t op, unsigned long long* out) {
    C_TRY* out = STATS(st)->GetComputeCycles(op);
    C_CATCH
}
int mlsl_statistics_get_total_isolation_comm_cycles(mlsl_statistics st, unsigned long long* out) {
    C_TRY* out = STATS(st)->GetTotalIsolationCommCycles();
    C_CATCH
}
int mlsl_statistics_get_total_comm_size(mlsl_statistics st, size_t* out) {
    C_TRY* out = STATS(st)->GetTotalCommSize();
    C_CATCH
}
int mlsl_statistics_get_total_comm_cycles(mlsl_statistics st, unsigned long long* out) {
    C_TRY* out = STATS(st)->GetTotalCommCycles();
    C_CATCH
}
int mlsl_statistics_get_total_compute_cycles(mlsl_statistics st, unsigned long long* out) {
    C_TRY* out = STATS(st)->GetTotalComputeCycles();
    C_CATCH
}
int mlsl_memcpy(void* dst, const void* src, size_t bytes) {
    C_TRY mlsl::Environment::GetEnv().Memcpy(dst, src, bytes);
    C_CATCH
}
int mlsl_statistics_get_comm_device_ns(mlsl_statistics st, size_t op,
                                       unsigned long long* out) {
    C_TRY* out = STATS(st)->GetCommDeviceNs(op);
    C_CATCH
}
int mlsl_statistics_get_total_comm_device_ns(mlsl_statistics st,
                                             unsigned long long* out) {
    C_TRY* out = STATS(st)->GetTotalCommDeviceNs();
    C_CATCH
}

/* ---- RMA windows ---- */
int mlsl_win_allocate(mlsl_distribution dist, mlsl_group g, size_t bytes,
                      mlsl_win* out) {
    C_TRY* out = new mlsl::RmaWindow(static_cast<mlsl::Distribution*>(dist),
                                     static_cast<mlsl::GroupKind>(g), bytes);
    C_CATCH
}
int mlsl_win_free(mlsl_win w) {
    C_TRY delete static_cast<mlsl::RmaWindow*>(w);
    C_CATCH
}
int mlsl_win_buffer(mlsl_win w, void** base, size_t* bytes) {
    C_TRY auto* win = static_cast<mlsl::RmaWindow*>(w);
    if (base) *base = win->Buffer();
    if (bytes) *bytes = win->Bytes();
    C_CATCH
}
int mlsl_win_put(mlsl_win w, const void* src, size_t bytes, size_t target,
                 size_t target_off) {
    C_TRY static_cast<mlsl::RmaWindow*>(w)->Put(src, bytes, target, target_off);
    C_CATCH
}
int mlsl_win_get(mlsl_win w, void* dst, size_t bytes, size_t target,
                 size_t target_off) {
    C_TRY static_cast<mlsl::RmaWindow*>(w)->Get(dst, bytes, target, target_off);
    C_CATCH
}
int mlsl_win_fence(mlsl_win w) {
    C_TRY static_cast<mlsl::RmaWindow*>(w)->Fence();
    C_CATCH
}


/* ---- native getters added for reference parity ---- */

int mlsl_activation_get_comm_buf(mlsl_activation a, void** out) {
    C_TRY* out = ACT(a)->GetCommBuf();
    C_CATCH
}
int mlsl_statistics_is_started(mlsl_statistics st, int* out) {
    C_TRY* out = STATS(st)->IsStarted() ? 1 : 0;
    C_CATCH
}
int mlsl_session_get_phase_type(mlsl_session s, mlsl_phase* out) {
    C_TRY* out = static_cast<mlsl_phase>(SES(s)->Phase());
    C_CATCH
}
int mlsl_operation_get_op_type(mlsl_operation o, mlsl_op_type* out) {
    C_TRY* out = static_cast<mlsl_op_type>(OP(o)->Kind());
    C_CATCH
}
int mlsl_operation_get_session(mlsl_operation o, mlsl_session* out) {
    C_TRY* out = OP(o)->GetSession();
    C_CATCH
}
int mlsl_operation_has_parameter_sets(mlsl_operation o, int* out) {
    C_TRY* out = OP(o)->GetParameterSetCount() > 0 ? 1 : 0;
    C_CATCH
}
int mlsl_get_quant_params(size_t* block_elems) {
    C_TRY* block_elems = Environment::GetEnv().GetQuantizationParams().block_elems;
    C_CATCH
}

/* ---- reference-name compatibility layer (exact include/mlsl.h
 * signatures; forwards to the native functions above). The environment
 * handle is a token for the singleton. ---- */

static int g_env_token;

int mlsl_environment_get_env(mlsl_environment* env) {
    C_TRY* env = &g_env_token;
    C_CATCH
}
int mlsl_environment_get_version(int* version) { return mlsl_get_version(version); }
int mlsl_environment_configure(mlsl_environment, const char* config) {
    return mlsl_configure(config);
}
int mlsl_environment_init(mlsl_environment, int*, char***) { return mlsl_init(-1, -1); }
int mlsl_environment_finalize(mlsl_environment) { return mlsl_finalize(); }
int mlsl_environment_is_initialized(mlsl_environment, int* is_initialized) {
    return mlsl_initialized(is_initialized);
}
int mlsl_environment_get_process_idx(mlsl_environment, size_t* process_idx) {
    return mlsl_rank(process_idx);
}
int mlsl_environment_get_process_count(mlsl_environment, size_t* process_count) {
    return mlsl_world_size(process_count);
}
int mlsl_environment_create_session(mlsl_environment, mlsl_phase phase_type,
                                    mlsl_session* session) {
    return mlsl_session_create(phase_type, session);
}
int mlsl_environment_delete_session(mlsl_environment, mlsl_session session) {
    return mlsl_session_free(session);
}
int mlsl_environment_create_distribution(mlsl_environment, size_t data_partitions,
                                         size_t model_partitions, mlsl_distribution* dist) {
    return mlsl_distribution_create(data_partitions, model_partitions, dist);
}
int mlsl_environment_delete_distribution(mlsl_environment, mlsl_distribution dist) {
    return mlsl_distribution_free(dist);
}
int mlsl_environment_wait(mlsl_environment, mlsl_comm_req req) {
    void* result = nullptr;
    return mlsl_wait(req, &result);
}
int mlsl_environment_test(mlsl_environment, mlsl_comm_req req, int* is_completed) {
    void* result = nullptr;
    return mlsl_test(req, is_completed, &result);
}
int mlsl_environment_alloc(mlsl_environment, size_t size, size_t alignment, void** ptr) {
    return mlsl_alloc(size, alignment, ptr);
}
int mlsl_environment_free(mlsl_environment, void* ptr) { return mlsl_dealloc(ptr); }
int mlsl_environment_set_quantization_params(mlsl_environment, mlsl_quant_params_t* params) {
    C_TRY
    if (!params || params->elem_in_block == 0)
        throw Error("quant params: elem_in_block must be > 0");
    QuantParams qp;
    qp.block_elems = params->elem_in_block;
    if (params->lib_path && params->lib_path[0]) {
        // dlopen'd compression plugin (reference quant/quant.c): the host
        // compressed path calls these three functions; block_size declares
        // the plugin's wire-block bytes.
        qp.lib_path = params->lib_path;
        if (params->quant_buffer_func_name)
            qp.quant_fn = params->quant_buffer_func_name;
        if (params->dequant_buffer_func_name)
            qp.dequant_fn = params->dequant_buffer_func_name;
        if (params->reduce_sum_func_name)
            qp.reduce_fn = params->reduce_sum_func_name;
        if (params->block_size) qp.block_bytes = params->block_size;
    }
    Environment::GetEnv().SetQuantizationParams(qp);
    C_CATCH
}
int mlsl_environment_get_quantization_params(mlsl_environment, mlsl_quant_params_t* params) {
    C_TRY
    const QuantParams& qp = Environment::GetEnv().GetQuantizationParams();
    params->lib_path = nullptr;
    params->quant_buffer_func_name = nullptr;
    params->dequant_buffer_func_name = nullptr;
    params->reduce_sum_func_name = nullptr;
    params->elem_in_block = qp.block_elems;
    params->block_size = qp.WireBlockBytes();
    C_CATCH
}
int mlsl_distribution_get_process_count(mlsl_distribution dist, mlsl_group group_type,
                                        size_t* process_count) {
    return mlsl_distribution_process_count(dist, group_type, process_count);
}
int mlsl_distribution_get_process_idx(mlsl_distribution dist, mlsl_group group_type,
                                      size_t* process_idx) {
    return mlsl_distribution_process_idx(dist, group_type, process_idx);
}
int mlsl_session_create_operation_reg_info(mlsl_session session, mlsl_op_type op_type,
                                           mlsl_op_reg_info* reg_info) {
    return mlsl_session_create_op_reg_info(session, op_type, reg_info);
}
int mlsl_session_delete_operation_reg_info(mlsl_session session, mlsl_op_reg_info reg_info) {
    return mlsl_session_delete_op_reg_info(session, reg_info);
}
int mlsl_session_add_operation_with_distribution(mlsl_session session,
                                                 mlsl_op_reg_info reg_info,
                                                 mlsl_distribution dist, size_t* op_idx) {
    return mlsl_session_add_operation(session, reg_info, dist, op_idx);
}
int mlsl_operation_reg_info_set_name(mlsl_op_reg_info reg_info, const char* name) {
    return mlsl_op_reg_info_set_name(reg_info, name);
}
int mlsl_operation_reg_info_add_input(mlsl_op_reg_info reg_info, size_t fm_count,
                                      size_t fm_size, mlsl_data_type dtype) {
    size_t idx = 0;
    return mlsl_op_reg_info_add_input(reg_info, fm_count, fm_size, dtype, &idx);
}
int mlsl_operation_reg_info_add_output(mlsl_op_reg_info reg_info, size_t fm_count,
                                       size_t fm_size, mlsl_data_type dtype) {
    size_t idx = 0;
    return mlsl_op_reg_info_add_output(reg_info, fm_count, fm_size, dtype, &idx);
}
int mlsl_operation_reg_info_add_parameter_set(mlsl_op_reg_info reg_info, size_t kernel_count,
                                              size_t kernel_size, mlsl_data_type dtype,
                                              int dist_update) {
    size_t idx = 0;
    return mlsl_op_reg_info_add_parameter_set(reg_info, kernel_count, kernel_size, dtype,
                                              dist_update, MLSL_CT_NONE, &idx);
}
int mlsl_operation_reg_info_add_parameter_set_with_compress(
    mlsl_op_reg_info reg_info, size_t kernel_count, size_t kernel_size,
    mlsl_data_type dtype, int dist_update, mlsl_compression compress_type) {
    size_t idx = 0;
    return mlsl_op_reg_info_add_parameter_set(reg_info, kernel_count, kernel_size, dtype,
                                              dist_update, compress_type, &idx);
}
int mlsl_operation_reg_info_validate(mlsl_op_reg_info reg_info, mlsl_distribution dist) {
    return mlsl_op_reg_info_validate(reg_info, dist);
}

}  // extern "C"
