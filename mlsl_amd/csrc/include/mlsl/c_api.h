/* mlsl_amd flat C API — capability parity with reference include/mlsl.h
 * (~95 mlsl_* functions over opaque handles, c_bind.cpp TRY_CATCH_RETURN
 * error-code convention). Backing implementation is the MI355X-native C++
 * core; this header is consumed by the Python ctypes binding and by C
 * frameworks. */
#ifndef MLSL_AMD_C_API_H
#define MLSL_AMD_C_API_H

#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

#define MLSL_SUCCESS 0
#define MLSL_FAILURE -1

typedef void* mlsl_distribution;
typedef void* mlsl_request;
typedef void* mlsl_session;
typedef void* mlsl_operation;
typedef void* mlsl_op_reg_info;
typedef void* mlsl_activation;
typedef void* mlsl_parameter_set;
typedef void* mlsl_statistics;
typedef void* mlsl_comm_block_info;

/* enums match core/types.hpp */
typedef enum { MLSL_DT_F32 = 0, MLSL_DT_F64 = 1, MLSL_DT_U8 = 2, MLSL_DT_BF16 = 3,
               MLSL_DT_F16 = 4, MLSL_DT_I32 = 5, MLSL_DT_I64 = 6 } mlsl_data_type;
typedef enum { MLSL_RT_SUM = 0, MLSL_RT_MIN = 1, MLSL_RT_MAX = 2 } mlsl_reduction;
typedef enum { MLSL_GT_DATA = 0, MLSL_GT_MODEL = 1, MLSL_GT_GLOBAL = 2 } mlsl_group;
typedef enum { MLSL_OT_CC = 0, MLSL_OT_BIAS = 1, MLSL_OT_ACT = 2, MLSL_OT_POOL = 3,
               MLSL_OT_SPLIT = 4, MLSL_OT_CONCAT = 5, MLSL_OT_BCAST = 6,
               MLSL_OT_REDUCE = 7, MLSL_OT_DATA = 8, MLSL_OT_EVAL = 9 } mlsl_op_type;
typedef enum { MLSL_PT_TRAIN = 0, MLSL_PT_TEST = 1 } mlsl_phase;
typedef enum { MLSL_CT_NONE = 0, MLSL_CT_QUANT_INT8 = 1 } mlsl_compression;

/* last error message for the calling thread (valid until next API call) */
const char* mlsl_last_error(void);

/* environment */
int mlsl_init(int rank, int size);  /* -1,-1: from env (RANK/WORLD_SIZE) */
int mlsl_finalize(void);
int mlsl_configure(const char* config);  /* "color=N" tenant re-split */
int mlsl_initialized(int* out);
int mlsl_get_version(int* out);
int mlsl_rank(size_t* out);
int mlsl_world_size(size_t* out);
int mlsl_alloc(size_t sz, size_t align, void** out);
int mlsl_dealloc(void* ptr);
int mlsl_wait(mlsl_request req, void** result);
int mlsl_test(mlsl_request req, int* done, void** result);
int mlsl_set_quant_params(size_t block_elems);

/* distribution */
int mlsl_distribution_create(size_t data_parts, size_t model_parts, mlsl_distribution* out);
int mlsl_distribution_create_with_colors(int data_color, int model_color, mlsl_distribution* out);
int mlsl_distribution_free(mlsl_distribution d);
int mlsl_distribution_process_idx(mlsl_distribution d, mlsl_group g, size_t* out);
int mlsl_distribution_process_count(mlsl_distribution d, mlsl_group g, size_t* out);
int mlsl_distribution_barrier(mlsl_distribution d, mlsl_group g);
int mlsl_distribution_bcast(mlsl_distribution d, void* buf, size_t count,
                            mlsl_data_type dt, size_t root, mlsl_group g, mlsl_request* out);
int mlsl_distribution_reduce(mlsl_distribution d, const void* sbuf, void* rbuf, size_t count,
                             mlsl_data_type dt, mlsl_reduction op, size_t root, mlsl_group g,
                             mlsl_request* out);
int mlsl_distribution_all_reduce(mlsl_distribution d, const void* sbuf, void* rbuf, size_t count,
                                 mlsl_data_type dt, mlsl_reduction op, mlsl_group g,
                                 mlsl_request* out);
int mlsl_distribution_all_to_all(mlsl_distribution d, const void* sbuf, size_t send_count,
                                 void* rbuf, mlsl_data_type dt, mlsl_group g, mlsl_request* out);
int mlsl_distribution_all_to_allv(mlsl_distribution d, const void* sbuf, const size_t* scnt,
                                  const size_t* soff, void* rbuf, const size_t* rcnt,
                                  const size_t* roff, mlsl_data_type dt, mlsl_group g,
                                  mlsl_request* out);
int mlsl_distribution_gather(mlsl_distribution d, const void* sbuf, size_t send_count, void* rbuf,
                             mlsl_data_type dt, size_t root, mlsl_group g, mlsl_request* out);
int mlsl_distribution_all_gather(mlsl_distribution d, const void* sbuf, size_t send_count,
                                 void* rbuf, mlsl_data_type dt, mlsl_group g, mlsl_request* out);
int mlsl_distribution_all_gatherv(mlsl_distribution d, const void* sbuf, size_t send_count,
                                  void* rbuf, const size_t* rcnt, mlsl_data_type dt, mlsl_group g,
                                  mlsl_request* out);
int mlsl_distribution_scatter(mlsl_distribution d, const void* sbuf, void* rbuf, size_t recv_count,
                              mlsl_data_type dt, size_t root, mlsl_group g, mlsl_request* out);
int mlsl_distribution_reduce_scatter(mlsl_distribution d, const void* sbuf, void* rbuf,
                                     size_t recv_count, mlsl_data_type dt, mlsl_reduction op,
                                     mlsl_group g, mlsl_request* out);

int mlsl_distribution_send_recv_list(mlsl_distribution d, const void* sbuf, void* rbuf,
                                     const size_t* peers, const size_t* soffs,
                                     const size_t* scnts, const size_t* roffs,
                                     const size_t* rcnts, size_t npairs,
                                     mlsl_data_type dt, mlsl_group g, mlsl_request* out);

/* persistent requests: describe once, Start/Wait/Test every iteration
 * (the reference's Session-level persistent-request contract exposed for
 * generic collectives; hot loops skip per-call planning/allocation). */
int mlsl_persistent_all_reduce(mlsl_distribution d, size_t count, mlsl_data_type dt,
                               mlsl_reduction op, mlsl_group g, int quantized,
                               mlsl_request* out);
int mlsl_persistent_reduce_scatter(mlsl_distribution d, size_t recv_count,
                                   mlsl_data_type dt, mlsl_reduction op, mlsl_group g,
                                   mlsl_request* out);
int mlsl_persistent_all_gather(mlsl_distribution d, size_t send_count, mlsl_data_type dt,
                               mlsl_group g, mlsl_request* out);
int mlsl_persistent_all_to_all(mlsl_distribution d, size_t send_count, mlsl_data_type dt,
                               mlsl_group g, mlsl_request* out);
int mlsl_request_start(mlsl_request req, const void* sbuf, void* rbuf);
int mlsl_request_wait(mlsl_request req, void** result);
int mlsl_request_test(mlsl_request req, int* done);
int mlsl_request_destroy(mlsl_request req);

/* session / planner (reference mlsl.hpp:731-795) */
int mlsl_session_create(mlsl_phase phase, mlsl_session* out);
int mlsl_session_free(mlsl_session s);
int mlsl_session_set_global_minibatch_size(mlsl_session s, size_t mb);
int mlsl_session_get_global_minibatch_size(mlsl_session s, size_t* out);
int mlsl_session_create_op_reg_info(mlsl_session s, mlsl_op_type ot, mlsl_op_reg_info* out);
int mlsl_session_delete_op_reg_info(mlsl_session s, mlsl_op_reg_info info);
int mlsl_session_add_operation(mlsl_session s, mlsl_op_reg_info info, mlsl_distribution d,
                               size_t* out_idx);
int mlsl_session_remove_operations(mlsl_session s);
int mlsl_session_get_operation_count(mlsl_session s, size_t* out);
int mlsl_session_get_operation(mlsl_session s, size_t idx, mlsl_operation* out);
int mlsl_session_commit(mlsl_session s);
int mlsl_session_get_stats(mlsl_session s, mlsl_statistics* out);

/* operation registration info (reference mlsl.hpp:510-556) */
int mlsl_op_reg_info_set_name(mlsl_op_reg_info i, const char* name);
int mlsl_op_reg_info_add_input(mlsl_op_reg_info i, size_t fm_count, size_t fm_size,
                               mlsl_data_type dt, size_t* out_idx);
int mlsl_op_reg_info_add_output(mlsl_op_reg_info i, size_t fm_count, size_t fm_size,
                                mlsl_data_type dt, size_t* out_idx);
int mlsl_op_reg_info_add_parameter_set(mlsl_op_reg_info i, size_t kernel_count, size_t kernel_size,
                                       mlsl_data_type dt, int distributed_update,
                                       mlsl_compression compress, size_t* out_idx);
int mlsl_op_reg_info_validate(mlsl_op_reg_info i, mlsl_distribution d);

/* operation (reference mlsl.hpp:564-645) */
int mlsl_operation_set_distribution(mlsl_operation o, mlsl_distribution d);
int mlsl_operation_get_distribution(mlsl_operation o, mlsl_distribution* out);
int mlsl_operation_set_prev(mlsl_operation o, mlsl_operation prev, size_t act_idx,
                            size_t prev_out_idx);
int mlsl_operation_set_next(mlsl_operation o, mlsl_operation next, size_t act_idx,
                            size_t next_in_idx);
int mlsl_operation_get_name(mlsl_operation o, const char** out);
int mlsl_operation_get_global_minibatch_size(mlsl_operation o, size_t* out);
int mlsl_operation_get_local_minibatch_size(mlsl_operation o, size_t* out);
int mlsl_operation_get_global_minibatch_offset(mlsl_operation o, size_t* out);
int mlsl_operation_get_input_count(mlsl_operation o, size_t* out);
int mlsl_operation_get_input(mlsl_operation o, size_t idx, mlsl_activation* out);
int mlsl_operation_get_output_count(mlsl_operation o, size_t* out);
int mlsl_operation_get_output(mlsl_operation o, size_t idx, mlsl_activation* out);
int mlsl_operation_get_parameter_set_count(mlsl_operation o, size_t* out);
int mlsl_operation_get_parameter_set(mlsl_operation o, size_t idx, mlsl_parameter_set* out);

/* activation (reference mlsl.hpp:210-268) */
int mlsl_activation_get_global_fm_count(mlsl_activation a, size_t* out);
int mlsl_activation_get_global_fm_offset(mlsl_activation a, size_t* out);
int mlsl_activation_get_local_fm_count(mlsl_activation a, size_t* out);
int mlsl_activation_get_fm_size(mlsl_activation a, size_t* out);
int mlsl_activation_get_data_type(mlsl_activation a, mlsl_data_type* out);
int mlsl_activation_get_comm_buf_size(mlsl_activation a, size_t* out);
int mlsl_activation_get_pack_block_count(mlsl_activation a, size_t* out);
int mlsl_activation_get_unpack_block_count(mlsl_activation a, size_t* out);
int mlsl_activation_get_pack_block(mlsl_activation a, size_t idx, mlsl_comm_block_info* out);
int mlsl_activation_get_unpack_block(mlsl_activation a, size_t idx, mlsl_comm_block_info* out);
int mlsl_activation_start_comm(mlsl_activation a, void* buf);
int mlsl_activation_wait_comm(mlsl_activation a, void** out);

/* comm block info (reference mlsl.hpp:177-203) */
int mlsl_comm_block_info_get_mb_offset(mlsl_comm_block_info b, size_t* out);
int mlsl_comm_block_info_get_mb_count(mlsl_comm_block_info b, size_t* out);
int mlsl_comm_block_info_get_fm_offset(mlsl_comm_block_info b, size_t* out);
int mlsl_comm_block_info_get_fm_count(mlsl_comm_block_info b, size_t* out);
int mlsl_comm_block_info_get_fm_size(mlsl_comm_block_info b, size_t* out);
int mlsl_comm_block_info_get_data_type(mlsl_comm_block_info b, mlsl_data_type* out);
int mlsl_comm_block_info_get_buf_offset(mlsl_comm_block_info b, size_t* out);

/* parameter set (reference mlsl.hpp:276-340) */
int mlsl_parameter_set_get_global_kernel_count(mlsl_parameter_set p, size_t* out);
int mlsl_parameter_set_get_global_kernel_offset(mlsl_parameter_set p, size_t* out);
int mlsl_parameter_set_get_local_kernel_count(mlsl_parameter_set p, size_t* out);
int mlsl_parameter_set_get_owned_kernel_count(mlsl_parameter_set p, size_t* out);
int mlsl_parameter_set_get_owned_kernel_offset(mlsl_parameter_set p, size_t* out);
int mlsl_parameter_set_get_kernel_size(mlsl_parameter_set p, size_t* out);
int mlsl_parameter_set_get_data_type(mlsl_parameter_set p, mlsl_data_type* out);
int mlsl_parameter_set_is_distributed_update(mlsl_parameter_set p, int* out);
int mlsl_parameter_set_start_gradient_comm(mlsl_parameter_set p, void* buf);
int mlsl_parameter_set_wait_gradient_comm(mlsl_parameter_set p, void** out);
int mlsl_parameter_set_test_gradient_comm(mlsl_parameter_set p, int* done, void** out);
int mlsl_parameter_set_start_increment_comm(mlsl_parameter_set p, void* buf);
int mlsl_parameter_set_wait_increment_comm(mlsl_parameter_set p, void** out);

/* statistics (reference mlsl.hpp:651-726) */
int mlsl_statistics_start(mlsl_statistics st);
int mlsl_statistics_stop(mlsl_statistics st);
int mlsl_statistics_reset(mlsl_statistics st);
int mlsl_statistics_is_enabled(mlsl_statistics st, int* out);
int mlsl_statistics_print(mlsl_statistics st);
int mlsl_statistics_get_isolation_comm_cycles(mlsl_statistics st, size_t op_idx,
                                              unsigned long long* out);
int mlsl_statistics_get_comm_size(mlsl_statistics st, size_t op_idx, size_t* out);
int mlsl_statistics_get_comm_cycles(mlsl_statistics st, size_t op_idx, unsigned long long* out);
int mlsl_statistics_get_compute_cycles(mlsl_statistics st, size_t op_idx, unsigned long long* out);
int mlsl_statistics_get_total_isolation_comm_cycles(mlsl_statistics st, unsigned long long* out);
int mlsl_statistics_get_total_comm_size(mlsl_statistics st, size_t* out);
int mlsl_statistics_get_total_comm_cycles(mlsl_statistics st, unsigned long long* out);
int mlsl_statistics_get_total_compute_cycles(mlsl_statistics st, unsigned long long* out);

/* ---- native getters added for reference parity ---- */
int mlsl_activation_get_comm_buf(mlsl_activation a, void** out);  /* library-owned, lazy */
int mlsl_statistics_is_started(mlsl_statistics st, int* out);
int mlsl_session_get_phase_type(mlsl_session s, mlsl_phase* out);
int mlsl_operation_get_op_type(mlsl_operation o, mlsl_op_type* out);
int mlsl_operation_get_session(mlsl_operation o, mlsl_session* out);
int mlsl_operation_has_parameter_sets(mlsl_operation o, int* out);
int mlsl_get_quant_params(size_t* block_elems);

/* ---- reference-name compatibility layer (drop-in for include/mlsl.h) ----
 * Exact reference signatures so migrating C code links unchanged.
 * mlsl_environment is a token handle for the singleton environment;
 * mlsl_quant_params_t mirrors the reference struct (the dlopen'd plugin
 * fields are accepted and ignored: quantization kernels are built in). */
typedef void* mlsl_environment;
typedef void* mlsl_comm_req;
typedef struct {
    char* lib_path;                  /* ignored: built-in CDNA4/CPU kernels */
    char* quant_buffer_func_name;    /* ignored */
    char* dequant_buffer_func_name;  /* ignored */
    char* reduce_sum_func_name;      /* ignored */
    size_t block_size;               /* wire block bytes (elem_in_block + 8) */
    size_t elem_in_block;            /* elements per quantization block */
} mlsl_quant_params_t;

int mlsl_environment_get_env(mlsl_environment* env);
int mlsl_environment_get_version(int* version);
int mlsl_environment_configure(mlsl_environment env, const char* config);
int mlsl_environment_init(mlsl_environment env, int* argc, char** argv[]);
int mlsl_environment_finalize(mlsl_environment env);
int mlsl_environment_is_initialized(mlsl_environment env, int* is_initialized);
int mlsl_environment_get_process_idx(mlsl_environment env, size_t* process_idx);
int mlsl_environment_get_process_count(mlsl_environment env, size_t* process_count);
int mlsl_environment_create_session(mlsl_environment env, mlsl_phase phase_type, mlsl_session* session);
int mlsl_environment_delete_session(mlsl_environment env, mlsl_session session);
int mlsl_environment_create_distribution(mlsl_environment env, size_t data_partitions, size_t model_partitions, mlsl_distribution* dist);
int mlsl_environment_delete_distribution(mlsl_environment env, mlsl_distribution dist);
int mlsl_environment_wait(mlsl_environment env, mlsl_comm_req req);
int mlsl_environment_test(mlsl_environment env, mlsl_comm_req req, int* is_completed);
int mlsl_environment_alloc(mlsl_environment env, size_t size, size_t alignment, void** ptr);
int mlsl_environment_free(mlsl_environment env, void* ptr);
int mlsl_environment_set_quantization_params(mlsl_environment env, mlsl_quant_params_t* params);
int mlsl_environment_get_quantization_params(mlsl_environment env, mlsl_quant_params_t* params);
int mlsl_distribution_get_process_count(mlsl_distribution dist, mlsl_group group_type, size_t* process_count);
int mlsl_distribution_get_process_idx(mlsl_distribution dist, mlsl_group group_type, size_t* process_idx);
int mlsl_session_create_operation_reg_info(mlsl_session session, mlsl_op_type op_type, mlsl_op_reg_info* reg_info);
int mlsl_session_delete_operation_reg_info(mlsl_session session, mlsl_op_reg_info reg_info);
int mlsl_session_add_operation_with_distribution(mlsl_session session, mlsl_op_reg_info reg_info, mlsl_distribution dist, size_t* op_idx);
int mlsl_operation_reg_info_set_name(mlsl_op_reg_info reg_info, const char* name);
int mlsl_operation_reg_info_add_input(mlsl_op_reg_info reg_info, size_t fm_count, size_t fm_size, mlsl_data_type dtype);
int mlsl_operation_reg_info_add_output(mlsl_op_reg_info reg_info, size_t fm_count, size_t fm_size, mlsl_data_type dtype);
int mlsl_operation_reg_info_add_parameter_set(mlsl_op_reg_info reg_info, size_t kernel_count, size_t kernel_size, mlsl_data_type dtype, int dist_update);
int mlsl_operation_reg_info_add_parameter_set_with_compress(mlsl_op_reg_info reg_info, size_t kernel_count, size_t kernel_size, mlsl_data_type dtype, int dist_update, mlsl_compression compress_type);
int mlsl_operation_reg_info_validate(mlsl_op_reg_info reg_info, mlsl_distribution dist);

/* ---- one-sided RMA windows (fence-epoch semantics; see dl/rma.hpp) ---- */
typedef void* mlsl_win;
/* Collective over dist's group: allocate a `bytes` window on every member. */
int mlsl_win_allocate(mlsl_distribution dist, mlsl_group group_type,
                      size_t bytes, mlsl_win* out);
int mlsl_win_free(mlsl_win w);
/* Local window memory (HBM in device mode — use mlsl_memcpy to fill/read). */
int mlsl_win_buffer(mlsl_win w, void** base, size_t* bytes);
/* Nonblocking; complete at the next mlsl_win_fence. */
int mlsl_win_put(mlsl_win w, const void* src, size_t bytes, size_t target,
                 size_t target_off);
int mlsl_win_get(mlsl_win w, void* dst, size_t bytes, size_t target,
                 size_t target_off);
/* Collective: applies the epoch's puts, then serves its gets. */
int mlsl_win_fence(mlsl_win w);

#ifdef __cplusplus
}
#endif

#endif /* MLSL_AMD_C_API_H */
