"""ctypes loader and signatures for libmlsl_amd.so.

Python binding layer (capability parity with the reference's ctypes binding
include/mlsl/mlsl.py over include/mlsl.h), Python 3 + numpy/torch-aware.
"""
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libmlsl_amd.so")

_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            raise RuntimeError(
                f"libmlsl_amd.so not found at {_LIB_PATH}; run `make lib` "
                "(or python -c 'import __graft_entry__; __graft_entry__.build()')")
        _lib = ctypes.CDLL(_LIB_PATH, mode=ctypes.RTLD_GLOBAL)
        _declare(_lib)
    return _lib


c_size_t = ctypes.c_size_t
c_void_p = ctypes.c_void_p
c_int = ctypes.c_int
c_char_p = ctypes.c_char_p
c_ull = ctypes.c_ulonglong


def _declare(L):
    def sig(name, argtypes, restype=c_int):
        fn = getattr(L, name)
        fn.argtypes = argtypes
        fn.restype = restype

    sig("mlsl_last_error", [], c_char_p)
    sig("mlsl_init", [c_int, c_int])
    sig("mlsl_finalize", [])
    sig("mlsl_initialized", [ctypes.POINTER(c_int)])
    sig("mlsl_get_version", [ctypes.POINTER(c_int)])
    sig("mlsl_rank", [ctypes.POINTER(c_size_t)])
    sig("mlsl_world_size", [ctypes.POINTER(c_size_t)])
    sig("mlsl_alloc", [c_size_t, c_size_t, ctypes.POINTER(c_void_p)])
    sig("mlsl_dealloc", [c_void_p])
    sig("mlsl_wait", [c_void_p, ctypes.POINTER(c_void_p)])
    sig("mlsl_test", [c_void_p, ctypes.POINTER(c_int), ctypes.POINTER(c_void_p)])
    sig("mlsl_set_quant_params", [c_size_t])
    sig("mlsl_memcpy", [c_void_p, c_void_p, c_size_t])

    sig("mlsl_win_allocate", [c_void_p, c_int, c_size_t,
                              ctypes.POINTER(c_void_p)])
    sig("mlsl_win_free", [c_void_p])
    sig("mlsl_win_buffer", [c_void_p, ctypes.POINTER(c_void_p),
                            ctypes.POINTER(c_size_t)])
    sig("mlsl_win_put", [c_void_p, c_void_p, c_size_t, c_size_t, c_size_t])
    sig("mlsl_win_get", [c_void_p, c_void_p, c_size_t, c_size_t, c_size_t])
    sig("mlsl_win_fence", [c_void_p])

    P = ctypes.POINTER
    sig("mlsl_distribution_create", [c_size_t, c_size_t, P(c_void_p)])
    sig("mlsl_distribution_create_with_colors", [c_int, c_int, P(c_void_p)])
    sig("mlsl_distribution_free", [c_void_p])
    sig("mlsl_distribution_process_idx", [c_void_p, c_int, P(c_size_t)])
    sig("mlsl_distribution_process_count", [c_void_p, c_int, P(c_size_t)])
    sig("mlsl_distribution_barrier", [c_void_p, c_int])
    sig("mlsl_distribution_bcast",
        [c_void_p, c_void_p, c_size_t, c_int, c_size_t, c_int, P(c_void_p)])
    sig("mlsl_distribution_reduce",
        [c_void_p, c_void_p, c_void_p, c_size_t, c_int, c_int, c_size_t, c_int, P(c_void_p)])
    sig("mlsl_distribution_all_reduce",
        [c_void_p, c_void_p, c_void_p, c_size_t, c_int, c_int, c_int, P(c_void_p)])
    sig("mlsl_distribution_all_to_all",
        [c_void_p, c_void_p, c_size_t, c_void_p, c_int, c_int, P(c_void_p)])
    sig("mlsl_distribution_all_to_allv",
        [c_void_p, c_void_p, P(c_size_t), P(c_size_t), c_void_p, P(c_size_t), P(c_size_t),
         c_int, c_int, P(c_void_p)])
    sig("mlsl_distribution_gather",
        [c_void_p, c_void_p, c_size_t, c_void_p, c_int, c_size_t, c_int, P(c_void_p)])
    sig("mlsl_distribution_all_gather",
        [c_void_p, c_void_p, c_size_t, c_void_p, c_int, c_int, P(c_void_p)])
    sig("mlsl_distribution_all_gatherv",
        [c_void_p, c_void_p, c_size_t, c_void_p, P(c_size_t), c_int, c_int, P(c_void_p)])
    sig("mlsl_distribution_scatter",
        [c_void_p, c_void_p, c_void_p, c_size_t, c_int, c_size_t, c_int, P(c_void_p)])
    sig("mlsl_distribution_reduce_scatter",
        [c_void_p, c_void_p, c_void_p, c_size_t, c_int, c_int, c_int, P(c_void_p)])

    sig("mlsl_session_create", [c_int, P(c_void_p)])
    sig("mlsl_session_free", [c_void_p])
    sig("mlsl_session_set_global_minibatch_size", [c_void_p, c_size_t])
    sig("mlsl_session_get_global_minibatch_size", [c_void_p, P(c_size_t)])
    sig("mlsl_session_create_op_reg_info", [c_void_p, c_int, P(c_void_p)])
    sig("mlsl_session_delete_op_reg_info", [c_void_p, c_void_p])
    sig("mlsl_session_add_operation", [c_void_p, c_void_p, c_void_p, P(c_size_t)])
    sig("mlsl_session_remove_operations", [c_void_p])
    sig("mlsl_session_get_operation_count", [c_void_p, P(c_size_t)])
    sig("mlsl_session_get_operation", [c_void_p, c_size_t, P(c_void_p)])
    sig("mlsl_session_commit", [c_void_p])
    sig("mlsl_session_get_stats", [c_void_p, P(c_void_p)])

    sig("mlsl_op_reg_info_set_name", [c_void_p, c_char_p])
    sig("mlsl_op_reg_info_add_input", [c_void_p, c_size_t, c_size_t, c_int, P(c_size_t)])
    sig("mlsl_op_reg_info_add_output", [c_void_p, c_size_t, c_size_t, c_int, P(c_size_t)])
    sig("mlsl_op_reg_info_add_parameter_set",
        [c_void_p, c_size_t, c_size_t, c_int, c_int, c_int, P(c_size_t)])
    sig("mlsl_op_reg_info_validate", [c_void_p, c_void_p])

    sig("mlsl_operation_set_distribution", [c_void_p, c_void_p])
    sig("mlsl_operation_get_distribution", [c_void_p, P(c_void_p)])
    sig("mlsl_operation_set_prev", [c_void_p, c_void_p, c_size_t, c_size_t])
    sig("mlsl_operation_set_next", [c_void_p, c_void_p, c_size_t, c_size_t])
    sig("mlsl_operation_get_name", [c_void_p, P(c_char_p)])
    for f in ("global_minibatch_size", "local_minibatch_size", "global_minibatch_offset",
              "input_count", "output_count", "parameter_set_count"):
        sig(f"mlsl_operation_get_{f}", [c_void_p, P(c_size_t)])
    sig("mlsl_operation_get_input", [c_void_p, c_size_t, P(c_void_p)])
    sig("mlsl_operation_get_output", [c_void_p, c_size_t, P(c_void_p)])
    sig("mlsl_operation_get_parameter_set", [c_void_p, c_size_t, P(c_void_p)])

    for f in ("global_fm_count", "global_fm_offset", "local_fm_count", "fm_size",
              "comm_buf_size", "pack_block_count", "unpack_block_count"):
        sig(f"mlsl_activation_get_{f}", [c_void_p, P(c_size_t)])
    sig("mlsl_activation_get_data_type", [c_void_p, P(c_int)])
    sig("mlsl_activation_get_pack_block", [c_void_p, c_size_t, P(c_void_p)])
    sig("mlsl_activation_get_unpack_block", [c_void_p, c_size_t, P(c_void_p)])
    sig("mlsl_activation_start_comm", [c_void_p, c_void_p])
    sig("mlsl_activation_wait_comm", [c_void_p, P(c_void_p)])

    for f in ("mb_offset", "mb_count", "fm_offset", "fm_count", "fm_size", "buf_offset"):
        sig(f"mlsl_comm_block_info_get_{f}", [c_void_p, P(c_size_t)])
    sig("mlsl_comm_block_info_get_data_type", [c_void_p, P(c_int)])

    for f in ("global_kernel_count", "global_kernel_offset", "local_kernel_count",
              "owned_kernel_count", "owned_kernel_offset", "kernel_size"):
        sig(f"mlsl_parameter_set_get_{f}", [c_void_p, P(c_size_t)])
    sig("mlsl_parameter_set_get_data_type", [c_void_p, P(c_int)])
    sig("mlsl_parameter_set_is_distributed_update", [c_void_p, P(c_int)])
    sig("mlsl_parameter_set_start_gradient_comm", [c_void_p, c_void_p])
    sig("mlsl_parameter_set_wait_gradient_comm", [c_void_p, P(c_void_p)])
    sig("mlsl_parameter_set_test_gradient_comm", [c_void_p, P(c_int), P(c_void_p)])
    sig("mlsl_parameter_set_start_increment_comm", [c_void_p, c_void_p])
    sig("mlsl_parameter_set_wait_increment_comm", [c_void_p, P(c_void_p)])

    for f in ("start", "stop", "reset", "print"):
        sig(f"mlsl_statistics_{f}", [c_void_p])
    sig("mlsl_statistics_is_enabled", [c_void_p, P(c_int)])
    sig("mlsl_statistics_get_isolation_comm_cycles", [c_void_p, c_size_t, P(c_ull)])
    sig("mlsl_statistics_get_comm_size", [c_void_p, c_size_t, P(c_size_t)])
    sig("mlsl_statistics_get_comm_cycles", [c_void_p, c_size_t, P(c_ull)])
    sig("mlsl_statistics_get_compute_cycles", [c_void_p, c_size_t, P(c_ull)])
    sig("mlsl_statistics_get_total_isolation_comm_cycles", [c_void_p, P(c_ull)])
    sig("mlsl_statistics_get_total_comm_size", [c_void_p, P(c_size_t)])
    sig("mlsl_statistics_get_total_comm_cycles", [c_void_p, P(c_ull)])
    sig("mlsl_statistics_get_total_compute_cycles", [c_void_p, P(c_ull)])


class MlslError(RuntimeError):
    pass


def check(rc):
    if rc != 0:
        raise MlslError(lib().mlsl_last_error().decode("utf-8", "replace"))
