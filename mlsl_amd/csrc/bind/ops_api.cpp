// C API for the standalone CDNA4 kernels (device-pointer in/out, used by
// the Python ops layer and GPU numerics tests). Synchronous variants run on
// the null stream and synchronize; the engine uses the stream variants
// internally.
#include <hip/hip_runtime.h>

#include "../comm/context.hpp"
#include "../comm/device_comm.hpp"
#include "../core/types.hpp"
#include "../hip/kernels.hpp"
#include "../include/mlsl/c_api.h"

using namespace mlsl;

extern "C" void mlsl_set_last_error_impl(const char* msg);

extern "C" {

#define OPS_TRY try {
#define OPS_CATCH                                                              \
    return MLSL_SUCCESS;                                                       \
    } catch (const std::exception& e) {                                        \
        mlsl_set_last_error_impl(e.what());                                    \
        return MLSL_FAILURE;                                                   \
    }

int mlsl_hip_device_count(int* out) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) n = 0;
    *out = n;
    return MLSL_SUCCESS;
}

int mlsl_set_compute_stream(void* stream) {
    OPS_TRY
    mlsl::DeviceRuntime* rt = mlsl::Context::Get().Device();
    if (rt) rt->SetComputeStream(stream);
    OPS_CATCH
}

int mlsl_hip_synchronize(void) {
    OPS_TRY if (hipDeviceSynchronize() != hipSuccess)
        throw Error("hipDeviceSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_reduce(void* dst, const void* src, size_t count, int dt, int op) {
    OPS_TRY LaunchReduce(dst, src, count, static_cast<DataType>(dt),
                         static_cast<ReduceOp>(op), nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_reduce_nt(void* dst, const void* src, size_t count) {
    OPS_TRY LaunchReduceNT(dst, src, count, nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_reduce_nt2(void* dst, const void* src, size_t count) {
    OPS_TRY LaunchReduceNT2(dst, src, count, nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_copy(void* dst, const void* src, size_t bytes) {
    OPS_TRY LaunchCopy(dst, src, bytes, nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_copy_variant(void* dst, const void* src, size_t bytes, int nt) {
    OPS_TRY LaunchCopyVariant(dst, src, bytes, nt != 0, nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_quantize(const void* in, void* err, void* wire, size_t count,
                      size_t block, int dt, int use_err) {
    OPS_TRY LaunchQuantize(in, err, wire, count, block, static_cast<DataType>(dt),
                           use_err != 0, nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_dequantize(const void* wire, void* out, size_t count, size_t block, int dt) {
    OPS_TRY LaunchDequantize(wire, out, count, block, static_cast<DataType>(dt), nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_quantize_f32_nt(const void* in, void* err, void* wire, size_t count,
                             size_t block) {
    OPS_TRY LaunchQuantizeF32NT(in, err, wire, count, block, nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_dequantize_nt(const void* wire, void* out, size_t count, size_t block, int dt) {
    OPS_TRY LaunchDequantizeNT(wire, out, count, block, static_cast<DataType>(dt), nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_quant_accum(void* acc, const void* wire, size_t count, size_t block) {
    OPS_TRY LaunchQuantAccum(acc, wire, count, block, nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_pack(const void* src, void* dst, size_t mb_off, size_t mb_cnt,
                  size_t fm_off, size_t fm_cnt, size_t fm_size, size_t buf_off,
                  size_t local_fm, size_t local_mb, int dt) {
    OPS_TRY PackBlockDesc d{mb_off, mb_cnt, fm_off, fm_cnt, fm_size, buf_off,
                            local_fm, local_mb};
    LaunchPack(src, dst, d, static_cast<DataType>(dt), nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

int mlsl_hip_unpack(const void* src, void* dst, size_t mb_off, size_t mb_cnt,
                    size_t fm_off, size_t fm_cnt, size_t fm_size, size_t buf_off,
                    size_t local_fm, size_t local_mb, int dt) {
    OPS_TRY PackBlockDesc d{mb_off, mb_cnt, fm_off, fm_cnt, fm_size, buf_off,
                            local_fm, local_mb};
    LaunchUnpack(src, dst, d, static_cast<DataType>(dt), nullptr);
    if (hipStreamSynchronize(nullptr) != hipSuccess)
        throw Error("hipStreamSynchronize failed");
    OPS_CATCH
}

}  // extern "C"
