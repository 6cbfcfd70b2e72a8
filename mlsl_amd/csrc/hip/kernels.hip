// CDNA4 (gfx950) kernels for the communication hot path: local reductions
// for the schedule-driven collectives, int8 block quantization with error
// feedback, and pack/unpack between layer layouts and comm buffers.
//
// Design per the MI355X rules: 64-wide wavefronts, 256-thread blocks,
// 16-byte vectorized accesses where alignment permits, grid capped at
// ~8 blocks/CU with grid-stride loops (memory-bound ops: HBM3E is the
// roofline, not VALU).
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <algorithm>

#include "../core/log.hpp"
#include "kernels.hpp"

namespace mlsl {

#define HIP_CHECK(cmd)                                                        \
    do {                                                                      \
        hipError_t e_ = (cmd);                                                \
        if (e_ != hipSuccess)                                                 \
            MLSL_THROW(std::string("HIP error: ") + hipGetErrorString(e_));   \
    } while (0)

namespace {

constexpr int kBlock = 256;
// 256 CUs × 8 blocks/CU: enough residency to cover HBM latency without
// oversubscribing the launch path.
constexpr int kMaxGrid = 2048;

inline int GridFor(size_t work_items) {
    size_t g = (work_items + kBlock - 1) / kBlock;
    if (g > kMaxGrid) g = kMaxGrid;
    if (g == 0) g = 1;
    return static_cast<int>(g);
}

template <typename T, ReduceOp OP>
__device__ __forceinline__ T Apply(T a, T b) {
    if constexpr (OP == ReduceOp::SUM) return a + b;
    if constexpr (OP == ReduceOp::MIN) return a < b ? a : b;
    return a > b ? a : b;
}

// ---- f32: float4-vectorized in-place reduce, 2-deep unroll for MLP ----
template <ReduceOp OP>
__global__ void ReduceF32Kernel(float* __restrict__ dst,
                                const float* __restrict__ src, size_t n) {
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    const size_t n4 = n / 4;
    const float4* s4 = reinterpret_cast<const float4*>(src);
    float4* d4 = reinterpret_cast<float4*>(dst);
    size_t i = tid;
    for (; i + stride < n4; i += 2 * stride) {
        float4 a0 = d4[i], b0 = s4[i];
        float4 a1 = d4[i + stride], b1 = s4[i + stride];
        a0.x = Apply<float, OP>(a0.x, b0.x);
        a0.y = Apply<float, OP>(a0.y, b0.y);
        a0.z = Apply<float, OP>(a0.z, b0.z);
        a0.w = Apply<float, OP>(a0.w, b0.w);
        a1.x = Apply<float, OP>(a1.x, b1.x);
        a1.y = Apply<float, OP>(a1.y, b1.y);
        a1.z = Apply<float, OP>(a1.z, b1.z);
        a1.w = Apply<float, OP>(a1.w, b1.w);
        d4[i] = a0;
        d4[i + stride] = a1;
    }
    for (; i < n4; i += stride) {
        float4 a = d4[i], b = s4[i];
        a.x = Apply<float, OP>(a.x, b.x);
        a.y = Apply<float, OP>(a.y, b.y);
        a.z = Apply<float, OP>(a.z, b.z);
        a.w = Apply<float, OP>(a.w, b.w);
        d4[i] = a;
    }
    for (size_t j = n4 * 4 + tid; j < n; j += stride)
        dst[j] = Apply<float, OP>(dst[j], src[j]);
}

// NT variant: nontemporal loads/stores (bypass L2 retention) — measured
// A/B against the default policy on pure streaming reductions.
using float4_ev = __attribute__((ext_vector_type(4))) float;

template <ReduceOp OP>
__global__ void ReduceF32NTKernel(float* __restrict__ dst,
                                  const float* __restrict__ src, size_t n) {
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    const size_t n4 = n / 4;
    const float4_ev* s4 = reinterpret_cast<const float4_ev*>(src);
    float4_ev* d4 = reinterpret_cast<float4_ev*>(dst);
    for (size_t i = tid; i < n4; i += stride) {
        float4_ev a = __builtin_nontemporal_load(d4 + i);
        float4_ev b = __builtin_nontemporal_load(s4 + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) a[j] = Apply<float, OP>(a[j], b[j]);
        __builtin_nontemporal_store(a, d4 + i);
    }
    for (size_t j = n4 * 4 + tid; j < n; j += stride)
        dst[j] = Apply<float, OP>(dst[j], src[j]);
}

// 32 B/lane variant (two float4 per iteration, consecutive): A/B hook.
template <ReduceOp OP>
__global__ void ReduceF32NT2Kernel(float* __restrict__ dst,
                                   const float* __restrict__ src, size_t n) {
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    const size_t n8 = n / 8;
    const float4_ev* s4 = reinterpret_cast<const float4_ev*>(src);
    float4_ev* d4 = reinterpret_cast<float4_ev*>(dst);
    for (size_t i = tid; i < n8; i += stride) {
        float4_ev a0 = __builtin_nontemporal_load(d4 + 2 * i);
        float4_ev a1 = __builtin_nontemporal_load(d4 + 2 * i + 1);
        float4_ev b0 = __builtin_nontemporal_load(s4 + 2 * i);
        float4_ev b1 = __builtin_nontemporal_load(s4 + 2 * i + 1);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            a0[j] = Apply<float, OP>(a0[j], b0[j]);
            a1[j] = Apply<float, OP>(a1[j], b1[j]);
        }
        __builtin_nontemporal_store(a0, d4 + 2 * i);
        __builtin_nontemporal_store(a1, d4 + 2 * i + 1);
    }
    for (size_t j = n8 * 8 + tid; j < n; j += stride)
        dst[j] = Apply<float, OP>(dst[j], src[j]);
}

// ---- generic scalar fallback (f64/i32/i64/u8) ----
template <typename T, ReduceOp OP>
__global__ void ReduceScalarKernel(T* __restrict__ dst, const T* __restrict__ src,
                                   size_t n) {
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    for (size_t i = tid; i < n; i += stride)
        dst[i] = Apply<T, OP>(dst[i], src[i]);
}

// ---- bf16: 8-element (16 B) vectorized via short4 pairs ----
using ushort4_t = __attribute__((ext_vector_type(4))) unsigned short;

template <ReduceOp OP>
__global__ void ReduceBf16Kernel(unsigned short* __restrict__ dst,
                                 const unsigned short* __restrict__ src, size_t n) {
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    const size_t n4 = n / 4;
    const ushort4_t* s4 = reinterpret_cast<const ushort4_t*>(src);
    ushort4_t* d4 = reinterpret_cast<ushort4_t*>(dst);
    for (size_t i = tid; i < n4; i += stride) {
        ushort4_t a = d4[i], b = s4[i];
        ushort4_t r;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            float fa = __bfloat162float(__hip_bfloat16_raw{a[j]});
            float fb = __bfloat162float(__hip_bfloat16_raw{b[j]});
            __hip_bfloat16 hr = __float2bfloat16(Apply<float, OP>(fa, fb));
            r[j] = reinterpret_cast<unsigned short&>(hr);
        }
        d4[i] = r;
    }
    for (size_t i = n4 * 4 + tid; i < n; i += stride) {
        float fa = __bfloat162float(__hip_bfloat16_raw{dst[i]});
        float fb = __bfloat162float(__hip_bfloat16_raw{src[i]});
        __hip_bfloat16 hr = __float2bfloat16(Apply<float, OP>(fa, fb));
        dst[i] = reinterpret_cast<unsigned short&>(hr);
    }
}

using ushort8_ev = __attribute__((ext_vector_type(8))) unsigned short;

template <ReduceOp OP>
__global__ void ReduceBf16NTKernel(unsigned short* __restrict__ dst,
                                   const unsigned short* __restrict__ src, size_t n) {
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    const size_t n8 = n / 8;
    const ushort8_ev* s8 = reinterpret_cast<const ushort8_ev*>(src);
    ushort8_ev* d8 = reinterpret_cast<ushort8_ev*>(dst);
    for (size_t i = tid; i < n8; i += stride) {
        ushort8_ev a = __builtin_nontemporal_load(d8 + i);
        ushort8_ev b = __builtin_nontemporal_load(s8 + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float fa = __bfloat162float(__hip_bfloat16_raw{a[j]});
            float fb = __bfloat162float(__hip_bfloat16_raw{b[j]});
            __hip_bfloat16 hr = __float2bfloat16(Apply<float, OP>(fa, fb));
            a[j] = reinterpret_cast<unsigned short&>(hr);
        }
        __builtin_nontemporal_store(a, d8 + i);
    }
    for (size_t i = n8 * 8 + tid; i < n; i += stride) {
        float fa = __bfloat162float(__hip_bfloat16_raw{dst[i]});
        float fb = __bfloat162float(__hip_bfloat16_raw{src[i]});
        __hip_bfloat16 hr = __float2bfloat16(Apply<float, OP>(fa, fb));
        dst[i] = reinterpret_cast<unsigned short&>(hr);
    }
}

template <ReduceOp OP>
__global__ void ReduceF16Kernel(__half* __restrict__ dst,
                                const __half* __restrict__ src, size_t n) {
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    for (size_t i = tid; i < n; i += stride) {
        float fa = __half2float(dst[i]);
        float fb = __half2float(src[i]);
        dst[i] = __float2half(Apply<float, OP>(fa, fb));
    }
}

// Runtime op -> compile-time instantiation dispatch.
#define MLSL_LAUNCH_BY_OP(KER, DST, SRC, N, OP, STREAM)                               \
    do {                                                                              \
        const int grid_ = GridFor((N) / 4 + 1);                                       \
        switch (OP) {                                                                 \
            case ReduceOp::SUM:                                                       \
                KER<ReduceOp::SUM><<<dim3(grid_), dim3(kBlock), 0, (STREAM)>>>(       \
                    (DST), (SRC), (N));                                               \
                break;                                                                \
            case ReduceOp::MIN:                                                       \
                KER<ReduceOp::MIN><<<dim3(grid_), dim3(kBlock), 0, (STREAM)>>>(       \
                    (DST), (SRC), (N));                                               \
                break;                                                                \
            case ReduceOp::MAX:                                                       \
                KER<ReduceOp::MAX><<<dim3(grid_), dim3(kBlock), 0, (STREAM)>>>(       \
                    (DST), (SRC), (N));                                               \
                break;                                                                \
        }                                                                             \
    } while (0)

// Half-wave (32-lane) max: __shfl_xor with width 32 stays inside each half.
__device__ __forceinline__ float HalfWaveMax(float m) {
#pragma unroll
    for (int off = 16; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off, 32));
    return m;
}

template <typename T>
void LaunchScalarByOp(T* dst, const T* src, size_t n, ReduceOp op, hipStream_t stream) {
    const int grid = GridFor(n);
    switch (op) {
        case ReduceOp::SUM:
            hipLaunchKernelGGL((ReduceScalarKernel<T, ReduceOp::SUM>), dim3(grid), dim3(kBlock), 0, stream, dst, src, n);
            break;
        case ReduceOp::MIN:
            hipLaunchKernelGGL((ReduceScalarKernel<T, ReduceOp::MIN>), dim3(grid), dim3(kBlock), 0, stream, dst, src, n);
            break;
        case ReduceOp::MAX:
            hipLaunchKernelGGL((ReduceScalarKernel<T, ReduceOp::MAX>), dim3(grid), dim3(kBlock), 0, stream, dst, src, n);
            break;
    }
}

}  // namespace

void LaunchReduce(void* dst, const void* src, size_t count, DataType dt,
                  ReduceOp op, hipStream_t stream) {
    // Nontemporal path for f32 streams past L2 reach: measured 4.2 -> 5.3
    // TB/s on 256 MiB (docs/BENCHMARKS.md); bf16 measured slower with NT.
    const bool nt = count * DtypeSize(dt) >= (16u << 20);
    switch (dt) {
        case DataType::F32:
            if (nt) {
                MLSL_LAUNCH_BY_OP(ReduceF32NTKernel, static_cast<float*>(dst),
                                  static_cast<const float*>(src), count, op, stream);
            } else {
                MLSL_LAUNCH_BY_OP(ReduceF32Kernel, static_cast<float*>(dst),
                                  static_cast<const float*>(src), count, op, stream);
            }
            break;
        case DataType::BF16:
            // NT measured slower for bf16 (5.64 -> 5.01 TB/s): keep default
            MLSL_LAUNCH_BY_OP(ReduceBf16Kernel, static_cast<unsigned short*>(dst),
                              static_cast<const unsigned short*>(src), count, op, stream);
            break;
        case DataType::F16:
            MLSL_LAUNCH_BY_OP(ReduceF16Kernel, static_cast<__half*>(dst),
                              static_cast<const __half*>(src), count, op, stream);
            break;
        case DataType::F64:
            LaunchScalarByOp(static_cast<double*>(dst), static_cast<const double*>(src), count, op, stream);
            break;
        case DataType::U8:
            LaunchScalarByOp(static_cast<uint8_t*>(dst), static_cast<const uint8_t*>(src), count, op, stream);
            break;
        case DataType::I32:
            LaunchScalarByOp(static_cast<int32_t*>(dst), static_cast<const int32_t*>(src), count, op, stream);
            break;
        case DataType::I64:
            LaunchScalarByOp(static_cast<int64_t*>(dst), static_cast<const int64_t*>(src), count, op, stream);
            break;
    }
    HIP_CHECK(hipGetLastError());
}

void LaunchReduceNT(void* dst, const void* src, size_t count, hipStream_t stream) {
    MLSL_LAUNCH_BY_OP(ReduceF32NTKernel, static_cast<float*>(dst),
                      static_cast<const float*>(src), count, ReduceOp::SUM, stream);
    HIP_CHECK(hipGetLastError());
}

void LaunchReduceNT2(void* dst, const void* src, size_t count, hipStream_t stream) {
    MLSL_LAUNCH_BY_OP(ReduceF32NT2Kernel, static_cast<float*>(dst),
                      static_cast<const float*>(src), count, ReduceOp::SUM, stream);
    HIP_CHECK(hipGetLastError());
}

namespace {

using uint4_ev = __attribute__((ext_vector_type(4))) unsigned int;

// Streaming copy: 16-B lanes, 2-deep unroll. NT template arm bypasses L2
// retention (each XCD's L2 is private; retention buys nothing for a pure
// copy) — A/B-measured against plain accesses.
template <bool NT>
__global__ void CopyNTKernel(uint4_ev* __restrict__ dst,
                             const uint4_ev* __restrict__ src, size_t n16) {
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    size_t i = tid;
    for (; i + stride < n16; i += 2 * stride) {
        uint4_ev a, b;
        if constexpr (NT) {
            a = __builtin_nontemporal_load(src + i);
            b = __builtin_nontemporal_load(src + i + stride);
            __builtin_nontemporal_store(a, dst + i);
            __builtin_nontemporal_store(b, dst + i + stride);
        } else {
            a = src[i];
            b = src[i + stride];
            dst[i] = a;
            dst[i + stride] = b;
        }
    }
    for (; i < n16; i += stride) {
        if constexpr (NT)
            __builtin_nontemporal_store(__builtin_nontemporal_load(src + i), dst + i);
        else
            dst[i] = src[i];
    }
}

}  // namespace

void LaunchCopyVariant(void* dst, const void* src, size_t bytes, bool nt,
                       hipStream_t stream) {
    const uintptr_t d = reinterpret_cast<uintptr_t>(dst);
    const uintptr_t s = reinterpret_cast<uintptr_t>(src);
    // Kernel path: 16-B aligned and big enough that launch cost (~5 us)
    // amortizes against the bandwidth win over the blit path.
    if (((d | s | bytes) & 15) == 0 && bytes >= (1u << 20)) {
        const size_t n16 = bytes / 16;
        if (nt)
            CopyNTKernel<true><<<dim3(GridFor(n16)), dim3(kBlock), 0, stream>>>(
                static_cast<uint4_ev*>(dst), static_cast<const uint4_ev*>(src), n16);
        else
            CopyNTKernel<false><<<dim3(GridFor(n16)), dim3(kBlock), 0, stream>>>(
                static_cast<uint4_ev*>(dst), static_cast<const uint4_ev*>(src), n16);
        HIP_CHECK(hipGetLastError());
        return;
    }
    HIP_CHECK(hipMemcpyAsync(dst, src, bytes, hipMemcpyDeviceToDevice, stream));
}

void LaunchCopy(void* dst, const void* src, size_t bytes, hipStream_t stream) {
    LaunchCopyVariant(dst, src, bytes, /*nt=*/true, stream);
}

void LaunchReduceOut(void* dst, const void* a, const void* b, size_t count,
                     DataType dt, ReduceOp op, hipStream_t stream) {
    // dst = a; dst += b  (two passes is fine: memory-bound and rarely used;
    // the in-place variant is the hot one).
    HIP_CHECK(hipMemcpyAsync(dst, a, count * DtypeSize(dt), hipMemcpyDeviceToDevice, stream));
    LaunchReduce(dst, b, count, dt, op, stream);
}

// ---------------------------------------------------------------------------
// int8 block quantization with error feedback.
// Wire block: [float scale][float reserved][int8 x block_elems], 8-byte header.

namespace {

__device__ __forceinline__ float LoadAsF32(const float* p, size_t i) { return p[i]; }
__device__ __forceinline__ float LoadAsF32(const unsigned short* p, size_t i) {
    return __bfloat162float(__hip_bfloat16_raw{p[i]});
}
__device__ __forceinline__ void StoreFromF32(float* p, size_t i, float v) { p[i] = v; }
__device__ __forceinline__ void StoreFromF32(unsigned short* p, size_t i, float v) {
    __hip_bfloat16 h = __float2bfloat16(v);
    p[i] = reinterpret_cast<unsigned short&>(h);
}

// Wave-per-block quantization: each 64-lane wavefront owns one wire block
// (4 waves per 256-thread workgroup, grid-strided over blocks). The scale
// reduction is a pure cross-lane shuffle — no LDS, no __syncthreads — so
// small blocks stay launch- and HBM-bound, not barrier-bound.
__device__ __forceinline__ float WaveMax(float m) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_down(m, off, 64));
    return __shfl(m, 0, 64);
}

template <typename T, bool USE_ERR>
__global__ void QuantizeKernel(const T* __restrict__ in, T* __restrict__ err,
                               uint8_t* __restrict__ wire, size_t count,
                               size_t block_elems) {
    const size_t nblocks = (count + block_elems - 1) / block_elems;
    const int lane = threadIdx.x & 63;
    const size_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const size_t wstride = (gridDim.x * blockDim.x) >> 6;
    const bool vec4 = (block_elems & 3) == 0;
    for (size_t blk = wave; blk < nblocks; blk += wstride) {
        const size_t base = blk * block_elems;
        const size_t n = min(block_elems, count - base);
        uint8_t* wblock = wire + blk * (block_elems + 8);
        float* hdr = reinterpret_cast<float*>(wblock);
        int8_t* payload = reinterpret_cast<int8_t*>(wblock + 8);

        float m = 0.f;
        if (vec4) {
            for (size_t i = lane * 4; i + 3 < n; i += 256) {
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    float v = LoadAsF32(in, base + i + j);
                    if (USE_ERR) v += LoadAsF32(err, base + i + j);
                    m = fmaxf(m, fabsf(v));
                }
            }
            for (size_t i = (n & ~size_t(3)) + lane; i < n; i += 64) {
                float v = LoadAsF32(in, base + i);
                if (USE_ERR) v += LoadAsF32(err, base + i);
                m = fmaxf(m, fabsf(v));
            }
        } else {
            for (size_t i = lane; i < n; i += 64) {
                float v = LoadAsF32(in, base + i);
                if (USE_ERR) v += LoadAsF32(err, base + i);
                m = fmaxf(m, fabsf(v));
            }
        }
        m = WaveMax(m);
        const float scale = m > 0.f ? m / 127.f : 1.f;
        if (lane == 0) {
            hdr[0] = scale;
            hdr[1] = 0.f;
        }
        const float inv = 1.f / scale;
        if (vec4) {
            int32_t* p4 = reinterpret_cast<int32_t*>(payload);
            for (size_t i = lane * 4; i + 3 < n; i += 256) {
                int32_t packed = 0;
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    float v = LoadAsF32(in, base + i + j);
                    if (USE_ERR) v += LoadAsF32(err, base + i + j);
                    float q = nearbyintf(v * inv);
                    q = fminf(127.f, fmaxf(-127.f, q));
                    packed |= (static_cast<int32_t>(q) & 0xff) << (8 * j);
                    if (USE_ERR) StoreFromF32(err, base + i + j, v - q * scale);
                }
                p4[i >> 2] = packed;
            }
            for (size_t i = (n & ~size_t(3)) + lane; i < n; i += 64) {
                float v = LoadAsF32(in, base + i);
                if (USE_ERR) v += LoadAsF32(err, base + i);
                float q = nearbyintf(v * inv);
                q = fminf(127.f, fmaxf(-127.f, q));
                payload[i] = static_cast<int8_t>(q);
                if (USE_ERR) StoreFromF32(err, base + i, v - q * scale);
            }
        } else {
            for (size_t i = lane; i < n; i += 64) {
                float v = LoadAsF32(in, base + i);
                if (USE_ERR) v += LoadAsF32(err, base + i);
                float q = nearbyintf(v * inv);
                q = fminf(127.f, fmaxf(-127.f, q));
                payload[i] = static_cast<int8_t>(q);
                if (USE_ERR) StoreFromF32(err, base + i, v - q * scale);
            }
        }
        for (size_t i = n + lane; i < block_elems; i += 64) payload[i] = 0;
    }
}

template <typename T, bool NT = false>
__global__ void DequantizeKernel(const uint8_t* __restrict__ wire, T* __restrict__ out,
                                 size_t count, size_t block_elems) {
    const size_t nblocks = (count + block_elems - 1) / block_elems;
    const int lane = threadIdx.x & 63;
    const size_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const size_t wstride = (gridDim.x * blockDim.x) >> 6;
    // 16-B stores need a 16-B-aligned output base (sliced tensors may not be)
    const bool vec4 = (block_elems & 3) == 0 &&
                      (reinterpret_cast<uintptr_t>(out) & 15) == 0;
    for (size_t blk = wave; blk < nblocks; blk += wstride) {
        const size_t base = blk * block_elems;
        const size_t n = min(block_elems, count - base);
        const uint8_t* wblock = wire + blk * (block_elems + 8);
        const float scale = reinterpret_cast<const float*>(wblock)[0];
        const int8_t* payload = reinterpret_cast<const int8_t*>(wblock + 8);
        if (vec4) {
            const int32_t* p4 = reinterpret_cast<const int32_t*>(payload);
            for (size_t i = lane * 4; i + 3 < n; i += 256) {
                const int32_t packed =
                    NT ? __builtin_nontemporal_load(p4 + (i >> 2)) : p4[i >> 2];
                if constexpr (sizeof(T) == 4) {
                    // f32: one 16-B store per lane pass
                    float4_ev v;
#pragma unroll
                    for (int j = 0; j < 4; ++j)
                        v[j] = static_cast<int8_t>((packed >> (8 * j)) & 0xff) * scale;
                    float4_ev* dst4 = reinterpret_cast<float4_ev*>(
                        reinterpret_cast<float*>(out) + base + i);
                    if constexpr (NT) __builtin_nontemporal_store(v, dst4);
                    else *dst4 = v;
                } else {
#pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        const int8_t q = static_cast<int8_t>((packed >> (8 * j)) & 0xff);
                        StoreFromF32(out, base + i + j, static_cast<float>(q) * scale);
                    }
                }
            }
            for (size_t i = (n & ~size_t(3)) + lane; i < n; i += 64)
                StoreFromF32(out, base + i, static_cast<float>(payload[i]) * scale);
        } else {
            for (size_t i = lane; i < n; i += 64)
                StoreFromF32(out, base + i, static_cast<float>(payload[i]) * scale);
        }
    }
}

// bf16 fast path (block_elems == 256, whole blocks): one 64-lane wave owns
// TWO wire blocks — each 32-lane half owns one block at 8 elements (16 B)
// per lane, doubling the per-lane access width over the generic path
// (bf16 at 4 elems/lane is only 8-B loads). Scale reduction is a half-wave
// shuffle; measured A/B against the generic kernel in docs/BENCHMARKS.md.
using ushort8_t = __attribute__((ext_vector_type(8))) unsigned short;
using uint2_ev = __attribute__((ext_vector_type(2))) unsigned int;

template <bool USE_ERR>
__global__ void QuantizeBf16x2Kernel(const unsigned short* __restrict__ in,
                                     unsigned short* __restrict__ err,
                                     uint8_t* __restrict__ wire, size_t nblocks) {
    constexpr size_t kBE = 256, kWB = kBE + 8;
    const int lane = threadIdx.x & 63;
    const int half = lane >> 5, sub = lane & 31;
    const size_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const size_t wstride = (gridDim.x * blockDim.x) >> 6;
    const size_t npairs = (nblocks + 1) / 2;
    for (size_t pair = wave; pair < npairs; pair += wstride) {
        const size_t blk = pair * 2 + half;
        if (blk >= nblocks) continue;
        const size_t base = blk * kBE + sub * 8;
        float v[8];
        {
            ushort8_t a = *reinterpret_cast<const ushort8_t*>(in + base);
            if (USE_ERR) {
                ushort8_t e = *reinterpret_cast<const ushort8_t*>(err + base);
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    v[j] = __bfloat162float(__hip_bfloat16_raw{a[j]}) +
                           __bfloat162float(__hip_bfloat16_raw{e[j]});
            } else {
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    v[j] = __bfloat162float(__hip_bfloat16_raw{a[j]});
            }
        }
        float m = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j) m = fmaxf(m, fabsf(v[j]));
        m = HalfWaveMax(m);
        const float scale = m > 0.f ? m / 127.f : 1.f;
        const float inv = 1.f / scale;
        uint8_t* wblock = wire + blk * kWB;
        if (sub == 0) {
            float* hdr = reinterpret_cast<float*>(wblock);
            hdr[0] = scale;
            hdr[1] = 0.f;
        }
        uint2_ev packed{0, 0};
        ushort8_t res;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float q = nearbyintf(v[j] * inv);
            q = fminf(127.f, fmaxf(-127.f, q));
            packed[j >> 2] |= (static_cast<unsigned int>(static_cast<int32_t>(q)) & 0xffu)
                              << (8 * (j & 3));
            if (USE_ERR) {
                __hip_bfloat16 hr = __float2bfloat16(v[j] - q * scale);
                res[j] = reinterpret_cast<unsigned short&>(hr);
            }
        }
        *reinterpret_cast<uint2_ev*>(wblock + 8 + sub * 8) = packed;
        if (USE_ERR) *reinterpret_cast<ushort8_t*>(err + base) = res;
    }
}

__global__ void DequantizeBf16x2Kernel(const uint8_t* __restrict__ wire,
                                       unsigned short* __restrict__ out,
                                       size_t nblocks) {
    constexpr size_t kBE = 256, kWB = kBE + 8;
    const int lane = threadIdx.x & 63;
    const int half = lane >> 5, sub = lane & 31;
    const size_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const size_t wstride = (gridDim.x * blockDim.x) >> 6;
    const size_t npairs = (nblocks + 1) / 2;
    for (size_t pair = wave; pair < npairs; pair += wstride) {
        const size_t blk = pair * 2 + half;
        if (blk >= nblocks) continue;
        const uint8_t* wblock = wire + blk * kWB;
        const float scale = reinterpret_cast<const float*>(wblock)[0];
        const uint2_ev packed =
            *reinterpret_cast<const uint2_ev*>(wblock + 8 + sub * 8);
        ushort8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int8_t q = static_cast<int8_t>((packed[j >> 2] >> (8 * (j & 3))) & 0xff);
            __hip_bfloat16 h = __float2bfloat16(static_cast<float>(q) * scale);
            o[j] = reinterpret_cast<unsigned short&>(h);
        }
        *reinterpret_cast<ushort8_t*>(out + blk * kBE + sub * 8) = o;
    }
}

// f32 variants of the same two-blocks-per-wave layout: 8 f32 (2x16 B) per
// lane. Doubles per-lane access width and ILP over the 4-elems/lane
// generic kernel; dequant uses NT stores (launcher gates on message size).
template <bool USE_ERR, bool NT_ST = false>
__global__ void QuantizeF32x2Kernel(const float* __restrict__ in,
                                    float* __restrict__ err,
                                    uint8_t* __restrict__ wire, size_t nblocks) {
    constexpr size_t kBE = 256, kWB = kBE + 8;
    const int lane = threadIdx.x & 63;
    const int half = lane >> 5, sub = lane & 31;
    const size_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const size_t wstride = (gridDim.x * blockDim.x) >> 6;
    const size_t npairs = (nblocks + 1) / 2;
    for (size_t pair = wave; pair < npairs; pair += wstride) {
        const size_t blk = pair * 2 + half;
        if (blk >= nblocks) continue;
        const size_t base = blk * kBE + sub * 8;
        float v[8];
        {
            const float4_ev a0 = *reinterpret_cast<const float4_ev*>(in + base);
            const float4_ev a1 = *reinterpret_cast<const float4_ev*>(in + base + 4);
#pragma unroll
            for (int j = 0; j < 4; ++j) { v[j] = a0[j]; v[4 + j] = a1[j]; }
            if (USE_ERR) {
                const float4_ev e0 = *reinterpret_cast<const float4_ev*>(err + base);
                const float4_ev e1 = *reinterpret_cast<const float4_ev*>(err + base + 4);
#pragma unroll
                for (int j = 0; j < 4; ++j) { v[j] += e0[j]; v[4 + j] += e1[j]; }
            }
        }
        float m = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j) m = fmaxf(m, fabsf(v[j]));
        m = HalfWaveMax(m);
        const float scale = m > 0.f ? m / 127.f : 1.f;
        const float inv = 1.f / scale;
        uint8_t* wblock = wire + blk * kWB;
        if (sub == 0) {
            float* hdr = reinterpret_cast<float*>(wblock);
            hdr[0] = scale;
            hdr[1] = 0.f;
        }
        uint2_ev packed{0, 0};
        float4_ev r0, r1;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float q = nearbyintf(v[j] * inv);
            q = fminf(127.f, fmaxf(-127.f, q));
            packed[j >> 2] |= (static_cast<unsigned int>(static_cast<int32_t>(q)) & 0xffu)
                              << (8 * (j & 3));
            if (USE_ERR) {
                const float r = v[j] - q * scale;
                if (j < 4) r0[j] = r;
                else r1[j - 4] = r;
            }
        }
        if constexpr (NT_ST) {
            __builtin_nontemporal_store(packed,
                reinterpret_cast<uint2_ev*>(wblock + 8 + sub * 8));
            if (USE_ERR) {
                __builtin_nontemporal_store(r0, reinterpret_cast<float4_ev*>(err + base));
                __builtin_nontemporal_store(r1, reinterpret_cast<float4_ev*>(err + base + 4));
            }
        } else {
            *reinterpret_cast<uint2_ev*>(wblock + 8 + sub * 8) = packed;
            if (USE_ERR) {
                *reinterpret_cast<float4_ev*>(err + base) = r0;
                *reinterpret_cast<float4_ev*>(err + base + 4) = r1;
            }
        }
    }
}

__global__ void DequantizeF32x2Kernel(const uint8_t* __restrict__ wire,
                                      float* __restrict__ out, size_t nblocks) {
    constexpr size_t kBE = 256, kWB = kBE + 8;
    const int lane = threadIdx.x & 63;
    const int half = lane >> 5, sub = lane & 31;
    const size_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const size_t wstride = (gridDim.x * blockDim.x) >> 6;
    const size_t npairs = (nblocks + 1) / 2;
    for (size_t pair = wave; pair < npairs; pair += wstride) {
        const size_t blk = pair * 2 + half;
        if (blk >= nblocks) continue;
        const uint8_t* wblock = wire + blk * kWB;
        const float scale = reinterpret_cast<const float*>(wblock)[0];
        // Split-half layout: lane handles elems [sub*4, +3] and
        // [128 + sub*4, +3] — each of the two 16-B stores is contiguous
        // across the 32 lanes (PMC showed the interleaved 32-B-stride
        // variant emitting +15% write traffic from split cache lines).
        const uint32_t* p4 = reinterpret_cast<const uint32_t*>(wblock + 8);
        const uint32_t pk0 = __builtin_nontemporal_load(p4 + sub);
        const uint32_t pk1 = __builtin_nontemporal_load(p4 + 32 + sub);
        float4_ev o0, o1;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            o0[j] = static_cast<int8_t>((pk0 >> (8 * j)) & 0xff) * scale;
            o1[j] = static_cast<int8_t>((pk1 >> (8 * j)) & 0xff) * scale;
        }
        float* obase = out + blk * kBE;
        __builtin_nontemporal_store(o0, reinterpret_cast<float4_ev*>(obase + sub * 4));
        __builtin_nontemporal_store(o1, reinterpret_cast<float4_ev*>(obase + 128 + sub * 4));
    }
}

// acc_wire += wire in the compressed domain: dequant both, sum, requant with
// a fresh scale (the reference's external reduce_sum hook, quant/quant.c:89).
//
// Fast path (block_elems == 256, full block): one wave owns one block and
// each lane owns exactly 4 elements, so the dequantized sums live in
// registers across the max-reduce and the requant — one load pair, one
// store, half the VALU of the generic two-pass version below.
__global__ void QuantAccum256Kernel(uint8_t* __restrict__ acc,
                                    const uint8_t* __restrict__ in,
                                    size_t nblocks) {
    constexpr size_t kBlockElems = 256;
    constexpr size_t kWireBlock = kBlockElems + 8;
    const int lane = threadIdx.x & 63;
    const size_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const size_t wstride = (gridDim.x * blockDim.x) >> 6;
    for (size_t blk = wave; blk < nblocks; blk += wstride) {
        uint8_t* ablock = acc + blk * kWireBlock;
        const uint8_t* iblock = in + blk * kWireBlock;
        float* ahdr = reinterpret_cast<float*>(ablock);
        const float as = ahdr[0];
        const float is = reinterpret_cast<const float*>(iblock)[0];
        int32_t* a4 = reinterpret_cast<int32_t*>(ablock + 8);
        const int32_t* i4 = reinterpret_cast<const int32_t*>(iblock + 8);
        const int32_t pa = a4[lane], pb = i4[lane];
        float v[4];
        float m = 0.f;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            v[j] = static_cast<int8_t>((pa >> (8 * j)) & 0xff) * as +
                   static_cast<int8_t>((pb >> (8 * j)) & 0xff) * is;
            m = fmaxf(m, fabsf(v[j]));
        }
        m = WaveMax(m);
        const float ns = m > 0.f ? m / 127.f : 1.f;
        const float inv = 1.f / ns;
        int32_t packed = 0;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            float q = nearbyintf(v[j] * inv);
            q = fminf(127.f, fmaxf(-127.f, q));
            packed |= (static_cast<int32_t>(q) & 0xff) << (8 * j);
        }
        a4[lane] = packed;
        if (lane == 0) ahdr[0] = ns;
    }
}

__device__ __forceinline__ void QuantAccumBody(uint8_t* __restrict__ acc,
                                               const uint8_t* __restrict__ in,
                                               size_t count, size_t block_elems) {
    const size_t nblocks = (count + block_elems - 1) / block_elems;
    const int lane = threadIdx.x & 63;
    const size_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const size_t wstride = (gridDim.x * blockDim.x) >> 6;
    const bool vec4 = (block_elems & 3) == 0;
    for (size_t blk = wave; blk < nblocks; blk += wstride) {
        const size_t base = blk * block_elems;
        const size_t n = min(block_elems, count - base);
        uint8_t* ablock = acc + blk * (block_elems + 8);
        const uint8_t* iblock = in + blk * (block_elems + 8);
        float* ahdr = reinterpret_cast<float*>(ablock);
        const float as = ahdr[0];
        const float is = reinterpret_cast<const float*>(iblock)[0];
        int8_t* ap = reinterpret_cast<int8_t*>(ablock + 8);
        const int8_t* ip = reinterpret_cast<const int8_t*>(iblock + 8);

        float m = 0.f;
        if (vec4) {
            const int32_t* a4 = reinterpret_cast<const int32_t*>(ap);
            const int32_t* i4 = reinterpret_cast<const int32_t*>(ip);
            for (size_t i = lane * 4; i + 3 < n; i += 256) {
                const int32_t pa = a4[i >> 2], pb = i4[i >> 2];
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const float v = static_cast<int8_t>((pa >> (8 * j)) & 0xff) * as +
                                    static_cast<int8_t>((pb >> (8 * j)) & 0xff) * is;
                    m = fmaxf(m, fabsf(v));
                }
            }
            for (size_t i = (n & ~size_t(3)) + lane; i < n; i += 64)
                m = fmaxf(m, fabsf(ap[i] * as + ip[i] * is));
        } else {
            for (size_t i = lane; i < n; i += 64)
                m = fmaxf(m, fabsf(ap[i] * as + ip[i] * is));
        }
        m = WaveMax(m);
        const float ns = m > 0.f ? m / 127.f : 1.f;
        const float inv = 1.f / ns;
        if (vec4) {
            int32_t* a4 = reinterpret_cast<int32_t*>(ap);
            const int32_t* i4 = reinterpret_cast<const int32_t*>(ip);
            for (size_t i = lane * 4; i + 3 < n; i += 256) {
                const int32_t pa = a4[i >> 2], pb = i4[i >> 2];
                int32_t packed = 0;
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const float v = static_cast<int8_t>((pa >> (8 * j)) & 0xff) * as +
                                    static_cast<int8_t>((pb >> (8 * j)) & 0xff) * is;
                    float q = nearbyintf(v * inv);
                    q = fminf(127.f, fmaxf(-127.f, q));
                    packed |= (static_cast<int32_t>(q) & 0xff) << (8 * j);
                }
                a4[i >> 2] = packed;
            }
            for (size_t i = (n & ~size_t(3)) + lane; i < n; i += 64) {
                const float v = ap[i] * as + ip[i] * is;
                float q = nearbyintf(v * inv);
                ap[i] = static_cast<int8_t>(fminf(127.f, fmaxf(-127.f, q)));
            }
        } else {
            for (size_t i = lane; i < n; i += 64) {
                const float v = ap[i] * as + ip[i] * is;
                float q = nearbyintf(v * inv);
                ap[i] = static_cast<int8_t>(fminf(127.f, fmaxf(-127.f, q)));
            }
        }
        if (lane == 0) ahdr[0] = ns;
    }
}

__global__ void QuantAccumKernel(uint8_t* __restrict__ acc,
                                 const uint8_t* __restrict__ in, size_t count,
                                 size_t block_elems) {
    QuantAccumBody(acc, in, count, block_elems);
}

}  // namespace

void LaunchQuantize(const void* in, void* err, void* wire, size_t count,
                    size_t block_elems, DataType dt, bool use_err,
                    hipStream_t stream) {
    const size_t nblocks = (count + block_elems - 1) / block_elems;
    // 4 waves per 256-thread workgroup, grid-strided over wire blocks.
    dim3 grid(static_cast<uint32_t>(std::min<size_t>((nblocks + 3) / 4, kMaxGrid)));
    if (dt == DataType::F32) {
        const bool fast = block_elems == 256 && count % 256 == 0 &&
                          count * 4 >= (4u << 20) &&
                          (reinterpret_cast<uintptr_t>(in) & 15) == 0 &&
                          (reinterpret_cast<uintptr_t>(err) & 15) == 0 &&
                          (reinterpret_cast<uintptr_t>(wire) & 7) == 0;
        if (fast) {
            const size_t npairs = (nblocks + 1) / 2;
            dim3 g2(static_cast<uint32_t>(std::min<size_t>((npairs + 3) / 4, kMaxGrid)));
            if (use_err)
                hipLaunchKernelGGL((QuantizeF32x2Kernel<true>), g2, dim3(kBlock), 0, stream,
                                   static_cast<const float*>(in), static_cast<float*>(err),
                                   static_cast<uint8_t*>(wire), nblocks);
            else
                hipLaunchKernelGGL((QuantizeF32x2Kernel<false>), g2, dim3(kBlock), 0, stream,
                                   static_cast<const float*>(in), static_cast<float*>(err),
                                   static_cast<uint8_t*>(wire), nblocks);
        } else if (use_err)
            hipLaunchKernelGGL((QuantizeKernel<float, true>), grid, dim3(kBlock), 0, stream,
                               static_cast<const float*>(in), static_cast<float*>(err),
                               static_cast<uint8_t*>(wire), count, block_elems);
        else
            hipLaunchKernelGGL((QuantizeKernel<float, false>), grid, dim3(kBlock), 0, stream,
                               static_cast<const float*>(in), static_cast<float*>(err),
                               static_cast<uint8_t*>(wire), count, block_elems);
    } else if (dt == DataType::BF16) {
        // two-blocks-per-wave fast path: whole 256-elem blocks, 16-B
        // aligned bf16 buffers (see QuantizeBf16x2Kernel).
        const bool fast = block_elems == 256 && count % 256 == 0 &&
                          (reinterpret_cast<uintptr_t>(in) & 15) == 0 &&
                          (reinterpret_cast<uintptr_t>(err) & 15) == 0 &&
                          (reinterpret_cast<uintptr_t>(wire) & 7) == 0;
        if (fast) {
            const size_t npairs = (nblocks + 1) / 2;
            dim3 g2(static_cast<uint32_t>(std::min<size_t>((npairs + 3) / 4, kMaxGrid)));
            if (use_err)
                hipLaunchKernelGGL((QuantizeBf16x2Kernel<true>), g2, dim3(kBlock), 0, stream,
                                   static_cast<const unsigned short*>(in),
                                   static_cast<unsigned short*>(err),
                                   static_cast<uint8_t*>(wire), nblocks);
            else
                hipLaunchKernelGGL((QuantizeBf16x2Kernel<false>), g2, dim3(kBlock), 0, stream,
                                   static_cast<const unsigned short*>(in),
                                   static_cast<unsigned short*>(err),
                                   static_cast<uint8_t*>(wire), nblocks);
        } else if (use_err)
            hipLaunchKernelGGL((QuantizeKernel<unsigned short, true>), grid, dim3(kBlock), 0, stream,
                               static_cast<const unsigned short*>(in),
                               static_cast<unsigned short*>(err),
                               static_cast<uint8_t*>(wire), count, block_elems);
        else
            hipLaunchKernelGGL((QuantizeKernel<unsigned short, false>), grid, dim3(kBlock), 0, stream,
                               static_cast<const unsigned short*>(in),
                               static_cast<unsigned short*>(err),
                               static_cast<uint8_t*>(wire), count, block_elems);
    } else {
        MLSL_THROW("quantization supports f32/bf16 only");
    }
    HIP_CHECK(hipGetLastError());
}

void LaunchDequantize(const void* wire, void* out, size_t count,
                      size_t block_elems, DataType dt, hipStream_t stream) {
    const size_t nblocks = (count + block_elems - 1) / block_elems;
    dim3 grid(static_cast<uint32_t>(std::min<size_t>((nblocks + 3) / 4, kMaxGrid)));
    if (dt == DataType::F32) {
        const bool fast = block_elems == 256 && count % 256 == 0 &&
                          count * 4 >= (4u << 20) &&
                          (reinterpret_cast<uintptr_t>(out) & 15) == 0 &&
                          (reinterpret_cast<uintptr_t>(wire) & 7) == 0;
        if (fast) {
            const size_t npairs = (nblocks + 1) / 2;
            dim3 g2(static_cast<uint32_t>(std::min<size_t>((npairs + 3) / 4, kMaxGrid)));
            hipLaunchKernelGGL(DequantizeF32x2Kernel, g2, dim3(kBlock), 0, stream,
                               static_cast<const uint8_t*>(wire),
                               static_cast<float*>(out), nblocks);
            HIP_CHECK(hipGetLastError());
            return;
        }
        // NT past L2 reach: measured 3.00 -> 3.17 TB/s effective at 256 MiB
        // (stream-once wire + output); small outputs keep L2 retention for
        // the consumer.
        if (count * sizeof(float) >= (16u << 20)) {
            hipLaunchKernelGGL((DequantizeKernel<float, true>), grid, dim3(kBlock), 0,
                               stream, static_cast<const uint8_t*>(wire),
                               static_cast<float*>(out), count, block_elems);
            HIP_CHECK(hipGetLastError());
            return;
        }
        hipLaunchKernelGGL((DequantizeKernel<float>), grid, dim3(kBlock), 0, stream,
                           static_cast<const uint8_t*>(wire), static_cast<float*>(out),
                           count, block_elems);
    } else if (dt == DataType::BF16) {
        const bool fast = block_elems == 256 && count % 256 == 0 &&
                          (reinterpret_cast<uintptr_t>(out) & 15) == 0 &&
                          (reinterpret_cast<uintptr_t>(wire) & 7) == 0;
        if (fast) {
            const size_t npairs = (nblocks + 1) / 2;
            dim3 g2(static_cast<uint32_t>(std::min<size_t>((npairs + 3) / 4, kMaxGrid)));
            hipLaunchKernelGGL(DequantizeBf16x2Kernel, g2, dim3(kBlock), 0, stream,
                               static_cast<const uint8_t*>(wire),
                               static_cast<unsigned short*>(out), nblocks);
            HIP_CHECK(hipGetLastError());
            return;
        }
        hipLaunchKernelGGL((DequantizeKernel<unsigned short>), grid, dim3(kBlock), 0, stream,
                           static_cast<const uint8_t*>(wire),
                           static_cast<unsigned short*>(out), count, block_elems);
    } else {
        MLSL_THROW("dequantization supports f32/bf16 only");
    }
    HIP_CHECK(hipGetLastError());
}

void LaunchQuantizeF32NT(const void* in, void* err, void* wire, size_t count,
                         size_t block_elems, hipStream_t stream) {
    // A/B hook: f32 two-blocks-per-wave with NT stores for wire+err.
    const size_t nblocks = (count + block_elems - 1) / block_elems;
    MLSL_CHECK(block_elems == 256 && count % 256 == 0, "NT A/B needs whole 256-blocks");
    const size_t npairs = (nblocks + 1) / 2;
    dim3 g2(static_cast<uint32_t>(std::min<size_t>((npairs + 3) / 4, kMaxGrid)));
    hipLaunchKernelGGL((QuantizeF32x2Kernel<true, true>), g2, dim3(kBlock), 0, stream,
                       static_cast<const float*>(in), static_cast<float*>(err),
                       static_cast<uint8_t*>(wire), nblocks);
    HIP_CHECK(hipGetLastError());
}

void LaunchDequantizeNT(const void* wire, void* out, size_t count,
                        size_t block_elems, DataType dt, hipStream_t stream) {
    // f32-only NT variant (A/B benchmark hook): stream-once wire reads and
    // output stores skip L2 retention.
    if (dt != DataType::F32) {
        LaunchDequantize(wire, out, count, block_elems, dt, stream);
        return;
    }
    const size_t nblocks = (count + block_elems - 1) / block_elems;
    dim3 grid(static_cast<uint32_t>(std::min<size_t>((nblocks + 3) / 4, kMaxGrid)));
    hipLaunchKernelGGL((DequantizeKernel<float, true>), grid, dim3(kBlock), 0, stream,
                       static_cast<const uint8_t*>(wire), static_cast<float*>(out),
                       count, block_elems);
    HIP_CHECK(hipGetLastError());
}

void LaunchQuantAccum(void* acc_wire, const void* wire, size_t count,
                      size_t block_elems, hipStream_t stream) {
    if (count == 0) return;
    const size_t nblocks = (count + block_elems - 1) / block_elems;
    dim3 grid(static_cast<uint32_t>(std::min<size_t>((nblocks + 3) / 4, kMaxGrid)));
    if (block_elems == 256 && count % 256 == 0) {
        hipLaunchKernelGGL(QuantAccum256Kernel, grid, dim3(kBlock), 0, stream,
                           static_cast<uint8_t*>(acc_wire),
                           static_cast<const uint8_t*>(wire), nblocks);
        HIP_CHECK(hipGetLastError());
        return;
    }
    hipLaunchKernelGGL(QuantAccumKernel, grid,
                       dim3(kBlock), 0, stream, static_cast<uint8_t*>(acc_wire),
                       static_cast<const uint8_t*>(wire), count, block_elems);
    HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// Pack/unpack between [mb][fm][fmSize] layer layout and comm-buffer blocks.

namespace {

template <typename T, bool PACK>
__global__ void PackKernel(const T* __restrict__ src, T* __restrict__ dst,
                           PackBlockDesc d) {
    const size_t total = d.mb_count * d.fm_count * d.fm_size;
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    for (size_t i = tid; i < total; i += stride) {
        const size_t k = i % d.fm_size;
        const size_t fm = (i / d.fm_size) % d.fm_count;
        const size_t mb = i / (d.fm_size * d.fm_count);
        const size_t layer_idx =
            ((d.mb_offset + mb) * d.local_fm_count + d.fm_offset + fm) * d.fm_size + k;
        const size_t buf_idx = d.buf_offset + i;
        if (PACK) dst[buf_idx] = src[layer_idx];
        else dst[layer_idx] = src[buf_idx];
    }
}

// Magic-number unsigned division: M = ceil(2^64 / d); for n < 2^32 the
// high half of n*M is exactly floor(n/d) (error term n*e/(d*2^64) < 2^-32).
// Replaces the per-element 64-bit div/mod pair that dominated the naive
// pack kernel (integer division has no hardware unit on CDNA4 — it
// expands to a long instruction sequence).
struct FastDiv {
    uint64_t M;     // ceil(2^64 / d); 0 means d == 1 (identity)
};

inline FastDiv MakeFastDiv(uint32_t d) {
    FastDiv r;
    r.M = d <= 1 ? 0 : (~0ull / d) + 1;
    return r;
}

__device__ __forceinline__ uint32_t FDiv(uint32_t n, FastDiv f) {
    if (f.M == 0) return n;
    return static_cast<uint32_t>(__umul64hi(static_cast<uint64_t>(n), f.M));
}

template <typename T, bool PACK, bool NT = false>
__global__ void PackFastKernel(const T* __restrict__ src, T* __restrict__ dst,
                               PackBlockDesc d, FastDiv dfs, FastDiv dfc,
                               uint32_t total) {
    const uint32_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t i = tid; i < total; i += stride) {
        const uint32_t row = FDiv(i, dfs);                    // i / fm_size
        const uint32_t k = i - row * static_cast<uint32_t>(d.fm_size);
        const uint32_t mb = FDiv(row, dfc);                   // row / fm_count
        const uint32_t fm = row - mb * static_cast<uint32_t>(d.fm_count);
        const size_t layer_idx =
            ((d.mb_offset + mb) * d.local_fm_count + d.fm_offset + fm) * d.fm_size + k;
        const size_t buf_idx = d.buf_offset + i;
        const T* s_ = PACK ? src + layer_idx : src + buf_idx;
        T* d_ = PACK ? dst + buf_idx : dst + layer_idx;
        if constexpr (NT) {
            if constexpr (sizeof(T) == 16) {
                // uint4 is a struct type; the NT builtins need an
                // ext-vector-compatible pointee.
                __builtin_nontemporal_store(
                    __builtin_nontemporal_load(reinterpret_cast<const uint4_ev*>(s_)),
                    reinterpret_cast<uint4_ev*>(d_));
            } else {
                __builtin_nontemporal_store(__builtin_nontemporal_load(s_), d_);
            }
        } else {
            *d_ = *s_;
        }
    }
}

}  // namespace

namespace {

// Row-contiguous blocks whose fm rows are 16-byte multiples vectorize to
// uint4 moves (4x fewer instructions; coalescing unchanged).
bool Pack16(const PackBlockDesc& d, size_t es, PackBlockDesc* out) {
    if ((d.fm_size * es) % 16 != 0 || (d.buf_offset * es) % 16 != 0) return false;
    *out = d;
    out->fm_size = d.fm_size * es / 16;
    out->buf_offset = d.buf_offset * es / 16;
    return true;
}

}  // namespace

namespace {

// Launch the magic-div kernel when totals fit 32 bits (any realistic
// activation block), else the size_t-div fallback.
template <typename T, bool PACK>
void LaunchPackTyped(const T* src, T* dst, const PackBlockDesc& d,
                     hipStream_t stream) {
    const size_t total = d.mb_count * d.fm_count * d.fm_size;
    const int grid = GridFor(total);
    if (total < (1ull << 32) && d.fm_size < (1ull << 32) &&
        d.fm_count < (1ull << 32)) {
        const FastDiv dfs = MakeFastDiv(static_cast<uint32_t>(d.fm_size));
        const FastDiv dfc = MakeFastDiv(static_cast<uint32_t>(d.fm_count));
        if (total * sizeof(T) >= (16u << 20))
            PackFastKernel<T, PACK, true><<<dim3(grid), dim3(kBlock), 0, stream>>>(
                src, dst, d, dfs, dfc, static_cast<uint32_t>(total));
        else
            PackFastKernel<T, PACK, false><<<dim3(grid), dim3(kBlock), 0, stream>>>(
                src, dst, d, dfs, dfc, static_cast<uint32_t>(total));
    } else {
        PackKernel<T, PACK><<<dim3(grid), dim3(kBlock), 0, stream>>>(src, dst, d);
    }
    HIP_CHECK(hipGetLastError());
}

}  // namespace

void LaunchPack(const void* src, void* dst, const PackBlockDesc& d, DataType dt,
                hipStream_t stream) {
    const size_t es = DtypeSize(dt);
    PackBlockDesc d16;
    if (Pack16(d, es, &d16)) {
        LaunchPackTyped<uint4, true>(static_cast<const uint4*>(src),
                                     static_cast<uint4*>(dst), d16, stream);
        return;
    }
    switch (DtypeSize(dt)) {
        case 4:
            LaunchPackTyped<uint32_t, true>(static_cast<const uint32_t*>(src),
                               static_cast<uint32_t*>(dst), d, stream);
            break;
        case 8:
            LaunchPackTyped<uint64_t, true>(static_cast<const uint64_t*>(src),
                               static_cast<uint64_t*>(dst), d, stream);
            break;
        case 2:
            LaunchPackTyped<uint16_t, true>(static_cast<const uint16_t*>(src),
                               static_cast<uint16_t*>(dst), d, stream);
            break;
        default:
            LaunchPackTyped<uint8_t, true>(static_cast<const uint8_t*>(src),
                               static_cast<uint8_t*>(dst), d, stream);
    }
    HIP_CHECK(hipGetLastError());
}

void LaunchUnpack(const void* src, void* dst, const PackBlockDesc& d, DataType dt,
                  hipStream_t stream) {
    const size_t es = DtypeSize(dt);
    PackBlockDesc d16;
    if (Pack16(d, es, &d16)) {
        LaunchPackTyped<uint4, false>(static_cast<const uint4*>(src),
                                      static_cast<uint4*>(dst), d16, stream);
        return;
    }
    switch (DtypeSize(dt)) {
        case 4:
            LaunchPackTyped<uint32_t, false>(static_cast<const uint32_t*>(src),
                               static_cast<uint32_t*>(dst), d, stream);
            break;
        case 8:
            LaunchPackTyped<uint64_t, false>(static_cast<const uint64_t*>(src),
                               static_cast<uint64_t*>(dst), d, stream);
            break;
        case 2:
            LaunchPackTyped<uint16_t, false>(static_cast<const uint16_t*>(src),
                               static_cast<uint16_t*>(dst), d, stream);
            break;
        default:
            LaunchPackTyped<uint8_t, false>(static_cast<const uint8_t*>(src),
                               static_cast<uint8_t*>(dst), d, stream);
    }
    HIP_CHECK(hipGetLastError());
}

// --- IPC p2p transport flag primitives (comm/p2p_transport.cpp) ---
// Mailboxes are monotonically increasing u64 sequence counters in the
// receiver's HBM window; the writer is a remote process on the same or a
// peer GPU, so both sides use system-scope atomics. The wait kernel is the
// stream-blocking primitive (one lane; s_sleep between polls) with TWO
// escape hatches so a lost peer can never wedge the GPU: a host-written
// abort word (pinned) and a wall-clock bound (s_memrealtime, 100 MHz).
// On escape it writes 1 to `status` (pinned) and returns — the host
// request machinery sees it and fails the request loudly.

namespace {

__global__ void WaitFlagKernel(const unsigned long long* __restrict__ mbox,
                               unsigned long long target,
                               const unsigned int* __restrict__ abort_word,
                               unsigned int* __restrict__ status,
                               unsigned long long max_ticks) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    const unsigned long long t0 = wall_clock64();
    while (__hip_atomic_load(mbox, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM) <
           target) {
        if (__hip_atomic_load(abort_word, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_SYSTEM) != 0 ||
            wall_clock64() - t0 > max_ticks) {
            __hip_atomic_store(status, 1u, __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_SYSTEM);
            return;
        }
        __builtin_amdgcn_s_sleep(64);
    }
    // acquire once satisfied (consumers in the NEXT kernel get their
    // visibility from the dispatch boundary; this orders same-stream work)
    (void)__hip_atomic_load(mbox, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
}

__global__ void SetFlagKernel(unsigned long long* __restrict__ mbox,
                              unsigned long long val) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __hip_atomic_store(mbox, val, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_SYSTEM);
    }
}

}  // namespace

// --- fused transport kernels: flag-wait prologue + payload work ---
// Every workgroup's thread 0 polls the (local-HBM) mailbox, then the
// workgroup proceeds — one kernel instead of wait-kernel + work-kernel.
// The publish (in_flag/ack) stays a separate 1-wg kernel so stream order
// provides the grid-completion barrier without cooperative launch.

namespace {

__device__ __forceinline__ bool PollGeq(const unsigned long long* mbox,
                                        unsigned long long target,
                                        const unsigned int* abort_word,
                                        unsigned int* status,
                                        unsigned long long max_ticks) {
    // Consumer recipe (MI355X_MICROARCH.md): RELAXED polls, then ONE
    // acquire once satisfied — polling with acquire loads is correct but
    // 2-3x slower per hop and cuts chip bandwidth with many pollers.
    __shared__ int ok;
    if (threadIdx.x == 0) {
        ok = 1;
        const unsigned long long t0 = wall_clock64();
        while (__hip_atomic_load(mbox, __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_SYSTEM) < target) {
            if (__hip_atomic_load(abort_word, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_SYSTEM) != 0 ||
                wall_clock64() - t0 > max_ticks) {
                __hip_atomic_store(status, 1u, __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_SYSTEM);
                ok = 0;
                break;
            }
            __builtin_amdgcn_s_sleep(64);
        }
        if (ok)
            (void)__hip_atomic_load(mbox, __ATOMIC_ACQUIRE,
                                    __HIP_MEMORY_SCOPE_SYSTEM);
    }
    __syncthreads();
    return ok != 0;
}

struct PollArgs {
    const unsigned long long* mbox;  // null = no wait
    unsigned long long target;
    const unsigned int* abort_word;
    unsigned int* status;
    unsigned long long max_ticks;
};

// byte copy, 16-B vectors + tail; optional poll prologue
__global__ void XferCopyKernel(uint8_t* __restrict__ dst,
                               const uint8_t* __restrict__ src, size_t bytes,
                               PollArgs pa) {
    if (pa.mbox &&
        !PollGeq(pa.mbox, pa.target, pa.abort_word, pa.status, pa.max_ticks))
        return;
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    const bool al = ((reinterpret_cast<uintptr_t>(dst) |
                      reinterpret_cast<uintptr_t>(src)) & 15) == 0;
    if (al) {
        const size_t n16 = bytes / 16;
        const uint4_ev* s = reinterpret_cast<const uint4_ev*>(src);
        uint4_ev* d = reinterpret_cast<uint4_ev*>(dst);
        // plain STORES on the hand-off path (nt stores don't probe the
        // consumer's cached lines); the local source may stream NT
        for (size_t i = tid; i < n16; i += stride)
            d[i] = __builtin_nontemporal_load(s + i);
        for (size_t j = n16 * 16 + tid; j < bytes; j += stride) dst[j] = src[j];
    } else {
        for (size_t j = tid; j < bytes; j += stride) dst[j] = src[j];
    }
}

// dst (op)= slot  |  dst = slot (op) other   — f32, vec4 when aligned
template <ReduceOp OP, bool OUT>
__global__ void XferReduceF32Kernel(float* __restrict__ dst,
                                    const float* __restrict__ slot,
                                    const float* __restrict__ other, size_t n,
                                    PollArgs pa) {
    if (pa.mbox &&
        !PollGeq(pa.mbox, pa.target, pa.abort_word, pa.status, pa.max_ticks))
        return;
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    const bool al = ((reinterpret_cast<uintptr_t>(dst) |
                      reinterpret_cast<uintptr_t>(slot) |
                      reinterpret_cast<uintptr_t>(other)) & 15) == 0;
    if (al) {
        const size_t n4 = n / 4;
        const float4_ev* s4 = reinterpret_cast<const float4_ev*>(slot);
        const float4_ev* o4 = reinterpret_cast<const float4_ev*>(other);
        float4_ev* d4 = reinterpret_cast<float4_ev*>(dst);
        for (size_t i = tid; i < n4; i += stride) {
            // plain accesses throughout (hand-off rule; the slot was
            // written by another process, the output feeds later sends)
            float4_ev a = OUT ? o4[i] : d4[i];
            float4_ev b = s4[i];
#pragma unroll
            for (int j = 0; j < 4; ++j) a[j] = Apply<float, OP>(a[j], b[j]);
            d4[i] = a;
        }
        for (size_t j = n4 * 4 + tid; j < n; j += stride)
            dst[j] = Apply<float, OP>(OUT ? other[j] : dst[j], slot[j]);
    } else {
        for (size_t j = tid; j < n; j += stride)
            dst[j] = Apply<float, OP>(OUT ? other[j] : dst[j], slot[j]);
    }
}

template <ReduceOp OP, bool OUT>
__global__ void XferReduceBf16Kernel(unsigned short* __restrict__ dst,
                                     const unsigned short* __restrict__ slot,
                                     const unsigned short* __restrict__ other,
                                     size_t n, PollArgs pa) {
    if (pa.mbox &&
        !PollGeq(pa.mbox, pa.target, pa.abort_word, pa.status, pa.max_ticks))
        return;
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    for (size_t j = tid; j < n; j += stride) {
        const float a = __bfloat162float(
            *reinterpret_cast<const __hip_bfloat16*>(OUT ? &other[j] : &dst[j]));
        const float b = __bfloat162float(
            *reinterpret_cast<const __hip_bfloat16*>(&slot[j]));
        const __hip_bfloat16 r = __float2bfloat16(Apply<float, OP>(a, b));
        dst[j] = *reinterpret_cast<const unsigned short*>(&r);
    }
}

}  // namespace

void LaunchXferCopy(void* dst, const void* src, size_t bytes,
                    const XferPoll* poll, hipStream_t stream) {
    PollArgs pa{};
    if (poll) {
        pa.mbox = static_cast<const unsigned long long*>(poll->mbox);
        pa.target = poll->target;
        pa.abort_word = static_cast<const unsigned int*>(poll->abort_word);
        pa.status = static_cast<unsigned int*>(poll->status);
        pa.max_ticks = poll->max_ticks;
    }
    hipLaunchKernelGGL(XferCopyKernel, dim3(GridFor(bytes / 16 + 1)),
                       dim3(kBlock), 0, stream, static_cast<uint8_t*>(dst),
                       static_cast<const uint8_t*>(src), bytes, pa);
    HIP_CHECK(hipGetLastError());
}

bool LaunchXferReduce(void* dst, const void* slot, const void* other, size_t n,
                      DataType dt, ReduceOp op, const XferPoll* poll,
                      hipStream_t stream) {
    PollArgs pa{};
    if (poll) {
        pa.mbox = static_cast<const unsigned long long*>(poll->mbox);
        pa.target = poll->target;
        pa.abort_word = static_cast<const unsigned int*>(poll->abort_word);
        pa.status = static_cast<unsigned int*>(poll->status);
        pa.max_ticks = poll->max_ticks;
    }
    const bool out = other != nullptr;
    const int grid = GridFor(n / 4 + 1);
#define XFER_DISPATCH(KERN, PT)                                               \
    do {                                                                      \
        if (out) {                                                            \
            switch (op) {                                                     \
                case ReduceOp::SUM:                                           \
                    hipLaunchKernelGGL((KERN<ReduceOp::SUM, true>),           \
                                       dim3(grid), dim3(kBlock), 0, stream,   \
                                       (PT*)dst, (const PT*)slot,             \
                                       (const PT*)other, n, pa);              \
                    break;                                                    \
                case ReduceOp::MIN:                                           \
                    hipLaunchKernelGGL((KERN<ReduceOp::MIN, true>),           \
                                       dim3(grid), dim3(kBlock), 0, stream,   \
                                       (PT*)dst, (const PT*)slot,             \
                                       (const PT*)other, n, pa);              \
                    break;                                                    \
                case ReduceOp::MAX:                                           \
                    hipLaunchKernelGGL((KERN<ReduceOp::MAX, true>),           \
                                       dim3(grid), dim3(kBlock), 0, stream,   \
                                       (PT*)dst, (const PT*)slot,             \
                                       (const PT*)other, n, pa);              \
                    break;                                                    \
            }                                                                 \
        } else {                                                              \
            switch (op) {                                                     \
                case ReduceOp::SUM:                                           \
                    hipLaunchKernelGGL((KERN<ReduceOp::SUM, false>),          \
                                       dim3(grid), dim3(kBlock), 0, stream,   \
                                       (PT*)dst, (const PT*)slot,             \
                                       (const PT*)other, n, pa);              \
                    break;                                                    \
                case ReduceOp::MIN:                                           \
                    hipLaunchKernelGGL((KERN<ReduceOp::MIN, false>),          \
                                       dim3(grid), dim3(kBlock), 0, stream,   \
                                       (PT*)dst, (const PT*)slot,             \
                                       (const PT*)other, n, pa);              \
                    break;                                                    \
                case ReduceOp::MAX:                                           \
                    hipLaunchKernelGGL((KERN<ReduceOp::MAX, false>),          \
                                       dim3(grid), dim3(kBlock), 0, stream,   \
                                       (PT*)dst, (const PT*)slot,             \
                                       (const PT*)other, n, pa);              \
                    break;                                                    \
            }                                                                 \
        }                                                                     \
    } while (0)
    switch (dt) {
        case DataType::F32:
            XFER_DISPATCH(XferReduceF32Kernel, float);
            break;
        case DataType::BF16:
            XFER_DISPATCH(XferReduceBf16Kernel, unsigned short);
            break;
        default:
            return false;  // caller falls back to wait + LaunchReduce
    }
#undef XFER_DISPATCH
    HIP_CHECK(hipGetLastError());
    return true;
}

// --- fully-fused small-message transport kernels ---
// One kernel per side per sub-message: poll prologue + payload + grid-
// completion counter + flag publish. The grid is FIXED at kFusedGrid
// workgroups so a spinning kernel can never starve the peer's consumer of
// CUs (the full-device fused-poll variant deadlocked two ranks; 2-4 ranks
// x a few 32-wg spinners always co-schedule on 256 CUs). Completion is
// detected with a monotonic per-edge counter: every workgroup does a
// system-scope acq_rel fetch-add; the one that observes target-1 publishes
// the flag with a system release store — which also orders every other
// workgroup's payload writes (their adds released them).

namespace {

constexpr unsigned kFusedGrid = 32;

__device__ __forceinline__ void FusedFinish(unsigned long long* ctr,
                                            unsigned long long target,
                                            unsigned long long* mbox,
                                            unsigned long long val) {
    __syncthreads();
    if (threadIdx.x == 0) {
        // ROCm 7.2 / gfx950 compiler hazard (MI355X_MICROARCH.md): when the
        // publishing wave's vmcnt scoreboard looks provably empty, the
        // compiler drops the s_waitcnt after the release's buffer_wbl2 and
        // the flag can overtake the payload write-back (~1e-4 stale under
        // load — observed as a zero arrival row in the ring RS). Inline asm
        // is invisible to that pass: force the wait before the release-add
        // and again before the publish store.
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        const unsigned long long prev = __hip_atomic_fetch_add(
            ctr, 1ull, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_SYSTEM);
        if (prev == target - 1) {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __hip_atomic_store(mbox, val, __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_SYSTEM);
        }
    }
}

// sender: [backpressure poll] + byte copy into the peer slot + publish
__global__ void XferSendFusedKernel(uint8_t* __restrict__ slot,
                                    const uint8_t* __restrict__ src,
                                    size_t bytes, PollArgs bp,
                                    unsigned long long* ctr,
                                    unsigned long long ctr_target,
                                    unsigned long long* in_mbox,
                                    unsigned long long seq) {
    if (bp.mbox &&
        !PollGeq(bp.mbox, bp.target, bp.abort_word, bp.status, bp.max_ticks))
        return;
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    const bool al = ((reinterpret_cast<uintptr_t>(slot) |
                      reinterpret_cast<uintptr_t>(src)) & 15) == 0;
    if (al) {
        const size_t n16 = bytes / 16;
        const uint4_ev* s = reinterpret_cast<const uint4_ev*>(src);
        uint4_ev* d = reinterpret_cast<uint4_ev*>(slot);
        // plain stores on the hand-off payload (see XferCopyKernel);
        // NT load of the local source is safe and streams past L2
        for (size_t i = tid; i < n16; i += stride)
            d[i] = __builtin_nontemporal_load(s + i);
        for (size_t j = n16 * 16 + tid; j < bytes; j += stride) slot[j] = src[j];
    } else {
        for (size_t j = tid; j < bytes; j += stride) slot[j] = src[j];
    }
    FusedFinish(ctr, ctr_target, in_mbox, seq);
}

// receiver: arrival poll + consume (copy / reduce / reduce-out) + ack
enum class FusedConsume : int { COPY = 0, REDUCE = 1, REDUCE_OUT = 2 };

template <typename T, ReduceOp OP, FusedConsume MODE>
__global__ void XferRecvFusedKernel(T* __restrict__ dst,
                                    const T* __restrict__ slot,
                                    const T* __restrict__ other, size_t n,
                                    PollArgs wp, unsigned long long* ctr,
                                    unsigned long long ctr_target,
                                    unsigned long long* ack_mbox,
                                    unsigned long long seq) {
    if (!PollGeq(wp.mbox, wp.target, wp.abort_word, wp.status, wp.max_ticks))
        return;
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    for (size_t j = tid; j < n; j += stride) {
        if constexpr (MODE == FusedConsume::COPY) {
            dst[j] = slot[j];
        } else if constexpr (MODE == FusedConsume::REDUCE) {
            if constexpr (sizeof(T) == 2) {
                // bf16 via float math (T = unsigned short container)
                const float a = __bfloat162float(
                    *reinterpret_cast<const __hip_bfloat16*>(&dst[j]));
                const float b = __bfloat162float(
                    *reinterpret_cast<const __hip_bfloat16*>(&slot[j]));
                const __hip_bfloat16 r = __float2bfloat16(Apply<float, OP>(a, b));
                dst[j] = *reinterpret_cast<const T*>(&r);
            } else {
                dst[j] = Apply<T, OP>(dst[j], slot[j]);
            }
        } else {
            if constexpr (sizeof(T) == 2) {
                const float a = __bfloat162float(
                    *reinterpret_cast<const __hip_bfloat16*>(&other[j]));
                const float b = __bfloat162float(
                    *reinterpret_cast<const __hip_bfloat16*>(&slot[j]));
                const __hip_bfloat16 r = __float2bfloat16(Apply<float, OP>(a, b));
                dst[j] = *reinterpret_cast<const T*>(&r);
            } else {
                dst[j] = Apply<T, OP>(other[j], slot[j]);
            }
        }
    }
    FusedFinish(ctr, ctr_target, ack_mbox, seq);
}

}  // namespace

void LaunchXferSendFused(void* slot, const void* src, size_t bytes,
                         const XferPoll* bp, void* ctr, uint64_t ctr_target,
                         void* in_mbox, uint64_t seq, hipStream_t stream) {
    PollArgs pa{};
    if (bp) {
        pa.mbox = static_cast<const unsigned long long*>(bp->mbox);
        pa.target = bp->target;
        pa.abort_word = static_cast<const unsigned int*>(bp->abort_word);
        pa.status = static_cast<unsigned int*>(bp->status);
        pa.max_ticks = bp->max_ticks;
    }
    hipLaunchKernelGGL(XferSendFusedKernel, dim3(kFusedGrid), dim3(kBlock), 0,
                       stream, static_cast<uint8_t*>(slot),
                       static_cast<const uint8_t*>(src), bytes, pa,
                       static_cast<unsigned long long*>(ctr), ctr_target,
                       static_cast<unsigned long long*>(in_mbox), seq);
    HIP_CHECK(hipGetLastError());
}

bool LaunchXferRecvFused(void* dst, const void* slot, const void* other,
                         size_t n, DataType dt, ReduceOp op, int mode,
                         const XferPoll* wp, void* ctr, uint64_t ctr_target,
                         void* ack_mbox, uint64_t seq, hipStream_t stream) {
    PollArgs pa{};
    pa.mbox = static_cast<const unsigned long long*>(wp->mbox);
    pa.target = wp->target;
    pa.abort_word = static_cast<const unsigned int*>(wp->abort_word);
    pa.status = static_cast<unsigned int*>(wp->status);
    pa.max_ticks = wp->max_ticks;
    const unsigned long long tgt = ctr_target;
    auto* c = static_cast<unsigned long long*>(ctr);
    auto* m = static_cast<unsigned long long*>(ack_mbox);
#define FUSED_LAUNCH(T, OPV)                                                  \
    do {                                                                      \
        if (mode == 0)                                                        \
            hipLaunchKernelGGL(                                               \
                (XferRecvFusedKernel<T, OPV, FusedConsume::COPY>),            \
                dim3(kFusedGrid), dim3(kBlock), 0, stream, (T*)dst,           \
                (const T*)slot, (const T*)other, n, pa, c, tgt, m, seq);      \
        else if (mode == 1)                                                   \
            hipLaunchKernelGGL(                                               \
                (XferRecvFusedKernel<T, OPV, FusedConsume::REDUCE>),          \
                dim3(kFusedGrid), dim3(kBlock), 0, stream, (T*)dst,           \
                (const T*)slot, (const T*)other, n, pa, c, tgt, m, seq);      \
        else                                                                  \
            hipLaunchKernelGGL(                                               \
                (XferRecvFusedKernel<T, OPV, FusedConsume::REDUCE_OUT>),      \
                dim3(kFusedGrid), dim3(kBlock), 0, stream, (T*)dst,           \
                (const T*)slot, (const T*)other, n, pa, c, tgt, m, seq);      \
    } while (0)
    if (mode == 0) {
        FUSED_LAUNCH(uint8_t, ReduceOp::SUM);  // byte copy, dtype-agnostic
        HIP_CHECK(hipGetLastError());
        return true;
    }
    switch (dt) {
        case DataType::F32:
            switch (op) {
                case ReduceOp::SUM: FUSED_LAUNCH(float, ReduceOp::SUM); break;
                case ReduceOp::MIN: FUSED_LAUNCH(float, ReduceOp::MIN); break;
                case ReduceOp::MAX: FUSED_LAUNCH(float, ReduceOp::MAX); break;
            }
            break;
        case DataType::BF16:
            switch (op) {
                case ReduceOp::SUM:
                    FUSED_LAUNCH(unsigned short, ReduceOp::SUM);
                    break;
                case ReduceOp::MIN:
                    FUSED_LAUNCH(unsigned short, ReduceOp::MIN);
                    break;
                case ReduceOp::MAX:
                    FUSED_LAUNCH(unsigned short, ReduceOp::MAX);
                    break;
            }
            break;
        default:
            return false;
    }
#undef FUSED_LAUNCH
    HIP_CHECK(hipGetLastError());
    return true;
}

// --- one-shot (direct) allreduce fan kernels ---
// Fan-out: ONE kernel pushes the payload to every peer's slot (4 wgs per
// peer, per-peer backpressure poll + publish). Fan-in: ONE kernel waits
// for all arrivals, then reduces them into dst in a single pass (P slot
// reads + 1 dst read + 1 dst write per element) and publishes every ack.
// Small-message allreduce becomes 2 transport kernels regardless of N.

namespace {

struct FanKernArgs {
    void* slot[8];
    unsigned long long* flag[8];
    unsigned long long flag_val[8];
    const unsigned long long* wait_mbox[8];
    unsigned long long wait_target[8];
    unsigned long long* ctr[8];
    unsigned long long ctr_target[8];
    int npeers;
};

__global__ void FanOutSendKernel(const uint8_t* __restrict__ src, size_t bytes,
                                 FanKernArgs fa, PollArgs ab, int wgs_per_peer) {
    const int p = blockIdx.x / wgs_per_peer;
    if (p >= fa.npeers) return;
    if (fa.wait_mbox[p] &&
        !PollGeq(fa.wait_mbox[p], fa.wait_target[p], ab.abort_word, ab.status,
                 ab.max_ticks))
        return;
    uint8_t* dst = static_cast<uint8_t*>(fa.slot[p]);
    const size_t tid =
        (blockIdx.x % wgs_per_peer) * blockDim.x + threadIdx.x;
    const size_t stride = static_cast<size_t>(wgs_per_peer) * blockDim.x;
    const bool al = ((reinterpret_cast<uintptr_t>(dst) |
                      reinterpret_cast<uintptr_t>(src)) & 15) == 0;
    if (al) {
        const size_t n16 = bytes / 16;
        const uint4_ev* s = reinterpret_cast<const uint4_ev*>(src);
        uint4_ev* d = reinterpret_cast<uint4_ev*>(dst);
        // PLAIN stores on the hand-off payload (MI355X_MICROARCH.md:
        // "never nt on hand-off stores" — nt bypasses the cache hierarchy
        // without probing, so a consumer-side cached line can shadow it);
        // NT load of the local source is safe.
        for (size_t i = tid; i < n16; i += stride)
            d[i] = __builtin_nontemporal_load(s + i);
        for (size_t j = n16 * 16 + tid; j < bytes; j += stride) dst[j] = src[j];
    } else {
        for (size_t j = tid; j < bytes; j += stride) dst[j] = src[j];
    }
    FusedFinish(fa.ctr[p], fa.ctr_target[p], fa.flag[p], fa.flag_val[p]);
}

template <typename T, ReduceOp OP>
__global__ void FanInReduceKernel(T* __restrict__ dst, size_t n,
                                  FanKernArgs fa, PollArgs ab,
                                  unsigned long long* ctr,
                                  unsigned long long ctr_target) {
    // wait every arrival (thread 0 polls each mailbox; abort-aware)
    __shared__ int ok;
    if (threadIdx.x == 0) {
        ok = 1;
        const unsigned long long t0 = wall_clock64();
        for (int p = 0; p < fa.npeers && ok; ++p) {
            while (__hip_atomic_load(fa.wait_mbox[p], __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_SYSTEM) <
                   fa.wait_target[p]) {
                if (__hip_atomic_load(ab.abort_word, __ATOMIC_RELAXED,
                                      __HIP_MEMORY_SCOPE_SYSTEM) != 0 ||
                    wall_clock64() - t0 > ab.max_ticks) {
                    __hip_atomic_store(ab.status, 1u, __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_SYSTEM);
                    ok = 0;
                    break;
                }
                __builtin_amdgcn_s_sleep(64);
            }
        }
        if (ok)
            (void)__hip_atomic_load(fa.wait_mbox[0], __ATOMIC_ACQUIRE,
                                    __HIP_MEMORY_SCOPE_SYSTEM);
    }
    __syncthreads();
    if (!ok) return;
    const size_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const size_t stride = gridDim.x * blockDim.x;
    for (size_t j = tid; j < n; j += stride) {
        if constexpr (sizeof(T) == 2) {
            float acc = __bfloat162float(
                *reinterpret_cast<const __hip_bfloat16*>(&dst[j]));
            for (int p = 0; p < fa.npeers; ++p) {
                const T* sl = static_cast<const T*>(fa.slot[p]);
                acc = Apply<float, OP>(
                    acc, __bfloat162float(
                             *reinterpret_cast<const __hip_bfloat16*>(&sl[j])));
            }
            const __hip_bfloat16 r = __float2bfloat16(acc);
            dst[j] = *reinterpret_cast<const T*>(&r);
        } else {
            T acc = dst[j];
            for (int p = 0; p < fa.npeers; ++p)
                acc = Apply<T, OP>(acc, static_cast<const T*>(fa.slot[p])[j]);
            dst[j] = acc;
        }
    }
    // single counter; the finisher publishes every ack (inline-asm waits:
    // see FusedFinish — the compiler may drop the post-wbl2 waitcnt)
    __syncthreads();
    if (threadIdx.x == 0) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        const unsigned long long prev = __hip_atomic_fetch_add(
            ctr, 1ull, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_SYSTEM);
        if (prev == ctr_target - 1) {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            for (int p = 0; p < fa.npeers; ++p)
                __hip_atomic_store(fa.flag[p], fa.flag_val[p], __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_SYSTEM);
        }
    }
}

// fused arrival-wait + compressed-domain accumulate (quantized ring)
__global__ void XferRecvQuantAccumKernel(uint8_t* __restrict__ acc,
                                         const uint8_t* __restrict__ slot,
                                         size_t count, size_t block_elems,
                                         PollArgs wp, unsigned long long* ctr,
                                         unsigned long long ctr_target,
                                         unsigned long long* ack_mbox,
                                         unsigned long long seq) {
    if (!PollGeq(wp.mbox, wp.target, wp.abort_word, wp.status, wp.max_ticks))
        return;
    QuantAccumBody(acc, slot, count, block_elems);
    FusedFinish(ctr, ctr_target, ack_mbox, seq);
}

FanKernArgs ToKernArgs(const FanPeer* peers, int np) {
    FanKernArgs fa{};
    fa.npeers = np;
    for (int p = 0; p < np; ++p) {
        fa.slot[p] = peers[p].slot;
        fa.flag[p] = static_cast<unsigned long long*>(peers[p].flag);
        fa.flag_val[p] = peers[p].flag_val;
        fa.wait_mbox[p] =
            static_cast<const unsigned long long*>(peers[p].wait_mbox);
        fa.wait_target[p] = peers[p].wait_target;
        fa.ctr[p] = static_cast<unsigned long long*>(peers[p].ctr);
        fa.ctr_target[p] = peers[p].ctr_target;
    }
    return fa;
}

}  // namespace

int FanOutWgsPerPeer(int npeers) {
    // Total spinner budget ~32 wgs per launch: one peer gets the full
    // bandwidth grid, many peers split it (>= 4 each).
    if (npeers <= 1) return 32;
    if (npeers <= 4) return 8;
    return 4;
}

void LaunchXferRecvQuantAccum(void* acc, const void* slot, size_t count,
                              size_t block_elems, const XferPoll* wp,
                              void* ctr, uint64_t ctr_target, void* ack_mbox,
                              uint64_t seq, hipStream_t stream) {
    PollArgs pa{};
    pa.mbox = static_cast<const unsigned long long*>(wp->mbox);
    pa.target = wp->target;
    pa.abort_word = static_cast<const unsigned int*>(wp->abort_word);
    pa.status = static_cast<unsigned int*>(wp->status);
    pa.max_ticks = wp->max_ticks;
    hipLaunchKernelGGL(XferRecvQuantAccumKernel, dim3(kFusedGrid), dim3(kBlock),
                       0, stream, static_cast<uint8_t*>(acc),
                       static_cast<const uint8_t*>(slot), count, block_elems,
                       pa, static_cast<unsigned long long*>(ctr), ctr_target,
                       static_cast<unsigned long long*>(ack_mbox), seq);
    HIP_CHECK(hipGetLastError());
}

void LaunchFanOutSend(const void* src, size_t bytes, const FanPeer* peers,
                      int npeers, const XferPoll* ab, hipStream_t stream) {
    PollArgs pa{};
    pa.abort_word = static_cast<const unsigned int*>(ab->abort_word);
    pa.status = static_cast<unsigned int*>(ab->status);
    pa.max_ticks = ab->max_ticks;
    FanKernArgs fa = ToKernArgs(peers, npeers);
    const int wpp = FanOutWgsPerPeer(npeers);
    hipLaunchKernelGGL(FanOutSendKernel, dim3(wpp * npeers), dim3(kBlock), 0,
                       stream, static_cast<const uint8_t*>(src), bytes, fa, pa,
                       wpp);
    HIP_CHECK(hipGetLastError());
}

bool LaunchFanInReduce(void* dst, size_t n, DataType dt, ReduceOp op,
                       const FanPeer* peers, int npeers, void* ctr,
                       uint64_t ctr_target, const XferPoll* ab,
                       hipStream_t stream) {
    PollArgs pa{};
    pa.abort_word = static_cast<const unsigned int*>(ab->abort_word);
    pa.status = static_cast<unsigned int*>(ab->status);
    pa.max_ticks = ab->max_ticks;
    FanKernArgs fa = ToKernArgs(peers, npeers);
    auto* c = static_cast<unsigned long long*>(ctr);
#define FANIN_LAUNCH(T, OPV)                                                      hipLaunchKernelGGL((FanInReduceKernel<T, OPV>), dim3(kFusedGrid),                                dim3(kBlock), 0, stream, (T*)dst, n, fa, pa, c,                               ctr_target)
    switch (dt) {
        case DataType::F32:
            switch (op) {
                case ReduceOp::SUM: FANIN_LAUNCH(float, ReduceOp::SUM); break;
                case ReduceOp::MIN: FANIN_LAUNCH(float, ReduceOp::MIN); break;
                case ReduceOp::MAX: FANIN_LAUNCH(float, ReduceOp::MAX); break;
            }
            break;
        case DataType::BF16:
            switch (op) {
                case ReduceOp::SUM:
                    FANIN_LAUNCH(unsigned short, ReduceOp::SUM);
                    break;
                case ReduceOp::MIN:
                    FANIN_LAUNCH(unsigned short, ReduceOp::MIN);
                    break;
                case ReduceOp::MAX:
                    FANIN_LAUNCH(unsigned short, ReduceOp::MAX);
                    break;
            }
            break;
        default:
            return false;
    }
#undef FANIN_LAUNCH
    HIP_CHECK(hipGetLastError());
    return true;
}

void LaunchWaitFlag(const void* mbox, uint64_t target, const void* abort_word,
                    void* status, uint64_t max_ticks, hipStream_t stream) {
    hipLaunchKernelGGL(WaitFlagKernel, dim3(1), dim3(1), 0, stream,
                       static_cast<const unsigned long long*>(mbox), target,
                       static_cast<const unsigned int*>(abort_word),
                       static_cast<unsigned int*>(status), max_ticks);
    HIP_CHECK(hipGetLastError());
}

void LaunchSetFlag(void* mbox, uint64_t val, hipStream_t stream) {
    hipLaunchKernelGGL(SetFlagKernel, dim3(1), dim3(1), 0, stream,
                       static_cast<unsigned long long*>(mbox), val);
    HIP_CHECK(hipGetLastError());
}

}  // namespace mlsl
