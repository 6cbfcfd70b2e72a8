// int8 block quantization with error feedback — host reference
// implementation, semantics-identical to the CDNA4 kernels
// (csrc/hip/kernels.hip). Contract parity with the reference's compression
// plugin interface (quant/quant.c:57-94: quantize/dequantize/reduce_sum over
// fixed-size blocks, error-feedback diff buffer per gradient buffer).
//
// Wire block layout: [f32 scale][f32 reserved][int8 x block_elems].
#pragma once

#include <cstddef>
#include <cstdint>

#include "../core/types.hpp"

namespace mlsl {

inline size_t QuantWireBytes(size_t count, size_t block) {
    return ((count + block - 1) / block) * (block + 8);
}

// in (+ err if non-null) -> wire; err updated to the residual.
void HostQuantize(const void* in, void* err, void* wire, size_t count,
                  size_t block, DataType dt, bool use_err);
void HostDequantize(const void* wire, void* out, size_t count, size_t block,
                    DataType dt);
// acc_wire += wire in the compressed domain (dequant-sum-requant per block).
void HostQuantAccum(void* acc_wire, const void* wire, size_t count, size_t block);

// dlopen'd compression plugin (reference quant/quant.c ABI — Intel
// DL-comp style). Loaded once per lib_path; used by the HOST compressed
// path when QuantParams.lib_path is set. Signatures quant/quant.c:57-65.
struct QuantPluginApi {
    int (*quant)(void* src, void* dst, size_t count, void* diff,
                 int src_data_type, size_t comp_ratio, int method);
    int (*dequant)(void* src, void* dst, size_t count);
    int (*reduce_sum)(const void* in, void* inout, size_t block_count);
};
// nullptr when qp.lib_path is empty; throws if the library or a symbol
// cannot be loaded (reference ASSERTs the same way, quant.c:112-126).
const QuantPluginApi* LoadQuantPlugin(const QuantParams& qp);

}  // namespace mlsl
