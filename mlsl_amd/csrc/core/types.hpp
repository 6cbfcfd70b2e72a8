// mlsl_amd — MI355X-native DL communication library.
// Core enums and POD types shared by every layer.
//
// Capability parity with intel/MLSL include/mlsl.hpp:88-171 (DataType,
// GroupType, OpType, ReductionType, CompressionType, QuantParams), re-designed
// for a GPU-resident library: adds BF16/FP16/INT dtypes that the MI355X compute
// path needs (the reference was fp32/fp64/byte only).
#pragma once

#include <cstddef>
#include <cstdint>
#include <stdexcept>
#include <string>

namespace mlsl {

enum class DataType : int {
    F32 = 0,
    F64 = 1,
    U8  = 2,
    BF16 = 3,
    F16 = 4,
    I32 = 5,
    I64 = 6,
};

inline size_t DtypeSize(DataType dt) {
    switch (dt) {
        case DataType::F32: return 4;
        case DataType::F64: return 8;
        case DataType::U8:  return 1;
        case DataType::BF16: return 2;
        case DataType::F16: return 2;
        case DataType::I32: return 4;
        case DataType::I64: return 8;
    }
    return 0;
}

inline const char* DtypeName(DataType dt) {
    switch (dt) {
        case DataType::F32: return "f32";
        case DataType::F64: return "f64";
        case DataType::U8:  return "u8";
        case DataType::BF16: return "bf16";
        case DataType::F16: return "f16";
        case DataType::I32: return "i32";
        case DataType::I64: return "i64";
    }
    return "?";
}

enum class ReduceOp : int {
    SUM = 0,
    MIN = 1,
    MAX = 2,
};

// Which sub-communicator of a Distribution a collective runs over
// (reference GroupType, include/mlsl.hpp:114-120).
enum class GroupKind : int {
    DATA = 0,
    MODEL = 1,
    GLOBAL = 2,
};

// Operation type for the DL-semantic planner (reference OpType,
// include/mlsl.hpp:136-148). CC = "compute-compute" layers whose
// model-parallel outputs are partial sums that need reduction; BIAS/ACT/etc.
// produce already-complete local feature maps.
enum class OpKind : int {
    CC = 0,    // fully-connected / conv: partial sums over the model group
    BIAS = 1,
    ACT = 2,
    POOL = 3,
    SPLIT = 4,
    CONCAT = 5,
    BCAST = 6,
    REDUCE = 7,
    DATA = 8,
    EVAL = 9,
};

enum class PhaseKind : int {
    TRAIN = 0,
    TEST = 1,
};

// Gradient compression for ParameterSet exchanges (reference
// CompressionType include/mlsl.hpp:150-158 + quant/quant.h contract).
enum class Compression : int {
    NONE = 0,
    QUANT_INT8 = 1,   // block int8 quantization with error feedback
};

// Block-quantization parameters (reference QuantParams
// include/mlsl.hpp:162-171). Built-in int8 HIP/CPU kernels by default; a
// dlopen'd user library (Intel DL-comp-style quantize/dequantize/
// reduce_sum, quant/quant.c ABI) takes over the HOST path when lib_path
// is set — the device path keeps the fused CDNA4 kernels.
struct QuantParams {
    size_t block_elems = 256;   // elements per quantization block
    std::string lib_path;       // dlopen plugin ("" = built-in kernels)
    std::string quant_fn = "dl_comp_compress_buffer";
    std::string dequant_fn = "dl_comp_decompress_buffer";
    std::string reduce_fn = "dl_comp_compressed_buffer_reduce_sum";
    size_t block_bytes = 0;     // plugin wire-block bytes (0 = derived)
    // On-wire block: plugin-declared size, else int8 payload + fp32 scale
    // + fp32 reserved.
    size_t WireBlockBytes() const {
        return block_bytes ? block_bytes : block_elems + 2 * sizeof(float);
    }
};

class Error : public std::runtime_error {
  public:
    explicit Error(const std::string& what) : std::runtime_error(what) {}
};

#define MLSL_THROW(msg)                                                     \
    do {                                                                    \
        throw ::mlsl::Error(std::string(__func__) + ": " + (msg));          \
    } while (0)

#define MLSL_CHECK(cond, msg)                                               \
    do {                                                                    \
        if (!(cond)) MLSL_THROW(std::string("check failed: ") + #cond +     \
                                " — " + (msg));                             \
    } while (0)

}  // namespace mlsl
