// Host-callable launchers for the CDNA4 HIP kernels (gfx950).
#pragma once

#include <hip/hip_runtime.h>

#include "../core/types.hpp"

namespace mlsl {

// dst[i] op= src[i] for count elements, on `stream`.
// Memory-bound: vectorized 16-byte accesses, grid-stride, XCD-friendly
// grid sizing (see kernels.hip).
void LaunchReduce(void* dst, const void* src, size_t count, DataType dt,
                  ReduceOp op, hipStream_t stream);

// dst = a + b elementwise into a third buffer (out-of-place variant used by
// fused pipelines).
// f32 SUM with nontemporal loads/stores (A/B benchmarking).
void LaunchReduceNT(void* dst, const void* src, size_t count, hipStream_t stream);
void LaunchReduceNT2(void* dst, const void* src, size_t count, hipStream_t stream);

void LaunchReduceOut(void* dst, const void* a, const void* b, size_t count,
                     DataType dt, ReduceOp op, hipStream_t stream);

// Streaming device-to-device copy. For large 16-byte-aligned transfers a
// nontemporal grid-stride kernel beats hipMemcpyAsync's blit path on HBM3E
// (same reason the NT reduce wins: no L2 retention for stream-once data);
// misaligned or small copies fall back to hipMemcpyAsync on `stream`.
void LaunchCopy(void* dst, const void* src, size_t bytes, hipStream_t stream);
void LaunchCopyVariant(void* dst, const void* src, size_t bytes, bool nt,
                       hipStream_t stream);

// --- int8 block quantization with error feedback (quant/quant.c contract,
//     fused into the allreduce path; see comm/quant.cpp) ---
// Wire block layout: [float scale][float reserved][int8 x block_elems].
// in: fp32/bf16 gradients; err: same shape residual carried across steps
// (error feedback); out: packed wire blocks.
void LaunchQuantize(const void* in, void* err, void* wire, size_t count,
                    size_t block_elems, DataType dt, bool use_err,
                    hipStream_t stream);
// Dequantize wire blocks into out (fp32/bf16).
void LaunchDequantize(const void* wire, void* out, size_t count,
                      size_t block_elems, DataType dt, hipStream_t stream);
void LaunchQuantizeF32NT(const void* in, void* err, void* wire, size_t count,
                         size_t block_elems, hipStream_t stream);
void LaunchDequantizeNT(const void* wire, void* out, size_t count,
                        size_t block_elems, DataType dt, hipStream_t stream);
// Compressed-domain accumulate: acc_wire += wire (dequant-sum-requant per
// block, matching the reference's reduce_sum plugin hook quant/quant.c:89-94).
void LaunchQuantAccum(void* acc_wire, const void* wire, size_t count,
                      size_t block_elems, hipStream_t stream);

// Strided pack/unpack between layer layout [mb][fm][fmSize] and a contiguous
// comm buffer block (CommBlockInfo contract; reference computes the shapes,
// the consumer does the copy — tests/examples/mlsl_test/mlsl_test.cpp:214-254.
// Here it is a library kernel so GPU consumers get a fused path).
struct PackBlockDesc {
    size_t mb_offset, mb_count;
    size_t fm_offset, fm_count;
    size_t fm_size;          // elements per feature map
    size_t buf_offset;       // element offset in the comm buffer
    size_t local_fm_count;   // feature maps per sample in the local layout
    size_t local_mb_count;   // samples in the local layout
};
void LaunchPack(const void* src, void* dst, const PackBlockDesc& d, DataType dt,
                hipStream_t stream);
void LaunchUnpack(const void* src, void* dst, const PackBlockDesc& d, DataType dt,
                  hipStream_t stream);

// --- fused transport kernels (IPC p2p path) ---
// Optional flag-wait prologue: every workgroup polls *mbox >= target
// before touching the payload (abort word + wall-clock escape as in
// LaunchWaitFlag). Fusing the wait into the payload kernel removes two
// kernel launches per transport sub-message.
struct XferPoll {
    const void* mbox;
    uint64_t target;
    const void* abort_word;
    void* status;
    uint64_t max_ticks;
};
// Streaming byte copy with optional poll prologue (NT, 16-B lanes).
void LaunchXferCopy(void* dst, const void* src, size_t bytes,
                    const XferPoll* poll, hipStream_t stream);
// dst op= slot (other==null) or dst = slot op other, with optional poll.
// Returns false when dtype isn't covered (caller uses LaunchWaitFlag +
// LaunchReduce instead).
bool LaunchXferReduce(void* dst, const void* slot, const void* other, size_t n,
                      DataType dt, ReduceOp op, const XferPoll* poll,
                      hipStream_t stream);

// Fully-fused small-message transport kernels: poll + payload + grid-
// completion counter + flag publish, in ONE launch with a bounded grid
// (spinners can never starve the peer's kernels). ctr_target is the
// ABSOLUTE cumulative workgroup-add count for that counter (the host
// accumulates kXferFusedGrid / kFanWgsPerPeer per emitted kernel, so
// kernels with different grids can share an edge counter).
// Recv `mode`: 0 copy (n = BYTES), 1 reduce-into, 2 reduce-out (n =
// elements); returns false for uncovered dtypes.
constexpr uint32_t kXferFusedGrid = 32;   // wgs per fused send/recv kernel
// wgs per peer in the fan-out: full grid for one peer, split for many
int FanOutWgsPerPeer(int npeers);
void LaunchXferSendFused(void* slot, const void* src, size_t bytes,
                         const XferPoll* bp, void* ctr, uint64_t ctr_target,
                         void* in_mbox, uint64_t seq, hipStream_t stream);
bool LaunchXferRecvFused(void* dst, const void* slot, const void* other,
                         size_t n, DataType dt, ReduceOp op, int mode,
                         const XferPoll* wp, void* ctr, uint64_t ctr_target,
                         void* ack_mbox, uint64_t seq, hipStream_t stream);

// Fused arrival-wait + compressed-domain block accumulate for the
// quantized ring's consume step (count in ELEMENTS of the quant blocks).
void LaunchXferRecvQuantAccum(void* acc, const void* slot, size_t count,
                              size_t block_elems, const XferPoll* wp,
                              void* ctr, uint64_t ctr_target, void* ack_mbox,
                              uint64_t seq, hipStream_t stream);

// One-shot (direct) allreduce fan kernels: up to 8 peers per launch.
// Fan-out pushes the payload into every peer slot (per-peer backpressure
// + publish); fan-in waits all arrivals, reduces them into dst in one
// pass, and publishes every ack (uses the dedicated `ctr`).
struct FanPeer {
    void* slot;
    void* flag;            // publish target (in_flag for send, ack for recv)
    uint64_t flag_val;
    const void* wait_mbox; // backpressure (send, nullable) / arrival (recv)
    uint64_t wait_target;
    void* ctr;             // per-peer counter (fan-out only)
    uint64_t ctr_target;   // absolute adds target
};
void LaunchFanOutSend(const void* src, size_t bytes, const FanPeer* peers,
                      int npeers, const XferPoll* abort_info,
                      hipStream_t stream);
bool LaunchFanInReduce(void* dst, size_t n, DataType dt, ReduceOp op,
                       const FanPeer* peers, int npeers, void* ctr,
                       uint64_t ctr_target, const XferPoll* abort_info,
                       hipStream_t stream);

// --- IPC p2p transport flag primitives ---
// Stream-blocking wait until *mbox >= target (system-scope acquire), with a
// host abort word and a wall-clock bound (ticks of the 100 MHz constant
// clock); on abort/timeout writes 1 to `status` (pinned) and returns.
void LaunchWaitFlag(const void* mbox, uint64_t target, const void* abort_word,
                    void* status, uint64_t max_ticks, hipStream_t stream);
// System-scope release store of a new sequence value.
void LaunchSetFlag(void* mbox, uint64_t val, hipStream_t stream);

}  // namespace mlsl
