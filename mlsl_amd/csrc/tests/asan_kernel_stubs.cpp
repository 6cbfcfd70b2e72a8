// Host-only kernel-launcher stubs for the ASan build of the full API
// selftest: the CPU/TCP matrix never launches device kernels, but
// device_comm.cpp references the symbols. Any call is a hard error.
#include "../core/types.hpp"
#include "../hip/kernels.hpp"

namespace mlsl {
#define STUB(...) { MLSL_THROW("kernel launcher called in host-only ASan build"); }
void LaunchReduce(void*, const void*, size_t, DataType, ReduceOp, hipStream_t) STUB()
void LaunchReduceNT(void*, const void*, size_t, hipStream_t) STUB()
void LaunchReduceNT2(void*, const void*, size_t, hipStream_t) STUB()
void LaunchReduceOut(void*, const void*, const void*, size_t, DataType, ReduceOp, hipStream_t) STUB()
void LaunchCopy(void*, const void*, size_t, hipStream_t) STUB()
void LaunchCopyVariant(void*, const void*, size_t, bool, hipStream_t) STUB()
void LaunchQuantize(const void*, void*, void*, size_t, size_t, DataType, bool, hipStream_t) STUB()
void LaunchQuantizeF32NT(const void*, void*, void*, size_t, size_t, hipStream_t) STUB()
void LaunchDequantize(const void*, void*, size_t, size_t, DataType, hipStream_t) STUB()
void LaunchDequantizeNT(const void*, void*, size_t, size_t, DataType, hipStream_t) STUB()
void LaunchQuantAccum(void*, const void*, size_t, size_t, hipStream_t) STUB()
void LaunchPack(const void*, void*, const PackBlockDesc&, DataType, hipStream_t) STUB()
void LaunchUnpack(const void*, void*, const PackBlockDesc&, DataType, hipStream_t) STUB()
void LaunchWaitFlag(const void*, uint64_t, const void*, void*, uint64_t, hipStream_t) STUB()
void LaunchSetFlag(void*, uint64_t, hipStream_t) STUB()
void LaunchXferCopy(void*, const void*, size_t, const XferPoll*, hipStream_t) STUB()
void LaunchXferSendFused(void*, const void*, size_t, const XferPoll*, void*, uint64_t, void*, uint64_t, hipStream_t) STUB()
bool LaunchXferRecvFused(void*, const void*, const void*, size_t, DataType, ReduceOp, int, const XferPoll*, void*, uint64_t, void*, uint64_t, hipStream_t) STUB()
int FanOutWgsPerPeer(int npeers) { return npeers <= 1 ? 32 : npeers <= 4 ? 8 : 4; }
void LaunchXferRecvQuantAccum(void*, const void*, size_t, size_t, const XferPoll*, void*, uint64_t, void*, uint64_t, hipStream_t) STUB()
void LaunchFanOutSend(const void*, size_t, const FanPeer*, int, const XferPoll*, hipStream_t) STUB()
bool LaunchFanInReduce(void*, size_t, DataType, ReduceOp, const FanPeer*, int, void*, uint64_t, const XferPoll*, hipStream_t) STUB()
bool LaunchXferReduce(void*, const void*, const void*, size_t, DataType, ReduceOp, const XferPoll*, hipStream_t) STUB()
}  // namespace mlsl
