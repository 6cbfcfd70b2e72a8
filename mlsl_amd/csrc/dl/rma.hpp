// One-sided RMA windows with fence-epoch semantics.
//
// Capability parity with the reference's compile-gated MPI RMA endpoint
// support (eplib/cqueue.c:2099-2156 Win_create/Put/Get/fence dispatch,
// ENABLE_MPIRMA_ENDPOINTS; eplib/window.c window table) — re-designed for
// MI355X instead of translated: the window is HBM from the runtime pool
// (Environment::Alloc — an IPC-shareable hipMalloc region in device mode),
// and an epoch's Put/Get traffic drains through the SAME schedule executor
// as every other collective (one AlltoAll + up to three AlltoAllv per
// Fence), so it runs identically over the host TCP mesh, RCCL/xGMI, and
// the shared-device IPC window transport with no new transport code.
//
// Semantics (MPI_Win_fence-like, active target synchronization):
//   - Put/Get are nonblocking and complete at the next Fence().
//   - At Fence, all puts of the epoch are applied to the target windows
//     first, then gets read the post-put window contents (deterministic:
//     puts-then-gets; concurrent puts to overlapping ranges have
//     last-source-rank-wins order, which we define rather than leave
//     undefined as MPI does).
//   - Fence is collective over the window's group and also orders local
//     window loads/stores (it completes all device work of the epoch).
#pragma once

#include <cstddef>
#include <cstdint>
#include <vector>

#include "../include/mlsl/mlsl.hpp"

namespace mlsl {

class RmaWindow {
  public:
    // Collective over d's group g: every member allocates `bytes` of
    // window memory (HBM in device mode). Non-trivial groups only.
    RmaWindow(Distribution* d, GroupKind g, size_t bytes);
    ~RmaWindow();

    void* Buffer() const { return base_; }
    size_t Bytes() const { return bytes_; }
    size_t GroupRank() const { return rank_; }
    size_t GroupSize() const { return size_; }

    // Stage `len` bytes from src into target's window at byte offset
    // target_off. src may be host or device memory (staged immediately —
    // the caller may reuse src right after the call returns).
    void Put(const void* src, size_t len, size_t target, size_t target_off);
    // Read `len` bytes from target's window at target_off into dst at the
    // next Fence (dst must stay valid until then).
    void Get(void* dst, size_t len, size_t target, size_t target_off);
    // Collective epoch completion: applies all puts, serves all gets.
    void Fence();

  private:
    struct PutRec { size_t target, off, len; void* stage; };
    struct GetRec { size_t target, off, len; void* dst; };

    Distribution* dist_;
    GroupKind group_;
    void* base_ = nullptr;
    size_t bytes_ = 0;
    size_t rank_ = 0, size_ = 1;
    std::vector<PutRec> puts_;
    std::vector<GetRec> gets_;
};

}  // namespace mlsl
