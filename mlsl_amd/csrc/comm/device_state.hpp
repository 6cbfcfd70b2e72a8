// Private device-side execution state of a CommRequest (HIP types; included
// only by .cpp files compiled with hipcc).
#pragma once

#include <hip/hip_runtime.h>

#include <vector>

namespace mlsl {

struct DeviceReqState {
    bool issued = false;
    // Records the caller's compute stream at Start; channel streams wait on
    // it so collectives order after the producer kernels (torch default
    // stream unless mlsl_set_compute_stream was called).
    hipEvent_t dep_event = nullptr;
    // One completion event per channel used by this request.
    std::vector<hipEvent_t> events;
    // hipEvent-backed device timing (MLSL_STATS): events are created
    // timing-capable, t0 is recorded at issue after the producer
    // dependency, and comm time = max over channels of elapsed(t0, ev).
    // The reference attributed comm time host-side via rdtsc deltas
    // (src/mlsl_impl_stats.cpp:564-668), which misattributes overlapped
    // async comm to the Wait call — event deltas measure the GPU truth.
    hipEvent_t t0_event = nullptr;
    bool timed = false;
    // Persistent device scratch (allocated at Setup in device mode).
    void* tmp_dev = nullptr;
    size_t tmp_bytes = 0;
    // Host-buffer staging (reference ReplaceIn/ReplaceOut,
    // src/comm_ep.cpp:363-566): user buffers that are not device memory are
    // staged through persistent HBM buffers around the collective. Pageable
    // user memory additionally bounces through persistent PINNED host
    // buffers (hipHostMalloc) in chunks, so the H2D/D2H legs are true
    // async DMA overlapping the host-side memcpys — the reference's
    // registered-shm-heap staging, MI355X edition.
    void* stage_send = nullptr;
    void* stage_recv = nullptr;
    size_t stage_send_bytes = 0, stage_recv_bytes = 0;
    bool recv_staged = false;
    void* pin_send = nullptr;
    void* pin_recv = nullptr;
    size_t pin_send_bytes = 0, pin_recv_bytes = 0;
    // D2H landed in pin_recv; the final pinned->user memcpy runs on the
    // host after the completion events fire.
    bool unstage_pending = false;
    size_t unstage_bytes = 0;
    // hipGraph replay (MLSL_USE_GRAPHS): the issue sequence captured once,
    // replayed on subsequent Starts with the same buffers.
    hipGraphExec_t graph_exec = nullptr;
    const void* captured_sbuf = nullptr;
    void* captured_rbuf = nullptr;
    bool graph_failed = false;   // capture failed once -> stay eager
    ~DeviceReqState();
};

}  // namespace mlsl
