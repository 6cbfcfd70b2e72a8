// Device (RCCL/HIP) transport interface. Implemented in device_comm.cpp;
// only reached when Context initializes in device mode (a visible MI355X).
// The GPU analog of the reference's process-mode backend (src/comm_ep.cpp):
// per-group RCCL communicators (one per channel) on dedicated HIP streams,
// fused RCCL collectives as the baseline path and schedule-driven
// send/recv + local-reduce-kernel pipelines as the custom path.
#pragma once

#include <cstddef>
#include <string>

namespace mlsl {

class CommRequest;
class ProcessGroup;
struct DeviceReqState;

class DeviceRuntime {
  public:
    virtual ~DeviceRuntime() = default;
    virtual int DeviceId() const = 0;
    virtual void* AllocDevice(size_t bytes) = 0;
    virtual void FreeDevice(void* p) = 0;
    virtual void Synchronize() = 0;
    // Create the per-group device communicators (collective over group).
    virtual void EnsureGroupComms(ProcessGroup* g) = 0;
    virtual std::string Name() const = 0;
    // Producer-ordering stream (0 = HIP legacy default stream). Collectives
    // issued after SetComputeStream order behind work on that stream.
    virtual void SetComputeStream(void* stream) = 0;
    virtual void* ComputeStream() const = 0;
    // True when multi-rank groups use the IPC window transport instead of
    // RCCL comms (MLSL_DEVICE_TRANSPORT, or auto-detected when several
    // ranks share one device). Decided once at context init, identically
    // on every rank.
    virtual bool UsesP2p() const = 0;
};

// Factory: returns nullptr when no HIP device is visible.
DeviceRuntime* CreateDeviceRuntime();

// Request hooks (called from CommRequest/Engine).
void DeviceSetupRequest(CommRequest* req, DeviceReqState& st);
bool DeviceAdvanceRequest(CommRequest* req, DeviceReqState& st);
// hipEvent comm time of the last issued request (ms); -1 when not timed
// (MLSL_STATS off) or not yet complete. Valid after completion.
double DeviceRequestCommMs(DeviceReqState& st);

}  // namespace mlsl
