// System/topology detection + auto-configuration.
// Reference analog: src/sysinfo.{hpp,cpp} (CPU/NIC detection feeding
// AutoConfig, mlsl.cpp:649-682) — MI355X-native version probes the GPU and
// the xGMI peer-to-peer topology instead of NIC types.
#pragma once

#include <string>
#include <vector>

namespace mlsl {

struct GpuInfo {
    int device_count = 0;
    std::string arch;            // e.g. gfx950
    size_t hbm_bytes = 0;
    int cu_count = 0;
    // p2p[i][j]: peer access i -> j available (xGMI reachable)
    std::vector<std::vector<bool>> p2p;
};

struct CpuInfo {
    std::string model;
    int cores = 0;
};

const GpuInfo& GetGpuInfo();   // probed once, cached
const CpuInfo& GetCpuInfo();

// AutoConfig (reference mlsl.cpp:649-682): derive channel count / chunking
// defaults from the topology when the user did not set them explicitly.
// Called from Context::Init before the engine starts.
void AutoConfig();

}  // namespace mlsl
