#include "sysinfo.hpp"

#include <hip/hip_runtime.h>

#include <cstdlib>
#include <cstring>
#include <fstream>
#include <mutex>

#include "config.hpp"
#include "log.hpp"

namespace mlsl {

static GpuInfo g_gpu;
static CpuInfo g_cpu;
static std::once_flag g_probe_once;

static void Probe() {
    // CPU from /proc/cpuinfo (reference sysinfo.cpp:85-120 pattern).
    std::ifstream f("/proc/cpuinfo");
    std::string line;
    int cores = 0;
    while (std::getline(f, line)) {
        if (line.rfind("model name", 0) == 0 && g_cpu.model.empty()) {
            auto pos = line.find(':');
            if (pos != std::string::npos) g_cpu.model = line.substr(pos + 2);
        }
        if (line.rfind("processor", 0) == 0) ++cores;
    }
    g_cpu.cores = cores;

    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) n = 0;
    g_gpu.device_count = n;
    if (n > 0) {
        hipDeviceProp_t prop;
        if (hipGetDeviceProperties(&prop, 0) == hipSuccess) {
            g_gpu.arch = prop.gcnArchName;
            g_gpu.hbm_bytes = prop.totalGlobalMem;
            g_gpu.cu_count = prop.multiProcessorCount;
        }
        g_gpu.p2p.assign(n, std::vector<bool>(n, false));
        for (int i = 0; i < n; ++i)
            for (int j = 0; j < n; ++j) {
                if (i == j) continue;
                int can = 0;
                if (hipDeviceCanAccessPeer(&can, i, j) == hipSuccess)
                    g_gpu.p2p[i][j] = can != 0;
            }
        MLSL_LOG(INFO, "sysinfo: %d GPU(s), arch=%s, %zu GB HBM, %d CUs",
                 n, g_gpu.arch.c_str(), g_gpu.hbm_bytes >> 30, g_gpu.cu_count);
    }
}

const GpuInfo& GetGpuInfo() {
    std::call_once(g_probe_once, Probe);
    return g_gpu;
}

const CpuInfo& GetCpuInfo() {
    std::call_once(g_probe_once, Probe);
    return g_cpu;
}

void AutoConfig() {
    Config& cfg = GlobalConfig();
    const bool user_set_channels = std::getenv("MLSL_NUM_CHANNELS") != nullptr ||
                                   std::getenv("MLSL_NUM_SERVERS") != nullptr;
    const GpuInfo& gpu = GetGpuInfo();
    if (!user_set_channels && gpu.device_count > 0) {
        // xGMI is point-to-point: 7 links per GPU on an 8-GPU MI355X node.
        // A single RCCL communicator already schedules multiple internal
        // channels, so the default stays 1; large-message fan-out across
        // extra comms is an explicit opt-in (MLSL_NUM_CHANNELS>1) measured
        // per workload rather than guessed here.
        cfg.num_channels = 1;
    }
    MLSL_LOG(DEBUG, "autoconfig: channels=%zu (user_set=%d)", cfg.num_channels,
             (int)user_set_channels);
}

}  // namespace mlsl
