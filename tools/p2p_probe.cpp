// Capability probe for the IPC device p2p transport (run on a GPU box):
//   1. hipIpcGetMemHandle/hipIpcOpenMemHandle across two processes on ONE
//      device (dmabuf IPC mode, HSA_ENABLE_IPC_MODE_LEGACY=0).
//   2. Cross-process payload+flag ordering: child copies a payload into the
//      parent's IPC-mapped buffer then sets a flag with a system-scope
//      release store; parent's stream blocks on a wait kernel until the
//      flag lands, then verifies the payload.
// NOTE: the child is fork+EXEC (a fresh process image) — HIP cannot be used
// in a plain fork()ed child of an initialized HIP process (HSA does not
// survive fork; the first probe version hung exactly there).
// Build: hipcc -O2 --offload-arch=gfx950 tools/p2p_probe.cpp -o build/p2p_probe
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstring>
#include <sys/wait.h>
#include <unistd.h>

#define CK(cmd)                                                               \
    do {                                                                      \
        hipError_t e_ = (cmd);                                                \
        if (e_ != hipSuccess) {                                               \
            std::printf("FAIL %s:%d %s: %s\n", __FILE__, __LINE__, #cmd,      \
                        hipGetErrorString(e_));                               \
            return 1;                                                         \
        }                                                                     \
    } while (0)

__global__ void fill_kernel(float* payload, size_t n) {
    const size_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) payload[i] = static_cast<float>(i) * 0.5f + 1.0f;
}

__global__ void set_flag(unsigned long long* flag, unsigned long long val) {
    if (threadIdx.x == 0 && blockIdx.x == 0)
        __hip_atomic_store(flag, val, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

__global__ void wait_flag_geq(unsigned long long* flag, unsigned long long target,
                              int* timed_out, unsigned long long max_ticks) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        const unsigned long long t0 = wall_clock64();
        while (__hip_atomic_load(flag, __ATOMIC_ACQUIRE,
                                 __HIP_MEMORY_SCOPE_SYSTEM) < target) {
            if (wall_clock64() - t0 > max_ticks) {
                *timed_out = 1;
                return;
            }
            __builtin_amdgcn_s_sleep(32);
        }
    }
}

static const char* kHandleFile = "/tmp/p2p_probe_handles.bin";
static const size_t N = 1 << 20;

static int ChildMain() {
    std::printf("child: start\n");
    CK(hipSetDevice(0));
    hipIpcMemHandle_t hp{}, hf{};
    FILE* f = nullptr;
    for (int i = 0; i < 100 && !f; ++i) {
        f = std::fopen(kHandleFile, "rb");
        if (!f) usleep(100000);
    }
    if (!f || std::fread(&hp, sizeof(hp), 1, f) != 1 ||
        std::fread(&hf, sizeof(hf), 1, f) != 1) {
        std::printf("child: FAIL reading handles\n");
        return 1;
    }
    std::fclose(f);
    void *cp = nullptr, *cf = nullptr;
    CK(hipIpcOpenMemHandle(&cp, hp, hipIpcMemLazyEnablePeerAccess));
    CK(hipIpcOpenMemHandle(&cf, hf, hipIpcMemLazyEnablePeerAccess));
    std::printf("child: IPC open OK\n");
    hipStream_t s;
    CK(hipStreamCreate(&s));
    float* src;
    CK(hipMalloc(&src, N * sizeof(float)));
    hipLaunchKernelGGL(fill_kernel, dim3((N + 255) / 256), dim3(256), 0, s, src, N);
    CK(hipMemcpyAsync(cp, src, N * sizeof(float), hipMemcpyDeviceToDevice, s));
    hipLaunchKernelGGL(set_flag, dim3(1), dim3(1), 0, s,
                       static_cast<unsigned long long*>(cf), 42ull);
    CK(hipStreamSynchronize(s));
    std::printf("child: wrote payload + flag\n");
    CK(hipIpcCloseMemHandle(cp));
    CK(hipIpcCloseMemHandle(cf));
    (void)hipFree(src);
    return 0;
}

int main(int argc, char** argv) {
    setvbuf(stdout, nullptr, _IONBF, 0);
    if (argc > 1 && std::strcmp(argv[1], "child") == 0) return ChildMain();

    int dev_count = 0;
    CK(hipGetDeviceCount(&dev_count));
    std::printf("devices: %d\n", dev_count);
    CK(hipSetDevice(0));
    int wv = 0;
    (void)hipDeviceGetAttribute(&wv, hipDeviceAttributeCanUseStreamWaitValue, 0);
    std::printf("CanUseStreamWaitValue attr: %d\n", wv);

    float* payload;
    unsigned long long* flag;
    CK(hipMalloc(&payload, N * sizeof(float)));
    CK(hipMalloc(&flag, sizeof(unsigned long long)));
    CK(hipMemset(payload, 0, N * sizeof(float)));
    CK(hipMemset(flag, 0, sizeof(unsigned long long)));
    hipIpcMemHandle_t hp{}, hf{};
    CK(hipIpcGetMemHandle(&hp, payload));
    CK(hipIpcGetMemHandle(&hf, flag));
    FILE* f = std::fopen(kHandleFile, "wb");
    if (!f) return 1;
    std::fwrite(&hp, sizeof(hp), 1, f);
    std::fwrite(&hf, sizeof(hf), 1, f);
    std::fclose(f);
    std::printf("parent: IPC handles written\n");

    pid_t pid = fork();
    if (pid == 0) {
        // fresh image: HIP re-initializes cleanly in the exec'd child
        execl(argv[0], argv[0], "child", static_cast<char*>(nullptr));
        _exit(127);
    }
    // enqueue the wait BEFORE the child writes (tests real cross-process
    // stream blocking), with a 60 s bound (100 MHz wall clock)
    hipStream_t s;
    CK(hipStreamCreate(&s));
    int* timed_out;
    CK(hipHostMalloc(reinterpret_cast<void**>(&timed_out), sizeof(int)));
    *timed_out = 0;
    hipLaunchKernelGGL(wait_flag_geq, dim3(1), dim3(1), 0, s, flag, 42ull,
                       timed_out, 6000000000ull);
    float* host = new float[N];
    CK(hipMemcpyAsync(host, payload, N * sizeof(float), hipMemcpyDeviceToHost, s));
    CK(hipStreamSynchronize(s));
    int status = 0;
    waitpid(pid, &status, 0);
    if (*timed_out) {
        std::printf("FAIL: wait kernel timed out\n");
        return 1;
    }
    size_t bad = 0;
    for (size_t i = 0; i < N; ++i)
        if (host[i] != static_cast<float>(i) * 0.5f + 1.0f) ++bad;
    std::printf("payload check: %zu bad of %zu; child rc=%d\n", bad, N,
                WEXITSTATUS(status));
    std::printf(bad == 0 && WEXITSTATUS(status) == 0 ? "P2P_PROBE PASSED\n"
                                                     : "P2P_PROBE FAILED\n");
    return bad == 0 && WEXITSTATUS(status) == 0 ? 0 : 1;
}
