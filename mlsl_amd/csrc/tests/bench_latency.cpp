// Native latency/bandwidth microbench over the persistent-request engine
// through the C API: no Python/ctypes in the loop. One JSON line per
// message size.
//
// Usage:  RANK/WORLD_SIZE/MASTER_ADDR env as usual;
//         ./bench_latency [iters] [warmup]
// Buffers come from mlsl_alloc (HBM in device mode, so the device path
// runs when a GPU is visible; host TCP otherwise).
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>

#include <hip/hip_runtime.h>

#include "../include/mlsl/c_api.h"

#define CK(call)                                                              \
    do {                                                                      \
        if ((call) != MLSL_SUCCESS) {                                         \
            std::fprintf(stderr, "FAILED %s: %s\n", #call, mlsl_last_error());\
            return 1;                                                         \
        }                                                                     \
    } while (0)

int main(int argc, char** argv) {
    const int iters = argc > 1 ? std::atoi(argv[1]) : 50;
    const int warmup = argc > 2 ? std::atoi(argv[2]) : 10;

    CK(mlsl_init(-1, -1));
    size_t rank = 0, size = 0;
    CK(mlsl_rank(&rank));
    CK(mlsl_world_size(&size));
    mlsl_distribution dist = nullptr;
    CK(mlsl_distribution_create(size, 1, &dist));

    const size_t sizes_b[] = {4096, 65536, 1048576, 16u << 20, 256u << 20};
    for (size_t bytes : sizes_b) {
        const size_t count = bytes / 4;
        void *sbuf = nullptr, *rbuf = nullptr;
        CK(mlsl_alloc(bytes, 64, &sbuf));
        CK(mlsl_alloc(bytes, 64, &rbuf));
        // mlsl_alloc returns HBM in device mode: touch via hipMemset (it
        // falls back to plain memset semantics for host pointers too).
        hipPointerAttribute_t attr;
        if (hipPointerGetAttributes(&attr, rbuf) == hipSuccess &&
            attr.type == hipMemoryTypeDevice) {
            (void)hipMemset(sbuf, 0, bytes);
            (void)hipMemset(rbuf, 0, bytes);
            (void)hipDeviceSynchronize();
        } else {
            std::memset(sbuf, 0, bytes);
            std::memset(rbuf, 0, bytes);
        }
        mlsl_request req = nullptr;
        CK(mlsl_persistent_all_reduce(dist, count, MLSL_DT_F32, MLSL_RT_SUM,
                                      MLSL_GT_DATA, /*quantized=*/0, &req));
        void* result = nullptr;
        for (int i = 0; i < warmup; ++i) {
            CK(mlsl_request_start(req, sbuf, rbuf));
            CK(mlsl_request_wait(req, &result));
        }
        CK(mlsl_distribution_barrier(dist, MLSL_GT_GLOBAL));
        const auto t0 = std::chrono::steady_clock::now();
        for (int i = 0; i < iters; ++i) {
            CK(mlsl_request_start(req, sbuf, rbuf));
            CK(mlsl_request_wait(req, &result));
        }
        const double dt =
            std::chrono::duration<double>(std::chrono::steady_clock::now() - t0)
                .count() /
            iters;
        if (rank == 0)
            std::printf("{\"bench\": \"cpp_latency\", \"world\": %zu, "
                        "\"bytes\": %zu, \"lat_us\": %.2f, \"algbw_GBps\": %.3f}\n",
                        size, bytes, dt * 1e6, bytes / dt / 1e9);
        CK(mlsl_request_destroy(req));
        CK(mlsl_dealloc(sbuf));
        CK(mlsl_dealloc(rbuf));
    }
    CK(mlsl_distribution_free(dist));
    CK(mlsl_finalize());
    return 0;
}
