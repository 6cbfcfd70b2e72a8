#include <hip/hip_runtime.h>
#include <chrono>
#include <cstdio>
__global__ void Empty() {}
#define T(label, expr, N)                                                     \
    {                                                                         \
        auto t0 = std::chrono::steady_clock::now();                           \
        for (int i = 0; i < N; ++i) { expr; }                                 \
        double us = std::chrono::duration<double>(                            \
                        std::chrono::steady_clock::now() - t0).count() * 1e6 / N; \
        std::printf("%-34s %8.2f us\n", label, us);                           \
    }
int main() {
    hipStream_t s; hipStreamCreateWithFlags(&s, hipStreamNonBlocking);
    hipEvent_t e, e2;
    hipEventCreateWithFlags(&e, hipEventDisableTiming);
    hipEventCreateWithFlags(&e2, hipEventDisableTiming);
    hipLaunchKernelGGL(Empty, dim3(1), dim3(64), 0, s); hipDeviceSynchronize();
    const int N = 2000;
    T("eventRecord(null stream)", hipEventRecord(e, nullptr), N);
    hipDeviceSynchronize();
    T("eventRecord(side stream)", hipEventRecord(e, s), N);
    hipDeviceSynchronize();
    T("streamWaitEvent", hipStreamWaitEvent(s, e, 0), N);
    hipDeviceSynchronize();
    T("empty kernel launch (side)", hipLaunchKernelGGL(Empty, dim3(1), dim3(64), 0, s), N);
    hipDeviceSynchronize();
    T("eventQuery (ready)", hipEventQuery(e), N);
    T("record+query until done (pair)",
      { hipEventRecord(e2, s); while (hipEventQuery(e2) != hipSuccess) {} }, 500);
    hipDeviceSynchronize();
    // full emulation of our Start+Wait handshake
    T("full handshake emulation",
      { hipEventRecord(e, nullptr); hipStreamWaitEvent(s, e, 0);
        hipLaunchKernelGGL(Empty, dim3(1), dim3(64), 0, s);
        hipEventRecord(e2, s); while (hipEventQuery(e2) != hipSuccess) {} }, 500);
    // short-path emulation: launch on null + record + query
    T("short-path emulation",
      { hipLaunchKernelGGL(Empty, dim3(1), dim3(64), 0, nullptr);
        hipEventRecord(e2, nullptr); while (hipEventQuery(e2) != hipSuccess) {} }, 500);
    return 0;
}
