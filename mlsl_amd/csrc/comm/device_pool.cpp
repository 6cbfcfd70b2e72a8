#include "device_pool.hpp"

#include <hip/hip_runtime.h>

#include "../core/log.hpp"
#include "../core/types.hpp"

namespace mlsl {

DevicePool::DevicePool(size_t max_cached_bytes) : max_cached_(max_cached_bytes) {}

DevicePool::~DevicePool() { Trim(); }

size_t DevicePool::Bucket(size_t bytes) const {
    // Small: next power of two >= 512 B (wasted tail is cheap against
    // 288 GB; exact-size reuse hits the common persistent case).
    // Large (>=64 MB): 2 MB granularity — pow2 would waste up to 2x on
    // the multi-hundred-MB quantized-wire and staging buffers.
    if (bytes >= (64u << 20)) return (bytes + (2u << 20) - 1) & ~size_t((2u << 20) - 1);
    size_t b = 512;
    while (b < bytes) b <<= 1;
    return b;
}

void* DevicePool::Alloc(size_t bytes) {
    const size_t b = Bucket(bytes);
    {
        std::lock_guard<std::mutex> lk(mu_);
        auto it = free_.find(b);
        if (it != free_.end() && !it->second.empty()) {
            void* p = it->second.back();
            it->second.pop_back();
            cached_bytes_ -= b;
            return p;
        }
    }
    void* p = nullptr;
    hipError_t e = hipMalloc(&p, b);
    if (e != hipSuccess) {
        // OOM: drop the cache and retry once
        (void)hipGetLastError();
        Trim();
        e = hipMalloc(&p, b);
        if (e != hipSuccess)
            MLSL_THROW(std::string("hipMalloc failed: ") + hipGetErrorString(e));
    }
    std::lock_guard<std::mutex> lk(mu_);
    sizes_[p] = b;
    return p;
}

void DevicePool::Free(void* p) {
    if (!p) return;
    size_t b = 0;
    {
        std::lock_guard<std::mutex> lk(mu_);
        auto it = sizes_.find(p);
        if (it == sizes_.end()) {
            // not pool-owned (pre-pool allocation): release directly
            (void)hipFree(p);
            return;
        }
        b = it->second;
        if (max_cached_ == 0 || cached_bytes_ + b <= max_cached_) {
            free_[b].push_back(p);
            cached_bytes_ += b;
            return;
        }
        sizes_.erase(it);
    }
    (void)hipFree(p);
}

void DevicePool::Trim() {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& kv : free_)
        for (void* p : kv.second) {
            sizes_.erase(p);
            (void)hipFree(p);
        }
    free_.clear();
    cached_bytes_ = 0;
}

}  // namespace mlsl
