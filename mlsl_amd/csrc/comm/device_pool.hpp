// HBM caching allocator — the MI355X re-design of the reference's
// shared-memory heap (eplib/memory.c + vendored dlmalloc): request scratch,
// staging buffers and Environment::Alloc allocations come from size-bucketed
// free lists over hipMalloc'd blocks, so steady-state Start/Setup never pays
// allocator latency. 288 GB of HBM3E makes "keep it cached" the right
// default; MLSL_HEAP_SIZE_MB caps the cached (free) bytes.
#pragma once

#include <cstddef>
#include <map>
#include <mutex>
#include <unordered_map>
#include <vector>

namespace mlsl {

class DevicePool {
  public:
    explicit DevicePool(size_t max_cached_bytes);
    ~DevicePool();

    void* Alloc(size_t bytes);
    void Free(void* p);
    void Trim();  // release all cached blocks
    size_t CachedBytes() const { return cached_bytes_; }

  private:
    size_t Bucket(size_t bytes) const;

    mutable std::mutex mu_;
    size_t max_cached_ = 0;
    size_t cached_bytes_ = 0;
    // bucket size -> free blocks of exactly that size
    std::map<size_t, std::vector<void*>> free_;
    std::unordered_map<void*, size_t> sizes_;  // live + cached block sizes
};

}  // namespace mlsl
