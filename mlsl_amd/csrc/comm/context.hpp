// Global communication context — the backing of Environment::Init/Finalize
// (reference src/mlsl.cpp:684-745 + CommInit src/comm_ep.cpp:1496-1750),
// MPI-free: TCP bootstrap + mesh + progress engine (+ device transport when
// a GPU is present).
#pragma once

#include <memory>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "../core/config.hpp"
#include "bootstrap.hpp"
#include "engine.hpp"
#include "group.hpp"
#include "mesh.hpp"

namespace mlsl {

class DeviceRuntime;  // RCCL/HIP side (device_comm.hpp); null in host mode

class Context {
  public:
    static Context& Get();
    static bool Initialized();

    void Init(int rank = -1, int size = -1);
    void Finalize();

    // Multi-tenant re-split (reference Environment::Configure "color=N",
    // src/mlsl.cpp:620-647): ranks with the same color form their own
    // world; Rank/Size and new groups are tenant-relative afterwards.
    void Configure(int tenant_color);

    int Rank() const { return rank_; }
    int Size() const { return size_; }
    int BootRank() const { return boot_rank_; }
    int BootSize() const { return boot_size_; }
    bool DeviceMode() const { return device_mode_; }
    int DeviceId() const { return device_id_; }

    ProcessGroup* World() { return world_; }
    ProcessGroup* Self() { return self_; }
    // Collective over the world: every rank passes a color; ranks with the
    // same color (>= 0) form a group ordered by world rank.
    ProcessGroup* CreateGroup(int color);
    void FreeGroup(ProcessGroup* g);

    Bootstrap* Boot() { return boot_.get(); }
    Mesh* GetMesh() { return mesh_.get(); }
    Engine* GetEngine() { return engine_.get(); }
    DeviceRuntime* Device() { return device_.get(); }

    // Registered-buffer allocation (Environment::Alloc/Free analog;
    // reference CommAlloc, src/comm_ep.cpp:1838-1860). Device mode: HBM pool;
    // host mode: aligned host memory. Registered with the pointer checker.
    void* Alloc(size_t size, size_t alignment);
    void Free(void* ptr);
    bool CheckBuffer(const void* ptr, size_t bytes) const;

  private:
    Context() = default;

    bool initialized_ = false;
    long init_pid_ = 0;  // fork guard: Finalize is a no-op in child pids
    int rank_ = 0, size_ = 1;           // tenant-relative after Configure
    int boot_rank_ = 0, boot_size_ = 1; // transport-wide
    int tenant_color_ = 0;
    bool device_mode_ = false;
    int device_id_ = -1;

    std::unique_ptr<Bootstrap> boot_;
    std::unique_ptr<Mesh> mesh_;
    std::unique_ptr<Engine> engine_;
    std::unique_ptr<DeviceRuntime> device_;

    std::vector<std::unique_ptr<ProcessGroup>> groups_;
    ProcessGroup* world_ = nullptr;
    ProcessGroup* self_ = nullptr;
    int next_group_uid_ = 0;

    struct AllocRec { size_t bytes; bool device; };
    std::unordered_map<const void*, AllocRec> allocs_;
    mutable std::mutex alloc_mu_;
};

}  // namespace mlsl
