// Process groups: ordered world-rank subsets with group-consistent ids.
// Reference analog: ProcessGroup + CreateProcessGroup(color)
// (src/comm.hpp:33-46, src/comm_ep.cpp:1821-1827). Color-split ordering
// matches MPI_Comm_split with key = world rank.
#pragma once

#include <atomic>
#include <cstdint>
#include <mutex>
#include <unordered_map>
#include <vector>

namespace mlsl {

class DeviceComm;

class ProcessGroup {
  public:
    ProcessGroup(int uid, std::vector<int> ranks, int my_world_rank);
    ~ProcessGroup();

    int Uid() const { return uid_; }
    int Size() const { return static_cast<int>(ranks_.size()); }
    int MyIdx() const { return my_idx_; }          // -1 if not a member
    bool IsMember() const { return my_idx_ >= 0; }
    int WorldRank(int group_idx) const { return ranks_[group_idx]; }
    const std::vector<int>& Ranks() const { return ranks_; }

    // Thread-safe counter draw (application threads may Start requests
    // concurrently). NOTE: like NCCL, collectives on ONE group must still
    // be issued in the same order on every rank — concurrent threads
    // sharing a group need their own ordering (or their own groups); the
    // atomicity here only guarantees distinct tags, not cross-rank order.
    uint32_t NextFlow() { return flow_seq_.fetch_add(1, std::memory_order_relaxed); }

    // Per-directed-edge message sequence for point-to-point (SRLIST) tags:
    // sender counts messages it sent TO peer, receiver counts messages it
    // received FROM peer — the two counters advance in lock-step per edge
    // (NCCL-style p2p matching), independent of any other group activity.
    uint32_t NextSendSeq(int peer) {
        std::lock_guard<std::mutex> lk(seq_mu_);
        return send_seq_[peer]++;
    }
    uint32_t NextRecvSeq(int peer) {
        std::lock_guard<std::mutex> lk(seq_mu_);
        return recv_seq_[peer]++;
    }

    // Device-side communicators (RCCL comms + streams per channel), created
    // lazily by the device transport. Owned here so persistent requests can
    // share them.
    DeviceComm* Device() const { return device_; }
    void SetDevice(DeviceComm* d) { device_ = d; }

  private:
    int uid_;
    std::vector<int> ranks_;
    int my_idx_ = -1;
    std::atomic<uint32_t> flow_seq_{0};
    std::mutex seq_mu_;
    std::unordered_map<int, uint32_t> send_seq_, recv_seq_;
    DeviceComm* device_ = nullptr;
};

}  // namespace mlsl
