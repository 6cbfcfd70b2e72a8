// IPC device p2p transport: cross-process HBM windows + stream-ordered
// flag synchronization, the schedule executor's second backend next to
// RCCL send/recv.
//
// Why it exists: RCCL refuses two ranks of one communicator on the same
// device ("Duplicate GPU detected"), so the hand-written ring/RHD/alltoall
// schedules could never be exercised at n>1 on a single-GPU box — and on a
// full 8-GPU node this transport gives US the per-link scheduling instead
// of delegating to ncclSend/Recv. Design (MI355X-first, no MPI/NCCL):
//
//   - Every group member allocates a WINDOW in its own HBM and shares it
//     with the other members via hipIpcMemHandle (dmabuf IPC). The window
//     holds, per (source peer, lane): kSlots staging slots plus two u64
//     mailboxes — in_flag (written by the source when a slot-message
//     lands) and ack_flag (written by the source as CONSUMER, acknowledging
//     slot-messages I pushed to it).
//   - A send enqueues, on the sender's stream: [backpressure wait on
//     ack >= seq-kSlots] -> DMA copy into the receiver's slot (xGMI peer
//     copy on multi-GPU, local D2D on one device) -> release-store of
//     in_flag = seq. A recv enqueues: wait in_flag >= seq -> consume the
//     slot (fused local reduce where the schedule allows, else copy) ->
//     release-store of the sender's ack_flag.
//   - All polling is on LOCAL HBM (the wait kernel spins in the
//     receiver's own window); remote traffic is payload DMA plus one
//     8-byte flag store per slot-message — the right shape for xGMI's
//     point-to-point links.
//   - Slot granularity (MLSL_P2P_SLOT_MB) pipelines transport against
//     compute: the sender's DMA of sub-message m+1 overlaps the
//     receiver's reduce of sub-message m, and kSlots of backpressure keep
//     multiple messages in flight per edge (the resumable-phase analog of
//     eplib/allreduce_pr.c:69-269).
//   - Safety: every wait kernel carries a host abort word and a wall-clock
//     bound; a dead peer fails the request loudly instead of wedging the
//     stream (reference had no failure detection at all — SURVEY.md 5.3).
//
// Reference analogs: the shm command-queue + proxy-server transport
// (eplib/cqueue.c, server.c) re-imagined as HBM windows + stream-resident
// progress; GET_EP_PAYLOAD endpoint chunking (src/comm_ep.cpp:99-115) maps
// to lanes; the registered shm heap (eplib/memory.c) to the IPC windows.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <memory>
#include <vector>

#include "schedule.hpp"

namespace mlsl {

class ProcessGroup;
class CommRequest;
struct ChunkExec;

class P2pGroup {
  public:
    // Collective over the WORLD bootstrap (every rank calls; only members
    // allocate + map). nlanes includes the priority lane when enabled.
    static std::unique_ptr<P2pGroup> Create(ProcessGroup* g, size_t nlanes,
                                            size_t nslots, size_t slot_bytes);
    ~P2pGroup();

    size_t Lanes() const { return nlanes_; }
    size_t SlotBytes() const { return slot_bytes_; }

    // Enqueue one schedule-step send/recv pair set onto `stream`.
    // `sbase`/`rbase`/`tmp` resolve the chunk's SEND/RECV/TMP spaces.
    // Throws on malformed schedules; device-side failures surface through
    // CheckHealthy().
    void IssueSchedule(CommRequest* req, ChunkExec& ce, size_t lane,
                       hipStream_t stream, const uint8_t* sbase,
                       uint8_t* rbase, uint8_t* tmp);

    // Host-side failure check: true while no wait kernel has aborted or
    // timed out. Called from the request poll loop.
    bool Healthy() const;
    // Raise the abort word (teardown / timeout): unblocks every in-flight
    // wait kernel on this group.
    void Abort();

  private:
    P2pGroup() = default;

    // window geometry
    uint8_t* MySlot(int src, size_t lane, size_t slot) const;
    uint8_t* PeerSlot(int peer, size_t lane, size_t slot) const;  // src = me
    void* MyInFlag(int src, size_t lane) const;
    void* MyAckFlag(int peer, size_t lane) const;
    void* PeerInFlag(int peer, size_t lane) const;   // src = me, in peer window
    void* PeerAckFlag(int peer, size_t lane) const;  // me acking peer's data

    int gsize_ = 0, my_idx_ = 0;
    size_t nlanes_ = 1, nslots_ = 4, slot_bytes_ = 0;
    size_t flags_bytes_ = 0, win_bytes_ = 0;
    uint8_t* my_base_ = nullptr;
    std::vector<uint8_t*> peer_base_;   // [gsize]; self = my_base_, else IPC-mapped
    std::vector<uint64_t> sent_, rcvd_; // [peer * nlanes + lane]
    // fused small-message kernels: monotonic grid-completion counters
    // (device) + matching host-side op counts, per edge-lane-direction
    // layout: [send edge ctrs | recv edge ctrs | per-lane fan-in ctrs]
    uint64_t* ctr_dev_ = nullptr;       // [2 * gsize * nlanes + nlanes]
    // host-side ABSOLUTE workgroup-add accumulators per counter
    std::vector<uint64_t> fused_sent_, fused_rcvd_, fanin_adds_;
    uint32_t* abort_host_ = nullptr;    // pinned, read by wait kernels
    uint32_t* status_host_ = nullptr;   // pinned, set by aborted wait kernels
    uint64_t max_ticks_ = 0;            // wall-clock bound for waits
    // hipStreamWriteValue64 works on this pool's window memory (probed at
    // Create): flag publishes/acks become queue packets instead of 1-wg
    // kernel launches. Waits stay as kernels — the HW wait packet cannot
    // be aborted, so a dead peer would wedge the queue forever.
    bool hw_write_ = false;
    void PublishFlag(void* mbox, uint64_t val, hipStream_t s);
    friend class P2pGroupTestPeek;
};

}  // namespace mlsl
