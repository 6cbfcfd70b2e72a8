// Collective schedules: every collective is compiled into a per-rank list of
// phased steps (send/recv ranges + local reduce/copy) over a process group.
//
// This is the MI355X-native generalization of the reference's resumable
// Rabenseifner allreduce state machine (eplib/allreduce_pr.c:69-343): instead
// of one hand-written allreduce, ALL collectives are explicit phase programs,
// so the same schedule runs (a) nonblocking on the host progress thread over
// TCP sockets (CPU tests, eplib-server analog) and (b) enqueued onto HIP
// streams as RCCL send/recv groups + local-reduce HIP kernels over xGMI.
// Resumability (phase counter + byte cursors) is what lets the progress
// engine interleave many requests and prioritize the newest
// (MLSL_MSG_PRIORITY analog).
//
// Algorithms: ring reduce-scatter/all-gather (bandwidth-optimal per xGMI
// link), recursive halving/doubling (Rabenseifner) for latency/pow2,
// binomial trees for bcast/reduce, pairwise exchange for alltoall(v),
// dissemination for barrier. Reference collective inventory:
// src/comm_ep.cpp:568-1378, src/comm_handoff.cpp:299-786.
#pragma once

#include <cstdint>
#include <vector>

#include "../core/types.hpp"

namespace mlsl {

// Which buffer a range lives in. SEND/RECV are the user buffers passed to
// Start(); TMP is the request's persistent scratch (sized by Schedule).
enum class Space : uint8_t { SEND = 0, RECV = 1, TMP = 2 };

struct BufRef {
    Space space = Space::TMP;
    size_t off = 0;     // bytes
    size_t bytes = 0;
};

// One step: an optional send, an optional recv (matched by (peer, phase)),
// and an optional local op that runs after the step's recv completes.
struct Step {
    int phase = 0;
    int send_peer = -1;   // group-rank; -1 = no send this step
    BufRef send;
    int recv_peer = -1;
    BufRef recv;
    enum class LocalOp : uint8_t { NONE = 0, REDUCE = 1, COPY = 2 };
    LocalOp local = LocalOp::NONE;
    BufRef local_src;     // REDUCE: dst += src ; COPY: dst = src
    BufRef local_dst;
};

struct Schedule {
    std::vector<Step> steps;   // sorted by phase
    int num_phases = 0;
    size_t tmp_bytes = 0;
    DataType dtype = DataType::F32;
    ReduceOp rop = ReduceOp::SUM;
    // >0: REDUCE local ops are compressed-domain int8 block accumulations
    // (QuantAccum) over wire blocks of this many elements; buffer refs are
    // then in wire bytes (unit = quant_block + 8).
    size_t quant_block = 0;
    // Direct (one-shot) allreduce shape: phase 1 is "send TMP(0,B) to all
    // peers, receive each peer's TMP slot, reduce all into RECV(0,B)" —
    // executors may batch it into fan-out/fan-in kernels.
    bool one_shot = false;
    // Where the result lives after the final phase (returned by Wait()).
    BufRef result;

    void AddStep(Step s) {
        steps.push_back(s);
        if (s.phase + 1 > num_phases) num_phases = s.phase + 1;
    }
};

// Equal-count partition helpers: segment i of `count` elements over `parts`.
size_t SegOffset(size_t count, size_t parts, size_t i);
size_t SegCount(size_t count, size_t parts, size_t i);

// Schedule builders. `rank`/`size` are group-local. Counts are in elements
// of `dt`. All builders support in-place (caller passes same pointer for
// SEND and RECV at execution time; builders emit SEND-space reads only in
// phase 0 positions that are safe, or route through TMP).
// `stride` rotates the ring (next = rank+stride mod size; must be coprime
// with size): different channels run different strides so their rings
// traverse disjoint xGMI point-to-point links on a fully-connected node.
Schedule BuildAllReduceRing(int rank, int size, size_t count, DataType dt, ReduceOp op,
                            int stride = 1);
// Stride assignment for channel c: 1, size-1, 3, size-3, ... (full-duplex
// pairs first, then fresh links), cycling.
int RingStrideForChannel(size_t channel, int size);
// Ring allreduce over opaque fixed-size units (used by the quantized path:
// unit = one int8 wire block of quant_block elements + 8-byte header).
Schedule BuildAllReduceRingUnits(int rank, int size, size_t units, size_t unit_bytes,
                                 size_t quant_block);
Schedule BuildAllReduceRHD(int rank, int size, size_t count, DataType dt, ReduceOp op);
// One-shot exchange + local reduce (latency-optimal on full-mesh xGMI;
// (N-1)x wire cost — small messages only).
Schedule BuildAllReduceDirect(int rank, int size, size_t count, DataType dt,
                              ReduceOp op);
Schedule BuildReduceScatter(int rank, int size, size_t recv_count, DataType dt, ReduceOp op);
// Chunked variants (channel fan-out for the non-elementwise-splittable
// ops; reference endpoint split, src/comm_ep.cpp:598-736): offsets are
// absolute in the user buffers, so the executor passes elem_off = 0.
Schedule BuildReduceScatterChunk(int rank, int size, size_t total, size_t off,
                                 size_t cnt, DataType dt, ReduceOp op);
Schedule BuildAllGathervChunk(int rank, int size,
                              const std::vector<size_t>& recv_counts,
                              size_t chunk_idx, size_t nchunks, DataType dt);
Schedule BuildAlltoAllvChunk(int rank, int size,
                             const std::vector<size_t>& send_counts,
                             const std::vector<size_t>& send_offs,
                             const std::vector<size_t>& recv_counts,
                             const std::vector<size_t>& recv_offs,
                             size_t chunk_idx, size_t nchunks, DataType dt);
Schedule BuildAllGather(int rank, int size, size_t send_count, DataType dt);
Schedule BuildAllGatherv(int rank, int size, const std::vector<size_t>& recv_counts, DataType dt);
Schedule BuildBcast(int rank, int size, size_t count, DataType dt, int root);
Schedule BuildReduce(int rank, int size, size_t count, DataType dt, ReduceOp op, int root);
Schedule BuildGather(int rank, int size, size_t send_count, DataType dt, int root);
Schedule BuildScatter(int rank, int size, size_t recv_count, DataType dt, int root);
Schedule BuildAlltoAll(int rank, int size, size_t send_count, DataType dt);
Schedule BuildAlltoAllv(int rank, int size,
                        const std::vector<size_t>& send_counts,
                        const std::vector<size_t>& send_offs,
                        const std::vector<size_t>& recv_counts,
                        const std::vector<size_t>& recv_offs, DataType dt);
Schedule BuildBarrier(int rank, int size);

// Neighbor send/recv lists (reference CommOpSRList, src/comm.hpp:212-248 —
// declared there but never implemented; here it is first-class so
// ring-attention-style neighbor exchanges can be composed).
struct SRPair { int peer; size_t send_off, send_count, recv_off, recv_count; };
Schedule BuildSendRecvList(int rank, int size, const std::vector<SRPair>& pairs, DataType dt);

// CPU reference executor: runs `size` ranks' schedules against in-memory
// buffers, delivering messages instantly. Used by unit tests to validate
// schedule semantics without sockets or GPUs, and by the planner's
// isolation-statistics dry-run mode.
void SimulateSchedules(const std::vector<Schedule>& per_rank,
                       std::vector<std::vector<uint8_t>>& send_bufs,
                       std::vector<std::vector<uint8_t>>& recv_bufs);

// Local reduction dst += src for `count` elements (host path; the device
// path is hip/kernels.hip::LaunchReduce).
void HostReduce(void* dst, const void* src, size_t count, DataType dt, ReduceOp op);

}  // namespace mlsl
