// Typed runtime configuration, populated from MLSL_* environment variables.
// Capability parity with reference src/env.{hpp,cpp} plus the ~25 knobs the
// reference parses inside its comm backends (src/comm_ep.cpp:43-97,1543-1699,
// src/comm_handoff.cpp:68-107) — here in ONE typed struct.
#pragma once

#include <cstddef>
#include <string>

namespace mlsl {

enum class ProgressMode : int {
    THREAD = 0,  // dedicated host progress thread (eplib ep_server analog)
    INLINE = 1,  // collectives issued on the calling thread ("thread mode" analog)
};

enum class AllReduceAlgo : int {
    AUTO = 0,
    FUSED = 1,   // transport-native fused collective (RCCL ncclAllReduce)
    RING = 2,    // our chunked ring schedule
    RHD = 3,     // recursive-halving/doubling (Rabenseifner) schedule
    DIRECT = 4,  // one-shot exchange + local reduce (small msgs, full mesh)
};

struct Config {
    int log_level = 0;             // MLSL_LOG_LEVEL
    bool stats = false;            // MLSL_STATS
    ProgressMode progress = ProgressMode::THREAD;  // MLSL_PROGRESS=thread|inline
    // Channel parallelism: number of parallel comm channels (streams/comms)
    // a large message is chunked over. The xGMI analog of the reference's
    // endpoint servers (MLSL_NUM_SERVERS, default 4, src/comm_ep.cpp:123).
    size_t num_channels = 1;       // MLSL_NUM_CHANNELS (alias MLSL_NUM_SERVERS)
    size_t max_short_msg = 65536;  // MLSL_MAX_SHORT_MSG_SIZE (bytes): world-1
                                   // messages at or below issue on the
                                   // compute stream (no dep-event handshake)
    size_t large_msg_mb = 128;     // MLSL_LARGE_MSG_SIZE_MB: chunk-harder threshold
    size_t large_msg_chunks = 4;   // MLSL_LARGE_MSG_CHUNKS: extra chunks per channel
    AllReduceAlgo allreduce_algo = AllReduceAlgo::AUTO;  // MLSL_ALLREDUCE_ALGO
    size_t msg_priority_threshold = 10000;  // MLSL_MSG_PRIORITY_THRESHOLD bytes
    bool msg_priority = false;     // MLSL_MSG_PRIORITY: newest-first scheduling
    size_t quant_block = 256;      // MLSL_QUANT_BLOCK: elems per int8 quant block
    size_t heap_mb = 0;            // MLSL_HEAP_SIZE_MB: device pool pre-reserve
    bool check_pointers = false;   // MLSL_CHECK_POINTERS: validate collective bufs
    std::string transport = "auto";  // MLSL_TRANSPORT=auto|tcp|rccl
    // Device transport for multi-rank groups: RCCL comms, or the IPC HBM
    // window transport (comm/p2p_transport.hpp). auto = RCCL, switching to
    // p2p when several ranks share one device (RCCL refuses that layout).
    std::string device_transport = "auto";  // MLSL_DEVICE_TRANSPORT=auto|rccl|p2p
    size_t p2p_slot_mb = 16;       // MLSL_P2P_SLOT_MB: staging slot size
    size_t p2p_slots = 4;          // MLSL_P2P_SLOTS: in-flight slots per edge-lane
    // Pageable-host staging copies (reference MLSL_USE_COPY_THREADS /
    // MLSL_COPY_THREADS / MLSL_COPY_THRESHOLD, src/comm_ep.cpp:357-361):
    // user<->pinned memcpys above the threshold split across threads.
    size_t copy_threads = 4;       // MLSL_COPY_THREADS
    size_t copy_threshold = 4u << 20;  // MLSL_COPY_THRESHOLD (bytes)
    // Pin the progress thread to a CPU core (reference MLSL_SERVER_AFFINITY,
    // eplib/server.c:63-81 pinned each ep_server). -1 = unpinned.
    int server_affinity = -1;      // MLSL_SERVER_AFFINITY
    int timeout_sec = 300;         // MLSL_TIMEOUT: bootstrap/collective timeout
    bool use_graphs = false;       // MLSL_USE_GRAPHS: hipGraph replay of
                                   // persistent device requests (single
                                   // channel; eager fallback on capture
                                   // failure)

    static Config FromEnv();
    void Dump() const;  // rank-0 dump of effective values (ref comm_ep.cpp:1701)
};

Config& GlobalConfig();

}  // namespace mlsl
