#include "config.hpp"

#include <cstdlib>
#include <cstring>

#include "log.hpp"

namespace mlsl {

static size_t EnvSize(const char* name, size_t dflt) {
    if (const char* e = std::getenv(name)) {
        long long v = std::atoll(e);
        if (v >= 0) return static_cast<size_t>(v);
    }
    return dflt;
}

static bool EnvBool(const char* name, bool dflt) {
    if (const char* e = std::getenv(name)) return std::atoi(e) != 0;
    return dflt;
}

Config Config::FromEnv() {
    Config c;
    c.log_level = static_cast<int>(EnvSize("MLSL_LOG_LEVEL", 0));
    c.stats = EnvBool("MLSL_STATS", false);

    if (const char* e = std::getenv("MLSL_PROGRESS")) {
        if (!std::strcmp(e, "inline")) c.progress = ProgressMode::INLINE;
        else c.progress = ProgressMode::THREAD;
    }
    // MLSL_NUM_SERVERS kept as a compatibility alias for users of the
    // reference library; MLSL_NUM_CHANNELS is the native name.
    c.num_channels = EnvSize("MLSL_NUM_CHANNELS", EnvSize("MLSL_NUM_SERVERS", 1));
    if (c.num_channels < 1) c.num_channels = 1;
    if (c.num_channels > 16) c.num_channels = 16;
    c.max_short_msg = EnvSize("MLSL_MAX_SHORT_MSG_SIZE", 65536);
    c.large_msg_mb = EnvSize("MLSL_LARGE_MSG_SIZE_MB", 128);
    c.large_msg_chunks = EnvSize("MLSL_LARGE_MSG_CHUNKS", 4);
    if (const char* e = std::getenv("MLSL_ALLREDUCE_ALGO")) {
        if (!std::strcmp(e, "fused")) c.allreduce_algo = AllReduceAlgo::FUSED;
        else if (!std::strcmp(e, "ring")) c.allreduce_algo = AllReduceAlgo::RING;
        else if (!std::strcmp(e, "rhd")) c.allreduce_algo = AllReduceAlgo::RHD;
        else if (!std::strcmp(e, "direct")) c.allreduce_algo = AllReduceAlgo::DIRECT;
        else c.allreduce_algo = AllReduceAlgo::AUTO;
    }
    c.msg_priority = EnvBool("MLSL_MSG_PRIORITY", false);
    c.msg_priority_threshold = EnvSize("MLSL_MSG_PRIORITY_THRESHOLD", 10000);
    c.quant_block = EnvSize("MLSL_QUANT_BLOCK", 256);
    c.heap_mb = EnvSize("MLSL_HEAP_SIZE_MB", 0);
    c.check_pointers = EnvBool("MLSL_CHECK_POINTERS", false);
    if (const char* e = std::getenv("MLSL_TRANSPORT")) c.transport = e;
    if (const char* e = std::getenv("MLSL_DEVICE_TRANSPORT")) c.device_transport = e;
    c.p2p_slot_mb = EnvSize("MLSL_P2P_SLOT_MB", 16);
    if (c.p2p_slot_mb < 1) c.p2p_slot_mb = 1;
    c.p2p_slots = EnvSize("MLSL_P2P_SLOTS", 4);
    if (c.p2p_slots < 2) c.p2p_slots = 2;
    c.copy_threads = EnvSize("MLSL_COPY_THREADS", 4);
    if (c.copy_threads < 1) c.copy_threads = 1;
    if (c.copy_threads > 16) c.copy_threads = 16;
    c.copy_threshold = EnvSize("MLSL_COPY_THRESHOLD", 4u << 20);
    if (const char* e = std::getenv("MLSL_SERVER_AFFINITY"))
        c.server_affinity = std::atoi(e);  // first core of the ref's list form
    c.timeout_sec = static_cast<int>(EnvSize("MLSL_TIMEOUT", 300));
    c.use_graphs = EnvBool("MLSL_USE_GRAPHS", false);
    return c;
}

void Config::Dump() const {
    MLSL_LOG(INFO, "config: log_level=%d stats=%d progress=%s channels=%zu "
             "large_msg_mb=%zu large_msg_chunks=%zu algo=%d priority=%d(%zuB) "
             "quant_block=%zu heap_mb=%zu chkp=%d transport=%s timeout=%ds",
             log_level, (int)stats,
             progress == ProgressMode::THREAD ? "thread" : "inline",
             num_channels, large_msg_mb, large_msg_chunks, (int)allreduce_algo,
             (int)msg_priority, msg_priority_threshold, quant_block, heap_mb,
             (int)check_pointers, transport.c_str(), timeout_sec);
}

Config& GlobalConfig() {
    static Config cfg = Config::FromEnv();
    return cfg;
}

}  // namespace mlsl
