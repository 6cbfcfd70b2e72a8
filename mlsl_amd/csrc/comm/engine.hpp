// Async progress engine: the MI355X-native redesign of eplib's endpoint
// servers (eplib/server.c, cqueue.c). Instead of proxy MPI processes fed by
// shared-memory rings, a host progress THREAD owns the TCP mesh / HIP
// streams and advances resumable request state machines; the API thread
// talks to it through a lock-free SPSC command ring (cqueue analog) with a
// mutex fallback for multi-threaded producers.
//
// MLSL_PROGRESS=inline gives the reference's "thread mode" analog: work is
// issued on the calling thread (device mode stays async via streams).
// MLSL_MSG_PRIORITY=1 makes the progress loop scan newest-first above the
// size threshold (eplib msg_priority_mode analog, allreduce_pr.c:69-81).
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <mutex>
#include <thread>
#include <vector>

#include "../core/config.hpp"

namespace mlsl {

class CommRequest;
class Mesh;

// Fixed-size lock-free single-producer/single-consumer ring (the cqueue_t
// analog, eplib/cqueue.h:95-183 — 1000-entry cache-line ring; ours carries
// pointers, payloads stay in the request).
class SpscRing {
  public:
    static constexpr size_t kCap = 1024;
    bool Push(CommRequest* r) {
        const uint64_t t = tail_.load(std::memory_order_relaxed);
        if (t - head_.load(std::memory_order_acquire) >= kCap) return false;
        slots_[t % kCap] = r;
        tail_.store(t + 1, std::memory_order_release);
        return true;
    }
    bool Empty() const {
        return head_.load(std::memory_order_acquire) ==
               tail_.load(std::memory_order_acquire);
    }
    CommRequest* Pop() {
        const uint64_t h = head_.load(std::memory_order_relaxed);
        if (h == tail_.load(std::memory_order_acquire)) return nullptr;
        CommRequest* r = slots_[h % kCap];
        head_.store(h + 1, std::memory_order_release);
        return r;
    }

  private:
    alignas(64) std::atomic<uint64_t> head_{0};
    alignas(64) std::atomic<uint64_t> tail_{0};
    alignas(64) CommRequest* slots_[kCap] = {};
};

class Engine {
  public:
    Engine(Mesh* mesh, ProgressMode mode, bool device_mode);
    ~Engine();

    // Hand a started request to the progress engine.
    void Submit(CommRequest* req);
    // Block until the request completes; progress is driven by the engine
    // thread (or by this caller in inline mode).
    void WaitFor(CommRequest* req);
    // Non-blocking completion probe.
    bool TestFor(CommRequest* req);

    uint64_t NextSeqno() { return seqno_++; }

  private:
    void Loop();
    bool AdvanceOne(CommRequest* req);
    void DrainInbox();
    bool ProgressAll();  // true if it drained, advanced or completed work

    Mesh* mesh_;
    ProgressMode mode_;
    bool device_mode_;
    std::atomic<bool> stop_{false};
    std::thread thread_;

    SpscRing ring_;
    std::mutex inbox_mu_;            // fallback for multi-producer submit
    std::deque<CommRequest*> inbox_overflow_;
    std::atomic<uint64_t> seqno_{1};

    std::vector<CommRequest*> active_;
    std::mutex done_mu_;
    std::condition_variable done_cv_;

    // Deep-idle parking: after the hot spin window the loop waits on this
    // condvar (bounded) instead of sleeping blind; Submit nudges it so
    // wake-up latency is a notify, not a sleep quantum (the reference's
    // ep_server spins 100% forever — we spin hot briefly, then park).
    std::mutex idle_mu_;
    std::condition_variable idle_cv_;
    std::atomic<bool> deep_idle_{false};

    friend class CommRequest;
    void NotifyDone();
};

}  // namespace mlsl
