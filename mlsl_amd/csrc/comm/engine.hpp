// Async progress engine: the MI355X-native redesign of eplib's endpoint
// servers (eplib/server.c, cqueue.c). Instead of proxy MPI processes fed by
// shared-memory rings, a host progress THREAD owns the TCP mesh / HIP
// streams and advances resumable request state machines; the API thread
// talks to it through a lock-free MPSC command ring (cqueue analog; any
// application thread may submit) with a mutex-guarded overflow deque.
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

// Fixed-size lock-free multi-producer/single-consumer ring (the cqueue_t
// analog, eplib/cqueue.h:95-183 — 1000-entry cache-line ring; ours carries
// pointers, payloads stay in the request). Multi-producer because any
// application thread may Start() a request concurrently (torch autograd
// hooks, user threads); per-slot sequence numbers (Vyukov bounded queue)
// make Push safe from any thread while Pop stays single-consumer (the
// progress thread).
class MpscRing {
  public:
    static constexpr size_t kCap = 1024;
    MpscRing() {
        for (size_t i = 0; i < kCap; ++i)
            slots_[i].seq.store(i, std::memory_order_relaxed);
    }
    bool Push(CommRequest* r) {
        uint64_t t = tail_.load(std::memory_order_relaxed);
        for (;;) {
            Slot& s = slots_[t % kCap];
            const uint64_t seq = s.seq.load(std::memory_order_acquire);
            const int64_t dif = static_cast<int64_t>(seq) - static_cast<int64_t>(t);
            if (dif == 0) {
                if (tail_.compare_exchange_weak(t, t + 1,
                                                std::memory_order_relaxed))
                {
                    s.req = r;
                    s.seq.store(t + 1, std::memory_order_release);
                    return true;
                }
                // CAS failure reloaded t; retry.
            } else if (dif < 0) {
                return false;  // full
            } else {
                t = tail_.load(std::memory_order_relaxed);
            }
        }
    }
    bool Empty() const {
        return head_.load(std::memory_order_acquire) ==
               tail_.load(std::memory_order_acquire);
    }
    CommRequest* Pop() {
        const uint64_t h = head_.load(std::memory_order_relaxed);
        Slot& s = slots_[h % kCap];
        const uint64_t seq = s.seq.load(std::memory_order_acquire);
        if (static_cast<int64_t>(seq) - static_cast<int64_t>(h + 1) < 0)
            return nullptr;
        CommRequest* r = s.req;
        s.seq.store(h + kCap, std::memory_order_release);
        head_.store(h + 1, std::memory_order_release);
        return r;
    }

  private:
    struct Slot {
        std::atomic<uint64_t> seq{0};
        CommRequest* req = nullptr;
    };
    alignas(64) std::atomic<uint64_t> head_{0};
    alignas(64) std::atomic<uint64_t> tail_{0};
    alignas(64) Slot slots_[kCap];
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

    MpscRing ring_;
    std::mutex inbox_mu_;            // overflow path when the ring is full
    std::deque<CommRequest*> inbox_overflow_;
    std::atomic<bool> overflow_pending_{false};
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
