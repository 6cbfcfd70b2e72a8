#include "engine.hpp"

#include <pthread.h>
#include <sched.h>

#include <algorithm>
#include <chrono>

#include "../core/log.hpp"
#include "../core/types.hpp"
#include "mesh.hpp"
#include "request.hpp"

namespace mlsl {

Engine::Engine(Mesh* mesh, ProgressMode mode, bool device_mode)
    : mesh_(mesh), mode_(mode), device_mode_(device_mode) {
    if (mode_ == ProgressMode::THREAD) {
        thread_ = std::thread([this]() { Loop(); });
    }
}

Engine::~Engine() {
    stop_.store(true, std::memory_order_release);
    {
        // Pairs with the Loop's untimed idle wait: the store above is
        // re-checked under idle_mu_, so this notify can't be lost.
        std::lock_guard<std::mutex> lk(idle_mu_);
        idle_cv_.notify_one();
    }
    if (thread_.joinable()) thread_.join();
}

static uint64_t NowNs() {
    return std::chrono::duration_cast<std::chrono::nanoseconds>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
}

void Engine::Submit(CommRequest* req) {
    req->start_seqno_ = NextSeqno();
    req->start_ns_ = NowNs();
    if (mode_ == ProgressMode::INLINE) {
        // "Thread mode" analog: drive the request to completion on the
        // calling thread (device mode still returns early: streams are
        // asynchronous and AdvanceDevice only enqueues + polls).
        req->state_.store(ReqState::ACTIVE, std::memory_order_release);
        try {
            if (device_mode_) {
                // Issue now; completion is polled in Wait/Test.
                if (req->AdvanceDevice()) req->MarkDone();
                else active_.push_back(req);
            } else {
                while (!req->AdvanceHost(mesh_)) {
                    if (mesh_) mesh_->Progress();
                }
                req->MarkDone();
            }
        } catch (const std::exception& e) {
            req->MarkFailed(e.what());
        }
        return;
    }
    // Progress-thread mode: hand off through the lock-free MPSC ring
    // (overflow to the locked deque only when the ring is full).
    if (!ring_.Push(req)) {
        std::lock_guard<std::mutex> lk(inbox_mu_);
        inbox_overflow_.push_back(req);
        overflow_pending_.store(true, std::memory_order_release);
    }
    // Dekker pairing with the Loop's park decision: the fence orders the
    // push before the deep_idle_ read in the seq_cst fence order, so either
    // the parking thread's predicate sees the push, or we see deep_idle_
    // and notify. (The loop has the matching fence before its predicate.)
    std::atomic_thread_fence(std::memory_order_seq_cst);
    if (deep_idle_.load(std::memory_order_acquire)) {
        // Pairs with the Loop's check-then-wait under idle_mu_: taking the
        // lock here means the loop either saw the push or is parked and
        // gets the notify.
        std::lock_guard<std::mutex> lk(idle_mu_);
        idle_cv_.notify_one();
    }
}

void Engine::DrainInbox() {
    while (CommRequest* r = ring_.Pop()) {
        r->state_.store(ReqState::ACTIVE, std::memory_order_release);
        active_.push_back(r);
    }
    if (overflow_pending_.load(std::memory_order_acquire)) {
        std::lock_guard<std::mutex> lk(inbox_mu_);
        for (CommRequest* r : inbox_overflow_) {
            r->state_.store(ReqState::ACTIVE, std::memory_order_release);
            active_.push_back(r);
        }
        inbox_overflow_.clear();
        overflow_pending_.store(false, std::memory_order_release);
    }
    // Newest-first priority above the size threshold (reference
    // MLSL_MSG_PRIORITY head-first scan, eplib/allreduce_pr.c:69-81 +
    // env.h:59): large fresh gradients overtake older bulk transfers.
    const Config& cfg = GlobalConfig();
    if (cfg.msg_priority && active_.size() > 1) {
        std::stable_sort(active_.begin(), active_.end(),
                         [&](CommRequest* a, CommRequest* b) {
                             const bool pa = a->MessageBytes() >= cfg.msg_priority_threshold;
                             const bool pb = b->MessageBytes() >= cfg.msg_priority_threshold;
                             if (pa != pb) return pa > pb;
                             if (pa) return a->StartSeqno() > b->StartSeqno();
                             return a->StartSeqno() < b->StartSeqno();
                         });
    }
}

bool Engine::AdvanceOne(CommRequest* req) {
    try {
        const bool done = device_mode_ ? req->AdvanceDevice() : req->AdvanceHost(mesh_);
        if (done) {
            req->MarkDone();
            return true;
        }
        // Failure detection: a collective stuck past MLSL_TIMEOUT (peer
        // died, mismatched schedule) fails loudly instead of hanging Wait
        // forever (reference had none of this — SURVEY.md 5.3).
        const int tmo = GlobalConfig().timeout_sec;
        if (tmo > 0 && NowNs() - req->start_ns_ > static_cast<uint64_t>(tmo) * 1000000000ull) {
            req->MarkFailed(std::string("timeout after ") + std::to_string(tmo) +
                            "s in " + CollOpName(req->Spec().op));
            return true;
        }
        return false;
    } catch (const std::exception& e) {
        MLSL_LOG(ERROR, "request %s failed in progress engine: %s",
                 CollOpName(req->Spec().op), e.what());
        req->MarkFailed(e.what());
        return true;
    }
}

bool Engine::ProgressAll() {
    const size_t before_drain = active_.size();
    DrainInbox();
    bool did_work = active_.size() != before_drain;
    if (mesh_) mesh_->Progress();
    bool any_done = false;
    for (size_t i = 0; i < active_.size();) {
        if (AdvanceOne(active_[i])) {
            active_.erase(active_.begin() + static_cast<long>(i));
            any_done = true;
        } else {
            ++i;
        }
    }
    if (any_done) NotifyDone();
    return did_work || any_done || !active_.empty();
}

void Engine::Loop() {
    // MLSL_SERVER_AFFINITY: pin the progress thread (the reference pinned
    // each ep_server process to a core, eplib/server.c:63-81). Keeps the
    // poll loop off the cores running the framework's compute threads.
    const int aff = GlobalConfig().server_affinity;
    if (aff >= 0) {
        cpu_set_t set;
        CPU_ZERO(&set);
        CPU_SET(static_cast<unsigned>(aff) % CPU_SETSIZE, &set);
        if (pthread_setaffinity_np(pthread_self(), sizeof(set), &set) != 0)
            MLSL_LOG(ERROR, "MLSL_SERVER_AFFINITY=%d: pinning failed", aff);
        else
            MLSL_LOG(DEBUG, "progress thread pinned to core %d", aff);
    }
    int idle_spins = 0;
    while (!stop_.load(std::memory_order_acquire)) {
        bool did_work = false;
        try {
            did_work = ProgressAll();
        } catch (const std::exception& e) {
            // Transport-level failure (not attributable to one request):
            // fail everything in flight rather than terminating the process.
            MLSL_LOG(ERROR, "progress engine transport failure: %s", e.what());
            for (CommRequest* r : active_) r->MarkFailed(e.what());
            active_.clear();
            NotifyDone();
        }
        if (!did_work) {
            // Hot for the first ~4096 polls (sub-ms window covering the
            // gaps of a busy training loop), then park on the condvar.
            // Untimed wait: the protocol is lost-wakeup-free (Submit and
            // the destructor notify under idle_mu_ AFTER their stores, and
            // the predicate re-checks under the same lock). A timed
            // wait_for would also work but trips a known TSan false
            // positive (libstdc++'s pthread_cond_clockwait is not
            // intercepted by gcc-11 libtsan — spurious "double lock").
            if (++idle_spins > 4096) {
                std::unique_lock<std::mutex> lk(idle_mu_);
                deep_idle_.store(true, std::memory_order_release);
                std::atomic_thread_fence(std::memory_order_seq_cst);
                idle_cv_.wait(lk, [&]() {
                    return !ring_.Empty() ||
                           overflow_pending_.load(std::memory_order_acquire) ||
                           stop_.load(std::memory_order_acquire);
                });
                deep_idle_.store(false, std::memory_order_release);
                idle_spins = 0;
            }
        } else {
            idle_spins = 0;
        }
    }
    // Drain what we can so Finalize doesn't strand requests.
    ProgressAll();
}

void Engine::NotifyDone() {
    std::lock_guard<std::mutex> lk(done_mu_);
    done_cv_.notify_all();
}

void Engine::WaitFor(CommRequest* req) {
    if (mode_ == ProgressMode::INLINE) {
        // Inline: caller drives progress until done.
        while (true) {
            ReqState st = req->State();
            if (st == ReqState::DONE || st == ReqState::FAILED || st == ReqState::IDLE)
                return;
            // Only device requests can still be pending here.
            auto it = std::find(active_.begin(), active_.end(), req);
            if (it != active_.end() && AdvanceOne(req))
                active_.erase(std::find(active_.begin(), active_.end(), req));
        }
    }
    // Spin briefly (hot path: sub-ms collectives), then block on the condvar.
    for (int i = 0; i < 4096; ++i) {
        ReqState st = req->State();
        if (st == ReqState::DONE || st == ReqState::FAILED || st == ReqState::IDLE) return;
    }
    std::unique_lock<std::mutex> lk(done_mu_);
    done_cv_.wait(lk, [&]() {
        ReqState st = req->State();
        return st == ReqState::DONE || st == ReqState::FAILED || st == ReqState::IDLE;
    });
}

bool Engine::TestFor(CommRequest* req) {
    if (mode_ == ProgressMode::INLINE) {
        auto it = std::find(active_.begin(), active_.end(), req);
        if (it != active_.end()) {
            if (AdvanceOne(req)) {
                active_.erase(std::find(active_.begin(), active_.end(), req));
                return true;
            }
            return false;
        }
    }
    ReqState st = req->State();
    return st == ReqState::DONE || st == ReqState::FAILED || st == ReqState::IDLE;
}

}  // namespace mlsl
