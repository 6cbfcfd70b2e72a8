// RMA window implementation — see rma.hpp for the design contract.
#include "rma.hpp"

#include <cstring>

#include "../core/log.hpp"

namespace mlsl {

namespace {

Environment& Env() { return Environment::GetEnv(); }

// Little-endian u64 pair {off,len} — the wire meta for one put/get record.
struct WireMeta {
    uint64_t off;
    uint64_t len;
};

}  // namespace

RmaWindow::RmaWindow(Distribution* d, GroupKind g, size_t bytes)
    : dist_(d), group_(g), bytes_(bytes) {
    MLSL_CHECK(d != nullptr, "RmaWindow: null distribution");
    MLSL_CHECK(bytes > 0, "RmaWindow: zero-byte window");
    rank_ = d->GetProcessIdx(g);
    size_ = d->GetProcessCount(g);
    base_ = Env().Alloc(bytes);
    // Deterministic start: zero the window (coherently — host BAR stores
    // to HBM are not; Environment::Memcpy is the sanctioned path).
    std::vector<uint8_t> zero(bytes, 0);
    Env().Memcpy(base_, zero.data(), bytes);
    // Window creation is collective; leave with every member's window live.
    dist_->Barrier(group_);
}

RmaWindow::~RmaWindow() {
    for (auto& p : puts_)
        if (p.stage) Env().Free(p.stage);
    if (base_) Env().Free(base_);
}

void RmaWindow::Put(const void* src, size_t len, size_t target,
                    size_t target_off) {
    if (len == 0) return;
    MLSL_CHECK(target < size_, "RmaWindow::Put: target out of range");
    MLSL_CHECK(target_off + len <= bytes_,
               "RmaWindow::Put: range exceeds window (windows are symmetric)");
    void* stage = Env().Alloc(len);
    Env().Memcpy(stage, src, len);
    puts_.push_back(PutRec{target, target_off, len, stage});
}

void RmaWindow::Get(void* dst, size_t len, size_t target, size_t target_off) {
    if (len == 0) return;
    MLSL_CHECK(target < size_, "RmaWindow::Get: target out of range");
    MLSL_CHECK(target_off + len <= bytes_,
               "RmaWindow::Get: range exceeds window (windows are symmetric)");
    gets_.push_back(GetRec{target, target_off, len, dst});
}

void RmaWindow::Fence() {
    const size_t N = size_;
    // ---- epoch header exchange: per-peer {put_blob_bytes, nput, nget, 0}.
    std::vector<uint64_t> put_blob(N, 0), nput(N, 0), nget(N, 0);
    for (const auto& p : puts_) {
        put_blob[p.target] += sizeof(WireMeta) + p.len;
        nput[p.target]++;
    }
    for (const auto& g : gets_) nget[g.target]++;
    std::vector<uint64_t> sh(N * 4, 0), rh(N * 4, 0);
    for (size_t t = 0; t < N; ++t) {
        sh[t * 4 + 0] = put_blob[t];
        sh[t * 4 + 1] = nput[t];
        sh[t * 4 + 2] = nget[t];
    }
    Env().Wait(dist_->AlltoAll(sh.data(), 4, rh.data(), DataType::I64, group_));

    auto v_exchange = [&](void* sbuf, const std::vector<size_t>& scnt,
                          const std::vector<size_t>& rcnt, void** rbuf_out) {
        std::vector<size_t> soff(N, 0), roff(N, 0);
        size_t stot = 0, rtot = 0;
        for (size_t t = 0; t < N; ++t) {
            soff[t] = stot;
            roff[t] = rtot;
            stot += scnt[t];
            rtot += rcnt[t];
        }
        void* rbuf = Env().Alloc(rtot ? rtot : 1);
        Env().Wait(dist_->AlltoAllv(sbuf, scnt.data(), soff.data(), rbuf,
                                    rcnt.data(), roff.data(), DataType::U8,
                                    group_));
        *rbuf_out = rbuf;
    };

    // ---- phase 2: deliver puts.
    // Send blob to target t: [nput[t] x WireMeta][payloads, same order].
    std::vector<size_t> p_scnt(N), p_rcnt(N), p_soff(N);
    size_t p_stot = 0;
    for (size_t t = 0; t < N; ++t) {
        p_scnt[t] = static_cast<size_t>(put_blob[t]);
        p_rcnt[t] = static_cast<size_t>(rh[t * 4 + 0]);
        p_soff[t] = p_stot;
        p_stot += p_scnt[t];
    }
    void* p_sbuf = Env().Alloc(p_stot ? p_stot : 1);
    {
        // Metas first (host-assembled per target, one coherent copy each),
        // then payloads staged at Put time (device-side copies in device
        // mode — the payload never bounces through the host).
        std::vector<std::vector<WireMeta>> metas(N);
        std::vector<size_t> pay_cur(N, 0);
        for (size_t t = 0; t < N; ++t) {
            metas[t].reserve(static_cast<size_t>(nput[t]));
            pay_cur[t] = p_soff[t] + sizeof(WireMeta) * nput[t];
        }
        for (const auto& p : puts_)
            metas[p.target].push_back(WireMeta{p.off, p.len});
        for (size_t t = 0; t < N; ++t)
            if (!metas[t].empty())
                Env().Memcpy(static_cast<char*>(p_sbuf) + p_soff[t],
                             metas[t].data(),
                             metas[t].size() * sizeof(WireMeta));
        for (const auto& p : puts_) {
            Env().Memcpy(static_cast<char*>(p_sbuf) + pay_cur[p.target],
                         p.stage, p.len);
            pay_cur[p.target] += p.len;
        }
    }
    void* p_rbuf = nullptr;
    v_exchange(p_sbuf, p_scnt, p_rcnt, &p_rbuf);
    // Apply received puts to the local window (source-rank order).
    {
        size_t cursor = 0;
        for (size_t s = 0; s < N; ++s) {
            const size_t blob = static_cast<size_t>(rh[s * 4 + 0]);
            const size_t n = static_cast<size_t>(rh[s * 4 + 1]);
            if (!blob) continue;
            std::vector<WireMeta> metas(n);
            Env().Memcpy(metas.data(), static_cast<char*>(p_rbuf) + cursor,
                         n * sizeof(WireMeta));
            size_t pay = cursor + n * sizeof(WireMeta);
            for (const auto& m : metas) {
                MLSL_CHECK(m.off + m.len <= bytes_,
                           "RmaWindow: received put exceeds window");
                Env().Memcpy(static_cast<char*>(base_) + m.off,
                             static_cast<char*>(p_rbuf) + pay,
                             static_cast<size_t>(m.len));
                pay += m.len;
            }
            cursor += blob;
        }
    }
    Env().Free(p_sbuf);
    Env().Free(p_rbuf);

    // ---- phase 3: get requests ({off,len} metas only).
    std::vector<size_t> g_scnt(N), g_rcnt(N), g_soff(N);
    size_t g_stot = 0;
    for (size_t t = 0; t < N; ++t) {
        g_scnt[t] = static_cast<size_t>(nget[t]) * sizeof(WireMeta);
        g_rcnt[t] = static_cast<size_t>(rh[t * 4 + 2]) * sizeof(WireMeta);
        g_soff[t] = g_stot;
        g_stot += g_scnt[t];
    }
    void* g_sbuf = Env().Alloc(g_stot ? g_stot : 1);
    {
        std::vector<std::vector<WireMeta>> metas(N);
        for (const auto& g : gets_)
            metas[g.target].push_back(WireMeta{g.off, g.len});
        for (size_t t = 0; t < N; ++t)
            if (!metas[t].empty())
                Env().Memcpy(static_cast<char*>(g_sbuf) + g_soff[t],
                             metas[t].data(),
                             metas[t].size() * sizeof(WireMeta));
    }
    void* g_rbuf = nullptr;
    v_exchange(g_sbuf, g_scnt, g_rcnt, &g_rbuf);
    Env().Free(g_sbuf);

    // ---- phase 4: get responses (window contents AFTER this epoch's puts).
    std::vector<std::vector<WireMeta>> reqs(N);
    std::vector<size_t> r_scnt(N, 0), r_rcnt(N, 0), r_soff(N);
    {
        size_t cursor = 0;
        for (size_t s = 0; s < N; ++s) {
            const size_t n = static_cast<size_t>(rh[s * 4 + 2]);
            reqs[s].resize(n);
            if (n) {
                Env().Memcpy(reqs[s].data(),
                             static_cast<char*>(g_rbuf) + cursor,
                             n * sizeof(WireMeta));
                for (const auto& m : reqs[s]) {
                    MLSL_CHECK(m.off + m.len <= bytes_,
                               "RmaWindow: received get exceeds window");
                    r_scnt[s] += static_cast<size_t>(m.len);
                }
            }
            cursor += n * sizeof(WireMeta);
        }
    }
    Env().Free(g_rbuf);
    for (const auto& g : gets_) r_rcnt[g.target] += g.len;
    size_t r_stot = 0;
    for (size_t t = 0; t < N; ++t) {
        r_soff[t] = r_stot;
        r_stot += r_scnt[t];
    }
    void* r_sbuf = Env().Alloc(r_stot ? r_stot : 1);
    for (size_t s = 0; s < N; ++s) {
        size_t cur = r_soff[s];
        for (const auto& m : reqs[s]) {
            Env().Memcpy(static_cast<char*>(r_sbuf) + cur,
                         static_cast<char*>(base_) + m.off,
                         static_cast<size_t>(m.len));
            cur += m.len;
        }
    }
    void* r_rbuf = nullptr;
    v_exchange(r_sbuf, r_scnt, r_rcnt, &r_rbuf);
    Env().Free(r_sbuf);
    // Scatter responses into the user dst pointers: per-target segments
    // arrive in my submission order (both sides preserve record order).
    {
        std::vector<size_t> roff(N, 0), cur(N, 0);
        size_t rtot = 0;
        for (size_t t = 0; t < N; ++t) {
            roff[t] = rtot;
            rtot += r_rcnt[t];
        }
        for (const auto& g : gets_) {
            Env().Memcpy(g.dst,
                         static_cast<char*>(r_rbuf) + roff[g.target] +
                             cur[g.target],
                         g.len);
            cur[g.target] += g.len;
        }
    }
    Env().Free(r_rbuf);

    // ---- epoch reset.
    for (auto& p : puts_) Env().Free(p.stage);
    puts_.clear();
    gets_.clear();
}

}  // namespace mlsl
