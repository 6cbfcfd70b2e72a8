// Pure-host unit test of every collective schedule via the in-memory
// simulator: analytic expected values over group sizes 1..9 (incl. non
// power-of-two), run by pytest (tests/test_schedules.py). Exit 0 = PASS.
#include <cmath>
#include <cstdio>
#include <cstring>
#include <vector>

#include "../comm/schedule.hpp"

using namespace mlsl;

static int g_failures = 0;

#define EXPECT(cond, ...)                                                     \
    do {                                                                      \
        if (!(cond)) {                                                        \
            std::printf("FAIL %s:%d: ", __FILE__, __LINE__);                  \
            std::printf(__VA_ARGS__);                                         \
            std::printf("\n");                                                \
            ++g_failures;                                                     \
        }                                                                     \
    } while (0)

static std::vector<float> F(const std::vector<uint8_t>& b) {
    std::vector<float> f(b.size() / 4);
    std::memcpy(f.data(), b.data(), b.size());
    return f;
}

static void FillF(std::vector<uint8_t>& b, size_t count, float base) {
    b.resize(count * 4);
    float* p = reinterpret_cast<float*>(b.data());
    for (size_t i = 0; i < count; ++i) p[i] = base + static_cast<float>(i);
}

static void TestAllReduce(int N, size_t count, bool rhd) {
    std::vector<Schedule> sch(N);
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        sch[r] = rhd ? BuildAllReduceRHD(r, N, count, DataType::F32, ReduceOp::SUM)
                     : BuildAllReduceRing(r, N, count, DataType::F32, ReduceOp::SUM);
        FillF(sbuf[r], count, static_cast<float>(r));
        rbuf[r].assign(count * 4, 0);
    }
    SimulateSchedules(sch, sbuf, rbuf);
    // sum over ranks of (r + i) = N*i + N(N-1)/2
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (size_t i = 0; i < count; ++i) {
            float want = static_cast<float>(N) * i + N * (N - 1) / 2.0f;
            EXPECT(f[i] == want, "%s allreduce N=%d rank=%d i=%zu got %f want %f",
                   rhd ? "rhd" : "ring", N, r, i, f[i], want);
            if (f[i] != want) return;
        }
    }
}

static void TestAllReduceDirect(int N, size_t count) {
    std::vector<Schedule> sch(N);
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        sch[r] = BuildAllReduceDirect(r, N, count, DataType::F32, ReduceOp::SUM);
        FillF(sbuf[r], count, static_cast<float>(r));
        rbuf[r].assign(count * 4, 0);
    }
    SimulateSchedules(sch, sbuf, rbuf);
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (size_t i = 0; i < count; ++i) {
            float want = static_cast<float>(N) * i + N * (N - 1) / 2.0f;
            EXPECT(f[i] == want, "direct allreduce N=%d rank=%d i=%zu got %f want %f",
                   N, r, i, f[i], want);
            if (f[i] != want) return;
        }
    }
}

static void TestAllReduceRingStride(int N, size_t count, int stride) {
    std::vector<Schedule> sch(N);
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        sch[r] = BuildAllReduceRing(r, N, count, DataType::F32, ReduceOp::SUM,
                                    stride);
        FillF(sbuf[r], count, static_cast<float>(r));
        rbuf[r].assign(count * 4, 0);
    }
    SimulateSchedules(sch, sbuf, rbuf);
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (size_t i = 0; i < count; ++i) {
            float want = static_cast<float>(N) * i + N * (N - 1) / 2.0f;
            EXPECT(f[i] == want,
                   "stride-%d ring allreduce N=%d rank=%d i=%zu got %f want %f",
                   stride, N, r, i, f[i], want);
            if (f[i] != want) return;
        }
    }
}

static void TestAllReduceMax(int N, size_t count) {
    std::vector<Schedule> sch(N);
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        sch[r] = BuildAllReduceRing(r, N, count, DataType::F32, ReduceOp::MAX);
        FillF(sbuf[r], count, static_cast<float>(r * 100));
        rbuf[r].assign(count * 4, 0);
    }
    SimulateSchedules(sch, sbuf, rbuf);
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (size_t i = 0; i < count; ++i) {
            float want = static_cast<float>((N - 1) * 100) + i;
            EXPECT(f[i] == want, "max allreduce N=%d rank=%d i=%zu got %f want %f", N,
                   r, i, f[i], want);
            if (f[i] != want) return;
        }
    }
}

static void TestReduceScatter(int N, size_t per) {
    std::vector<Schedule> sch(N);
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        sch[r] = BuildReduceScatter(r, N, per, DataType::F32, ReduceOp::SUM);
        FillF(sbuf[r], per * N, static_cast<float>(r));
        rbuf[r].assign(per * 4, 0);
    }
    SimulateSchedules(sch, sbuf, rbuf);
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (size_t i = 0; i < per; ++i) {
            float want = static_cast<float>(N) * (r * per + i) + N * (N - 1) / 2.0f;
            EXPECT(f[i] == want, "rs N=%d rank=%d i=%zu got %f want %f", N, r, i, f[i],
                   want);
            if (f[i] != want) return;
        }
    }
}

static void TestAllGather(int N, size_t per) {
    std::vector<Schedule> sch(N);
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        sch[r] = BuildAllGather(r, N, per, DataType::F32);
        FillF(sbuf[r], per, static_cast<float>(r * 1000));
        rbuf[r].assign(per * N * 4, 0);
    }
    SimulateSchedules(sch, sbuf, rbuf);
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (int j = 0; j < N; ++j)
            for (size_t i = 0; i < per; ++i) {
                float want = static_cast<float>(j * 1000) + i;
                EXPECT(f[j * per + i] == want, "ag N=%d rank=%d seg=%d i=%zu", N, r, j, i);
                if (f[j * per + i] != want) return;
            }
    }
}

static void TestAllGatherv(int N) {
    std::vector<size_t> counts;
    for (int i = 0; i < N; ++i) counts.push_back(3 + 2 * i);
    size_t total = 0;
    for (auto c : counts) total += c;
    std::vector<Schedule> sch(N);
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        sch[r] = BuildAllGatherv(r, N, counts, DataType::F32);
        FillF(sbuf[r], counts[r], static_cast<float>(r * 1000));
        rbuf[r].assign(total * 4, 0);
    }
    SimulateSchedules(sch, sbuf, rbuf);
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        size_t off = 0;
        for (int j = 0; j < N; ++j) {
            for (size_t i = 0; i < counts[j]; ++i) {
                float want = static_cast<float>(j * 1000) + i;
                EXPECT(f[off + i] == want, "agv N=%d rank=%d seg=%d i=%zu", N, r, j, i);
                if (f[off + i] != want) return;
            }
            off += counts[j];
        }
    }
}

// Chunked builders (channel fan-out): running every chunk's schedule over
// the SAME user buffers must reproduce the unchunked op exactly (absolute
// offsets, elem_off = 0 contract — request.cpp BuildChunks).
static void TestReduceScatterChunked(int N, size_t per, size_t nchunks) {
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        FillF(sbuf[r], per * N, static_cast<float>(r));
        rbuf[r].assign(per * 4, 0);
    }
    for (size_t c = 0; c < nchunks; ++c) {
        std::vector<Schedule> sch(N);
        for (int r = 0; r < N; ++r)
            sch[r] = BuildReduceScatterChunk(r, N, per, SegOffset(per, nchunks, c),
                                             SegCount(per, nchunks, c),
                                             DataType::F32, ReduceOp::SUM);
        SimulateSchedules(sch, sbuf, rbuf);
    }
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (size_t i = 0; i < per; ++i) {
            float want = static_cast<float>(N) * (r * per + i) + N * (N - 1) / 2.0f;
            EXPECT(f[i] == want, "rs-chunk N=%d k=%zu rank=%d i=%zu got %f want %f",
                   N, nchunks, r, i, f[i], want);
            if (f[i] != want) return;
        }
    }
}

static void TestAllGathervChunked(int N, size_t nchunks) {
    std::vector<size_t> counts;
    for (int i = 0; i < N; ++i) counts.push_back(3 + 2 * i);
    size_t total = 0;
    for (auto c : counts) total += c;
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        FillF(sbuf[r], counts[r], static_cast<float>(r * 1000));
        rbuf[r].assign(total * 4, 0);
    }
    for (size_t c = 0; c < nchunks; ++c) {
        std::vector<Schedule> sch(N);
        for (int r = 0; r < N; ++r)
            sch[r] = BuildAllGathervChunk(r, N, counts, c, nchunks, DataType::F32);
        SimulateSchedules(sch, sbuf, rbuf);
    }
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        size_t off = 0;
        for (int j = 0; j < N; ++j) {
            for (size_t i = 0; i < counts[j]; ++i) {
                float want = static_cast<float>(j * 1000) + i;
                EXPECT(f[off + i] == want, "agv-chunk N=%d k=%zu rank=%d seg=%d i=%zu",
                       N, nchunks, r, j, i);
                if (f[off + i] != want) return;
            }
            off += counts[j];
        }
    }
}

static void TestAlltoAllvChunked(int N, size_t nchunks) {
    // Heterogeneous counts: block (r -> j) has 2 + ((r + j) % 3) elements.
    std::vector<std::vector<size_t>> scnt(N), soff(N), rcnt(N), roff(N);
    for (int r = 0; r < N; ++r) {
        scnt[r].resize(N);
        soff[r].resize(N);
        rcnt[r].resize(N);
        roff[r].resize(N);
        size_t so = 0, ro = 0;
        for (int j = 0; j < N; ++j) {
            scnt[r][j] = 2 + static_cast<size_t>((r + j) % 3);
            soff[r][j] = so;
            so += scnt[r][j];
            rcnt[r][j] = 2 + static_cast<size_t>((j + r) % 3);
            roff[r][j] = ro;
            ro += rcnt[r][j];
        }
    }
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        size_t stot = soff[r][N - 1] + scnt[r][N - 1];
        size_t rtot = roff[r][N - 1] + rcnt[r][N - 1];
        sbuf[r].resize(stot * 4);
        float* p = reinterpret_cast<float*>(sbuf[r].data());
        for (int j = 0; j < N; ++j)
            for (size_t i = 0; i < scnt[r][j]; ++i)
                p[soff[r][j] + i] = r * 10000 + j * 100 + static_cast<float>(i);
        rbuf[r].assign(rtot * 4, 0);
    }
    for (size_t c = 0; c < nchunks; ++c) {
        std::vector<Schedule> sch(N);
        for (int r = 0; r < N; ++r)
            sch[r] = BuildAlltoAllvChunk(r, N, scnt[r], soff[r], rcnt[r], roff[r],
                                         c, nchunks, DataType::F32);
        SimulateSchedules(sch, sbuf, rbuf);
    }
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (int j = 0; j < N; ++j)
            for (size_t i = 0; i < rcnt[r][j]; ++i) {
                float want = j * 10000 + r * 100 + static_cast<float>(i);
                EXPECT(f[roff[r][j] + i] == want,
                       "a2av-chunk N=%d k=%zu rank=%d from=%d i=%zu got %f",
                       N, nchunks, r, j, i, f[roff[r][j] + i]);
                if (f[roff[r][j] + i] != want) return;
            }
    }
}

static void TestBcastReduce(int N, size_t count, int root) {
    {
        std::vector<Schedule> sch(N);
        std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
        for (int r = 0; r < N; ++r) {
            sch[r] = BuildBcast(r, N, count, DataType::F32, root);
            sbuf[r].assign(count * 4, 0);
            if (r == root) FillF(rbuf[r], count, 7.0f);
            else rbuf[r].assign(count * 4, 0);
        }
        SimulateSchedules(sch, sbuf, rbuf);
        for (int r = 0; r < N; ++r) {
            auto f = F(rbuf[r]);
            for (size_t i = 0; i < count; ++i) {
                EXPECT(f[i] == 7.0f + i, "bcast N=%d root=%d rank=%d i=%zu got %f", N,
                       root, r, i, f[i]);
                if (f[i] != 7.0f + i) return;
            }
        }
    }
    {
        std::vector<Schedule> sch(N);
        std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
        for (int r = 0; r < N; ++r) {
            sch[r] = BuildReduce(r, N, count, DataType::F32, ReduceOp::SUM, root);
            FillF(sbuf[r], count, static_cast<float>(r));
            rbuf[r].assign(count * 4, 0);
        }
        SimulateSchedules(sch, sbuf, rbuf);
        auto f = F(rbuf[root]);
        for (size_t i = 0; i < count; ++i) {
            float want = static_cast<float>(N) * i + N * (N - 1) / 2.0f;
            EXPECT(f[i] == want, "reduce N=%d root=%d i=%zu got %f want %f", N, root, i,
                   f[i], want);
            if (f[i] != want) return;
        }
    }
}

static void TestGatherScatter(int N, size_t per, int root) {
    {
        std::vector<Schedule> sch(N);
        std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
        for (int r = 0; r < N; ++r) {
            sch[r] = BuildGather(r, N, per, DataType::F32, root);
            FillF(sbuf[r], per, static_cast<float>(r * 10));
            rbuf[r].assign(per * N * 4, 0);
        }
        SimulateSchedules(sch, sbuf, rbuf);
        auto f = F(rbuf[root]);
        for (int j = 0; j < N; ++j)
            for (size_t i = 0; i < per; ++i) {
                float want = static_cast<float>(j * 10) + i;
                EXPECT(f[j * per + i] == want, "gather N=%d root=%d seg=%d", N, root, j);
                if (f[j * per + i] != want) return;
            }
    }
    {
        std::vector<Schedule> sch(N);
        std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
        for (int r = 0; r < N; ++r) {
            sch[r] = BuildScatter(r, N, per, DataType::F32, root);
            if (r == root) FillF(sbuf[r], per * N, 0.0f);
            else sbuf[r].assign(per * N * 4, 0);
            rbuf[r].assign(per * 4, 0);
        }
        SimulateSchedules(sch, sbuf, rbuf);
        for (int r = 0; r < N; ++r) {
            auto f = F(rbuf[r]);
            for (size_t i = 0; i < per; ++i) {
                float want = static_cast<float>(r * per + i);
                EXPECT(f[i] == want, "scatter N=%d root=%d rank=%d i=%zu", N, root, r, i);
                if (f[i] != want) return;
            }
        }
    }
}

static void TestAlltoAll(int N, size_t per) {
    std::vector<Schedule> sch(N);
    std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
    for (int r = 0; r < N; ++r) {
        sch[r] = BuildAlltoAll(r, N, per, DataType::F32);
        // element (dest j, i) = r*10000 + j*100 + i
        sbuf[r].resize(per * N * 4);
        float* p = reinterpret_cast<float*>(sbuf[r].data());
        for (int j = 0; j < N; ++j)
            for (size_t i = 0; i < per; ++i) p[j * per + i] = r * 10000 + j * 100 + i;
        rbuf[r].assign(per * N * 4, 0);
    }
    SimulateSchedules(sch, sbuf, rbuf);
    for (int r = 0; r < N; ++r) {
        auto f = F(rbuf[r]);
        for (int j = 0; j < N; ++j)
            for (size_t i = 0; i < per; ++i) {
                float want = j * 10000 + r * 100 + static_cast<float>(i);
                EXPECT(f[j * per + i] == want, "a2a N=%d rank=%d from=%d i=%zu got %f",
                       N, r, j, i, f[j * per + i]);
                if (f[j * per + i] != want) return;
            }
    }
}

static void TestBarrierAndSRList(int N) {
    {
        std::vector<Schedule> sch(N);
        std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
        for (int r = 0; r < N; ++r) sch[r] = BuildBarrier(r, N);
        SimulateSchedules(sch, sbuf, rbuf);  // just must not throw/mismatch
    }
    if (N > 1) {
        // neighbor ring exchange: send my value to next, recv from prev.
        std::vector<Schedule> sch(N);
        std::vector<std::vector<uint8_t>> sbuf(N), rbuf(N);
        for (int r = 0; r < N; ++r) {
            std::vector<SRPair> pairs;
            pairs.push_back(SRPair{(r + 1) % N, 0, 4, 0, 0});
            pairs.push_back(SRPair{(r - 1 + N) % N, 0, 0, 0, 4});
            sch[r] = BuildSendRecvList(r, N, pairs, DataType::F32);
            FillF(sbuf[r], 4, static_cast<float>(r * 50));
            rbuf[r].assign(16, 0);
        }
        SimulateSchedules(sch, sbuf, rbuf);
        for (int r = 0; r < N; ++r) {
            auto f = F(rbuf[r]);
            int prev = (r - 1 + N) % N;
            for (size_t i = 0; i < 4; ++i) {
                EXPECT(f[i] == prev * 50 + static_cast<float>(i), "srlist N=%d rank=%d",
                       N, r);
                if (f[i] != prev * 50 + static_cast<float>(i)) return;
            }
        }
    }
}

int main() {
    for (int N : {1, 2, 3, 4, 5, 7, 8, 9}) {
        for (size_t count : {1ul, 5ul, 64ul, 1000ul}) {
            TestAllReduce(N, count, false);
            if ((N & (N - 1)) == 0) TestAllReduce(N, count, true);
        }
        TestAllReduceMax(N, 100);
        TestAllReduceDirect(N, 73);
        // Rotated rings (multi-xGMI-link channel fan-out): every coprime
        // stride must produce the identical allreduce.
        for (int s = 1; s < N; ++s) {
            bool cop = true;
            for (int d = 2; d <= s; ++d)
                if (s % d == 0 && N % d == 0) cop = false;
            if (cop) TestAllReduceRingStride(N, 64, s);
        }
        for (size_t ch = 0; ch < 8; ++ch)
            EXPECT(RingStrideForChannel(ch, N) >= 1, "stride channel %zu", ch);
        TestReduceScatter(N, 17);
        TestAllGather(N, 9);
        TestAllGatherv(N);
        for (size_t k : {2ul, 3ul, 5ul}) {
            TestReduceScatterChunked(N, 17, k);
            TestAllGathervChunked(N, k);
            TestAlltoAllvChunked(N, k);
        }
        for (int root : {0, N - 1}) {
            TestBcastReduce(N, 33, root);
            TestGatherScatter(N, 5, root);
        }
        TestAlltoAll(N, 6);
        TestBarrierAndSRList(N);
    }
    if (g_failures == 0) {
        std::printf("SCHEDULE SELFTEST PASSED\n");
        return 0;
    }
    std::printf("SCHEDULE SELFTEST: %d failures\n", g_failures);
    return 1;
}
