// Concurrent-submit stress: N application threads Start()/Wait() their own
// requests simultaneously, exercising the MPSC command ring (engine.hpp)
// from multiple producers. Built under TSan (make tsan-submit) this is the
// regression test for the round-1 submit race: the old SPSC ring raced on
// tail_ when two threads Start()ed concurrently.
//
// Env: RANK/WORLD_SIZE/MLSL_PORT (tests/mp.py convention), THREADS, ITERS.
#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <thread>
#include <vector>

#include "../include/mlsl/mlsl.hpp"

using namespace mlsl;

static int EnvInt(const char* n, int d) {
    const char* e = std::getenv(n);
    return e ? std::atoi(e) : d;
}

int main(int argc, char** argv) {
    Environment& env = Environment::GetEnv();
    env.Init(&argc, &argv);
    const size_t size = env.GetProcessCount();
    const int nthreads = EnvInt("THREADS", 4);
    const int iters = EnvInt("ITERS", 50);

    // One Distribution PER THREAD: collectives on one group must be issued
    // in the same order on every rank, which concurrent threads cannot
    // guarantee — per-thread groups make the cross-rank matching safe while
    // the Submit()s still race into the shared MPSC ring. Color-created
    // groups are fresh objects even when they span the whole world
    // (Distribution(size,1) would alias the shared world group).
    std::vector<Distribution*> dists;
    for (int t = 0; t < nthreads; ++t)
        dists.push_back(env.CreateDistributionWithColors(
            /*data_color=*/0, /*model_color=*/static_cast<int>(env.GetProcessIdx())));
    std::atomic<int> fails{0};

    std::vector<std::thread> ts;
    for (int t = 0; t < nthreads; ++t) {
        ts.emplace_back([&, t]() {
            const size_t count = 64 + static_cast<size_t>(t) * 17;
            std::vector<float> in(count), out(count);
            for (int it = 0; it < iters; ++it) {
                for (size_t i = 0; i < count; ++i)
                    in[i] = static_cast<float>(i + t) + it;
                CommRequest* r = dists[t]->AllReduce(in.data(), out.data(), count,
                                                     DataType::F32, ReduceOp::SUM,
                                                     GroupKind::DATA);
                float* res = static_cast<float*>(env.Wait(r));
                for (size_t i = 0; i < count; ++i) {
                    const float want = size * (static_cast<float>(i + t) + it);
                    if (res[i] != want) {
                        std::printf("FAIL t=%d it=%d i=%zu: %f vs %f\n", t, it,
                                    i, res[i], want);
                        fails.fetch_add(1);
                        return;
                    }
                }
            }
        });
    }
    for (auto& th : ts) th.join();
    env.Finalize();
    if (fails.load() == 0) {
        std::printf("PASSED\n");
        return 0;
    }
    return 1;
}
