// Minimal correctness sample (reference mlsl_to_oneccl/mlsl_sample.cpp:17-63
// behavior): AllReduce of COUNT fp32 over the world; rank r contributes its
// rank index, every element must equal (size-1)*size/2. Prints PASSED.
//
// Build: make samples      Run (2 ranks, no MPI needed):
//   RANK=0 WORLD_SIZE=2 MLSL_TRANSPORT=tcp ./build/mlsl_sample &
//   RANK=1 WORLD_SIZE=2 MLSL_TRANSPORT=tcp ./build/mlsl_sample
#include <cstdio>
#include <vector>

#include "mlsl/mlsl.hpp"

static const size_t kCount = 128;

int main(int argc, char** argv) {
    using namespace mlsl;
    Environment& env = Environment::GetEnv();
    env.Init(&argc, &argv);

    const size_t rank = env.GetProcessIdx();
    const size_t size = env.GetProcessCount();

    Distribution* dist = env.CreateDistribution(size, 1);
    std::vector<float> buf(kCount, static_cast<float>(rank));

    CommRequest* req = dist->AllReduce(buf.data(), buf.data(), kCount,
                                       DataType::F32, ReduceOp::SUM,
                                       GroupKind::DATA);
    env.Wait(req);

    const float expected = (size - 1) * size / 2.0f;
    size_t bad = 0;
    for (size_t i = 0; i < kCount; ++i)
        if (buf[i] != expected) ++bad;

    if (bad == 0) {
        std::printf("[%zu/%zu] PASSED (value %.1f)\n", rank, size, expected);
    } else {
        std::printf("[%zu/%zu] FAILED: %zu wrong elements\n", rank, size, bad);
    }
    env.DeleteDistribution(dist);
    env.Finalize();
    return bad == 0 ? 0 : 1;
}
