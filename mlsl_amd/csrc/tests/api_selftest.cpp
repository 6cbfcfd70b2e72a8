// End-to-end C++ API test: the 2-layer synthetic CC net over the reference
// run matrix (tests/examples/mlsl_test/mlsl_test.cpp protocol), exercising
// Environment / Distribution / Session / Operation / Activation /
// ParameterSet directly in C++ with analytic expected values.
//
// Env: RANK/WORLD_SIZE/MLSL_PORT (see tests/mp.py), MP=model_parts,
// DIST_UPDATE=0|1. Exit 0 = PASS.
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

#include "../dl/rma.hpp"
#include "../dl/session.hpp"
#include "../include/mlsl/mlsl.hpp"

using namespace mlsl;

static int g_fail = 0;
#define CHECK_EQ(a, b, ...)                                                   \
    do {                                                                      \
        if (!((a) == (b))) {                                                  \
            std::printf("FAIL %d: ", __LINE__);                               \
            std::printf(__VA_ARGS__);                                         \
            std::printf("\n");                                                \
            ++g_fail;                                                         \
        }                                                                     \
    } while (0)

static int EnvInt(const char* n, int d) {
    const char* e = std::getenv(n);
    return e ? std::atoi(e) : d;
}

int main(int argc, char** argv) {
    Environment& env = Environment::GetEnv();
    env.Init(&argc, &argv);
    const size_t size = env.GetProcessCount();
    const size_t mp = static_cast<size_t>(EnvInt("MP", 1));
    const bool du = EnvInt("DIST_UPDATE", 0) != 0;
    const size_t dp = size / mp;
    const size_t S = 5, F0 = 4 * mp, F1 = 8 * mp, KS = 3;
    const size_t MB = 4 * dp;

    Distribution* dist = env.CreateDistribution(dp, mp);
    Session* sess = env.CreateSession();
    sess->SetGlobalMinibatchSize(MB);

    auto mkop = [&](const char* name, size_t fin, size_t fout) {
        OperationRegInfo* info = sess->CreateOperationRegInfo(OpKind::CC);
        info->SetName(name);
        info->AddInput(fin, S, DataType::F32);
        info->AddOutput(fout, S, DataType::F32);
        info->AddParameterSet(fin * fout, KS, DataType::F32, du);
        info->Validate(dist);
        return sess->GetOperation(sess->AddOperation(info, dist));
    };
    Operation* op0 = mkop("fc0", F0, F1);
    Operation* op1 = mkop("fc1", F1, F0);
    op0->SetNext(op1, 0, 0);
    sess->Commit();

    Activation* out0 = op0->GetOutput(0);
    Activation* in1 = op1->GetInput(0);
    ParameterSet* ps = op0->GetParameterSet(0);
    const size_t lmb = op0->GetLocalMinibatchSize();
    const size_t didx = dist->GetProcessIdx(GroupKind::DATA);
    const size_t midx = dist->GetProcessIdx(GroupKind::MODEL);
    const size_t f1l = in1->GetLocalFmCount();

    std::vector<float> comm0(out0->GetCommBufSize() / 4 + 1);
    std::vector<float> comm1(in1->GetCommBufSize() / 4 + 1);

    for (int it = 0; it < 2; ++it) {
        if (mp > 1) {
            // forward: pack partials partial(b,fm,k) = fm*100 + k + b + m
            for (size_t i = 0; i < out0->GetPackBlockCount(); ++i) {
                const CommBlockInfo* b = out0->GetPackBlock(i);
                for (size_t mb = 0; mb < b->GetMbCount(); ++mb)
                    for (size_t fm = 0; fm < b->GetFmCount(); ++fm)
                        for (size_t k = 0; k < S; ++k) {
                            const size_t gfm = b->GetFmOffset() + fm;
                            comm0[b->GetBufOffset() + (mb * b->GetFmCount() + fm) * S + k] =
                                gfm * 100.0f + k + (b->GetMbOffset() + mb) + it + midx;
                        }
            }
            out0->StartComm(comm0.data());
            float* res = static_cast<float*>(in1->WaitComm());
            CHECK_EQ(res != nullptr, true, "null fwd result");
            const CommBlockInfo* ub = in1->GetUnpackBlock(0);
            for (size_t mb = 0; mb < lmb && res; ++mb)
                for (size_t fm = 0; fm < f1l; ++fm)
                    for (size_t k = 0; k < S; ++k) {
                        const float got = res[ub->GetBufOffset() + (mb * f1l + fm) * S + k];
                        const size_t gfm = midx * f1l + fm;
                        const float want = mp * (gfm * 100.0f + k + mb + it) +
                                           mp * (mp - 1) / 2.0f;
                        if (got != want) {
                            CHECK_EQ(got, want, "fwd mismatch mb=%zu fm=%zu k=%zu: %f vs %f",
                                     mb, fm, k, got, want);
                            goto fwd_done;
                        }
                    }
        fwd_done:
            // backward: grads g(b,fm,k) = gfm*7 + k + b + it, AllGather back
            {
                const CommBlockInfo* pb = in1->GetPackBlock(0);
                for (size_t mb = 0; mb < lmb; ++mb)
                    for (size_t fm = 0; fm < f1l; ++fm)
                        for (size_t k = 0; k < S; ++k) {
                            const size_t gfm = midx * f1l + fm;
                            comm1[pb->GetBufOffset() + (mb * f1l + fm) * S + k] =
                                gfm * 7.0f + k + mb + it;
                        }
                in1->StartComm(comm1.data());
                float* bres = static_cast<float*>(out0->WaitComm());
                CHECK_EQ(bres != nullptr, true, "null bwd result");
                for (size_t i = 0; i < out0->GetUnpackBlockCount() && bres; ++i) {
                    const CommBlockInfo* b = out0->GetUnpackBlock(i);
                    for (size_t mb = 0; mb < b->GetMbCount(); ++mb)
                        for (size_t fm = 0; fm < b->GetFmCount(); ++fm)
                            for (size_t k = 0; k < S; ++k) {
                                const float got =
                                    bres[b->GetBufOffset() + (mb * b->GetFmCount() + fm) * S + k];
                                const size_t gfm = b->GetFmOffset() + fm;
                                const float want = gfm * 7.0f + k + (b->GetMbOffset() + mb) + it;
                                if (got != want) {
                                    CHECK_EQ(got, want, "bwd mismatch");
                                    goto bwd_done;
                                }
                            }
                }
            bwd_done:;
            }
        }

        // gradient exchange: grad(j,e) = (gk0+j)*10 + e + didx + it
        const size_t lk = ps->GetLocalKernelCount();
        const size_t gk0 = ps->GetGlobalKernelOffset();
        std::vector<float> grad(lk * KS);
        for (size_t j = 0; j < lk; ++j)
            for (size_t e = 0; e < KS; ++e)
                grad[j * KS + e] = (gk0 + j) * 10.0f + e + didx + it;
        ps->StartGradientComm(grad.data());
        float* g = static_cast<float*>(ps->WaitGradientComm());
        if (dp > 1) {
            const size_t n = du ? ps->GetOwnedKernelCount() : lk;
            const size_t joff = du ? ps->GetOwnedKernelOffset() : 0;
            for (size_t j = 0; j < n; ++j)
                for (size_t e = 0; e < KS; ++e) {
                    const float want =
                        dp * ((gk0 + joff + j) * 10.0f + e + it) + dp * (dp - 1) / 2.0f;
                    if (g[j * KS + e] != want) {
                        CHECK_EQ(g[j * KS + e], want, "grad mismatch j=%zu e=%zu", j, e);
                        goto grad_done;
                    }
                }
        grad_done:
            if (du) {
                // increments: inc(j,e) = (gk0+j)*3 + e; AllGather to full
                const size_t ok = ps->GetOwnedKernelCount(), oo = ps->GetOwnedKernelOffset();
                for (size_t j = 0; j < ok; ++j)
                    for (size_t e = 0; e < KS; ++e)
                        grad[(oo + j) * KS + e] = (gk0 + oo + j) * 3.0f + e + it;
                ps->StartIncrementComm(grad.data());
                float* inc = static_cast<float*>(ps->WaitIncrementComm());
                for (size_t j = 0; j < lk; ++j)
                    for (size_t e = 0; e < KS; ++e) {
                        const float want = (gk0 + j) * 3.0f + e + it;
                        if (inc[j * KS + e] != want) {
                            CHECK_EQ(inc[j * KS + e], want, "inc mismatch");
                            goto inc_done;
                        }
                    }
            inc_done:;
            }
        }
    }

    // One-sided RMA window over the data group (fence-epoch semantics):
    // every rank puts rank+1 into ITS slot of every member's window and
    // same-epoch-gets its right neighbor's own slot (must observe the put).
    {
        const size_t dn = dist->GetProcessCount(GroupKind::DATA);
        const size_t dr = dist->GetProcessIdx(GroupKind::DATA);
        const size_t n = 32;
        RmaWindow win(dist, GroupKind::DATA, dn * n * sizeof(float));
        std::vector<float> src(n, static_cast<float>(dr + 1));
        for (size_t t = 0; t < dn; ++t)
            win.Put(src.data(), n * sizeof(float), t, dr * n * sizeof(float));
        const size_t rt = (dr + 1) % dn;
        std::vector<float> peek(n, 0.0f);
        win.Get(peek.data(), n * sizeof(float), rt, rt * n * sizeof(float));
        win.Fence();
        CHECK_EQ(peek[0], static_cast<float>(rt + 1), "rma get-after-put %f",
                 peek[0]);
        CHECK_EQ(peek[n - 1], static_cast<float>(rt + 1), "rma get tail");
        std::vector<float> local(dn * n, 0.0f);
        env.Memcpy(local.data(), win.Buffer(), dn * n * sizeof(float));
        for (size_t s = 0; s < dn; ++s)
            if (local[s * n] != static_cast<float>(s + 1)) {
                CHECK_EQ(local[s * n], static_cast<float>(s + 1),
                         "rma window slot %zu", s);
                break;
            }
        win.Fence();  // empty epoch
    }

    // Quantized gradient allreduce (host int8 block ring with error
    // feedback) — lossy, so the check is tolerance-based: values are in
    // [-4, 4], block scale <= 4/127, accumulated re-quant error across a
    // dp-rank ring stays well under 0.5 while any missing/extra
    // contribution is >= 1.
    if (dp > 1) {
        Session* qs = env.CreateSession();
        qs->SetGlobalMinibatchSize(MB);
        OperationRegInfo* qi = qs->CreateOperationRegInfo(OpKind::CC);
        qi->SetName("qfc");
        qi->AddInput(F0, S, DataType::F32);
        qi->AddOutput(F1, S, DataType::F32);
        qi->AddParameterSet(512 * mp, 2, DataType::F32, false,
                            Compression::QUANT_INT8);
        qi->Validate(dist);
        Operation* qop = qs->GetOperation(qs->AddOperation(qi, dist));
        qs->Commit();
        ParameterSet* qp = qop->GetParameterSet(0);
        const size_t lk = qp->GetLocalKernelCount();
        const size_t gk0 = qp->GetGlobalKernelOffset();
        std::vector<float> g(lk * 2);
        for (size_t j = 0; j < lk; ++j)
            for (size_t e = 0; e < 2; ++e)
                g[j * 2 + e] = static_cast<float>(((gk0 + j + e + didx) % 9)) /
                                   2.0f - 2.0f;
        std::vector<float> want(lk * 2, 0.0f);
        for (size_t r = 0; r < dp; ++r)
            for (size_t j = 0; j < lk; ++j)
                for (size_t e = 0; e < 2; ++e)
                    want[j * 2 + e] +=
                        static_cast<float>(((gk0 + j + e + r) % 9)) / 2.0f - 2.0f;
        qp->StartGradientComm(g.data());
        float* qg = static_cast<float*>(qp->WaitGradientComm());
        CHECK_EQ(qg != nullptr, true, "null quant grad result");
        for (size_t i = 0; i < lk * 2 && qg; ++i)
            if (std::fabs(qg[i] - want[i]) > 0.5f) {
                CHECK_EQ(qg[i], want[i], "quant grad i=%zu got %f want %f", i,
                         qg[i], want[i]);
                break;
            }
        env.DeleteSession(qs);
    }

    env.DeleteSession(sess);
    env.DeleteDistribution(dist);
    if (g_fail == 0) std::printf("API SELFTEST PASSED\n");
    else std::printf("API SELFTEST: %d failures\n", g_fail);
    env.Finalize();
    return g_fail == 0 ? 0 : 1;
}
