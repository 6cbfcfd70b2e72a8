#include "schedule.hpp"

#include <algorithm>
#include <cstring>
#include <map>

#include "../core/log.hpp"

namespace mlsl {

size_t SegOffset(size_t count, size_t parts, size_t i) { return i * count / parts; }
size_t SegCount(size_t count, size_t parts, size_t i) {
    return (i + 1) * count / parts - i * count / parts;
}

namespace {

BufRef Ref(Space sp, size_t off, size_t bytes) { return BufRef{sp, off, bytes}; }

Step MakeCopy(int phase, BufRef src, BufRef dst) {
    Step s;
    s.phase = phase;
    s.local = Step::LocalOp::COPY;
    s.local_src = src;
    s.local_dst = dst;
    return s;
}

// Degenerate single-rank group: result = copy of SEND.
Schedule SelfOnly(size_t bytes, DataType dt, ReduceOp op) {
    Schedule sch;
    sch.dtype = dt;
    sch.rop = op;
    if (bytes) sch.AddStep(MakeCopy(0, Ref(Space::SEND, 0, bytes), Ref(Space::RECV, 0, bytes)));
    sch.result = Ref(Space::RECV, 0, bytes);
    return sch;
}

int CeilLog2(int n) {
    int l = 0;
    while ((1 << l) < n) ++l;
    return l;
}

int Gcd(int a, int b) {
    while (b) {
        int t = a % b;
        a = b;
        b = t;
    }
    return a;
}

}  // namespace

// Ring allreduce: bandwidth-optimal reduce-scatter + all-gather over the
// group ring; each phase moves one segment to the next neighbor. This is the
// per-xGMI-link-bound algorithm the chunk-over-channels layer parallelizes
// (reference analog: GET_EP_PAYLOAD endpoint fan-out, src/comm_ep.cpp:99-115).
//
// `stride` rotates the ring: next = rank+stride (mod N), for any stride
// coprime with N. Different channels run different strides, so on a
// fully-connected xGMI node each channel's ring traverses a DISJOINT set
// of point-to-point links (stride s uses edges (i, i+s)) — the multi-link
// analog of the reference's multi-endpoint fan-out. The schedule algebra
// is stride-invariant: walking the ring in generator order g^0, g^1, ...
// relabels which rank owns which segment but keeps the reduce-scatter /
// all-gather structure intact.
static Schedule RingAllReduceImpl(int rank, int size, size_t count, size_t es,
                                  DataType dt, ReduceOp op, int stride) {
    if (size == 1) return SelfOnly(count * es, dt, op);

    Schedule sch;
    sch.dtype = dt;
    sch.rop = op;
    const int N = size;
    // Position of this rank along the stride-ring: r such that
    // rank = (stride * r) mod N. Segments are owned by ring POSITION, so
    // the rest of the algorithm is the textbook stride-1 ring on positions.
    int r = 0;
    if (stride % N != 1) {
        for (int pos = 0, node = 0; pos < N; ++pos, node = (node + stride) % N)
            if (node == rank) {
                r = pos;
                break;
            }
    } else {
        r = rank;
    }
    auto segOff = [&](int i) { return SegOffset(count, N, (i % N + N) % N) * es; };
    auto segBytes = [&](int i) { return SegCount(count, N, (i % N + N) % N) * es; };
    size_t max_seg = 0;
    for (int i = 0; i < N; ++i) max_seg = std::max(max_seg, segBytes(i));

    // Phase 0: materialize the working copy in RECV (skipped in-place).
    sch.AddStep(MakeCopy(0, Ref(Space::SEND, 0, count * es), Ref(Space::RECV, 0, count * es)));

    const int next = (rank + stride) % N, prev = (rank - stride % N + N) % N;
    // Reduce-scatter phases.
    for (int p = 0; p < N - 1; ++p) {
        const int ssend = (r - p % N + N) % N;
        const int srecv = (r - p % N - 1 + 2 * N) % N;
        Step st;
        st.phase = 1 + p;
        st.send_peer = next;
        st.send = Ref(Space::RECV, segOff(ssend), segBytes(ssend));
        st.recv_peer = prev;
        st.recv = Ref(Space::TMP, 0, segBytes(srecv));
        st.local = Step::LocalOp::REDUCE;
        st.local_src = st.recv;
        st.local_dst = Ref(Space::RECV, segOff(srecv), segBytes(srecv));
        sch.AddStep(st);
    }
    // All-gather phases: rank r owns segment (r+1) complete after RS.
    for (int p = 0; p < N - 1; ++p) {
        const int ssend = (r + 1 - p % N + 2 * N) % N;
        const int srecv = (r - p % N + 2 * N) % N;
        Step st;
        st.phase = N + p;
        st.send_peer = next;
        st.send = Ref(Space::RECV, segOff(ssend), segBytes(ssend));
        st.recv_peer = prev;
        st.recv = Ref(Space::RECV, segOff(srecv), segBytes(srecv));
        sch.AddStep(st);
    }
    sch.tmp_bytes = max_seg;
    sch.result = Ref(Space::RECV, 0, count * es);
    return sch;
}

Schedule BuildAllReduceRing(int rank, int size, size_t count, DataType dt, ReduceOp op,
                            int stride) {
    MLSL_CHECK(stride >= 1 && Gcd(stride, size) == 1,
               "ring stride must be coprime with the group size");
    return RingAllReduceImpl(rank, size, count, DtypeSize(dt), dt, op, stride);
}

int RingStrideForChannel(size_t channel, int size) {
    if (size <= 2) return 1;
    // Enumerate strides coprime with N in the order 1, N-1, 3, N-3, 5, ...
    // — pairs (s, N-s) are the two directions of the same physical links,
    // so consecutive channels first exploit full duplex, then new links.
    std::vector<int> strides;
    for (int s = 1; 2 * s <= size; ++s) {
        if (Gcd(s, size) != 1) continue;
        strides.push_back(s);
        if (s != size - s) strides.push_back(size - s);
    }
    return strides[channel % strides.size()];
}

Schedule BuildAllReduceRingUnits(int rank, int size, size_t units, size_t unit_bytes,
                                 size_t quant_block) {
    Schedule sch = RingAllReduceImpl(rank, size, units, unit_bytes, DataType::U8,
                                     ReduceOp::SUM, 1);
    sch.quant_block = quant_block;
    return sch;
}

// Direct (one-shot) allreduce: every rank sends its whole buffer to every
// peer in ONE exchange phase and reduces the N-1 arrivals locally. On a
// fully-connected xGMI node this drives all N-1 point-to-point links
// simultaneously with a single communication round — the latency-optimal
// small-message algorithm for MI355X's topology (the ring needs 2(N-1)
// rounds; RHD needs 2·log2 N). Costs (N-1)·S wire bytes per rank, so it
// only pays below the bandwidth crossover. TMP carries a copy of the own
// contribution (in-place safety: sends never read a buffer the reduces
// mutate) plus one arrival slot per peer.
Schedule BuildAllReduceDirect(int rank, int size, size_t count, DataType dt,
                              ReduceOp op) {
    const size_t es = DtypeSize(dt), B = count * es;
    if (size == 1) return SelfOnly(B, dt, op);

    Schedule sch;
    sch.dtype = dt;
    sch.rop = op;
    const int N = size, r = rank;
    sch.tmp_bytes = static_cast<size_t>(N) * B;  // [own][N-1 arrivals]
    sch.AddStep(MakeCopy(0, Ref(Space::SEND, 0, B), Ref(Space::TMP, 0, B)));
    sch.AddStep(MakeCopy(0, Ref(Space::SEND, 0, B), Ref(Space::RECV, 0, B)));
    sch.one_shot = true;
    int slot = 1;
    for (int j = 1; j < N; ++j, ++slot) {
        const int peer = (r + j) % N;
        Step st;
        st.phase = 1;
        st.send_peer = peer;
        st.send = Ref(Space::TMP, 0, B);
        st.recv_peer = peer;
        st.recv = Ref(Space::TMP, static_cast<size_t>(slot) * B, B);
        st.local = Step::LocalOp::REDUCE;
        st.local_src = st.recv;
        st.local_dst = Ref(Space::RECV, 0, B);
        sch.AddStep(st);
    }
    sch.result = Ref(Space::RECV, 0, B);
    return sch;
}

// Recursive halving reduce-scatter + recursive doubling all-gather
// (Rabenseifner). Power-of-two groups only — the same restriction as the
// reference's priority allreduce (eplib/cqueue.c:1903-1904). Latency-optimal:
// 2·log2(N) phases.
Schedule BuildAllReduceRHD(int rank, int size, size_t count, DataType dt, ReduceOp op) {
    const size_t es = DtypeSize(dt);
    if (size == 1) return SelfOnly(count * es, dt, op);
    MLSL_CHECK((size & (size - 1)) == 0, "RHD allreduce needs power-of-two group");

    Schedule sch;
    sch.dtype = dt;
    sch.rop = op;
    const int N = size, r = rank, L = CeilLog2(N);

    sch.AddStep(MakeCopy(0, Ref(Space::SEND, 0, count * es), Ref(Space::RECV, 0, count * es)));

    // Halving: window [lo, lo+cnt) of elements this rank still reduces.
    size_t lo = 0, cnt = count;
    struct Level { size_t lo, cnt; bool kept_low; int partner; };
    std::vector<Level> trail;
    int phase = 1;
    for (int s = 0; s < L; ++s, ++phase) {
        const int dist = N >> (s + 1);
        const int partner = r ^ dist;
        const size_t h0 = cnt / 2, h1 = cnt - h0;
        const bool keep_low = (r & dist) == 0;
        const size_t keep_off = keep_low ? lo : lo + h0;
        const size_t keep_cnt = keep_low ? h0 : h1;
        const size_t send_off = keep_low ? lo + h0 : lo;
        const size_t send_cnt = keep_low ? h1 : h0;
        Step st;
        st.phase = phase;
        st.send_peer = partner;
        st.send = Ref(Space::RECV, send_off * es, send_cnt * es);
        st.recv_peer = partner;
        st.recv = Ref(Space::TMP, 0, keep_cnt * es);
        st.local = Step::LocalOp::REDUCE;
        st.local_src = st.recv;
        st.local_dst = Ref(Space::RECV, keep_off * es, keep_cnt * es);
        sch.AddStep(st);
        trail.push_back({lo, cnt, keep_low, partner});
        lo = keep_off;
        cnt = keep_cnt;
        sch.tmp_bytes = std::max(sch.tmp_bytes, keep_cnt * es);
    }
    // Doubling: unwind the trail, exchanging owned window for the sibling.
    for (int s = L - 1; s >= 0; --s, ++phase) {
        const Level& lv = trail[s];
        const size_t h0 = lv.cnt / 2, h1 = lv.cnt - h0;
        const size_t my_off = lv.kept_low ? lv.lo : lv.lo + h0;
        const size_t my_cnt = lv.kept_low ? h0 : h1;
        const size_t sib_off = lv.kept_low ? lv.lo + h0 : lv.lo;
        const size_t sib_cnt = lv.kept_low ? h1 : h0;
        Step st;
        st.phase = phase;
        st.send_peer = lv.partner;
        st.send = Ref(Space::RECV, my_off * es, my_cnt * es);
        st.recv_peer = lv.partner;
        st.recv = Ref(Space::RECV, sib_off * es, sib_cnt * es);
        sch.AddStep(st);
    }
    sch.result = Ref(Space::RECV, 0, count * es);
    return sch;
}

// Ring reduce-scatter: rank r ends with the full sum of segment r in RECV.
// Uniform segment size (MLSL semantics: recv_count elements per rank,
// reference src/comm.hpp AddReduceScatter).
Schedule BuildReduceScatter(int rank, int size, size_t recv_count, DataType dt, ReduceOp op) {
    const size_t es = DtypeSize(dt);
    const size_t segB = recv_count * es;
    if (size == 1) return SelfOnly(segB, dt, op);

    Schedule sch;
    sch.dtype = dt;
    sch.rop = op;
    const int N = size, r = rank;
    const int next = (r + 1) % N, prev = (r - 1 + N) % N;
    // Every phase receives into a TMP slot (two alternating), so RECV may
    // alias any SEND segment (in-place ZeRO-1 gradient sharding).
    const int slots = std::min(2, N - 1);
    sch.tmp_bytes = static_cast<size_t>(slots) * segB;

    for (int p = 0; p <= N - 2; ++p) {
        const int ssend = (r - p - 1 + 2 * N) % N;
        const int srecv = (r - p - 2 + 2 * N) % N;
        Step st;
        st.phase = p;
        st.send_peer = next;
        st.send = (p == 0) ? Ref(Space::SEND, ssend * segB, segB)
                           : Ref(Space::TMP, ((p - 1) % 2) * segB, segB);
        st.recv_peer = prev;
        st.recv = Ref(Space::TMP, (p % 2) * segB, segB);
        st.local = Step::LocalOp::REDUCE;
        st.local_src = Ref(Space::SEND, srecv * segB, segB);
        st.local_dst = st.recv;
        sch.AddStep(st);
    }
    sch.AddStep(MakeCopy(N - 1, Ref(Space::TMP, ((N - 2) % 2) * segB, segB),
                         Ref(Space::RECV, 0, segB)));
    sch.result = Ref(Space::RECV, 0, segB);
    return sch;
}

// Chunked reduce-scatter: the chunk covers output elements [off, off+cnt)
// of every rank's recv_count-long segment; SEND reads use the FULL-buffer
// row stride (total*es per rank-row) so independent chunks cover the whole
// exchange (reference endpoint split of RS: src/comm_ep.cpp:623-639).
Schedule BuildReduceScatterChunk(int rank, int size, size_t total, size_t off,
                                 size_t cnt, DataType dt, ReduceOp op) {
    const size_t es = DtypeSize(dt);
    const size_t segB = cnt * es;
    if (size == 1) {
        Schedule sch;
        sch.dtype = dt;
        sch.rop = op;
        if (segB)
            sch.AddStep(MakeCopy(0, Ref(Space::SEND, off * es, segB),
                                 Ref(Space::RECV, off * es, segB)));
        sch.result = Ref(Space::RECV, off * es, segB);
        return sch;
    }

    Schedule sch;
    sch.dtype = dt;
    sch.rop = op;
    const int N = size, r = rank;
    const int next = (r + 1) % N, prev = (r - 1 + N) % N;
    const int slots = std::min(2, N - 1);
    sch.tmp_bytes = static_cast<size_t>(slots) * segB;
    auto srow = [&](int i) {
        return Ref(Space::SEND, (static_cast<size_t>(i) * total + off) * es, segB);
    };
    for (int p = 0; p <= N - 2; ++p) {
        const int ssend = (r - p - 1 + 2 * N) % N;
        const int srecv = (r - p - 2 + 2 * N) % N;
        Step st;
        st.phase = p;
        st.send_peer = next;
        st.send = (p == 0) ? srow(ssend)
                           : Ref(Space::TMP, ((p - 1) % 2) * segB, segB);
        st.recv_peer = prev;
        st.recv = Ref(Space::TMP, (p % 2) * segB, segB);
        st.local = Step::LocalOp::REDUCE;
        st.local_src = srow(srecv);
        st.local_dst = st.recv;
        sch.AddStep(st);
    }
    sch.AddStep(MakeCopy(N - 1, Ref(Space::TMP, ((N - 2) % 2) * segB, segB),
                         Ref(Space::RECV, off * es, segB)));
    sch.result = Ref(Space::RECV, 0, total * es);
    return sch;
}

// Chunked all-gather(v): chunk c covers sub-range SegOffset/SegCount(rc[i],
// nchunks, c) of EVERY rank's block — the ring forwards only those slices
// (reference endpoint split of AG(v): src/comm_ep.cpp:598-622).
Schedule BuildAllGathervChunk(int rank, int size,
                              const std::vector<size_t>& recv_counts,
                              size_t chunk_idx, size_t nchunks, DataType dt) {
    const size_t es = DtypeSize(dt);
    MLSL_CHECK(recv_counts.size() == static_cast<size_t>(size),
               "recv_counts size mismatch");
    std::vector<size_t> offB(size + 1, 0);
    for (int i = 0; i < size; ++i) offB[i + 1] = offB[i] + recv_counts[i] * es;

    Schedule sch;
    sch.dtype = dt;
    const int N = size, r = rank;
    auto sub_off = [&](int i) {
        return SegOffset(recv_counts[(i % N + N) % N], nchunks, chunk_idx);
    };
    auto sub_cnt = [&](int i) {
        return SegCount(recv_counts[(i % N + N) % N], nchunks, chunk_idx);
    };
    auto seg = [&](int i) {
        int k = (i % N + N) % N;
        return Ref(Space::RECV, offB[k] + sub_off(k) * es, sub_cnt(k) * es);
    };
    if (size == 1) {
        if (sub_cnt(0))
            sch.AddStep(MakeCopy(0, Ref(Space::SEND, sub_off(0) * es,
                                        sub_cnt(0) * es), seg(0)));
        sch.result = Ref(Space::RECV, 0, offB[1]);
        return sch;
    }
    const int next = (r + 1) % N, prev = (r - 1 + N) % N;
    sch.AddStep(MakeCopy(0, Ref(Space::SEND, sub_off(r) * es, sub_cnt(r) * es),
                         seg(r)));
    for (int p = 1; p <= N - 1; ++p) {
        Step st;
        st.phase = p;
        st.send_peer = next;
        st.send = seg(r - p + 1);
        st.recv_peer = prev;
        st.recv = seg(r - p);
        sch.AddStep(st);
    }
    sch.result = Ref(Space::RECV, 0, offB[N]);
    return sch;
}

// Chunked alltoall(v): chunk c carries sub-range c of every pairwise block
// (reference MLSL_ALLTOALL_SPLIT, src/comm_ep.cpp:712-736).
Schedule BuildAlltoAllvChunk(int rank, int size,
                             const std::vector<size_t>& send_counts,
                             const std::vector<size_t>& send_offs,
                             const std::vector<size_t>& recv_counts,
                             const std::vector<size_t>& recv_offs,
                             size_t chunk_idx, size_t nchunks, DataType dt) {
    const size_t es = DtypeSize(dt);
    Schedule sch;
    sch.dtype = dt;
    const int N = size, r = rank;
    auto sref = [&](int i) {
        return Ref(Space::SEND,
                   (send_offs[i] + SegOffset(send_counts[i], nchunks, chunk_idx)) * es,
                   SegCount(send_counts[i], nchunks, chunk_idx) * es);
    };
    auto rref = [&](int i) {
        return Ref(Space::RECV,
                   (recv_offs[i] + SegOffset(recv_counts[i], nchunks, chunk_idx)) * es,
                   SegCount(recv_counts[i], nchunks, chunk_idx) * es);
    };
    sch.AddStep(MakeCopy(0, sref(r), rref(r)));
    for (int j = 1; j < N; ++j) {
        const int to = (r + j) % N, from = (r - j + N) % N;
        Step st;
        st.phase = j;
        st.send_peer = to;
        st.send = sref(to);
        st.recv_peer = from;
        st.recv = rref(from);
        sch.AddStep(st);
    }
    size_t total = 0;
    for (int i = 0; i < N; ++i)
        total = std::max(total, (recv_offs[i] + recv_counts[i]) * es);
    sch.result = Ref(Space::RECV, 0, total);
    return sch;
}

// Ring all-gather: phase 0 copies the local contribution into slot `rank`,
// then N-1 neighbor forwards.
Schedule BuildAllGather(int rank, int size, size_t send_count, DataType dt) {
    std::vector<size_t> counts(static_cast<size_t>(size), send_count);
    return BuildAllGatherv(rank, size, counts, dt);
}

Schedule BuildAllGatherv(int rank, int size, const std::vector<size_t>& recv_counts, DataType dt) {
    const size_t es = DtypeSize(dt);
    MLSL_CHECK(recv_counts.size() == static_cast<size_t>(size), "recv_counts size mismatch");
    if (size == 1) return SelfOnly(recv_counts[0] * es, dt, ReduceOp::SUM);

    std::vector<size_t> offB(size + 1, 0);
    for (int i = 0; i < size; ++i) offB[i + 1] = offB[i] + recv_counts[i] * es;

    Schedule sch;
    sch.dtype = dt;
    const int N = size, r = rank;
    const int next = (r + 1) % N, prev = (r - 1 + N) % N;
    auto seg = [&](int i) {
        int k = (i % N + N) % N;
        return Ref(Space::RECV, offB[k], offB[k + 1] - offB[k]);
    };
    sch.AddStep(MakeCopy(0, Ref(Space::SEND, 0, recv_counts[r] * es), seg(r)));
    for (int p = 1; p <= N - 1; ++p) {
        Step st;
        st.phase = p;
        st.send_peer = next;
        st.send = seg(r - p + 1);
        st.recv_peer = prev;
        st.recv = seg(r - p);
        sch.AddStep(st);
    }
    sch.result = Ref(Space::RECV, 0, offB[N]);
    return sch;
}

// Binomial-tree broadcast over the single user buffer (RECV space).
Schedule BuildBcast(int rank, int size, size_t count, DataType dt, int root) {
    const size_t es = DtypeSize(dt);
    Schedule sch;
    sch.dtype = dt;
    sch.result = Ref(Space::RECV, 0, count * es);
    if (size == 1) return sch;

    const int N = size, L = CeilLog2(N);
    const int rr = (rank - root + N) % N;
    auto abs = [&](int rel) { return (rel + root) % N; };
    // Rank rr receives at its lowest set bit's mask, then forwards at every
    // smaller mask (classic binomial: sender at mask m iff rr % 2m == 0).
    int phase = 0;
    for (int m = 1 << (L - 1); m >= 1; m >>= 1, ++phase) {
        Step st;
        st.phase = phase;
        if ((rr & (2 * m - 1)) == 0 && rr + m < N) {
            st.send_peer = abs(rr + m);
            st.send = Ref(Space::RECV, 0, count * es);
            sch.AddStep(st);
        } else if ((rr & (2 * m - 1)) == m) {
            st.recv_peer = abs(rr - m);
            st.recv = Ref(Space::RECV, 0, count * es);
            sch.AddStep(st);
        }
    }
    sch.num_phases = std::max(sch.num_phases, L);
    return sch;
}

// Binomial-tree reduce to root. Accumulator: RECV at root, TMP upper half
// elsewhere; incoming messages land in TMP lower half.
Schedule BuildReduce(int rank, int size, size_t count, DataType dt, ReduceOp op, int root) {
    const size_t es = DtypeSize(dt), B = count * es;
    if (size == 1) return SelfOnly(B, dt, op);

    Schedule sch;
    sch.dtype = dt;
    sch.rop = op;
    const int N = size;
    const int rr = (rank - root + N) % N;
    auto abs = [&](int rel) { return (rel + root) % N; };
    const bool is_root = (rr == 0);
    BufRef acc = is_root ? Ref(Space::RECV, 0, B) : Ref(Space::TMP, B, B);
    sch.tmp_bytes = is_root ? B : 2 * B;

    sch.AddStep(MakeCopy(0, Ref(Space::SEND, 0, B), acc));
    int phase = 1;
    for (int m = 1; m < N; m <<= 1, ++phase) {
        if (rr & m) {
            Step st;
            st.phase = phase;
            st.send_peer = abs(rr - m);
            st.send = acc;
            sch.AddStep(st);
            break;
        }
        if (rr + m < N) {
            Step st;
            st.phase = phase;
            st.recv_peer = abs(rr + m);
            st.recv = Ref(Space::TMP, 0, B);
            st.local = Step::LocalOp::REDUCE;
            st.local_src = st.recv;
            st.local_dst = acc;
            sch.AddStep(st);
        }
    }
    sch.num_phases = std::max(sch.num_phases, 1 + CeilLog2(N));
    sch.result = Ref(Space::RECV, 0, B);
    return sch;
}

// Direct gather: every non-root sends its block to root in one phase
// (reference does ring-ordered tagged Isend/Irecv, src/comm_ep.cpp:1072-1183).
Schedule BuildGather(int rank, int size, size_t send_count, DataType dt, int root) {
    const size_t es = DtypeSize(dt), B = send_count * es;
    Schedule sch;
    sch.dtype = dt;
    if (size == 1) return SelfOnly(B, dt, ReduceOp::SUM);

    if (rank == root) {
        sch.AddStep(MakeCopy(0, Ref(Space::SEND, 0, B), Ref(Space::RECV, root * B, B)));
        for (int j = 0; j < size; ++j) {
            if (j == root) continue;
            Step st;
            st.phase = 0;
            st.recv_peer = j;
            st.recv = Ref(Space::RECV, j * B, B);
            sch.AddStep(st);
        }
        sch.result = Ref(Space::RECV, 0, size * B);
    } else {
        Step st;
        st.phase = 0;
        st.send_peer = root;
        st.send = Ref(Space::SEND, 0, B);
        sch.AddStep(st);
        sch.result = Ref(Space::RECV, 0, 0);
    }
    return sch;
}

Schedule BuildScatter(int rank, int size, size_t recv_count, DataType dt, int root) {
    const size_t es = DtypeSize(dt), B = recv_count * es;
    Schedule sch;
    sch.dtype = dt;
    if (size == 1) return SelfOnly(B, dt, ReduceOp::SUM);

    if (rank == root) {
        sch.AddStep(MakeCopy(0, Ref(Space::SEND, root * B, B), Ref(Space::RECV, 0, B)));
        for (int j = 0; j < size; ++j) {
            if (j == root) continue;
            Step st;
            st.phase = 0;
            st.send_peer = j;
            st.send = Ref(Space::SEND, j * B, B);
            sch.AddStep(st);
        }
    } else {
        Step st;
        st.phase = 0;
        st.recv_peer = root;
        st.recv = Ref(Space::RECV, 0, B);
        sch.AddStep(st);
    }
    sch.result = Ref(Space::RECV, 0, B);
    return sch;
}

// Pairwise-exchange alltoall: phase j exchanges with ranks ±j (the
// reference's schedule, src/comm_ep.cpp:1188-1265, re-expressed).
Schedule BuildAlltoAll(int rank, int size, size_t send_count, DataType dt) {
    std::vector<size_t> cnt(static_cast<size_t>(size), send_count);
    std::vector<size_t> off(static_cast<size_t>(size));
    for (int i = 0; i < size; ++i) off[i] = i * send_count;
    return BuildAlltoAllv(rank, size, cnt, off, cnt, off, dt);
}

Schedule BuildAlltoAllv(int rank, int size,
                        const std::vector<size_t>& send_counts,
                        const std::vector<size_t>& send_offs,
                        const std::vector<size_t>& recv_counts,
                        const std::vector<size_t>& recv_offs, DataType dt) {
    const size_t es = DtypeSize(dt);
    Schedule sch;
    sch.dtype = dt;
    const int N = size, r = rank;
    sch.AddStep(MakeCopy(0, Ref(Space::SEND, send_offs[r] * es, send_counts[r] * es),
                         Ref(Space::RECV, recv_offs[r] * es, recv_counts[r] * es)));
    for (int j = 1; j < N; ++j) {
        const int to = (r + j) % N, from = (r - j + N) % N;
        Step st;
        st.phase = j;
        st.send_peer = to;
        st.send = Ref(Space::SEND, send_offs[to] * es, send_counts[to] * es);
        st.recv_peer = from;
        st.recv = Ref(Space::RECV, recv_offs[from] * es, recv_counts[from] * es);
        sch.AddStep(st);
    }
    size_t total = 0;
    for (int i = 0; i < N; ++i) total = std::max(total, (recv_offs[i] + recv_counts[i]) * es);
    sch.result = Ref(Space::RECV, 0, total);
    return sch;
}

// Dissemination barrier: log2(N) rounds of 1-byte tokens.
Schedule BuildBarrier(int rank, int size) {
    Schedule sch;
    sch.dtype = DataType::U8;
    if (size == 1) return sch;
    sch.tmp_bytes = 2;
    const int N = size, L = CeilLog2(N);
    for (int k = 0, phase = 0; k < L; ++k, ++phase) {
        const int d = 1 << k;
        Step st;
        st.phase = phase;
        st.send_peer = (rank + d) % N;
        st.send = Ref(Space::TMP, 0, 1);
        st.recv_peer = (rank - d + N) % N;
        st.recv = Ref(Space::TMP, 1, 1);
        sch.AddStep(st);
    }
    return sch;
}

Schedule BuildSendRecvList(int rank, int size, const std::vector<SRPair>& pairs, DataType dt) {
    const size_t es = DtypeSize(dt);
    Schedule sch;
    sch.dtype = dt;
    (void)rank;
    size_t out_end = 0;
    for (const auto& p : pairs) {
        MLSL_CHECK(p.peer >= 0 && p.peer < size, "SRList peer out of range");
        Step st;
        st.phase = 0;
        if (p.send_count) {
            st.send_peer = p.peer;
            st.send = Ref(Space::SEND, p.send_off * es, p.send_count * es);
        }
        if (p.recv_count) {
            st.recv_peer = p.peer;
            st.recv = Ref(Space::RECV, p.recv_off * es, p.recv_count * es);
            out_end = std::max(out_end, (p.recv_off + p.recv_count) * es);
        }
        sch.AddStep(st);
    }
    sch.result = Ref(Space::RECV, 0, out_end);
    return sch;
}

// ---------------------------------------------------------------------------
// Host-side reduction and the in-memory simulator.

namespace {

inline float Bf16ToF32(uint16_t h) {
    uint32_t u = static_cast<uint32_t>(h) << 16;
    float f;
    std::memcpy(&f, &u, 4);
    return f;
}

inline uint16_t F32ToBf16(float f) {
    uint32_t u;
    std::memcpy(&u, &f, 4);
    // round-to-nearest-even
    uint32_t lsb = (u >> 16) & 1;
    u += 0x7fffu + lsb;
    return static_cast<uint16_t>(u >> 16);
}

template <typename T>
void ReduceLoop(T* dst, const T* src, size_t n, ReduceOp op) {
    switch (op) {
        case ReduceOp::SUM:
            for (size_t i = 0; i < n; ++i) dst[i] = dst[i] + src[i];
            break;
        case ReduceOp::MIN:
            for (size_t i = 0; i < n; ++i) dst[i] = std::min(dst[i], src[i]);
            break;
        case ReduceOp::MAX:
            for (size_t i = 0; i < n; ++i) dst[i] = std::max(dst[i], src[i]);
            break;
    }
}

void ReduceBf16(uint16_t* dst, const uint16_t* src, size_t n, ReduceOp op) {
    for (size_t i = 0; i < n; ++i) {
        float a = Bf16ToF32(dst[i]), b = Bf16ToF32(src[i]);
        float r = 0.f;
        switch (op) {
            case ReduceOp::SUM: r = a + b; break;
            case ReduceOp::MIN: r = std::min(a, b); break;
            case ReduceOp::MAX: r = std::max(a, b); break;
        }
        dst[i] = F32ToBf16(r);
    }
}

}  // namespace

void HostReduce(void* dst, const void* src, size_t count, DataType dt, ReduceOp op) {
    switch (dt) {
        case DataType::F32:
            ReduceLoop(static_cast<float*>(dst), static_cast<const float*>(src), count, op);
            break;
        case DataType::F64:
            ReduceLoop(static_cast<double*>(dst), static_cast<const double*>(src), count, op);
            break;
        case DataType::U8:
            ReduceLoop(static_cast<uint8_t*>(dst), static_cast<const uint8_t*>(src), count, op);
            break;
        case DataType::I32:
            ReduceLoop(static_cast<int32_t*>(dst), static_cast<const int32_t*>(src), count, op);
            break;
        case DataType::I64:
            ReduceLoop(static_cast<int64_t*>(dst), static_cast<const int64_t*>(src), count, op);
            break;
        case DataType::BF16:
            ReduceBf16(static_cast<uint16_t*>(dst), static_cast<const uint16_t*>(src), count, op);
            break;
        case DataType::F16:
            MLSL_THROW("host f16 reduction not supported (GPU path only)");
    }
}

void SimulateSchedules(const std::vector<Schedule>& per_rank,
                       std::vector<std::vector<uint8_t>>& send_bufs,
                       std::vector<std::vector<uint8_t>>& recv_bufs) {
    const int N = static_cast<int>(per_rank.size());
    std::vector<std::vector<uint8_t>> tmp(N);
    int max_phase = 0;
    for (const auto& s : per_rank) {
        max_phase = std::max(max_phase, s.num_phases);
    }
    for (int r = 0; r < N; ++r) tmp[r].resize(per_rank[r].tmp_bytes);

    auto ptr = [&](int r, const BufRef& b) -> uint8_t* {
        switch (b.space) {
            case Space::SEND: return send_bufs[r].data() + b.off;
            case Space::RECV: return recv_bufs[r].data() + b.off;
            case Space::TMP: return tmp[r].data() + b.off;
        }
        return nullptr;
    };

    for (int phase = 0; phase < max_phase; ++phase) {
        // mailbox[(src,dst)] = payload
        std::map<std::pair<int, int>, std::vector<uint8_t>> mail;
        for (int r = 0; r < N; ++r)
            for (const auto& st : per_rank[r].steps)
                if (st.phase == phase && st.send_peer >= 0) {
                    const uint8_t* p = ptr(r, st.send);
                    mail[{r, st.send_peer}].assign(p, p + st.send.bytes);
                }
        for (int r = 0; r < N; ++r)
            for (const auto& st : per_rank[r].steps)
                if (st.phase == phase) {
                    if (st.recv_peer >= 0) {
                        auto it = mail.find({st.recv_peer, r});
                        MLSL_CHECK(it != mail.end(), "simulator: no matching send");
                        MLSL_CHECK(it->second.size() == st.recv.bytes,
                                   "simulator: size mismatch");
                        if (st.recv.bytes)  // memcpy(p, NULL, 0) is UB
                            std::memcpy(ptr(r, st.recv), it->second.data(),
                                        st.recv.bytes);
                        mail.erase(it);
                    }
                    if (st.local == Step::LocalOp::COPY) {
                        std::memmove(ptr(r, st.local_dst), ptr(r, st.local_src),
                                     st.local_src.bytes);
                    } else if (st.local == Step::LocalOp::REDUCE) {
                        HostReduce(ptr(r, st.local_dst), ptr(r, st.local_src),
                                   st.local_dst.bytes / DtypeSize(per_rank[r].dtype),
                                   per_rank[r].dtype, per_rank[r].rop);
                    }
                }
        MLSL_CHECK(mail.empty(), "simulator: unmatched sends in phase");
    }
}

}  // namespace mlsl
