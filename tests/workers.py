"""Worker entry points for multi-process tests (see tests/mp.py).

Each worker asserts analytic expected values; process exit code is the test
result. Run as: python -m tests.workers <worker_name>
"""
import os
import sys

import numpy as np


def _init():
    import mlsl_amd as mx
    mx.init()
    return mx, mx.rank(), mx.world_size()


def plumbing_allreduce(*, count=128):
    """The mlsl_sample check (mlsl_to_oneccl/mlsl_sample.cpp:40-55):
    AllReduce of `count` fp32, each rank contributes its idx;
    expected per element = (size-1)*size/2."""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    buf = np.full(count, float(rank), dtype=np.float32)
    req = d.all_reduce(buf, buf, count, op="sum", group="data")
    mx.wait(req)
    want = (size - 1) * size / 2.0
    assert np.all(buf == want), f"allreduce got {buf[:4]} want {want}"
    d.barrier("global")
    mx.finalize()


def collectives_sweep():
    """All 12 collectives with analytic values on the world group."""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    g = "data"

    # all_reduce sum + max, several sizes incl. chunked path
    for count in (1, 7, 1024, 50000):
        a = np.arange(count, dtype=np.float32) + rank
        out = np.zeros(count, dtype=np.float32)
        mx.wait(d.all_reduce(a, out, count, op="sum", group=g))
        want = size * np.arange(count, dtype=np.float32) + size * (size - 1) / 2.0
        assert np.allclose(out, want), f"allreduce count={count}"
    a = np.full(64, float(rank * 3), dtype=np.float32)
    mx.wait(d.all_reduce(a, a, 64, op="max", group=g))
    assert np.all(a == (size - 1) * 3.0), "allreduce max in-place"

    # f64 + i64
    a = np.arange(100, dtype=np.float64) * (rank + 1)
    out = np.zeros(100, dtype=np.float64)
    mx.wait(d.all_reduce(a, out, 100, op="sum", group=g))
    want = np.arange(100, dtype=np.float64) * (size * (size + 1) / 2.0)
    assert np.allclose(out, want), "f64 allreduce"
    a = np.arange(64, dtype=np.int64) + rank
    out = np.zeros(64, dtype=np.int64)
    mx.wait(d.all_reduce(a, out, 64, op="sum", group=g))
    assert np.all(out == size * np.arange(64) + size * (size - 1) // 2), "i64"

    # bcast
    b = np.zeros(33, dtype=np.float32)
    if rank == min(1, size - 1):
        b[:] = np.arange(33) + 5
    mx.wait(d.bcast(b, 33, root=min(1, size - 1), group=g))
    assert np.all(b == np.arange(33) + 5), "bcast"

    # reduce
    a = np.full(17, float(rank + 1), dtype=np.float32)
    out = np.zeros(17, dtype=np.float32)
    mx.wait(d.reduce(a, out, 17, op="sum", root=0, group=g))
    if rank == 0:
        assert np.all(out == size * (size + 1) / 2.0), "reduce"

    # reduce_scatter: rank r gets sum of segment r
    per = 13
    a = np.arange(per * size, dtype=np.float32) + rank
    out = np.zeros(per, dtype=np.float32)
    mx.wait(d.reduce_scatter(a, out, per, op="sum", group=g))
    seg = np.arange(rank * per, (rank + 1) * per, dtype=np.float32)
    want = size * seg + size * (size - 1) / 2.0
    assert np.allclose(out, want), "reduce_scatter"

    # all_gather
    a = np.full(9, float(rank * 100), dtype=np.float32) + np.arange(9, dtype=np.float32)
    out = np.zeros(9 * size, dtype=np.float32)
    mx.wait(d.all_gather(a, 9, out, group=g))
    for j in range(size):
        assert np.all(out[j * 9:(j + 1) * 9] == j * 100 + np.arange(9)), "all_gather"

    # all_gatherv
    counts = [3 + 2 * i for i in range(size)]
    a = np.full(counts[rank], float(rank), dtype=np.float32)
    out = np.zeros(sum(counts), dtype=np.float32)
    mx.wait(d.all_gatherv(a, counts[rank], out, counts, group=g))
    off = 0
    for j in range(size):
        assert np.all(out[off:off + counts[j]] == j), "all_gatherv"
        off += counts[j]

    # gather / scatter
    a = np.full(5, float(rank), dtype=np.float32)
    out = np.zeros(5 * size, dtype=np.float32)
    mx.wait(d.gather(a, 5, out, root=0, group=g))
    if rank == 0:
        for j in range(size):
            assert np.all(out[j * 5:(j + 1) * 5] == j), "gather"
    sb = np.arange(4 * size, dtype=np.float32) if rank == 0 else np.zeros(4 * size, np.float32)
    rb = np.zeros(4, dtype=np.float32)
    mx.wait(d.scatter(sb, rb, 4, root=0, group=g))
    assert np.all(rb == np.arange(rank * 4, rank * 4 + 4)), "scatter"

    # alltoall
    per = 6
    a = np.zeros(per * size, dtype=np.float32)
    for j in range(size):
        a[j * per:(j + 1) * per] = rank * 1000 + j * 10 + np.arange(per)
    out = np.zeros(per * size, dtype=np.float32)
    mx.wait(d.all_to_all(a, per, out, group=g))
    for j in range(size):
        assert np.all(out[j * per:(j + 1) * per] == j * 1000 + rank * 10 + np.arange(per)), "a2a"

    # alltoallv (ragged)
    scnt = [(rank + j) % size + 1 for j in range(size)]
    rcnt = [(j + rank) % size + 1 for j in range(size)]
    soff, acc = [], 0
    for c in scnt:
        soff.append(acc)
        acc += c
    roff, acc = [], 0
    for c in rcnt:
        roff.append(acc)
        acc += c
    a = np.zeros(sum(scnt), dtype=np.float32)
    for j in range(size):
        a[soff[j]:soff[j] + scnt[j]] = rank * 100 + j
    out = np.zeros(sum(rcnt), dtype=np.float32)
    mx.wait(d.all_to_allv(a, scnt, soff, out, rcnt, roff, group=g))
    for j in range(size):
        assert np.all(out[roff[j]:roff[j] + rcnt[j]] == j * 100 + rank), "a2av"

    # barrier + test() path
    d.barrier("global")
    a = np.full(1000, float(rank), dtype=np.float32)
    out = np.zeros(1000, dtype=np.float32)
    req = d.all_reduce(a, out, 1000, op="sum", group=g)
    import time
    done, _ = mx.test(req)
    t0 = time.time()
    while not done:
        assert time.time() - t0 < 60, "test() never completed"
        done, _ = mx.test(req)
    assert np.all(out == size * (size - 1) / 2.0 + 0 * out), "test-completed allreduce"

    mx.finalize()


def hybrid_grid():
    """Distribution grid math on a data x model grid (world must be 4):
    checks group indices/counts and per-group collectives."""
    mx, rank, size = _init()
    assert size == 4
    d = mx.Distribution(2, 2)
    # lId = rank % 4; iM = lId / 2 (data idx), iF = lId % 2 (model idx)
    assert d.process_count("data") == 2
    assert d.process_count("model") == 2
    assert d.process_idx("data") == rank // 2
    assert d.process_idx("model") == rank % 2

    # model-group allreduce: partners are {0,1} and {2,3}
    a = np.full(8, float(rank), dtype=np.float32)
    out = np.zeros(8, dtype=np.float32)
    mx.wait(d.all_reduce(a, out, 8, op="sum", group="model"))
    want = {0: 1.0, 1: 1.0, 2: 5.0, 3: 5.0}[rank]
    assert np.all(out == want), f"model group allreduce got {out[0]} want {want}"

    # data-group allreduce: partners are {0,2} and {1,3}
    mx.wait(d.all_reduce(a, out, 8, op="sum", group="data"))
    want = {0: 2.0, 1: 4.0, 2: 2.0, 3: 4.0}[rank]
    assert np.all(out == want), f"data group allreduce got {out[0]} want {want}"

    # colors variant: same grouping built by hand
    d2 = mx.Distribution(colors=(rank % 2, rank // 2))
    assert d2.process_count("data") == 2
    assert d2.process_count("model") == 2
    mx.wait(d2.all_reduce(a, out, 8, op="sum", group="model"))
    want = {0: 1.0, 1: 1.0, 2: 5.0, 3: 5.0}[rank]
    assert np.all(out == want), "colors model group"
    mx.finalize()


def srlist_ring():
    """SendRecvList ring: send my 8-element block to next, receive from
    prev (the ring-attention neighbor-exchange primitive)."""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    n = 8
    a = np.full(n, float(rank * 11), dtype=np.float32) + np.arange(n, dtype=np.float32)
    out = np.zeros(n, dtype=np.float32)
    if size == 1:
        mx.finalize()
        return
    pairs = [((rank + 1) % size, 0, n, 0, 0),
             ((rank - 1 + size) % size, 0, 0, 0, n)]
    mx.wait(d.send_recv_list(a, out, pairs, group="data"))
    prev = (rank - 1 + size) % size
    assert np.all(out == prev * 11 + np.arange(n)), f"srlist got {out[:3]}"
    mx.finalize()


def priority_and_inline():
    """MLSL_PROGRESS=inline and MLSL_MSG_PRIORITY paths still compute
    correct results under concurrent requests."""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    reqs = []
    bufs = []
    for k in range(8):
        n = 1000 * (k + 1)
        a = np.full(n, float(rank + k), dtype=np.float32)
        out = np.zeros(n, dtype=np.float32)
        reqs.append(d.all_reduce(a, out, n, op="sum", group="data"))
        bufs.append((a, out, k))
    for req, (a, out, k) in zip(reqs, bufs):
        mx.wait(req)
        want = size * k + size * (size - 1) / 2.0
        assert np.all(out == want), f"concurrent req {k}"
    mx.finalize()


WORKERS = {
    "plumbing_allreduce": plumbing_allreduce,
    "collectives_sweep": collectives_sweep,
    "hybrid_grid": hybrid_grid,
    "srlist_ring": srlist_ring,
    "priority_and_inline": priority_and_inline,
}




def quantized_allreduce():
    """Quantized gradient allreduce through the planner (compression=int8):
    result within per-block quantization error of the exact sum, and error
    feedback keeps the long-run average unbiased."""
    mx, rank, size = _init()
    dp = size
    s = mx.Session()
    s.set_global_minibatch_size(4 * dp)
    d = mx.Distribution(dp, 1)
    info = s.create_op_reg_info("cc")
    info.add_input(4, 2, "f32")
    info.add_output(4, 2, "f32")
    info.add_parameter_set(1000, 17, "f32", compression="int8")
    op = s.operation(s.add_operation(info, d))
    s.commit()
    ps = op.parameter_set(0)
    n = ps.local_kernel_count * ps.kernel_size

    rng = np.random.RandomState(123 + rank)
    acc_exact = np.zeros(n, dtype=np.float64)
    acc_quant = np.zeros(n, dtype=np.float64)
    for it in range(20):
        g_all = [np.random.RandomState(1000 + it * size + r).randn(n).astype(np.float32)
                 for r in range(size)]
        g = g_all[rank].copy()
        exact = np.sum(g_all, axis=0)
        ps.start_gradient_comm(g)
        ps.wait_gradient_comm()
        if size == 1:
            got = g
        else:
            got = g
        step = np.abs(exact).max() / 127.0
        err = np.abs(got - exact).max()
        # ring quantized allreduce: <= ~2 requant steps per hop
        assert err < max(4 * size * step, 1e-3), f"quant allreduce err {err} step {step}"
        acc_exact += exact
        acc_quant += got
    drift = np.abs(acc_exact - acc_quant).max()
    scale = np.abs(acc_exact).max() / 127.0
    assert drift < 20 * max(scale, 0.1), f"quant drift {drift}"
    del rng
    mx.finalize()


WORKERS["quantized_allreduce"] = quantized_allreduce


def py_sample():
    import runpy
    import sys as _sys
    _sys.argv = ["mlsl_sample.py"]
    runpy.run_path("samples/mlsl_sample.py", run_name="__main__")


WORKERS["py_sample"] = py_sample


def py_rma_sample():
    import runpy
    import sys as _sys
    _sys.argv = ["rma_sample.py"]
    runpy.run_path("samples/rma_sample.py", run_name="__main__")


WORKERS["py_rma_sample"] = py_rma_sample




def compat_shim():
    """Reference-style Python surface (mlsl_amd.compat) end to end."""
    from mlsl_amd.compat import MLSL, DataType, GroupType, ReductionType
    m = MLSL()
    m.Init()
    rank, size = m.GetProcessIdx(), m.GetProcessCount()
    dist = m.CreateDistribution(size, 1)
    buf = np.full(128, float(rank), dtype=np.float32)
    req = dist.AllReduce(buf, buf, 128, DataType.FLOAT, ReductionType.SUM,
                         GroupType.DATA)
    m.Wait(req)
    assert np.all(buf == (size - 1) * size / 2.0)
    m.DeleteDistribution(dist)
    m.Finalize()


WORKERS["compat_shim"] = compat_shim


def configure_tenants():
    """Environment.Configure("color=N"): 4 boot ranks split into two
    2-rank tenant worlds; collectives stay inside each tenant."""
    import mlsl_amd as mx
    boot_rank = int(os.environ["RANK"])
    mx.init()
    assert mx.world_size() == 4
    color = boot_rank // 2
    mx.configure(f"color={color}")
    assert mx.world_size() == 2, mx.world_size()
    assert mx.rank() == boot_rank % 2
    d = mx.Distribution(2, 1)
    a = np.full(16, float(boot_rank), dtype=np.float32)
    out = np.zeros(16, dtype=np.float32)
    mx.wait(d.all_reduce(a, out, 16, op="sum", group="data"))
    want = {0: 1.0, 1: 1.0, 2: 5.0, 3: 5.0}[boot_rank]
    assert np.all(out == want), f"tenant allreduce got {out[0]} want {want}"
    mx.finalize()


WORKERS["configure_tenants"] = configure_tenants


def rma_window():
    """One-sided RMA window (fence-epoch semantics): puts land in the
    target window, gets observe same-epoch puts, empty fences are legal."""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    n = 64  # floats per rank-slot
    win = d.win_allocate(size * n * 4, group="data")

    # Epoch 1: every rank puts `rank+1` into ITS slot of every window
    # (incl. its own). After the fence, slot s of every window holds s+1.
    src = np.full(n, float(rank + 1), dtype=np.float32)
    for t in range(size):
        win.put(src, n * 4, t, rank * n * 4)
    src[:] = -1.0  # staged at put time: mutation after put must not matter
    win.fence()
    local = np.zeros(size * n, dtype=np.float32)
    win.read(local, size * n * 4)
    for s in range(size):
        want = float(s + 1)
        got = local[s * n:(s + 1) * n]
        assert np.all(got == want), f"rank {rank} slot {s}: {got[:3]} != {want}"

    # Epoch 2: put a fresh value into the right neighbor's OWN slot and get
    # the same range from it in the SAME epoch — the get must observe the
    # epoch's put (puts-then-gets ordering).
    right = (rank + 1) % size
    left = (rank - 1) % size
    fresh = np.full(n, 100.0 + rank, dtype=np.float32)
    win.put(fresh, n * 4, right, right * n * 4)
    seen = np.zeros(n, dtype=np.float32)
    win.get(seen, n * 4, right, right * n * 4)
    win.fence()
    assert np.all(seen == 100.0 + rank), f"get-after-put saw {seen[:3]}"
    mine = np.zeros(n, dtype=np.float32)
    win.read(mine, n * 4, off=rank * n * 4)
    assert np.all(mine == 100.0 + left), f"own slot {mine[:3]} != {100.0 + left}"

    # Epoch 3: empty fence (collective no-op) is legal.
    win.fence()
    win.close()
    d.barrier("global")
    mx.finalize()


WORKERS["rma_window"] = rma_window


def fork_safety():
    """Finalize in a forked child is a no-op (reference src/mlsl.cpp:720-724):
    the child must not tear down the parent's sockets/progress thread."""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    pid = os.fork()
    if pid == 0:
        # Child: finalize must not close anything the parent owns.
        try:
            mx.finalize()
            os._exit(0)
        except BaseException:
            os._exit(1)
    _, status = os.waitpid(pid, 0)
    assert os.waitstatus_to_exitcode(status) == 0, "child finalize failed"
    # Parent: the transport must still be fully functional.
    buf = np.full(32, float(rank), dtype=np.float32)
    mx.wait(d.all_reduce(buf, buf, 32, op="sum", group="data"))
    want = (size - 1) * size / 2.0
    assert np.all(buf == want), f"post-fork allreduce got {buf[0]} want {want}"
    d.barrier("global")
    mx.finalize()


WORKERS["fork_safety"] = fork_safety


def fault_peer_death():
    """Failure detection (SURVEY.md 5.3 — absent in the reference): when a
    peer dies mid-collective the survivor gets a failed request (transport
    EOF or MLSL_TIMEOUT watchdog), never an infinite hang."""
    import mlsl_amd as mx
    from mlsl_amd import MlslError
    rank = int(os.environ["RANK"])
    mx.init()
    d = mx.Distribution(mx.world_size(), 1)
    if rank == 1:
        os._exit(0)  # die without participating
    a = np.ones(200000, dtype=np.float32)
    out = np.zeros_like(a)
    try:
        mx.wait(d.all_reduce(a, out, a.size, op="sum", group="data"))
    except MlslError as e:
        print(f"OK got expected failure: {e}", flush=True)
        os._exit(0)
    raise AssertionError("allreduce with dead peer did not fail")


WORKERS["fault_peer_death"] = fault_peer_death


def ddp_wrapper():
    """DistributedData (torch hook-driven bucketed allreduce) matches
    manual gradient averaging at world 2 on CPU tensors."""
    import torch
    import mlsl_amd as mx
    from mlsl_amd.parallel import DistributedData

    torch.manual_seed(42)  # same init on every rank pre-broadcast
    mx.init()
    rank, size = mx.rank(), mx.world_size()

    model = torch.nn.Sequential(
        torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 8))
    ref = torch.nn.Sequential(
        torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 8))
    ref.load_state_dict(model.state_dict())

    dd = DistributedData(model, bucket_mb=1)

    for it in range(3):
        xs = [torch.randn(16, 32, generator=torch.Generator().manual_seed(100 + it * size + r))
              for r in range(size)]
        # distributed: each rank its own shard
        model.zero_grad()
        model(xs[rank]).sum().backward()
        dd.finish_gradients()
        # reference: average of per-shard grads on the full data
        ref.zero_grad()
        for x in xs:
            ref(x).sum().backward()
        for p, q in zip(model.parameters(), ref.parameters()):
            want = q.grad / size
            assert torch.allclose(p.grad, want, atol=1e-5), f"it={it} grad mismatch"
    mx.finalize()


WORKERS["ddp_wrapper"] = ddp_wrapper


def seqpar_reshard():
    """Ulysses-style sequence<->head re-shard round trip (AlltoAll over the
    model group): seq_to_head then head_to_seq reproduces the input, and the
    head-sharded intermediate holds the right global slices."""
    import mlsl_amd as mx
    from mlsl_amd.parallel import seqpar
    mx.init()
    rank, size = mx.rank(), mx.world_size()
    d = mx.Distribution(1, size)
    B, S, H = 2, 4 * size, 6 * size
    Sl, Hl = S // size, H // size
    # global tensor value(b, s, h) = b*10000 + s*100 + h
    full = (np.arange(B)[:, None, None] * 10000 +
            np.arange(S)[None, :, None] * 100 +
            np.arange(H)[None, None, :]).astype(np.float32)
    mine = full[:, rank * Sl:(rank + 1) * Sl, :].copy()

    headed = seqpar.seq_to_head(d, mine)
    want = full[:, :, rank * Hl:(rank + 1) * Hl]
    assert headed.shape == (B, S, Hl), headed.shape
    assert np.array_equal(headed, want), "seq->head mismatch"

    back = seqpar.head_to_seq(d, headed)
    assert np.array_equal(back, mine), "head->seq roundtrip mismatch"

    # ring exchange: my block lands on the next rank
    blk = np.full((3, 5), float(rank), dtype=np.float32)
    got = seqpar.ring_exchange(d, blk)
    assert np.all(got == (rank - 1) % size), "ring exchange mismatch"
    mx.finalize()


WORKERS["seqpar_reshard"] = seqpar_reshard


def pointer_checker():
    """MLSL_CHECK_POINTERS=1: collectives accept registered Environment
    allocations and reject foreign buffers (reference pointer_checker)."""
    import ctypes
    import mlsl_amd as mx
    from mlsl_amd import MlslError
    mx.init()
    size = mx.world_size()
    d = mx.Distribution(size, 1)
    n = 1024
    ptr = mx.alloc(n * 4)
    buf = (ctypes.c_float * n).from_address(ptr)
    for i in range(n):
        buf[i] = float(mx.rank())
    mx.wait(d.all_reduce(ptr, ptr, n, dtype="f32", op="sum", group="data"))
    assert buf[0] == (size - 1) * size / 2.0
    # a numpy array is NOT a registered allocation -> must be rejected
    a = np.ones(16, dtype=np.float32)
    try:
        mx.wait(d.all_reduce(a, a, 16, op="sum", group="data"))
    except MlslError:
        mx.free(ptr)
        mx.finalize()
        return
    raise AssertionError("unregistered buffer was not rejected")


WORKERS["pointer_checker"] = pointer_checker


def edge_cases():
    """Zero-count and single-element collectives terminate correctly."""
    import mlsl_amd as mx
    mx.init()
    size = mx.world_size()
    d = mx.Distribution(size, 1)
    a = np.zeros(1, dtype=np.float32)
    out = np.zeros(1, dtype=np.float32)
    mx.wait(d.all_reduce(a, out, 0, op="sum", group="data"))
    a[0] = mx.rank()
    mx.wait(d.all_reduce(a, out, 1, op="sum", group="data"))
    assert out[0] == (size - 1) * size / 2.0
    mx.finalize()


WORKERS["edge_cases"] = edge_cases


def stress_random():
    """Randomized cross-group stress: a seeded, rank-consistent sequence of
    mixed collectives over world/data/model groups with varying sizes,
    each verified against a numpy oracle. Catches tag/flow-ordering bugs."""
    import mlsl_amd as mx
    mx.init()
    rank, size = mx.rank(), mx.world_size()
    mp = 2 if size % 2 == 0 and size > 1 else 1
    d = mx.Distribution(size // mp, mp)
    rng = np.random.RandomState(7)   # same sequence on every rank

    def group_info(gname):
        return d.process_idx(gname), d.process_count(gname)

    iters = int(os.environ.get("STRESS_ITERS", "120"))
    for it in range(iters):
        op = rng.choice(["allreduce", "bcast", "reduce_scatter", "allgather",
                         "alltoall", "reduce"])
        gname = rng.choice(["data", "model", "global"])
        n = int(rng.randint(1, 5000))
        gidx, gsz = group_info(gname)
        base = np.arange(n, dtype=np.float32) + it
        if op == "allreduce":
            a = base + gidx
            out = np.zeros_like(a)
            mx.wait(d.all_reduce(a, out, n, op="sum", group=gname))
            assert np.allclose(out, gsz * base + gsz * (gsz - 1) / 2.0), (it, op)
        elif op == "bcast":
            root = int(rng.randint(0, gsz))
            b = base.copy() if gidx == root else np.zeros_like(base)
            mx.wait(d.bcast(b, n, root=root, group=gname))
            assert np.array_equal(b, base), (it, op)
        elif op == "reduce":
            root = int(rng.randint(0, gsz))
            a = base + gidx
            out = np.zeros_like(a)
            mx.wait(d.reduce(a, out, n, op="sum", root=root, group=gname))
            if gidx == root:
                assert np.allclose(out, gsz * base + gsz * (gsz - 1) / 2.0), (it, op)
        elif op == "reduce_scatter":
            a = np.tile(base, gsz) + gidx
            out = np.zeros(n, dtype=np.float32)
            mx.wait(d.reduce_scatter(a, out, n, op="sum", group=gname))
            assert np.allclose(out, gsz * base + gsz * (gsz - 1) / 2.0), (it, op)
        elif op == "allgather":
            a = base + 1000 * gidx
            out = np.zeros(n * gsz, dtype=np.float32)
            mx.wait(d.all_gather(a, n, out, group=gname))
            for j in range(gsz):
                assert np.array_equal(out[j * n:(j + 1) * n], base + 1000 * j), (it, op)
        elif op == "alltoall":
            a = np.zeros(n * gsz, dtype=np.float32)
            for j in range(gsz):
                a[j * n:(j + 1) * n] = base + gidx * 100 + j
            out = np.zeros(n * gsz, dtype=np.float32)
            mx.wait(d.all_to_all(a, n, out, group=gname))
            for j in range(gsz):
                assert np.array_equal(out[j * n:(j + 1) * n], base + j * 100 + gidx), (it, op)
    mx.finalize()


WORKERS["stress_random"] = stress_random




def c_env_compat():
    """Reference-named C compatibility surface drives a full lifecycle:
    environment init -> alloc -> distribution -> session/planner commit ->
    gradient comm -> finalize, all through mlsl_environment_* /
    mlsl_operation_reg_info_* / mlsl_session_*_operation_reg_info names
    (reference include/mlsl.h signatures)."""
    import ctypes
    from ctypes import byref, c_void_p, c_size_t, c_int, c_char_p
    from mlsl_amd._lib import lib
    L = lib()

    def ck(rc):
        assert rc == 0, ctypes.string_at(L.mlsl_last_error()).decode()

    env = c_void_p()
    ck(L.mlsl_environment_get_env(byref(env)))
    ck(L.mlsl_environment_init(env, None, None))
    try:
        ini = c_int(0)
        ck(L.mlsl_environment_is_initialized(env, byref(ini)))
        assert ini.value == 1
        n = c_size_t(0)
        ck(L.mlsl_environment_get_process_count(env, byref(n)))
        size = n.value
        ck(L.mlsl_environment_get_process_idx(env, byref(n)))
        rank = n.value

        ptr = c_void_p()
        ck(L.mlsl_environment_alloc(env, 4096, 64, byref(ptr)))

        dist = c_void_p()
        ck(L.mlsl_environment_create_distribution(env, size, 1, byref(dist)))
        cnt = c_size_t(0)
        ck(L.mlsl_distribution_get_process_count(dist, 0, byref(cnt)))
        assert cnt.value == size

        # generic collective through the env-named wait
        buf = (ctypes.c_float * 128)(*([float(rank)] * 128))
        req = c_void_p()
        ck(L.mlsl_distribution_all_reduce(dist, buf, buf, 128, 0, 0, 0, byref(req)))
        ck(L.mlsl_environment_wait(env, req))
        want = (size - 1) * size / 2.0
        assert abs(buf[0] - want) < 1e-6 and abs(buf[127] - want) < 1e-6

        sess = c_void_p()
        ck(L.mlsl_environment_create_session(env, 0, byref(sess)))
        ck(L.mlsl_session_set_global_minibatch_size(sess, 8 * size))
        ri = c_void_p()
        ck(L.mlsl_session_create_operation_reg_info(sess, 0, byref(ri)))
        ck(L.mlsl_operation_reg_info_set_name(ri, b"fc1"))
        ck(L.mlsl_operation_reg_info_add_input(ri, 16, 4, 0))
        ck(L.mlsl_operation_reg_info_add_output(ri, 16, 4, 0))
        ck(L.mlsl_operation_reg_info_add_parameter_set(ri, 16, 9, 0, 0))
        idx = c_size_t(0)
        ck(L.mlsl_session_add_operation_with_distribution(sess, ri, dist, byref(idx)))
        ck(L.mlsl_session_delete_operation_reg_info(sess, ri))
        ck(L.mlsl_session_commit(sess))

        op = c_void_p()
        ck(L.mlsl_session_get_operation(sess, idx.value, byref(op)))
        hp = c_int(0)
        ck(L.mlsl_operation_has_parameter_sets(op, byref(hp)))
        assert hp.value == 1
        ot = c_int(-1)
        ck(L.mlsl_operation_get_op_type(op, byref(ot)))
        assert ot.value == 0
        ph = c_int(-1)
        ck(L.mlsl_session_get_phase_type(sess, byref(ph)))
        assert ph.value == 0

        ps = c_void_p()
        ck(L.mlsl_operation_get_parameter_set(op, 0, byref(ps)))
        import numpy as np
        g = np.full(16 * 9, float(rank + 1), dtype=np.float32)
        ck(L.mlsl_parameter_set_start_gradient_comm(
            ps, g.ctypes.data_as(c_void_p)))
        res = c_void_p()
        ck(L.mlsl_parameter_set_wait_gradient_comm(ps, byref(res)))
        got = np.ctypeslib.as_array(
            ctypes.cast(res, ctypes.POINTER(ctypes.c_float)), shape=(16 * 9,))
        assert np.allclose(got, size * (size + 1) / 2.0), got[:4]

        # activation comm buf: library-owned lazy buffer
        act = c_void_p()
        ck(L.mlsl_operation_get_input(op, 0, byref(act)))
        cb = c_void_p()
        ck(L.mlsl_activation_get_comm_buf(act, byref(cb)))
        cbs = c_size_t(0)
        ck(L.mlsl_activation_get_comm_buf_size(act, byref(cbs)))
        assert (cb.value is not None) == (cbs.value > 0)

        # quant params struct roundtrip
        class QP(ctypes.Structure):
            _fields_ = [("lib_path", c_char_p), ("quant", c_char_p),
                        ("dequant", c_char_p), ("reduce_sum", c_char_p),
                        ("block_size", c_size_t), ("elem_in_block", c_size_t)]
        qp = QP(None, None, None, None, 0, 128)
        ck(L.mlsl_environment_set_quantization_params(env, byref(qp)))
        qp2 = QP()
        ck(L.mlsl_environment_get_quantization_params(env, byref(qp2)))
        assert qp2.elem_in_block == 128 and qp2.block_size == 128 + 8
        qp.elem_in_block = 256
        ck(L.mlsl_environment_set_quantization_params(env, byref(qp)))

        st = c_void_p()
        ck(L.mlsl_session_get_stats(sess, byref(st)))
        started = c_int(-1)
        ck(L.mlsl_statistics_is_started(st, byref(started)))
        assert started.value in (0, 1)

        ck(L.mlsl_environment_delete_session(env, sess))
        ck(L.mlsl_environment_delete_distribution(env, dist))
        ck(L.mlsl_environment_free(env, ptr))
    finally:
        ck(L.mlsl_environment_finalize(env))


WORKERS["c_env_compat"] = c_env_compat




def train_ddp_sample():
    """samples/train_ddp.py as a worker: full DDP training loop, asserts
    cross-rank parameter identity (exit code is the verdict)."""
    import runpy
    import os
    os.environ["STEPS"] = "5"
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    runpy.run_path(os.path.join(repo, "samples", "train_ddp.py"),
                   run_name="__main__")


WORKERS["train_ddp_sample"] = train_ddp_sample




def quant_plugin():
    """dlopen'd compression plugin (reference quant/quant.c ABI): load the
    sample int8 plugin via the reference-named C API, run a quantized
    allreduce over the host transport, verify against the exact sum within
    the per-block quantization step."""
    import ctypes
    import subprocess
    from ctypes import byref, c_void_p, c_size_t, c_char_p
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    so = os.path.join(repo, "build", "libquant_plugin.so")
    if not os.path.exists(so):
        subprocess.run(["make", "quantplugin"], cwd=repo, check=True,
                       capture_output=True, timeout=300)
    from mlsl_amd._lib import lib
    L = lib()

    class QP(ctypes.Structure):
        _fields_ = [("lib_path", c_char_p), ("quant", c_char_p),
                    ("dequant", c_char_p), ("reduce_sum", c_char_p),
                    ("block_size", c_size_t), ("elem_in_block", c_size_t)]

    import mlsl_amd as mx
    mx.init()
    rank, size = mx.rank(), mx.world_size()
    qp = QP(so.encode(), None, None, None, 264, 256)
    env = c_void_p()
    assert L.mlsl_environment_get_env(byref(env)) == 0
    assert L.mlsl_environment_set_quantization_params(env, byref(qp)) == 0, \
        ctypes.string_at(L.mlsl_last_error()).decode()

    d = mx.Distribution(size, 1)
    n = 70000   # includes a partial tail block
    rng = np.random.RandomState(3 + rank)
    g = (rng.randn(n) * 2).astype(np.float32)
    all_g = [(np.random.RandomState(3 + r).randn(n) * 2).astype(np.float32)
             for r in range(size)]
    want = np.sum(all_g, axis=0)
    out = np.zeros_like(g)
    preq = mx.PersistentRequest(d, "all_reduce", n, dtype="f32", op="sum",
                                group="data", quantized=True)
    preq.start(g, out)
    preq.wait()
    preq.destroy()
    block = 256
    nb = (n + block - 1) // block
    pad = nb * block - n
    wb = np.pad(want, (0, pad)).reshape(nb, block)
    step = np.abs(wb).max(axis=1) / 127.0
    bound = np.repeat(step, block)[:n] * (2.0 * max(size, 2)) + 1e-5
    err = np.abs(out - want)
    assert (err <= bound).mean() > 0.999, (err.max(), bound.max())
    # restore defaults for any later use in this process
    qp2 = QP(None, None, None, None, 0, 256)
    L.mlsl_environment_set_quantization_params(env, byref(qp2))
    mx.finalize()


WORKERS["quant_plugin"] = quant_plugin




def zero1_sharded_opt():
    """ShardedOptimizer (ZeRO-1 / distributedUpdate analog): RS -> sharded
    AdamW step -> AG must match a full-size AdamW on the averaged grads."""
    import torch
    from mlsl_amd.parallel.zero1 import ShardedOptimizer
    import mlsl_amd as mx
    mx.init()
    rank, size = mx.rank(), mx.world_size()
    torch.manual_seed(99)
    model = torch.nn.Sequential(torch.nn.Linear(37, 53), torch.nn.Linear(53, 11))
    ref = torch.nn.Sequential(torch.nn.Linear(37, 53), torch.nn.Linear(53, 11))
    ref.load_state_dict(model.state_dict())

    d = mx.Distribution(size, 1)
    sopt = ShardedOptimizer(model.parameters(), torch.optim.AdamW, d,
                            reduce="rs", lr=1e-2)
    ref_opt = torch.optim.AdamW(ref.parameters(), lr=1e-2)

    for it in range(4):
        grads = []
        for i, p in enumerate(model.parameters()):
            g = torch.full_like(p, 0.01 * (it + 1) * (i + 1) * (rank + 1))
            p.grad = g
            # mean over ranks: 0.01*(it+1)*(i+1) * (1+..+size)/size
            grads.append(torch.full_like(p, 0.01 * (it + 1) * (i + 1) *
                                         (size + 1) / 2.0))
        for p, g in zip(ref.parameters(), grads):
            p.grad = g
        sopt.step()
        ref_opt.step()
        sopt.zero_grad()
        ref_opt.zero_grad()

    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, q.data, atol=1e-6), \
            (p.data - q.data).abs().max()
    # shard-local optimizer state only
    n_state = sum(v.numel() for st in sopt.opt.state.values()
                  for v in st.values() if torch.is_tensor(v))
    total = sum(p.numel() for p in model.parameters())
    assert n_state <= 2 * ((total + size - 1) // size) + 4, n_state
    mx.finalize()


WORKERS["zero1_sharded_opt"] = zero1_sharded_opt




def inplace_collectives():
    """NCCL-convention in-place buffers: allgather with sbuf = rbuf +
    rank*count, reduce_scatter with rbuf = sbuf + rank*count, bcast
    in-place. (Reference hard-part: in-place rules per op,
    src/comm_ep.cpp:623-736.)"""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    cnt = 1000

    # all_gather in-place: contribute from own slot of the full buffer
    full = np.zeros(size * cnt, dtype=np.float32)
    full[rank * cnt:(rank + 1) * cnt] = np.arange(cnt) + 100.0 * rank
    sb = full[rank * cnt:(rank + 1) * cnt]
    mx.wait(d.all_gather(sb, cnt, full, group="data"))
    for r in range(size):
        assert np.allclose(full[r * cnt:(r + 1) * cnt],
                           np.arange(cnt) + 100.0 * r), "allgather in-place"

    # reduce_scatter in-place: result lands in own slot of the send buffer
    send = np.tile(np.arange(size * cnt, dtype=np.float32), 1) + rank
    rb = send[rank * cnt:(rank + 1) * cnt]
    mx.wait(d.reduce_scatter(send, rb, cnt, op="sum", group="data"))
    want = size * (np.arange(cnt) + rank * cnt + 0.0) + size * (size - 1) / 2.0
    # careful: expected = sum over ranks of (idx_global + r) at slot rank
    idxg = np.arange(rank * cnt, (rank + 1) * cnt, dtype=np.float32)
    want = size * idxg + size * (size - 1) / 2.0
    assert np.allclose(rb, want), ("reduce_scatter in-place", rb[:4], want[:4])

    # bcast is naturally in-place
    b = (np.arange(cnt, dtype=np.float32) if rank == 0
         else np.zeros(cnt, dtype=np.float32))
    mx.wait(d.bcast(b, cnt, root=0, group="data"))
    assert np.allclose(b, np.arange(cnt)), "bcast in-place"
    mx.finalize()


WORKERS["inplace_collectives"] = inplace_collectives




def torch_backend():
    """torch.distributed backend "mlsl": init_process_group + the standard
    collective calls run over our engine with analytic expectations."""
    import torch
    import torch.distributed as td
    import mlsl_amd.torch_backend  # noqa: F401 (registers "mlsl")
    rank = int(os.environ["RANK"])
    size = int(os.environ["WORLD_SIZE"])
    # torch's env:// store needs its own port; our bootstrap uses MLSL_PORT
    os.environ.setdefault("MASTER_PORT",
                          str(int(os.environ.get("MLSL_PORT", "29600")) + 10))
    td.init_process_group(backend="mlsl", rank=rank, world_size=size)
    assert td.get_rank() == rank and td.get_world_size() == size

    t = torch.full((1000,), float(rank))
    td.all_reduce(t)
    assert torch.all(t == size * (size - 1) / 2.0), "all_reduce"

    t = torch.arange(64, dtype=torch.float32) if rank == 1 % size \
        else torch.zeros(64)
    td.broadcast(t, src=1 % size)
    assert torch.equal(t, torch.arange(64, dtype=torch.float32)), "broadcast"

    inp = torch.full((10,), float(rank))
    outs = [torch.zeros(10) for _ in range(size)]
    td.all_gather(outs, inp)
    for r in range(size):
        assert torch.all(outs[r] == r), "all_gather"

    big = torch.zeros(10 * size)
    td.all_gather_into_tensor(big, inp)
    for r in range(size):
        assert torch.all(big[r * 10:(r + 1) * 10] == r), "all_gather_into_tensor"

    src = torch.arange(size * 5, dtype=torch.float32) + rank
    out = torch.zeros(5)
    td.reduce_scatter_tensor(out, src)
    want = size * (torch.arange(5, dtype=torch.float32) + rank * 5) \
        + size * (size - 1) / 2.0
    # careful: slot for this rank
    want = size * (torch.arange(rank * 5, (rank + 1) * 5, dtype=torch.float32)) \
        + size * (size - 1) / 2.0
    assert torch.allclose(out, want), ("reduce_scatter_tensor", out, want)

    a2a_in = torch.arange(size * 3, dtype=torch.float32) + 100.0 * rank
    a2a_out = torch.zeros(size * 3)
    td.all_to_all_single(a2a_out, a2a_in)
    for r in range(size):
        want = torch.arange(rank * 3, rank * 3 + 3, dtype=torch.float32) + 100.0 * r
        assert torch.equal(a2a_out[r * 3:(r + 1) * 3], want), "all_to_all_single"

    td.barrier()

    # reduce / gather / scatter
    t = torch.full((50,), float(rank + 1))
    td.reduce(t, dst=0)
    if rank == 0:
        assert torch.all(t == size * (size + 1) / 2.0), "reduce"
    gin = torch.full((6,), float(rank))
    gouts = [torch.zeros(6) for _ in range(size)] if rank == 1 % size else None
    td.gather(gin, gouts, dst=1 % size)
    if rank == 1 % size:
        for r in range(size):
            assert torch.all(gouts[r] == r), "gather"
    sout = torch.zeros(4)
    sins = [torch.full((4,), float(r * 2)) for r in range(size)] if rank == 0 else None
    td.scatter(sout, sins, src=0)
    assert torch.all(sout == rank * 2), "scatter"

    # p2p ring: send to next, recv from prev (pipeline-parallel pattern)
    nxt, prv = (rank + 1) % size, (rank - 1) % size
    ps = torch.full((32,), float(rank))
    pr = torch.zeros(32)
    if rank % 2 == 0:
        td.send(ps, nxt)
        td.recv(pr, prv)
    else:
        td.recv(pr, prv)
        td.send(ps, nxt)
    assert torch.all(pr == prv), "p2p ring"

    # torch-native DDP over the mlsl backend (the C++ reducer drives
    # broadcast + bucketed allreduce through our ProcessGroup)
    torch.manual_seed(17)
    m = torch.nn.Linear(16, 4)
    ddp = torch.nn.parallel.DistributedDataParallel(m)
    x = torch.randn(8, 16, generator=torch.Generator().manual_seed(23 + rank))
    ddp(x).sum().backward()
    g0 = torch.cat([p.grad.flatten() for p in ddp.parameters()])
    lo, hi = g0.clone(), g0.clone()
    td.all_reduce(lo, op=td.ReduceOp.MIN)
    td.all_reduce(hi, op=td.ReduceOp.MAX)
    assert torch.allclose(lo, hi, atol=1e-6), "DDP grads differ across ranks"

    td.destroy_process_group()


WORKERS["torch_backend"] = torch_backend




def p2p_asymmetric():
    """Asymmetric point-to-point (rank0 sends 3 messages to rank1 while the
    rest of the group is idle) followed by a group collective: per-edge
    sequence tags must keep both the p2p matching and the later collective
    tags aligned (a group-flow-based tag scheme breaks exactly here)."""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    n = 512
    if rank == 0:
        for i in range(3):
            buf = np.full(n, 10.0 * (i + 1), dtype=np.float32)
            mx.wait(d.send_recv_list(buf, buf, [(1, 0, n, 0, 0)], group="data"))
    elif rank == 1:
        for i in range(3):
            out = np.zeros(n, dtype=np.float32)
            mx.wait(d.send_recv_list(out, out, [(0, 0, 0, 0, n)], group="data"))
            assert np.all(out == 10.0 * (i + 1)), (i, out[:3])
    # now a collective over the whole group: flows must still agree
    a = np.full(64, float(rank), dtype=np.float32)
    out = np.zeros_like(a)
    mx.wait(d.all_reduce(a, out, 64, op="sum", group="data"))
    assert np.all(out == size * (size - 1) / 2.0), "post-p2p allreduce"
    # interleave: p2p both directions with different counts again
    if rank == 0:
        got = np.zeros(n, dtype=np.float32)
        mx.wait(d.send_recv_list(got, got, [(1, 0, 0, 0, n)], group="data"))
        assert np.all(got == 77.0)
    elif rank == 1:
        msg = np.full(n, 77.0, dtype=np.float32)
        mx.wait(d.send_recv_list(msg, msg, [(0, 0, n, 0, 0)], group="data"))
    d.barrier("global")
    mx.finalize()


WORKERS["p2p_asymmetric"] = p2p_asymmetric




def p2p_stress():
    """Randomized asymmetric p2p + collective interleave: a seeded global
    schedule of directed messages (uneven per-rank counts, bursts that
    queue unexpected) with interspersed group collectives. Exercises the
    per-edge sequence tags under load."""
    mx, rank, size = _init()
    d = mx.Distribution(size, 1)
    rng = np.random.RandomState(13)   # same schedule on every rank
    iters = int(os.environ.get("P2P_STRESS_ITERS", "80"))
    for it in range(iters):
        kind = rng.choice(["msg", "burst", "coll"], p=[0.5, 0.2, 0.3])
        if kind in ("msg", "burst"):
            src = int(rng.randint(0, size))
            dst = int((src + 1 + rng.randint(0, size - 1)) % size)
            n = int(rng.randint(1, 3000))
            reps = int(rng.randint(2, 5)) if kind == "burst" else 1
            for k in range(reps):
                val = it * 1000.0 + k
                if rank == src:
                    buf = np.full(n, val, dtype=np.float32)
                    mx.wait(d.send_recv_list(buf, buf, [(dst, 0, n, 0, 0)],
                                             group="data"))
                elif rank == dst:
                    out = np.zeros(n, dtype=np.float32)
                    mx.wait(d.send_recv_list(out, out, [(src, 0, 0, 0, n)],
                                             group="data"))
                    assert np.all(out == val), (it, k, out[:3])
        else:
            n = int(rng.randint(1, 2000))
            a = np.arange(n, dtype=np.float32) + rank
            out = np.zeros_like(a)
            mx.wait(d.all_reduce(a, out, n, op="sum", group="data"))
            want = size * np.arange(n, dtype=np.float32) + size * (size - 1) / 2.0
            assert np.allclose(out, want), (it, "coll")
    d.barrier("global")
    mx.finalize()


WORKERS["p2p_stress"] = p2p_stress


def main():
    name = sys.argv[1]
    fn = WORKERS.get(name)
    if fn is None:
        # planner workers live in a separate module to keep this one lean
        from tests import workers_planner
        fn = workers_planner.WORKERS[name]
    fn()
    print(f"OK {name} rank={os.environ.get('RANK')}")


if __name__ == "__main__":
    main()
