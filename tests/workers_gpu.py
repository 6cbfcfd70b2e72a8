"""GPU multi-rank worker entry points (all ranks on ONE device).

RCCL refuses several ranks of one communicator on the same GPU, so these
worlds exercise the IPC window transport (csrc/comm/p2p_transport.cpp) —
the same schedule executor that runs over xGMI on a real 8-GPU node. The
reference bar is the 4-rank transport matrix of
/root/reference/tests/examples/mlsl_test/Makefile:59-107.

Run as: python -m tests.workers_gpu <worker_name>   (see test_gpu_multirank)
"""
import os
import sys


def _init():
    import torch
    import mlsl_amd as mx
    torch.cuda.set_device(0)
    mx.init()
    return mx, torch, mx.rank(), mx.world_size()


def _arange(torch, count, rank):
    return torch.arange(count, dtype=torch.float32, device="cuda") + rank


def gpu_collectives():
    """Every collective with analytic expected values on cuda tensors."""
    mx, torch, rank, size = _init()
    d = mx.Distribution(size, 1)
    g = "data"

    # allreduce sum, sizes spanning one-slot and multi-slot paths
    for count in (1, 257, 65536, 1 << 20):
        a = _arange(torch, count, rank)
        out = torch.empty_like(a)
        mx.wait(d.all_reduce(a, out, count, op="sum", group=g))
        torch.cuda.synchronize()
        want = size * torch.arange(count, dtype=torch.float32, device="cuda") \
            + size * (size - 1) / 2.0
        assert torch.allclose(out, want), ("allreduce", count, out[:4], want[:4])

    # in-place allreduce + max
    count = 4099
    a = _arange(torch, count, rank)
    mx.wait(d.all_reduce(a, a, count, op="sum", group=g))
    torch.cuda.synchronize()
    want = size * torch.arange(count, dtype=torch.float32, device="cuda") \
        + size * (size - 1) / 2.0
    assert torch.allclose(a, want), "in-place allreduce"
    a = _arange(torch, count, rank)
    mx.wait(d.all_reduce(a, a, count, op="max", group=g))
    torch.cuda.synchronize()
    wantm = torch.arange(count, dtype=torch.float32, device="cuda") + (size - 1)
    assert torch.allclose(a, wantm), "max allreduce"

    # bcast from root 0 and root size-1
    for root in (0, size - 1):
        count = 3333
        b = _arange(torch, count, rank) if rank == root \
            else torch.zeros(count, device="cuda")
        mx.wait(d.bcast(b, count, root=root, group=g))
        torch.cuda.synchronize()
        assert torch.allclose(b, _arange(torch, count, root)), ("bcast", root)

    # reduce to root
    root = 1 % size
    count = 2048
    a = _arange(torch, count, rank)
    out = torch.empty_like(a)
    mx.wait(d.reduce(a, out, count, op="sum", root=root, group=g))
    torch.cuda.synchronize()
    if rank == root:
        want = size * torch.arange(count, dtype=torch.float32, device="cuda") \
            + size * (size - 1) / 2.0
        assert torch.allclose(out, want), "reduce"

    # reduce_scatter: rank r gets sum over ranks of seg r
    per = 1536
    src = torch.cat([_arange(torch, per, rank) + i * 1000 for i in range(size)])
    out = torch.empty(per, device="cuda")
    mx.wait(d.reduce_scatter(src, out, per, op="sum", group=g))
    torch.cuda.synchronize()
    want = size * torch.arange(per, dtype=torch.float32, device="cuda") \
        + size * (size - 1) / 2.0 + size * rank * 1000
    assert torch.allclose(out, want), "reduce_scatter"

    # all_gather
    per = 777
    mine = _arange(torch, per, rank) * (rank + 1)
    flat = torch.empty(size * per, device="cuda")
    mx.wait(d.all_gather(mine, per, flat, group=g))
    torch.cuda.synchronize()
    for r in range(size):
        want = (_arange(torch, per, r)) * (r + 1)
        assert torch.allclose(flat[r * per:(r + 1) * per], want), ("allgather", r)

    # all_to_all: block j of rank i == i*100 + j
    per = 512
    src = torch.cat([torch.full((per,), float(rank * 100 + j), device="cuda")
                     for j in range(size)])
    dst = torch.empty_like(src)
    mx.wait(d.all_to_all(src, per, dst, group=g))
    torch.cuda.synchronize()
    for j in range(size):
        want = float(j * 100 + rank)
        assert torch.all(dst[j * per:(j + 1) * per] == want), ("alltoall", j)

    # gather / scatter
    root = 0
    per = 333
    mine = _arange(torch, per, rank) + rank * 7
    flat = torch.empty(size * per, device="cuda") if rank == root \
        else torch.empty(1, device="cuda")
    mx.wait(d.gather(mine, per, flat, root=root, group=g))
    torch.cuda.synchronize()
    if rank == root:
        for r in range(size):
            want = _arange(torch, per, r) + r * 7
            assert torch.allclose(flat[r * per:(r + 1) * per], want), ("gather", r)
    if rank == root:
        sflat = torch.cat([_arange(torch, per, r) * 2 for r in range(size)])
    else:
        sflat = torch.empty(1, device="cuda")
    rout = torch.empty(per, device="cuda")
    mx.wait(d.scatter(sflat, rout, per, root=root, group=g))
    torch.cuda.synchronize()
    assert torch.allclose(rout, _arange(torch, per, rank) * 2), "scatter"

    # dtype fallback paths on the device transport: f64 / i64 allreduce
    # ride the wait-kernel + generic-reduce consumers; bf16 the fused ones
    count = 2048
    a64 = torch.arange(count, dtype=torch.float64, device="cuda") + rank
    o64 = torch.empty_like(a64)
    mx.wait(d.all_reduce(a64, o64, count, op="sum", group=g))
    torch.cuda.synchronize()
    w64 = size * torch.arange(count, dtype=torch.float64, device="cuda") \
        + size * (size - 1) / 2.0
    assert torch.allclose(o64, w64), "f64 allreduce"
    ai = torch.arange(count, dtype=torch.int64, device="cuda") + rank
    oi = torch.empty_like(ai)
    mx.wait(d.all_reduce(ai, oi, count, op="sum", group=g))
    torch.cuda.synchronize()
    wi = size * torch.arange(count, dtype=torch.int64, device="cuda") \
        + size * (size - 1) // 2
    assert torch.equal(oi, wi), "i64 allreduce"
    ab = (torch.arange(count, dtype=torch.bfloat16, device="cuda") % 8) + rank
    ob = torch.empty_like(ab)
    mx.wait(d.all_reduce(ab, ob, count, op="sum", group=g))
    torch.cuda.synchronize()
    wb = size * ((torch.arange(count, dtype=torch.bfloat16, device="cuda") % 8)) \
        + size * (size - 1) / 2.0
    assert torch.allclose(ob.float(), wb.float()), "bf16 allreduce"

    # p2p ring over send_recv_list
    n = 1024
    dst_rank = (rank + 1) % size
    src_rank = (rank - 1 + size) % size
    sendv = torch.full((n,), float(rank), device="cuda")
    recvv = torch.empty(n, device="cuda")
    mx.wait(d.send_recv_list(sendv, recvv,
                             [(dst_rank, 0, n, 0, 0), (src_rank, 0, 0, 0, n)],
                             group=g))
    torch.cuda.synchronize()
    assert torch.all(recvv == float(src_rank)), "srlist ring"

    d.barrier(g)
    d.barrier("global")
    mx.finalize()


def gpu_allreduce_multislot():
    """Large message: segments span many transport slots (backpressure +
    interleave), both ring and RHD, in-place and out-of-place."""
    mx, torch, rank, size = _init()
    d = mx.Distribution(size, 1)
    count = 1 << 22  # 16 MiB fp32
    for _ in range(3):
        a = _arange(torch, count, rank)
        out = torch.empty_like(a)
        mx.wait(d.all_reduce(a, out, count, op="sum", group="data"))
        torch.cuda.synchronize()
        want = size * torch.arange(count, dtype=torch.float32, device="cuda") \
            + size * (size - 1) / 2.0
        assert torch.allclose(out, want), "multislot out-of-place"
        a = _arange(torch, count, rank)
        mx.wait(d.all_reduce(a, a, count, op="sum", group="data"))
        torch.cuda.synchronize()
        assert torch.allclose(a, want), "multislot in-place"
    d.barrier("global")
    mx.finalize()


def gpu_quant_allreduce():
    """int8-quantized allreduce over the compressed-domain ring (driver
    config 5 shape, small size): relative error bounded, error feedback on."""
    mx, torch, rank, size = _init()
    d = mx.Distribution(size, 1)
    count = 1 << 20
    torch.manual_seed(100 + rank)
    for dt, name in ((torch.float32, "f32"), (torch.bfloat16, "bf16")):
        a = torch.randn(count, dtype=dt, device="cuda")
        out = torch.empty_like(a)
        req = mx.PersistentRequest(d, "all_reduce", count, dtype=name,
                                   op="sum", group="data", quantized=True)
        req.start(a, out)
        req.wait()
        torch.cuda.synchronize()
        # exact sum via non-quantized path for comparison
        exact = torch.empty_like(a)
        mx.wait(d.all_reduce(a, exact, count, op="sum", dtype=name, group="data"))
        torch.cuda.synchronize()
        num = (out.float() - exact.float()).norm()
        den = exact.float().norm().clamp_min(1e-6)
        rel = (num / den).item()
        assert rel < 0.05, (name, rel)
        req.destroy()
    d.barrier("global")
    mx.finalize()


def gpu_hybrid_grid():
    """Distribution(dp, mp) sub-communicators on the p2p transport
    (driver config 4 shape at reduced scale)."""
    mx, torch, rank, size = _init()
    assert size % 2 == 0
    d = mx.Distribution(size // 2, 2)
    # model-group allreduce: partners are (rank//2)*2 and +1
    count = 8192
    a = _arange(torch, count, rank)
    out = torch.empty_like(a)
    mx.wait(d.all_reduce(a, out, count, op="sum", group="model"))
    torch.cuda.synchronize()
    base = (rank // 2) * 2
    want = 2 * torch.arange(count, dtype=torch.float32, device="cuda") \
        + base + base + 1
    assert torch.allclose(out, want), "model-group allreduce"
    # data-group allreduce
    a = _arange(torch, count, rank)
    mx.wait(d.all_reduce(a, a, count, op="sum", group="data"))
    torch.cuda.synchronize()
    dp = size // 2
    mates = [r for r in range(size) if r % 2 == rank % 2]
    want = dp * torch.arange(count, dtype=torch.float32, device="cuda") \
        + float(sum(mates))
    assert torch.allclose(a, want), "data-group allreduce"
    d.barrier("global")
    mx.finalize()


def gpu_stats_device_ns():
    """hipEvent-backed device Statistics: a committed 2-layer net at dp=2
    must attribute nonzero GPU comm nanoseconds to the gradient allreduce
    (host-rdtsc stats would hide overlapped comm in the Wait residue)."""
    import os
    os.environ["MLSL_STATS"] = "1"
    mx, torch, rank, size = _init()
    from mlsl_amd.models.synthetic import SyntheticNet
    net = SyntheticNet(size, 1, global_mb=4 * size, xp=torch, device="cuda")
    for _ in range(3):
        net.step()
    st = net.sess.stats
    assert st.enabled
    dev_ns = st.total_comm_device_ns
    assert dev_ns > 0, f"device comm ns not attributed: {dev_ns}"
    st.print()
    mx.finalize()


def gpu_stress():
    """Time-boxed randomized stress on the p2p transport: mixed ops, sizes
    and algorithms back-to-back (soak analog for the n>1 device paths).
    STRESS_SECONDS env bounds the duration (default 30)."""
    import time
    import random
    mx, torch, rank, size = _init()
    d = mx.Distribution(size, 1)
    secs = float(os.environ.get("STRESS_SECONDS", "30"))
    rng = random.Random(1234)  # same op sequence on every rank
    t_end = time.time() + secs
    iters = 0
    cont = torch.ones(1, device="cuda")
    while True:
        # Lockstep termination: ranks' clocks differ, and a rank that stops
        # while its peer issues one more collective desyncs the transport —
        # the continue flag is agreed via allreduce-min every iteration.
        cont.fill_(1.0 if time.time() < t_end else 0.0)
        mx.wait(d.all_reduce(cont, cont, 1, op="min", group="data"))
        torch.cuda.synchronize()
        if cont.item() == 0.0:
            break
        op = rng.choice(["ar", "ar_inplace", "rs", "ag", "a2a", "bcast",
                         "bar", "srl", "quant"])
        count = rng.choice([64, 4097, 1 << 16, 1 << 20])
        if op == "ar" or op == "ar_inplace":
            a = _arange(torch, count, rank)
            out = a if op == "ar_inplace" else torch.empty_like(a)
            mx.wait(d.all_reduce(a, out, count, op="sum", group="data"))
            torch.cuda.synchronize()
            want = size * torch.arange(count, dtype=torch.float32,
                                       device="cuda") + size * (size - 1) / 2.0
            assert torch.allclose(out, want), (iters, op, count)
        elif op == "rs":
            src = torch.cat([_arange(torch, count, rank) for _ in range(size)])
            out = torch.empty(count, device="cuda")
            mx.wait(d.reduce_scatter(src, out, count, op="sum", group="data"))
            torch.cuda.synchronize()
            want = size * torch.arange(count, dtype=torch.float32,
                                       device="cuda") + size * (size - 1) / 2.0
            assert torch.allclose(out, want), (iters, op, count)
        elif op == "ag":
            mine = _arange(torch, count, rank)
            flat = torch.empty(size * count, device="cuda")
            mx.wait(d.all_gather(mine, count, flat, group="data"))
            torch.cuda.synchronize()
            for r in range(size):
                assert torch.allclose(flat[r * count:(r + 1) * count],
                                      _arange(torch, count, r)), (iters, op, r)
        elif op == "a2a":
            src = torch.cat([torch.full((count,), float(rank * 10 + j),
                                        device="cuda") for j in range(size)])
            dst = torch.empty_like(src)
            mx.wait(d.all_to_all(src, count, dst, group="data"))
            torch.cuda.synchronize()
            for j in range(size):
                assert torch.all(dst[j * count:(j + 1) * count] ==
                                 float(j * 10 + rank)), (iters, op, j)
        elif op == "bcast":
            b = _arange(torch, count, 3) if rank == 0                 else torch.zeros(count, device="cuda")
            mx.wait(d.bcast(b, count, root=0, group="data"))
            torch.cuda.synchronize()
            assert torch.allclose(b, _arange(torch, count, 3)), (iters, op)
        elif op == "srl":
            dstp = (rank + 1) % size
            srcp = (rank - 1 + size) % size
            sv = torch.full((count,), float(rank * 3 + 1), device="cuda")
            rv = torch.empty(count, device="cuda")
            mx.wait(d.send_recv_list(sv, rv, [(dstp, 0, count, 0, 0),
                                              (srcp, 0, 0, 0, count)],
                                     group="data"))
            torch.cuda.synchronize()
            assert torch.all(rv == float(srcp * 3 + 1)), (iters, op)
        elif op == "quant":
            a = torch.randn(count, device="cuda")
            outq = torch.empty_like(a)
            req = mx.PersistentRequest(d, "all_reduce", count, dtype="f32",
                                       op="sum", group="data", quantized=True)
            req.start(a, outq)
            req.wait()
            torch.cuda.synchronize()
            exact = torch.empty_like(a)
            mx.wait(d.all_reduce(a, exact, count, op="sum", group="data"))
            torch.cuda.synchronize()
            rel = ((outq - exact).norm() / exact.norm().clamp_min(1e-6)).item()
            assert rel < 0.06, (iters, op, rel)
            req.destroy()
        else:
            d.barrier("data")
        iters += 1
    d.barrier("global")
    print(f"stress iters={iters}")
    mx.finalize()


def gpu_ddp():
    """DistributedData (bucketed DDP with autograd hooks) at world 2 on the
    device engine: gradients after finish_gradients() must equal the
    average of the per-rank local gradients, and match across ranks."""
    mx, torch, rank, size = _init()
    from mlsl_amd.parallel import DistributedData
    torch.manual_seed(7)  # same init on every rank (bcast also enforces it)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(),
        torch.nn.Linear(128, 32)).cuda()
    x = torch.full((16, 64), 1.0 + rank, device="cuda")

    # expected: average over ranks of the local gradient
    import copy
    ref = copy.deepcopy(model)
    ref(x).sum().backward()
    local_grads = [p.grad.detach().clone() for p in ref.parameters()]
    d = mx.Distribution(size, 1)
    for g in local_grads:
        flat = g.reshape(-1).contiguous()
        mx.wait(d.all_reduce(flat, flat, flat.numel(), op="sum", group="data"))
        torch.cuda.synchronize()
        g.copy_((flat / size).view_as(g))

    dd = DistributedData(model, dist=d)
    model(x).sum().backward()
    dd.finish_gradients()
    torch.cuda.synchronize()
    for p, want in zip(model.parameters(), local_grads):
        assert torch.allclose(p.grad, want, rtol=1e-5, atol=1e-5), \
            (p.shape, (p.grad - want).abs().max().item())
    mx.finalize()


def gpu_zero1():
    """ShardedOptimizer (ZeRO-1) at world 2 on the device engine:
    reduce-scatter of raw grads + owned-shard SGD + allgather of updated
    params must match a plain full-replica SGD step."""
    mx, torch, rank, size = _init()
    from mlsl_amd.parallel import ShardedOptimizer
    torch.manual_seed(5)
    model = torch.nn.Linear(96, 96).cuda()
    # plain reference: full allreduce-averaged grads + SGD on a clone
    import copy
    ref = copy.deepcopy(model)
    x = torch.full((8, 96), 1.0 + rank, device="cuda")
    ref(x).sum().backward()
    d = mx.Distribution(size, 1)
    for p in ref.parameters():
        flat = p.grad.detach().reshape(-1).contiguous()
        mx.wait(d.all_reduce(flat, flat, flat.numel(), op="sum", group="data"))
        torch.cuda.synchronize()
        p.grad.copy_((flat / size).view_as(p.grad))
    torch.optim.SGD(ref.parameters(), lr=0.1).step()

    opt = ShardedOptimizer(model.parameters(), torch.optim.SGD, dist=d,
                           reduce="rs", lr=0.1)
    model(x).sum().backward()
    opt.step()
    torch.cuda.synchronize()
    for p, w in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, w.data, rtol=1e-5, atol=1e-6), \
            (p.shape, (p.data - w.data).abs().max().item())
    mx.finalize()


def gpu_seqpar():
    """Ulysses-style sequence<->head resharding (alltoall) at world 2 on
    the device engine: seq_to_head . head_to_seq == identity."""
    mx, torch, rank, size = _init()
    from mlsl_amd.parallel import seqpar
    d = mx.Distribution(1, size)
    B, S, H = 2, 8 * size, 16 * size
    torch.manual_seed(20 + rank)
    x = torch.randn(B, S // size, H, device="cuda")
    y = seqpar.seq_to_head(d, x, group="model")
    torch.cuda.synchronize()
    assert y.shape == (B, S, H // size), y.shape
    z = seqpar.head_to_seq(d, y, group="model")
    torch.cuda.synchronize()
    assert z.shape == x.shape, z.shape
    assert torch.allclose(z, x), (z - x).abs().max().item()
    mx.finalize()


def gpu_rs_overlap():
    """Reduce-scatter with the reference's in-place layout (rbuf = sbuf +
    rank*seg, overlapping buffers) — the ZeRO-1 gradient exchange shape
    (grad_req Start(b, b + owned_off), session.cpp)."""
    mx, torch, rank, size = _init()
    d = mx.Distribution(size, 1)
    for per in (48, 1536, 65536):
        buf = torch.cat([_arange(torch, per, rank) + i * 1000
                         for i in range(size)])
        wanted = size * torch.arange(per, dtype=torch.float32, device="cuda")             + size * (size - 1) / 2.0 + size * rank * 1000
        rb = buf[rank * per:(rank + 1) * per]
        mx.wait(d.reduce_scatter(buf, rb, per, op="sum", group="data"))
        torch.cuda.synchronize()
        assert torch.allclose(rb, wanted),             (per, rank, rb[:4].tolist(), wanted[:4].tolist())
    d.barrier("global")
    mx.finalize()


def gpu_hybrid_rs():
    """Reduce-scatter with overlapping buffers on a COLOR-created data
    subgroup (Distribution(size//2, 2)) — the exact ZeRO-1 exchange of the
    hybrid e2e, minus the planner."""
    mx, torch, rank, size = _init()
    d = mx.Distribution(size // 2, 2)
    dp = size // 2
    didx = d.process_idx("data")
    for per in (48, 1536):
        buf = torch.cat([_arange(torch, per, didx) + i * 1000
                         for i in range(dp)])
        want = dp * torch.arange(per, dtype=torch.float32, device="cuda")             + dp * (dp - 1) / 2.0 + dp * didx * 1000
        rb = buf[didx * per:(didx + 1) * per]
        mx.wait(d.reduce_scatter(buf, rb, per, op="sum", group="data"))
        torch.cuda.synchronize()
        assert torch.allclose(rb, want),             (per, rank, didx, rb[:4].tolist(), want[:4].tolist())
        # and the increment allgather right after (inc_req shape)
        src = _arange(torch, per, didx) * 2
        flat = torch.empty(dp * per, device="cuda")
        mx.wait(d.all_gather(src, per, flat, group="data"))
        torch.cuda.synchronize()
        for r2 in range(dp):
            assert torch.allclose(flat[r2 * per:(r2 + 1) * per],
                                  _arange(torch, per, r2) * 2), (per, r2)
    d.barrier("global")
    mx.finalize()


def gpu_rma_window():
    """RMA window over DEVICE memory: puts from cuda tensors land in the
    peers' HBM windows, same-epoch gets observe them (fence-epoch
    semantics over the p2p/RCCL schedule executor)."""
    import torch
    mx, torch, rank, size = _init()
    d = mx.Distribution(size, 1)
    n = 1024
    win = d.win_allocate(size * n * 4, group="data")

    src = torch.full((n,), float(rank + 1), dtype=torch.float32, device="cuda")
    torch.cuda.synchronize()
    for t in range(size):
        win.put(src, n * 4, t, rank * n * 4)
    src.fill_(-1.0)  # staged at put time
    torch.cuda.synchronize()
    win.fence()
    local = torch.zeros(size * n, dtype=torch.float32, device="cuda")
    win.read(local, size * n * 4)
    torch.cuda.synchronize()
    for s in range(size):
        got = local[s * n:(s + 1) * n]
        assert torch.all(got == float(s + 1)), \
            f"rank {rank} slot {s}: {got[:3]} != {s + 1}"

    right = (rank + 1) % size
    left = (rank - 1) % size
    fresh = torch.full((n,), 100.0 + rank, dtype=torch.float32, device="cuda")
    seen = torch.zeros(n, dtype=torch.float32, device="cuda")
    torch.cuda.synchronize()
    win.put(fresh, n * 4, right, right * n * 4)
    win.get(seen, n * 4, right, right * n * 4)
    win.fence()
    torch.cuda.synchronize()
    assert torch.all(seen == 100.0 + rank), f"get-after-put saw {seen[:3]}"
    mine = torch.zeros(n, dtype=torch.float32, device="cuda")
    win.read(mine, n * 4, off=rank * n * 4)
    torch.cuda.synchronize()
    assert torch.all(mine == 100.0 + left), f"own slot {mine[:3]}"

    win.fence()  # empty epoch
    win.close()
    d.barrier("global")
    mx.finalize()


def gpu_configure_tenants():
    """Environment.Configure("color=N") in DEVICE mode: 4 boot ranks on one
    GPU split into two 2-rank tenant worlds; each tenant's p2p window group
    + allreduce stays inside the tenant. Regression cover for the
    boot-world-span Allgather buffers (bootstrap gathers write
    Boot()->Size() entries even after Configure shrinks ctx.Size())."""
    import torch
    import mlsl_amd as mx
    torch.cuda.set_device(0)
    boot_rank = int(os.environ["RANK"])
    mx.init()
    assert mx.world_size() == 4
    color = boot_rank // 2
    mx.configure(f"color={color}")
    assert mx.world_size() == 2, mx.world_size()
    assert mx.rank() == boot_rank % 2
    d = mx.Distribution(2, 1)
    a = torch.full((4096,), float(boot_rank), dtype=torch.float32, device="cuda")
    out = torch.zeros_like(a)
    mx.wait(d.all_reduce(a, out, 4096, op="sum", group="data"))
    torch.cuda.synchronize()
    want = {0: 1.0, 1: 1.0, 2: 5.0, 3: 5.0}[boot_rank]
    assert torch.all(out == want), f"tenant allreduce got {out[0]} want {want}"
    mx.finalize()


WORKERS = {
    "gpu_collectives": gpu_collectives,
    "gpu_configure_tenants": gpu_configure_tenants,
    "gpu_rma_window": gpu_rma_window,
    "gpu_rs_overlap": gpu_rs_overlap,
    "gpu_hybrid_rs": gpu_hybrid_rs,
    "gpu_ddp": gpu_ddp,
    "gpu_zero1": gpu_zero1,
    "gpu_seqpar": gpu_seqpar,
    "gpu_allreduce_multislot": gpu_allreduce_multislot,
    "gpu_quant_allreduce": gpu_quant_allreduce,
    "gpu_hybrid_grid": gpu_hybrid_grid,
    "gpu_stats_device_ns": gpu_stats_device_ns,
    "gpu_stress": gpu_stress,
}


def main():
    name = sys.argv[1]
    WORKERS[name]()
    print(f"OK {name} rank={os.environ.get('RANK')}")


if __name__ == "__main__":
    main()
