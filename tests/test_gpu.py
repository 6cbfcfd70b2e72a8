"""GPU tests (single MI355X, run via gpurun / the driver).

Kernel numerics vs plain torch fp32 references, device-side request paths
at world size 1, and the quantized-allreduce building blocks.
"""
import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@requires_gpu
class TestKernels:
    def setup_method(self, _):
        torch.cuda.set_device(0)
        torch.manual_seed(1234)

    @pytest.mark.parametrize("n", [1, 255, 4096, 1 << 20, (1 << 22) + 3])
    @pytest.mark.parametrize("op", ["sum", "min", "max"])
    def test_reduce_f32(self, n, op):
        from mlsl_amd import ops
        a = torch.randn(n, device="cuda")
        b = torch.randn(n, device="cuda")
        want = {"sum": a + b, "min": torch.minimum(a, b), "max": torch.maximum(a, b)}[op]
        ops.reduce_(a, b, n, dtype="f32", op=op)
        torch.cuda.synchronize()
        assert torch.allclose(a, want)

    @pytest.mark.parametrize("n", [4096, (1 << 20) + 17])
    def test_reduce_bf16(self, n):
        from mlsl_amd import ops
        a = torch.randn(n, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
        want = (a.float() + b.float()).to(torch.bfloat16)
        ops.reduce_(a, b, n, dtype="bf16", op="sum")
        torch.cuda.synchronize()
        assert torch.allclose(a.float(), want.float(), atol=0.0, rtol=0.0)

    @pytest.mark.parametrize("n", [4096])
    def test_reduce_f64_i32(self, n):
        from mlsl_amd import ops
        a = torch.randn(n, device="cuda", dtype=torch.float64)
        b = torch.randn(n, device="cuda", dtype=torch.float64)
        want = a + b
        ops.reduce_(a, b, n, dtype="f64", op="sum")
        a32 = torch.randint(-1000, 1000, (n,), device="cuda", dtype=torch.int32)
        b32 = torch.randint(-1000, 1000, (n,), device="cuda", dtype=torch.int32)
        want32 = a32 + b32
        ops.reduce_(a32, b32, n, dtype="i32", op="sum")
        torch.cuda.synchronize()
        assert torch.allclose(a, want)
        assert torch.equal(a32, want32)

    @pytest.mark.parametrize("n,block", [(1 << 20, 256), (100_000, 256), (5000, 512),
                                         (1 << 21, 256)])  # >=4MiB: f32 x2 fast path
    def test_quant_roundtrip_f32(self, n, block):
        from mlsl_amd import ops
        g = torch.randn(n, device="cuda") * 3.0
        wire = torch.empty(ops.wire_bytes(n, block), device="cuda", dtype=torch.uint8)
        out = torch.empty_like(g)
        err = torch.zeros_like(g)
        ops.quantize(g, wire, n, err=err, block=block)
        ops.dequantize(wire, out, n, block=block)
        torch.cuda.synchronize()
        # per-block error bound: half a quantization step of the block max
        gb = torch.nn.functional.pad(g, (0, (block - n % block) % block)).view(-1, block)
        scale = gb.abs().amax(dim=1, keepdim=True) / 127.0
        bound = (scale.expand_as(gb).reshape(-1)[:n] * 0.5) + 1e-7
        assert ((out - g).abs() <= bound + 1e-6).all()
        # error feedback: residual = input - quantized
        assert torch.allclose(g, out + err, atol=1e-5)

    @pytest.mark.parametrize("block,n", [(256, 1 << 18), (256, (1 << 18) - 128),
                                         (128, 1 << 16)])
    def test_quant_roundtrip_bf16(self, block, n):
        # (256, 256-multiple) exercises the two-blocks-per-wave bf16 fast
        # path; the others the generic kernel.
        from mlsl_amd import ops
        a = torch.randn(n, device="cuda", dtype=torch.bfloat16)
        err = torch.zeros(n, device="cuda", dtype=torch.bfloat16)
        wire = torch.empty(ops.wire_bytes(n, block), device="cuda",
                           dtype=torch.uint8)
        ops.quantize(a, wire, n, err=err, block=block, dtype="bf16")
        out = torch.empty_like(a)
        ops.dequantize(wire, out, n, block=block, dtype="bf16")
        torch.cuda.synchronize()
        af = a.float()
        # per-block scale: max abs / 127 -> error bounded by ~scale/2 (+
        # bf16 rounding); residual holds the rest
        nb = (n + block - 1) // block
        pad = nb * block - n
        blocks = torch.nn.functional.pad(af, (0, pad)).view(nb, block)
        step = blocks.abs().max(dim=1).values / 127.0
        err_bound = step.repeat_interleave(block)[:n] * 0.6 + 0.02 * af.abs()
        diff = (out.float() - af).abs()
        assert (diff <= err_bound + 1e-6).float().mean().item() > 0.999, \
            f"bf16 roundtrip error too large: {diff.max()}"
        # error feedback: residual + dequant ~= original
        rec = out.float() + err.float()
        assert (rec - af).abs().max().item() < 0.05 * af.abs().max().item() + 0.05

    def test_quant_error_feedback_accumulates(self):
        from mlsl_amd import ops
        n, block = 1 << 16, 256
        torch.manual_seed(7)
        g = torch.randn(n, device="cuda")
        wire = torch.empty(ops.wire_bytes(n, block), device="cuda", dtype=torch.uint8)
        out = torch.empty_like(g)
        err = torch.zeros_like(g)
        total_in, total_out = torch.zeros_like(g), torch.zeros_like(g)
        for _ in range(50):
            total_in += g
            ops.quantize(g, wire, n, err=err, block=block)
            ops.dequantize(wire, out, n, block=block)
            total_out += out
        torch.cuda.synchronize()
        # with error feedback the accumulated quantized stream tracks the
        # accumulated input to within ~1 quantization step (not 50 steps)
        scale = g.abs().max().item() / 127.0
        drift = (total_in - total_out - err).abs().max().item()
        assert drift < 3 * scale, f"error feedback drifting: {drift} vs step {scale}"

    @pytest.mark.parametrize("block", [256, 128])
    def test_quant_accum_compressed_domain(self, block):
        # block=256 exercises the register-resident wave-per-block fast
        # path; 128 the generic two-pass kernel.
        from mlsl_amd import ops
        n = 1 << 18
        a = torch.randn(n, device="cuda")
        b = torch.randn(n, device="cuda")
        wa = torch.empty(ops.wire_bytes(n, block), device="cuda", dtype=torch.uint8)
        wb = torch.empty_like(wa)
        ops.quantize(a, wa, n, block=block)
        ops.quantize(b, wb, n, block=block)
        ops.quant_accum(wa, wb, n, block=block)
        out = torch.empty_like(a)
        ops.dequantize(wa, out, n, block=block)
        torch.cuda.synchronize()
        want = a + b
        step = want.abs().max().item() / 127.0
        assert (out - want).abs().max().item() < 4 * step

    def test_pack_unpack(self):
        from mlsl_amd import ops
        mb, fm, s = 8, 16, 12
        src = torch.randn(mb * fm * s, device="cuda")
        dst = torch.zeros(4 * 8 * s, device="cuda")
        kw = dict(mb_offset=2, mb_count=4, fm_offset=4, fm_count=8, fm_size=s,
                  buf_offset=0, local_fm_count=fm, local_mb_count=mb, dtype="f32")
        ops.pack(src, dst, **kw)
        back = torch.zeros_like(src)
        ops.unpack(dst, back, **kw)
        torch.cuda.synchronize()
        v = src.view(mb, fm, s)[2:6, 4:12, :].reshape(-1)
        assert torch.equal(dst, v)
        assert torch.equal(back.view(mb, fm, s)[2:6, 4:12, :].reshape(-1), v)

    def test_reduce_bandwidth_sane(self):
        """The f32 reduce kernel should stream well above 1 TB/s on HBM3E
        (2 reads + 1 write per element; ceiling ~6.3 TB/s)."""
        from mlsl_amd import ops
        import time
        n = 1 << 26  # 256 MiB per buffer
        a = torch.randn(n, device="cuda")
        b = torch.randn(n, device="cuda")
        ops.reduce_(a, b, n)  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 10
        for _ in range(iters):
            ops.reduce_(a, b, n)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        tbs = 3 * 4 * n / dt / 1e12
        print(f"reduce f32 streaming: {tbs:.2f} TB/s")
        assert tbs > 1.0, f"reduce kernel too slow: {tbs} TB/s"


@requires_gpu
class TestDevicePaths:
    def test_world1_device_requests(self):
        import mlsl_amd as mx
        mx.init()
        try:
            d = mx.Distribution(1, 1)
            x = torch.randn(100_000, device="cuda")
            y = torch.zeros_like(x)
            mx.wait(d.all_reduce(x, y, x.numel(), op="sum", group="data"))
            torch.cuda.synchronize()
            assert torch.equal(x, y)
            # reduce_scatter / allgather / alltoall degenerate paths
            out = torch.zeros_like(x)
            mx.wait(d.reduce_scatter(x, out, x.numel(), op="sum", group="data"))
            torch.cuda.synchronize()
            assert torch.equal(out, x)
        finally:
            mx.finalize()

    def test_synthetic_net_device(self):
        import mlsl_amd as mx
        mx.init()
        try:
            from mlsl_amd.models.synthetic import SyntheticNet
            net = SyntheticNet(1, 1, global_mb=8, xp=torch, device="cuda")
            for it in range(3):
                net.step(it)
        finally:
            mx.finalize()


@requires_gpu
class TestHostStaging:
    def test_numpy_buffers_on_device_engine(self):
        """Host (numpy) buffers ride the GPU path via staging
        (ReplaceIn/Out analog)."""
        import numpy as np
        import mlsl_amd as mx
        mx.init()
        try:
            d = mx.Distribution(1, 1)
            a = np.random.randn(100_000).astype(np.float32)
            out = np.zeros_like(a)
            mx.wait(d.all_reduce(a, out, a.size, op="sum", group="data"))
            assert np.allclose(out, a)
        finally:
            mx.finalize()


@requires_gpu
class TestGraphReplay:
    def test_persistent_graph_replay(self):
        """MLSL_USE_GRAPHS=1: a persistent request is captured once and
        replayed; results stay correct as buffer contents change."""
        import os
        import mlsl_amd as mx
        os.environ["MLSL_USE_GRAPHS"] = "1"
        try:
            mx.init()
            d = mx.Distribution(1, 1)
            n = 1 << 16
            x = torch.zeros(n, device="cuda")
            y = torch.zeros(n, device="cuda")
            preq = mx.PersistentRequest(d, "all_reduce", n, dtype="f32",
                                        op="sum", group="data")
            for it in range(6):
                x.fill_(float(it + 1))
                preq.start(x, y)
                preq.wait()
                torch.cuda.synchronize()
                assert torch.all(y == it + 1), f"replay iter {it}"
            preq.destroy()
            mx.finalize()
        finally:
            os.environ.pop("MLSL_USE_GRAPHS", None)


@pytest.mark.gpu
class TestTorchWrappers:
    def test_torch_backend_on_device(self):
        import os
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29756")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        import torch.distributed as td
        import mlsl_amd.torch_backend  # noqa: F401
        td.init_process_group(backend="mlsl", rank=0, world_size=1)
        t = torch.randn(1 << 20, device="cuda")
        ref = t.clone()
        td.all_reduce(t)
        assert torch.allclose(t, ref)
        big = torch.zeros(1 << 20, device="cuda")
        td.all_gather_into_tensor(big, t)
        assert torch.allclose(big, t)
        td.barrier()
        td.destroy_process_group()
        import mlsl_amd as mx
        mx.finalize()

    def test_zero1_on_device(self):
        import mlsl_amd as mx
        from mlsl_amd.parallel.zero1 import ShardedOptimizer
        mx.init()
        torch.manual_seed(5)
        model = torch.nn.Linear(64, 64).cuda()
        d = mx.Distribution(1, 1)
        sopt = ShardedOptimizer(model.parameters(), torch.optim.SGD, d,
                                reduce="rs", lr=0.1)
        x = torch.randn(8, 64, device="cuda")
        loss = model(x).pow(2).mean()
        loss.backward()
        before = model.weight.detach().clone()
        sopt.step()
        torch.cuda.synchronize()
        assert not torch.equal(before, model.weight.detach())
        mx.finalize()


@pytest.mark.gpu
class TestStatsDevice:
    def test_statistics_on_device_path(self):
        # MLSL_STATS over the device engine: runtime + isolation counters
        # populate for a committed net (reference mlsl_test stats getters,
        # mlsl_test.cpp:679-685).
        import os
        os.environ["MLSL_STATS"] = "1"
        import mlsl_amd as mx
        try:
            mx.init()
            from mlsl_amd.models.synthetic import SyntheticNet
            net = SyntheticNet(1, 1, global_mb=8, xp=torch, device="cuda")
            for _ in range(3):
                net.step()
            st = net.sess.stats
            assert st.enabled
            assert st.total_comm_cycles + st.isolation_comm_cycles(0) >= 0
            assert st.total_comm_size >= 0
            st.print()
            mx.finalize()
        finally:
            os.environ.pop("MLSL_STATS", None)


@pytest.mark.gpu
class TestSoak:
    def test_no_device_memory_growth(self):
        """200 one-shot + 200 persistent request iterations must not grow
        device memory: scratch comes from the HBM pool, one-shot requests
        are freed at wait (RequestStorage removal), events are reused."""
        import mlsl_amd as mx
        import torch
        mx.init()
        d = mx.Distribution(1, 1)
        n = 1 << 20
        a = torch.randn(n, device="cuda")
        b = torch.empty_like(a)
        # warm every path once (pool high-water, events, staging)
        preq = mx.PersistentRequest(d, "all_reduce", n, dtype="f32", op="sum",
                                    group="data")
        qreq = mx.PersistentRequest(d, "all_reduce", n, dtype="f32", op="sum",
                                    group="data", quantized=True)
        for _ in range(3):
            mx.wait(d.all_reduce(a, b, n, op="sum", group="data"))
            preq.start(a, b); preq.wait()
            qreq.start(a, b); qreq.wait()
        torch.cuda.synchronize()
        free0, _ = torch.cuda.mem_get_info()
        for _ in range(200):
            mx.wait(d.all_reduce(a, b, n, op="sum", group="data"))
            preq.start(a, b); preq.wait()
            qreq.start(a, b); qreq.wait()
        torch.cuda.synchronize()
        free1, _ = torch.cuda.mem_get_info()
        grew = free0 - free1
        assert grew < (64 << 20), f"device memory grew {grew / 1e6:.1f} MB"
        preq.destroy(); qreq.destroy()
        mx.finalize()
