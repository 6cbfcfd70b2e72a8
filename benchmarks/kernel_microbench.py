#!/usr/bin/env python3
"""Single-GPU kernel microbenchmarks (run under rocprofv3 for the committed
profiles/ evidence): local reduce streaming bandwidth, int8 quantize /
dequantize / compressed-accumulate throughput, pack/unpack."""
import json
import sys
import time

import torch

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mlsl_amd import ops  # noqa: E402


def timed(fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    torch.cuda.set_device(0)
    n = 1 << 26  # 64M elements = 256 MiB fp32
    res = {}

    a = torch.randn(n, device="cuda")
    b = torch.randn(n, device="cuda")
    dt = timed(lambda: ops.reduce_(a, b, n))
    res["reduce_f32_256MiB_TBps"] = round(3 * 4 * n / dt / 1e12, 3)

    dt = timed(lambda: ops.reduce_nt(a, b, n))
    res["reduce_f32_nt_TBps"] = round(3 * 4 * n / dt / 1e12, 3)

    dt = timed(lambda: ops.reduce_nt2(a, b, n))
    res["reduce_f32_nt2_TBps"] = round(3 * 4 * n / dt / 1e12, 3)

    # streaming copy: NT kernel vs torch/HIP blit path (same 256 MiB)
    dt = timed(lambda: ops.copy(a, b, n * 4))
    res["copy_nt_256MiB_TBps"] = round(2 * 4 * n / dt / 1e12, 3)
    dt = timed(lambda: ops.copy_variant(a, b, n * 4, False))
    res["copy_plain_256MiB_TBps"] = round(2 * 4 * n / dt / 1e12, 3)

    # ops.* sync per call; make torch pay the same per-call sync for a
    # fair A/B.
    def _torch_copy():
        a.copy_(b)
        torch.cuda.synchronize()
    dt = timed(_torch_copy)
    res["copy_torch_256MiB_TBps"] = round(2 * 4 * n / dt / 1e12, 3)

    abf = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    bbf = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    dt = timed(lambda: ops.reduce_(abf, bbf, n, dtype="bf16"))
    res["reduce_bf16_128MiB_TBps"] = round(3 * 2 * n / dt / 1e12, 3)

    wire = torch.empty(ops.wire_bytes(n), device="cuda", dtype=torch.uint8)
    err = torch.zeros(n, device="cuda")
    dt = timed(lambda: ops.quantize(a, wire, n, err=err))
    # reads in+err (8B/elem), writes wire(~1B)+err(4B)
    res["quantize_f32_TBps"] = round((8 + 1 + 4) * n / dt / 1e12, 3)
    res["quantize_f32_Gelems_s"] = round(n / dt / 1e9, 2)

    dt = timed(lambda: ops.quantize_f32_nt(a, wire, n, err))
    res["quantize_f32_ntst_Gelems_s"] = round(n / dt / 1e9, 2)

    out = torch.empty_like(a)
    dt = timed(lambda: ops.dequantize(wire, out, n))
    res["dequantize_f32_TBps"] = round((1 + 4) * n / dt / 1e12, 3)

    dt = timed(lambda: ops.dequantize_nt(wire, out, n))
    res["dequantize_nt_f32_TBps"] = round((1 + 4) * n / dt / 1e12, 3)

    # bf16 quantize/dequantize (driver config 5 dtype; two-blocks-per-wave
    # fast path)
    nb2 = n  # 64M bf16 = 128 MiB
    abf2 = torch.randn(nb2, device="cuda", dtype=torch.bfloat16)
    errb = torch.zeros(nb2, device="cuda", dtype=torch.bfloat16)
    wireb = torch.empty(ops.wire_bytes(nb2), device="cuda", dtype=torch.uint8)
    dt = timed(lambda: ops.quantize(abf2, wireb, nb2, err=errb, dtype="bf16"))
    # reads in+err (4B/elem at HBM), writes wire(~1B)+err(2B)
    res["quantize_bf16_TBps"] = round((4 + 1 + 2) * nb2 / dt / 1e12, 3)
    res["quantize_bf16_Gelems_s"] = round(nb2 / dt / 1e9, 2)
    outb = torch.empty_like(abf2)
    dt = timed(lambda: ops.dequantize(wireb, outb, nb2, dtype="bf16"))
    res["dequantize_bf16_TBps"] = round((1 + 2) * nb2 / dt / 1e12, 3)

    wb = torch.empty_like(wire)
    ops.quantize(b, wb, n)
    dt = timed(lambda: ops.quant_accum(wire, wb, n))
    res["quant_accum_TBps"] = round(3 * 1 * n / dt / 1e12, 3)

    # pack: half the fms of a [mb][fm][s] block. Sized so the move is
    # ~128 MiB: at 16 MiB the old size, per-call sync+launch (~12 us)
    # dominated the ~7 us copy and under-reported bandwidth 2.5x.
    mb, fm, s = 64, 2048, 512
    src = torch.randn(mb * fm * s, device="cuda")
    dst = torch.empty(mb * (fm // 2) * s, device="cuda")
    kw = dict(mb_offset=0, mb_count=mb, fm_offset=fm // 4, fm_count=fm // 2,
              fm_size=s, buf_offset=0, local_fm_count=fm, local_mb_count=mb,
              dtype="f32")
    dt = timed(lambda: ops.pack(src, dst, **kw))
    res["pack_f32_TBps"] = round(2 * 4 * mb * (fm // 2) * s / dt / 1e12, 3)

    print(json.dumps(res, indent=1))


if __name__ == "__main__":
    main()
