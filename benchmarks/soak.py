#!/usr/bin/env python3
"""GPU soak: alternate exact / quantized persistent allreduces, one-shot
collectives and pack/unpack for N seconds; asserts no device-memory growth
and stable numerics. Catches async-ordering and scratch-reuse bugs that
single-shot tests miss."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import mlsl_amd as mx
from mlsl_amd import ops


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=20)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    mx.init()
    d = mx.Distribution(mx.world_size(), 1)
    n = 1 << 22
    a = torch.randn(n, device="cuda")
    b = torch.empty_like(a)
    abf = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    bbf = torch.empty_like(abf)
    preq = mx.PersistentRequest(d, "all_reduce", n, dtype="f32", op="sum",
                                group="data")
    qreq = mx.PersistentRequest(d, "all_reduce", n, dtype="bf16", op="sum",
                                group="data", quantized=True)
    src = torch.randn(32 * 64 * 128, device="cuda")
    dst = torch.empty(32 * 32 * 128, device="cuda")

    for _ in range(3):
        preq.start(a, b); preq.wait()
        qreq.start(abf, bbf); qreq.wait()
    torch.cuda.synchronize()
    free0, _ = torch.cuda.mem_get_info()

    iters = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < args.seconds:
        preq.start(a, b); preq.wait()
        qreq.start(abf, bbf); qreq.wait()
        mx.wait(d.all_reduce(a, b, n, op="max", group="data"))
        ops.pack(src, dst, mb_offset=0, mb_count=32, fm_offset=16,
                 fm_count=32, fm_size=128, buf_offset=0, local_fm_count=64,
                 local_mb_count=32, dtype="f32")
        iters += 1
        if iters % 50 == 0 and mx.world_size() == 1:
            # numerics spot check: world-1 exact allreduce copies
            assert torch.equal(a, b), "exact allreduce drifted"
    torch.cuda.synchronize()
    free1, _ = torch.cuda.mem_get_info()
    grew = (free0 - free1) / 1e6
    print({"soak_iters": iters, "seconds": round(time.perf_counter() - t0, 1),
           "mem_growth_mb": round(grew, 1)})
    assert grew < 64, f"device memory grew {grew} MB"
    mx.finalize()
    print("SOAK PASSED")


if __name__ == "__main__":
    main()
