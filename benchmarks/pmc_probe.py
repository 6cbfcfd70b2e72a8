#!/usr/bin/env python3
"""Minimal kernel-launch probe for rocprofv3 PMC counter collection.

Run (counters only — never combined with trace domains):
    rocprofv3 --pmc FETCH_SIZE -d out_fetch -- python benchmarks/pmc_probe.py
    rocprofv3 --pmc WRITE_SIZE -d out_write -- python benchmarks/pmc_probe.py
(FETCH_SIZE needs 3 TCC slots and WRITE_SIZE 2 — they do not fit in one
pass on gfx950, hence two runs.)

Launches each hot kernel a fixed number of times on a fixed 256 MiB f32
problem so per-dispatch counter rows can be matched to kernels by name and
compared against the analytic HBM byte counts:
    reduce NT   : read 2x256 MiB, write 256 MiB
    copy NT     : read 256 MiB, write 256 MiB
    quantize    : read in+err 512 MiB, write wire 264 MiB/4... (~66 MiB) + err 256 MiB
    dequantize  : read wire ~66 MiB, write 256 MiB
Note (MI355X_MICROARCH.md): on gfx950 FETCH_SIZE reads half the bytes a
wide coalesced stream actually fetches; hbm read bytes ~= 2*FETCH_SIZE.
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mlsl_amd import ops  # noqa: E402

ITERS = 5


def main():
    assert torch.cuda.is_available()
    torch.cuda.set_device(0)
    n = 1 << 26  # 64M f32 = 256 MiB
    a = torch.randn(n, device="cuda")
    b = torch.randn(n, device="cuda")
    wire = torch.empty(ops.wire_bytes(n), device="cuda", dtype=torch.uint8)
    err = torch.zeros(n, device="cuda")
    out = torch.empty_like(a)
    torch.cuda.synchronize()

    for _ in range(ITERS):
        ops.reduce_(a, b, n)          # ReduceF32NTKernel (>=16 MiB -> NT)
    for _ in range(ITERS):
        ops.copy(a, b, n * 4)         # CopyNTKernel
    for _ in range(ITERS):
        ops.quantize(a, wire, n, err=err)   # QuantizeKernel
    for _ in range(ITERS):
        ops.dequantize(wire, out, n)  # DequantizeF32x2Kernel
    # bf16 two-blocks-per-wave pair
    abf = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    errb = torch.zeros(n, device="cuda", dtype=torch.bfloat16)
    wireb = torch.empty(ops.wire_bytes(n), device="cuda", dtype=torch.uint8)
    outb = torch.empty_like(abf)
    for _ in range(ITERS):
        ops.quantize(abf, wireb, n, err=errb, dtype="bf16")  # QuantizeBf16x2Kernel
    for _ in range(ITERS):
        ops.dequantize(wireb, outb, n, dtype="bf16")  # DequantizeBf16x2Kernel
    # pack: half the fms of a 256 MiB block (PackFastKernel<uint4,NT>)
    mb, fm, sdim = 64, 2048, 512
    src = torch.randn(mb * fm * sdim, device="cuda")
    dst = torch.empty(mb * (fm // 2) * sdim, device="cuda")
    for _ in range(ITERS):
        ops.pack(src, dst, mb_offset=0, mb_count=mb, fm_offset=fm // 4,
                 fm_count=fm // 2, fm_size=sdim, buf_offset=0,
                 local_fm_count=fm, local_mb_count=mb, dtype="f32")
    torch.cuda.synchronize()
    print("pmc probe done")


if __name__ == "__main__":
    main()
