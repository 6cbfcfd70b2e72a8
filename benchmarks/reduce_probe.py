#!/usr/bin/env python3
"""Tiny probe for rocprofv3 PMC runs: a handful of reduce/quantize launches
on 256 MiB buffers (keep iteration count minimal — counter collection
serializes dispatches)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from mlsl_amd import ops  # noqa: E402


def main():
    torch.cuda.set_device(0)
    n = 1 << 26
    a = torch.randn(n, device="cuda")
    b = torch.randn(n, device="cuda")
    for _ in range(3):
        ops.reduce_(a, b, n)
    wire = torch.empty(ops.wire_bytes(n), device="cuda", dtype=torch.uint8)
    for _ in range(3):
        ops.quantize(a, wire, n)
    torch.cuda.synchronize()
    print("probe done")


if __name__ == "__main__":
    main()
