#!/usr/bin/env python3
"""Plumbing sample through the Python binding (reference analog:
tests/examples/mlsl_test/mlsl_test.py over include/mlsl/mlsl.py)."""
import numpy as np

import mlsl_amd as mx

COUNT = 128


def main():
    mx.init()
    rank, size = mx.rank(), mx.world_size()
    d = mx.Distribution(size, 1)
    buf = np.full(COUNT, float(rank), dtype=np.float32)
    mx.wait(d.all_reduce(buf, buf, COUNT, op="sum", group="data"))
    expected = (size - 1) * size / 2.0
    ok = bool(np.all(buf == expected))
    print(f"[{rank}/{size}] {'PASSED' if ok else 'FAILED'}")
    mx.finalize()
    return 0 if ok else 1


if __name__ == "__main__":
    raise SystemExit(main())
