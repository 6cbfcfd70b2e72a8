#!/usr/bin/env python3
"""One-sided RMA window sample: each rank puts its value into every
member's window and reads the full table back after the fence (the
fence-epoch semantics of Distribution.win_allocate / Win — see
docs/USER_GUIDE.md §2.6)."""
import numpy as np

import mlsl_amd as mx

N = 16  # floats per rank slot


def main():
    mx.init()
    rank, size = mx.rank(), mx.world_size()
    d = mx.Distribution(size, 1)
    win = d.win_allocate(size * N * 4, group="data")

    mine = np.full(N, float(rank + 1), dtype=np.float32)
    for target in range(size):
        win.put(mine, N * 4, target, target_off=rank * N * 4)
    win.fence()

    table = np.zeros(size * N, dtype=np.float32)
    win.read(table, size * N * 4)
    ok = all(bool(np.all(table[s * N:(s + 1) * N] == s + 1))
             for s in range(size))
    print(f"[{rank}/{size}] {'PASSED' if ok else 'FAILED'}")
    win.close()
    d.barrier("global")
    mx.finalize()
    return 0 if ok else 1


if __name__ == "__main__":
    raise SystemExit(main())
