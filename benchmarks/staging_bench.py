#!/usr/bin/env python3
"""Host-staging throughput (ReplaceIn/Out analog): allreduce of a pageable
numpy buffer through the device engine at world 1 — measures the
user->pinned->HBM->pinned->user staging pipeline (chunked, multi-threaded
memcpy, async DMA). Prints one JSON line per size."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import numpy as np
    import mlsl_amd as mx
    mx.init()
    d = mx.Distribution(mx.world_size(), 1)
    for mib in (16, 64, 256):
        count = mib * 1024 * 1024 // 4
        buf = np.random.randn(count).astype(np.float32)
        out = np.empty_like(buf)
        preq = mx.PersistentRequest(d, "all_reduce", count, dtype="f32",
                                    op="sum", group="data")
        for _ in range(3):
            preq.start(buf, out)
            preq.wait()
        iters = 10
        t0 = time.perf_counter()
        for _ in range(iters):
            preq.start(buf, out)
            preq.wait()
        dt = (time.perf_counter() - t0) / iters
        nbytes = count * 4
        print(json.dumps({
            "staged_mib": mib,
            "lat_ms": round(dt * 1e3, 3),
            "roundtrip_GBps": round(2 * nbytes / dt / 1e9, 3),
            "copy_threads": os.environ.get("MLSL_COPY_THREADS", "4"),
        }))
        assert np.allclose(out, buf), "staging roundtrip mismatch"
        preq.destroy()
    mx.finalize()


if __name__ == "__main__":
    main()
