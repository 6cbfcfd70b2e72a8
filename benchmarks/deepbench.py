#!/usr/bin/env python3
"""DeepBench-style allreduce benchmark (the reference's eplib README names
DeepBench as its benchmark workload): the classic Baidu DeepBench allreduce
message sizes, fp32, through the persistent-request engine.

Single process: degenerate local path. Multi-rank:
    torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 \
        benchmarks/deepbench.py
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# DeepBench allreduce problem sizes (elements, fp32)
SIZES = [100_000, 3_097_600, 4_194_304, 6_553_600, 16_777_217, 38_360_000,
         64_500_000]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--device", default="auto")
    args = ap.parse_args()

    torch = None
    use_cuda = False
    if args.device in ("auto", "cuda"):
        try:
            import torch as _t
            torch = _t
            use_cuda = torch.cuda.is_available()
        except ImportError:
            pass
    if not use_cuda:
        os.environ.setdefault("MLSL_TRANSPORT", "tcp")

    import mlsl_amd as mx
    mx.init()
    rank, size = mx.rank(), mx.world_size()
    d = mx.Distribution(size, 1)

    if use_cuda:
        local = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local % torch.cuda.device_count())

    rows = []
    for n in SIZES:
        if use_cuda:
            buf = torch.randn(n, dtype=torch.float32, device="cuda")
            out = torch.empty_like(buf)
        else:
            import numpy as np
            buf = np.random.randn(n).astype(np.float32)
            out = np.empty_like(buf)
        preq = mx.PersistentRequest(d, "all_reduce", n, dtype="f32", op="sum",
                                    group="data")
        for _ in range(args.warmup):
            preq.start(buf, out)
            preq.wait()
        if use_cuda:
            torch.cuda.synchronize()
        d.barrier("global")
        t0 = time.perf_counter()
        for _ in range(args.iters):
            preq.start(buf, out)
            preq.wait()
        if use_cuda:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        preq.destroy()
        bytes_ = n * 4
        # standard allreduce bus bandwidth convention
        busbw = (2 * (size - 1) / size) * bytes_ / dt / 1e9 if size > 1 \
            else bytes_ / dt / 1e9
        rows.append({"elems": n, "mb": round(bytes_ / 1e6, 1),
                     "lat_us": round(dt * 1e6, 1),
                     "algbw_GBps": round(bytes_ / dt / 1e9, 2),
                     "busbw_GBps": round(busbw, 2)})
    if rank == 0:
        for r in rows:
            print(json.dumps({"bench": "deepbench_allreduce", "world": size,
                              **r}))
    mx.finalize()


if __name__ == "__main__":
    main()
