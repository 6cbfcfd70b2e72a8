#!/usr/bin/env python3
"""AllReduce sweep 4KB-1GB: algorithmic bandwidth and
latency per message size through the mlsl_amd engine, any world size.
Launch with torchrun for N>1 ranks; prints one JSON line per size on rank 0.

Optional comparison point: --torch-dist also measures torch.distributed
(pure RCCL / gloo) on the same sizes.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--min-kb", type=int, default=4)
    ap.add_argument("--max-mb", type=int, default=1024)
    ap.add_argument("--max-kb", type=int, default=0,
                    help="ceiling in KiB (overrides --max-mb when > 0)")
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--device", default="auto")
    ap.add_argument("--algo", default=None)
    ap.add_argument("--torch-dist", action="store_true",
                    help="also measure torch.distributed allreduce")
    ap.add_argument("--out", default=None, help="write JSONL here (rank 0)")
    args = ap.parse_args()

    if args.algo:
        os.environ["MLSL_ALLREDUCE_ALGO"] = args.algo

    use_cuda = False
    torch = None
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
    if use_cuda:
        local = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local % torch.cuda.device_count())

    d = mx.Distribution(size, 1)
    lines = []

    tdist = None
    if args.torch_dist and size >= 1:
        import torch.distributed as td
        backend = "nccl" if use_cuda else "gloo"
        if size > 1:
            td.init_process_group(backend=backend)
        else:
            # world-1 comparison point (RCCL's own degenerate path)
            td.init_process_group(backend=backend, init_method="tcp://127.0.0.1:29712",
                                  rank=0, world_size=1)
        tdist = td

    sizes = []
    b = args.min_kb * 1024
    max_bytes = (args.max_kb * 1024) if args.max_kb > 0 else (args.max_mb * 1024 * 1024)
    while b <= max_bytes:
        sizes.append(b)
        b *= 4

    for nbytes in sizes:
        count = nbytes // 4
        if use_cuda:
            buf = torch.randn(count, device="cuda")
            out = torch.empty_like(buf)
        else:
            import numpy as np
            buf = np.random.randn(count).astype(np.float32)
            out = np.empty_like(buf)

        preq = mx.PersistentRequest(d, "all_reduce", count, dtype="f32",
                                    op="sum", group="data")

        def run():
            preq.start(buf, out)
            preq.wait()

        def sync():
            if use_cuda:
                torch.cuda.synchronize()

        for _ in range(args.warmup):
            run()
        sync()
        d.barrier("global")
        t0 = time.perf_counter()
        for _ in range(args.iters):
            run()
        sync()
        dt = (time.perf_counter() - t0) / args.iters

        rec = {
            "bytes": nbytes,
            "world": size,
            "lat_us": round(dt * 1e6, 2),
            "algbw_GBps": round(nbytes / dt / 1e9, 3),
            "busbw_GBps": round(2 * (size - 1) / max(size, 1) * nbytes / dt / 1e9, 3),
            "impl": "mlsl_amd",
            "algo": os.environ.get("MLSL_ALLREDUCE_ALGO", "auto"),
        }
        if rank == 0:
            print(json.dumps(rec))
            lines.append(rec)
        preq.destroy()

        if tdist is not None:
            # torch's all_reduce is in-place only; compose copy+allreduce so
            # both impls do the same out-of-place semantics (at world 1 the
            # in-place op is a no-op and would not be a comparison at all).
            tb = buf.clone() if use_cuda else buf.copy()

            def trun():
                # same round-trip semantics as preq.start+wait: out-of-place
                # result, host observes completion every iteration (torch's
                # all_reduce only enqueues; without the sync it would be
                # measuring enqueue pipelining against our blocking wait)
                if use_cuda:
                    tb.copy_(buf)
                else:
                    tb[:] = buf
                tdist.all_reduce(tb)
                sync()
            for _ in range(args.warmup):
                trun()
            sync()
            tdist.barrier()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                trun()
            sync()
            dt2 = (time.perf_counter() - t0) / args.iters
            rec2 = dict(rec, impl="torch.distributed", lat_us=round(dt2 * 1e6, 2),
                        algbw_GBps=round(nbytes / dt2 / 1e9, 3),
                        busbw_GBps=round(2 * (size - 1) / max(size, 1) * nbytes / dt2 / 1e9, 3))
            if rank == 0:
                print(json.dumps(rec2))
                lines.append(rec2)

    if rank == 0 and args.out:
        with open(args.out, "w") as f:
            for r in lines:
                f.write(json.dumps(r) + "\n")
    mx.finalize()


if __name__ == "__main__":
    main()
