#!/usr/bin/env python3
"""Flagship benchmark: fp32 AllReduce algorithmic bandwidth (BASELINE.json
metric: "allreduce algo-bandwidth (GB/s) + latency vs msg size, fp32,
1/2/4/8 ranks") on N GPUs of one node over RCCL/xGMI through the mlsl_amd
request engine.

Launched by the driver as either
    python bench.py --gpus 1 --steps K --warmup W
or  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...
(one rank per GPU; RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* read from the env).

One step = one out-of-place AllReduce of a fixed fp32 message (default
256 MiB). Metric definition (pinned; BASELINE.md and docs/BENCHMARKS.md
use the same formulas):

    algbw (the reported `value`) = message_bytes / t_step      [GB/s]
    busbw (auxiliary)            = 2*(n-1)/n * algbw           [GB/s]

t_step = max over ranks. algbw is the standard per-collective algorithmic
bandwidth (nccl-tests convention) — NOT multiplied by N. n_gpus=1
degenerates to the library's local-copy path (algbw = copy bandwidth) and
is reported as such in config.note.
"""
import argparse
import json
import os
import sys
import time


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--mbytes", type=int, default=256, help="message MiB (fp32)")
    ap.add_argument("--device", default="auto", choices=["auto", "cuda", "cpu"])
    ap.add_argument("--algo", default=None, help="MLSL_ALLREDUCE_ALGO override")
    args = ap.parse_args()

    if args.algo:
        os.environ["MLSL_ALLREDUCE_ALGO"] = args.algo

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world == 1 and args.gpus > 1:
        print("ERROR: --gpus > 1 requires torchrun (WORLD_SIZE env)", file=sys.stderr)
        sys.exit(2)

    use_cuda = args.device == "cuda"
    torch = None
    if args.device in ("auto", "cuda"):
        try:
            import torch as _torch
            torch = _torch
            use_cuda = torch.cuda.is_available()
        except ImportError:
            use_cuda = False
    if args.device == "cuda" and not use_cuda:
        print("ERROR: --device cuda but no GPU", file=sys.stderr)
        sys.exit(2)
    if not use_cuda:
        os.environ.setdefault("MLSL_TRANSPORT", "tcp")

    import mlsl_amd as mx
    mx.init()
    rank, size = mx.rank(), mx.world_size()
    count = args.mbytes * 1024 * 1024 // 4

    if use_cuda:
        local = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local % torch.cuda.device_count())
        buf = torch.randn(count, dtype=torch.float32, device="cuda")
        out = torch.empty_like(buf)
    else:
        import numpy as np
        buf = np.random.randn(count).astype(np.float32)
        out = np.empty_like(buf)

    d = mx.Distribution(size, 1)
    # Persistent request: described once, re-started every step (the
    # library's hot-loop contract; avoids per-step planning/allocation).
    preq = mx.PersistentRequest(d, "all_reduce", count, dtype="f32", op="sum",
                                group="data")

    def sync():
        if use_cuda:
            torch.cuda.synchronize()

    def one_step():
        # out-of-place: at n_gpus=1 this is a measured 2x256MiB HBM pass,
        # never a skipped no-op; at n>1 it is the standard allreduce shape.
        preq.start(buf, out)
        preq.wait()

    for _ in range(args.warmup):
        one_step()
    sync()
    d.barrier("global")
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    sync()
    elapsed = time.perf_counter() - t0
    d.barrier("global")
    sync()

    # max over ranks
    if use_cuda:
        tbuf = torch.tensor([elapsed], dtype=torch.float32, device="cuda")
    else:
        import numpy as np
        tbuf = np.array([elapsed], dtype=np.float32)
    mx.wait(d.all_reduce(tbuf, tbuf, 1, op="max", group="data"))
    elapsed_max = float(tbuf[0])

    ms_per_step = elapsed_max / args.steps * 1e3
    msg_bytes = count * 4
    # Pinned metric (see module docstring): algbw = S/t, busbw = 2(n-1)/n * algbw.
    algbw = msg_bytes / (elapsed_max / args.steps) / 1e9
    busbw = (2.0 * (size - 1) / size) * algbw if size > 1 else 0.0

    if rank == 0:
        out = {
            "metric": "allreduce_algbw_GBps",
            "value": round(algbw, 3),
            "unit": "GB/s",
            "n_gpus": size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "allreduce-sweep-headline",
                "busbw_GBps": round(busbw, 3),
                "message_mib": args.mbytes,
                "global_batch": None,
                "seq_len": None,
                "parallelism": f"dp{size}",
                "algo": os.environ.get("MLSL_ALLREDUCE_ALGO", "auto"),
                "note": ("n_gpus=1 is the degenerate local-copy path"
                         if size == 1 else
                         "fused RCCL over xGMI (distinct devices) or the "
                         "IPC window transport (ranks sharing a device)"),
            },
        }
        print(json.dumps(out))
    mx.finalize()


if __name__ == "__main__":
    main()
