#!/usr/bin/env python3
"""Driver config 5: int8-quantized allreduce of bf16/f32 gradients
(default 256 MiB) vs the uncompressed path — bandwidth + quantization error.
Launch with torchrun for N>1."""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mbytes", type=int, default=256)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "f32"])
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--device", default="auto")
    args = ap.parse_args()

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
    es = 2 if args.dtype == "bf16" else 4
    count = args.mbytes * 1024 * 1024 // es

    if use_cuda:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) %
                              torch.cuda.device_count())
        tdt = torch.bfloat16 if args.dtype == "bf16" else torch.float32
        g = (torch.randn(count, dtype=torch.float32, device="cuda") * 0.01).to(tdt)
        out_q = torch.empty_like(g)
        out_x = torch.empty_like(g)
    else:
        import numpy as np
        if args.dtype == "bf16":
            print("bf16 on CPU host path unsupported; use --dtype f32",
                  file=sys.stderr)
            args.dtype = "f32"
            es = 4
            count = args.mbytes * 1024 * 1024 // es
        g = (np.random.randn(count) * 0.01).astype(np.float32)
        out_q = np.empty_like(g)
        out_x = np.empty_like(g)

    d = mx.Distribution(size, 1)
    qreq = mx.PersistentRequest(d, "all_reduce", count, dtype=args.dtype,
                                op="sum", group="data", quantized=True)
    xreq = mx.PersistentRequest(d, "all_reduce", count, dtype=args.dtype,
                                op="sum", group="data")

    def sync():
        if use_cuda:
            torch.cuda.synchronize()

    def run(req, out):
        req.start(g, out)
        req.wait()

    results = {}
    for name, req, out in (("quantized", qreq, out_q), ("exact", xreq, out_x)):
        for _ in range(args.warmup):
            run(req, out)
        sync()
        d.barrier("global")
        t0 = time.perf_counter()
        for _ in range(args.iters):
            run(req, out)
        sync()
        dt = (time.perf_counter() - t0) / args.iters
        results[name] = {"ms": round(dt * 1e3, 3),
                         "algbw_GBps": round(count * es / dt / 1e9, 2)}

    # error of the quantized result vs the exact sum
    if use_cuda:
        err = (out_q.float() - out_x.float()).abs().max().item()
        step = out_x.float().abs().max().item() / 127.0
    else:
        import numpy as np
        err = float(np.abs(out_q - out_x).max())
        step = float(np.abs(out_x).max()) / 127.0

    if rank == 0:
        print(json.dumps({
            "config": "int8-quantized-allreduce",
            "dtype": args.dtype,
            "message_mib": args.mbytes,
            "world": size,
            **{f"{k}_{kk}": vv for k, v in results.items() for kk, vv in v.items()},
            "wire_compression": round(es / ((256 + 8) / 256), 2),
            "max_abs_err": round(err, 6),
            "per_block_step": round(step, 6),
        }))
    qreq.destroy()
    xreq.destroy()
    mx.finalize()


if __name__ == "__main__":
    main()
