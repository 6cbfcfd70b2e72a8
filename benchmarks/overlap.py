#!/usr/bin/env python3
"""Driver config 3: ResNet-50-shaped gradient buckets (~25.5M params fp32)
non-blocking AllReduce overlapped with a dummy backward, measuring exposed
communication time. Launch with torchrun for N>1.

Reports: compute-only time, comm-only time, overlapped step time, exposed
comm = overlapped - compute.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--bucket-mb", type=int, default=25)
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
    from mlsl_amd.models.synthetic import resnet50_buckets
    from mlsl_amd.parallel import GradBucketer

    mx.init()
    rank, size = mx.rank(), mx.world_size()
    if use_cuda:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) %
                              torch.cuda.device_count())

    counts = resnet50_buckets(bucket_mb=args.bucket_mb)
    if use_cuda:
        bufs = [torch.randn(c, device="cuda") for c in counts]
        # dummy backward: one GEMM per bucket (keeps the GPU busy per layer)
        a = torch.randn(2048, 2048, device="cuda")
        b = torch.randn(2048, 2048, device="cuda")

        def compute_layer():
            return a @ b
    else:
        import numpy as np
        bufs = [np.random.randn(c).astype(np.float32) for c in counts]
        a = np.random.randn(256, 256).astype(np.float32)
        b = np.random.randn(256, 256).astype(np.float32)

        def compute_layer():
            return a @ b

    d = mx.Distribution(size, 1)
    # out-of-place buckets: at world 1 an in-place allreduce is an identity
    # no-op, which would make "comm" vacuous — out-of-place keeps real data
    # motion at every world size (2x bucket bytes of HBM at n=1).
    if use_cuda:
        outs = [torch.empty_like(b) for b in bufs]
    else:
        outs = [np.empty_like(b) for b in bufs]
    buck = GradBucketer(d, bufs, outputs=outs)

    def sync():
        if use_cuda:
            torch.cuda.synchronize()

    def run(do_compute, do_comm):
        for i in reversed(range(len(bufs))):
            if do_compute:
                compute_layer()
            if do_comm:
                buck.start(i)
        if do_comm:
            buck.wait_all()
        sync()

    def measure(do_compute, do_comm):
        for _ in range(args.warmup):
            run(do_compute, do_comm)
        d.barrier("global")
        t0 = time.perf_counter()
        for _ in range(args.iters):
            run(do_compute, do_comm)
        return (time.perf_counter() - t0) / args.iters

    t_comp = measure(True, False)
    t_comm = measure(False, True)
    t_both = measure(True, True)

    if rank == 0:
        total_bytes = sum(counts) * 4
        print(json.dumps({
            "config": "resnet50-buckets-overlap",
            "world": size,
            "buckets": len(counts),
            "grad_bytes": total_bytes,
            "compute_ms": round(t_comp * 1e3, 3),
            "comm_ms": round(t_comm * 1e3, 3),
            "overlapped_ms": round(t_both * 1e3, 3),
            "exposed_comm_ms": round((t_both - t_comp) * 1e3, 3),
            "overlap_efficiency": round(
                1.0 - max(t_both - t_comp, 0.0) / max(t_comm, 1e-9), 3),
        }))
    mx.finalize()


if __name__ == "__main__":
    main()
