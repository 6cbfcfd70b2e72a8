#!/usr/bin/env python3
"""Spawn an N-rank allreduce sweep with ALL RANKS ON ONE GPU (IPC window
transport — RCCL refuses this layout, so this is the harness that measures
the hand-written schedules at n>1 on a single-GPU box).

    python benchmarks/sweep_multirank.py --world 2 --algo ring \
        --max-mb 256 --out gpurun_out/sweep_w2_ring.jsonl

Passes through MLSL_* env (channels, slots, priority) to the ranks.
"""
import argparse
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=2)
    ap.add_argument("--algo", default=None)
    ap.add_argument("--min-kb", type=int, default=4)
    ap.add_argument("--max-mb", type=int, default=256)
    ap.add_argument("--max-kb", type=int, default=0,
                    help="ceiling in KiB (overrides --max-mb when > 0)")
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--out", default=None)
    ap.add_argument("--timeout", type=int, default=600)
    args = ap.parse_args()

    port = free_port()
    cmd = [sys.executable, os.path.join(REPO, "benchmarks", "sweep.py"),
           "--min-kb", str(args.min_kb), "--max-mb", str(args.max_mb),
           "--max-kb", str(args.max_kb),
           "--iters", str(args.iters), "--warmup", str(args.warmup)]
    if args.algo:
        cmd += ["--algo", args.algo]
    rank_log = os.environ.get("SWEEP_RANK_LOG")  # prefix: live per-rank logs
    wrap0 = os.environ.get("SWEEP_WRAP0", "")    # e.g. "rocprofv3 --kernel-trace -d out --"
    procs = []
    logs = []
    for r in range(args.world):
        env = dict(os.environ)
        env.update({"RANK": str(r), "WORLD_SIZE": str(args.world),
                    "MASTER_ADDR": "127.0.0.1", "MLSL_PORT": str(port),
                    "PYTHONPATH": REPO, "MLSL_TIMEOUT": "120",
                    "PYTHONUNBUFFERED": "1"})
        env.pop("MLSL_TRANSPORT", None)
        rcmd = (wrap0.split() + cmd) if (r == 0 and wrap0) else cmd
        if rank_log:
            lf = open(f"{rank_log}.rank{r}.log", "w")
            logs.append(lf)
            procs.append(subprocess.Popen(rcmd, env=env, cwd=REPO, stdout=lf,
                                          stderr=subprocess.STDOUT, text=True))
        else:
            procs.append(subprocess.Popen(
                rcmd, env=env, cwd=REPO, stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT, text=True))
    rc = 0
    out0 = ""
    for r, p in enumerate(procs):
        try:
            out, _ = p.communicate(timeout=args.timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            print(f"TIMEOUT rank {r}", file=sys.stderr)
            sys.exit(3)
        if r == 0:
            out0 = out if out is not None else ""
        if p.returncode != 0:
            rc = p.returncode
            print(f"--- rank {r} rc={p.returncode} ---\n"
                  f"{(out or '(see rank log)')[-3000:]}", file=sys.stderr)
    for lf in logs:
        lf.close()
    if rank_log and not out0:
        with open(f"{rank_log}.rank0.log") as f:
            out0 = f.read()
    lines = [ln for ln in out0.splitlines() if ln.startswith("{")]
    for ln in lines:
        print(ln)
    if args.out and lines:
        with open(args.out, "a") as f:
            for ln in lines:
                f.write(ln + "\n")
    sys.exit(rc)


if __name__ == "__main__":
    main()
