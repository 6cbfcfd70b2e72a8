#!/usr/bin/env python3
"""Generic multi-rank-on-one-box launcher for the benchmark scripts:

    python benchmarks/mp_run.py --world 2 -- python benchmarks/overlap.py ...

Sets the bootstrap env (RANK/WORLD_SIZE/MASTER_ADDR/MLSL_PORT), leaves
MLSL_TRANSPORT unset (device mode on a GPU box; all ranks on one device
ride the IPC window transport), prints rank 0's stdout."""
import argparse
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=2)
    ap.add_argument("--timeout", type=int, default=300)
    ap.add_argument("cmd", nargs=argparse.REMAINDER)
    args = ap.parse_args()
    cmd = args.cmd
    if cmd and cmd[0] == "--":
        cmd = cmd[1:]
    if not cmd:
        print("no command", file=sys.stderr)
        sys.exit(2)

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = []
    for r in range(args.world):
        env = dict(os.environ)
        env.update({"RANK": str(r), "WORLD_SIZE": str(args.world),
                    "MASTER_ADDR": "127.0.0.1", "MLSL_PORT": str(port),
                    "PYTHONPATH": REPO, "PYTHONUNBUFFERED": "1",
                    "MLSL_TIMEOUT": "120"})
        env.pop("MLSL_TRANSPORT", None)
        procs.append(subprocess.Popen(cmd, env=env, cwd=REPO,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    rc = 0
    for r, p in enumerate(procs):
        try:
            out, _ = p.communicate(timeout=args.timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            print(f"TIMEOUT rank {r}", file=sys.stderr)
            sys.exit(3)
        if r == 0:
            print(out, end="")
        if p.returncode != 0:
            rc = p.returncode
            print(f"--- rank {r} rc={p.returncode} ---\n{out[-3000:]}",
                  file=sys.stderr)
    sys.exit(rc)


if __name__ == "__main__":
    main()
