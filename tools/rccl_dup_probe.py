#!/usr/bin/env python3
"""Probe: does RCCL accept 2 ranks in one communicator on ONE device?

Spawns world=2 through the mlsl_amd stack with both ranks on cuda:0
(device transport, fused RCCL path) and runs one small allreduce.
Prints DUP_OK / DUP_FAIL plus the error. Run on a GPU box:
    timeout 120 python tools/rccl_dup_probe.py
"""
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import os, sys
import torch
import mlsl_amd as mx
mx.init()
torch.cuda.set_device(0)
t = torch.ones(1024, dtype=torch.float32, device="cuda")
o = torch.empty_like(t)
d = mx.Distribution(mx.world_size(), 1)
mx.wait(d.all_reduce(t, o, 1024, op="sum", group="data"))
torch.cuda.synchronize()
assert o[0].item() == mx.world_size(), o[0].item()
print("RANK_OK", mx.rank())
mx.finalize()
"""


def main():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = []
    for r in range(2):
        env = dict(os.environ, RANK=str(r), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MLSL_PORT=str(port),
                   PYTHONPATH=REPO, MLSL_TIMEOUT="60")
        procs.append(subprocess.Popen([sys.executable, "-c", WORKER], env=env,
                                      cwd=REPO, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    ok = True
    for r, p in enumerate(procs):
        try:
            out, _ = p.communicate(timeout=100)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            out = "(timed out)"
        print(f"--- rank {r} rc={p.returncode} ---\n{out[-2000:]}")
        ok = ok and p.returncode == 0
    print("DUP_OK" if ok else "DUP_FAIL")


if __name__ == "__main__":
    main()
