"""Multi-process test runner: spawns N ranks of a worker function over the
native TCP transport on 127.0.0.1 (the reference's `mpiexec -n N` matrix,
without MPI)."""
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


def run_ranks(worker_name, world, timeout=180, extra_env=None):
    """Run tests.workers.<worker_name>(rank, world) in `world` processes.

    Raises on any nonzero exit; returns list of stdouts.
    """
    port = free_port()
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update({
            "RANK": str(r),
            "WORLD_SIZE": str(world),
            "MASTER_ADDR": "127.0.0.1",
            "MLSL_PORT": str(port),
            "MLSL_TRANSPORT": "tcp",
            "PYTHONPATH": REPO,
        })
        if extra_env:
            env.update(extra_env)
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "tests.workers", worker_name],
            env=env, cwd=REPO,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True))
    outs = []
    failed = []
    for r, p in enumerate(procs):
        try:
            out, _ = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            raise AssertionError(f"worker rank {r} timed out ({worker_name})")
        outs.append(out)
        if p.returncode != 0:
            failed.append((r, p.returncode, out))
    if failed:
        msgs = "\n".join(f"--- rank {r} rc={rc} ---\n{out}" for r, rc, out in failed)
        raise AssertionError(f"{worker_name} world={world} failed:\n{msgs}")
    return outs
