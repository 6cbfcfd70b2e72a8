"""Run the migration samples (C++ / C / Python) at world 2 over TCP —
driver config 1, through every language binding."""
import os
import subprocess

import pytest

from tests.mp import REPO, free_port, run_ranks


def _run_binary(path, world=2):
    if not os.path.exists(path):
        subprocess.run(["make", "samples"], cwd=REPO, check=True,
                       capture_output=True, timeout=600)
    port = free_port()
    procs = []
    for r in range(world):
        env = dict(os.environ, RANK=str(r), WORLD_SIZE=str(world),
                   MASTER_ADDR="127.0.0.1", MLSL_PORT=str(port),
                   MLSL_TRANSPORT="tcp")
        procs.append(subprocess.Popen([path], env=env, cwd=REPO,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    for r, p in enumerate(procs):
        out, _ = p.communicate(timeout=120)
        assert p.returncode == 0, f"rank {r}: {out}"
        assert "PASSED" in out


def test_cpp_sample():
    _run_binary(os.path.join(REPO, "build", "mlsl_sample"))


def test_c_sample():
    _run_binary(os.path.join(REPO, "build", "cmlsl_sample"))


def test_python_sample():
    run_ranks("py_sample", 2)


def test_python_rma_sample():
    run_ranks("py_rma_sample", 2)


import pytest


@pytest.mark.parametrize("mp,du", [(1, 0), (2, 0), (4, 1), (2, 1)])
def test_cpp_api_selftest(mp, du):
    """Full C++ API surface (Session/Operation/Activation/ParameterSet) at
    world 4 over TCP, reference run-matrix style."""
    path = os.path.join(REPO, "build", "api_selftest")
    if not os.path.exists(path):
        subprocess.run(["make", "apitest"], cwd=REPO, check=True,
                       capture_output=True, timeout=600)
    port = free_port()
    procs = []
    world = 4
    for r in range(world):
        env = dict(os.environ, RANK=str(r), WORLD_SIZE=str(world),
                   MASTER_ADDR="127.0.0.1", MLSL_PORT=str(port),
                   MLSL_TRANSPORT="tcp", MP=str(mp), DIST_UPDATE=str(du))
        procs.append(subprocess.Popen([path], env=env, cwd=REPO,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    for r, p in enumerate(procs):
        out, _ = p.communicate(timeout=120)
        assert p.returncode == 0, f"rank {r}: {out}"
        assert "PASSED" in out


def test_train_ddp_sample():
    run_ranks("train_ddp_sample", 2, timeout=240)
