"""Multi-rank-on-one-GPU tests: worlds 2 and 4 with every rank on device 0.

RCCL refuses this layout ("Duplicate GPU detected"), so the runtime
auto-selects the IPC window transport (csrc/comm/p2p_transport.cpp) — these
worlds execute the exact device code (schedules, flag kernels, slot
backpressure, lanes, quantized ring) that runs over xGMI on an 8-GPU node.
Reference bar: the 4-rank matrix of
/root/reference/tests/examples/mlsl_test/Makefile:59-107 on real transport.
"""
import os
import subprocess
import sys

import pytest

torch = pytest.importorskip("torch")

from tests.mp import free_port  # noqa: E402

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_gpu_ranks(worker, world, timeout=240, extra_env=None):
    port = free_port()
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update({
            "RANK": str(r),
            "WORLD_SIZE": str(world),
            "MASTER_ADDR": "127.0.0.1",
            "MLSL_PORT": str(port),
            "PYTHONPATH": REPO,
            "MLSL_TIMEOUT": "90",
        })
        env.pop("MLSL_TRANSPORT", None)  # device transport, not TCP
        if extra_env:
            env.update(extra_env)
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "tests.workers_gpu", worker],
            env=env, cwd=REPO,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True))
    outs, failed = [], []
    for r, p in enumerate(procs):
        try:
            out, _ = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            raise AssertionError(f"gpu worker rank {r} timed out ({worker})")
        outs.append(out)
        if p.returncode != 0:
            failed.append((r, p.returncode, out))
    if failed:
        msgs = "\n".join(f"--- rank {r} rc={rc} ---\n{out[-3000:]}"
                         for r, rc, out in failed)
        raise AssertionError(f"{worker} world={world} failed:\n{msgs}")
    return outs


@requires_gpu
@pytest.mark.parametrize("world", [2, 4, 8])
def test_collectives_multirank(world):
    # world 8 = the target node's group size: 7-peer fan kernels, the
    # full stride set {1,3,5,7}, RHD at log2(8) depth — all on one device
    run_gpu_ranks("gpu_collectives", world, timeout=360)


@requires_gpu
@pytest.mark.parametrize("algo", ["ring", "rhd"])
def test_allreduce_algos_multislot(algo):
    # 1 MiB slots force 8 sub-messages per 8 MiB segment: slot wrap +
    # backpressure + the interleaved phase loop all engage.
    run_gpu_ranks("gpu_allreduce_multislot", 2,
                  extra_env={"MLSL_ALLREDUCE_ALGO": algo,
                             "MLSL_P2P_SLOT_MB": "1"})


@requires_gpu
def test_allreduce_channels():
    run_gpu_ranks("gpu_allreduce_multislot", 2,
                  extra_env={"MLSL_NUM_CHANNELS": "2"})


@requires_gpu
def test_allreduce_priority_lane():
    run_gpu_ranks("gpu_allreduce_multislot", 2,
                  extra_env={"MLSL_MSG_PRIORITY": "1"})


@requires_gpu
def test_quant_allreduce_multirank():
    run_gpu_ranks("gpu_quant_allreduce", 2)


@requires_gpu
@pytest.mark.parametrize("world", [4, 8])
def test_hybrid_grid_multirank(world):
    run_gpu_ranks("gpu_hybrid_grid", world)


@requires_gpu
def test_stats_device_ns_multirank():
    run_gpu_ranks("gpu_stats_device_ns", 2)


@requires_gpu
def test_configure_tenants_device():
    run_gpu_ranks("gpu_configure_tenants", 4)


@requires_gpu
@pytest.mark.parametrize("world", [2, 4])
def test_rma_window_device(world):
    run_gpu_ranks("gpu_rma_window", world)


@requires_gpu
@pytest.mark.parametrize("mp,du,user_buf,quant", [
    (1, 0, 0, 0), (2, 1, 0, 0), (1, 0, 1, 0), (1, 0, 0, 1),
])
def test_e2e_device_multirank(mp, du, user_buf, quant):
    _e2e_device(4, mp, du, user_buf, quant)


@requires_gpu
def test_e2e_device_world8():
    # the target node's group size: dp=4 x mp=2 planner matrix on device
    _e2e_device(8, 2, 1, 0, 0)
    _e2e_device(8, 1, 0, 0, 1)


def _e2e_device(world, mp, du, user_buf, quant):
    """mlsl_test-equivalent epoch loop on the DEVICE engine (p2p transport,
    multi-rank-one-GPU): session planner + pack blocks + grad/inc exchange,
    with the user_buf toggle exercising pinned-host staging."""
    exe = os.path.join(REPO, "build", "mlsl_e2e")
    if not os.path.exists(exe):
        subprocess.run(["make", "e2e"], cwd=REPO, check=True,
                       capture_output=True, timeout=900)
    port = free_port()
    procs = []
    for r in range(world):
        env = dict(os.environ, RANK=str(r), WORLD_SIZE=str(world),
                   MASTER_ADDR="127.0.0.1", MLSL_PORT=str(port),
                   MP=str(mp), DIST_UPDATE=str(du), USER_BUF=str(user_buf),
                   QUANT=str(quant), MLSL_TIMEOUT="90")
        env.pop("MLSL_TRANSPORT", None)
        procs.append(subprocess.Popen([exe], env=env, cwd=REPO,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    for r, p in enumerate(procs):
        out, _ = p.communicate(timeout=240)
        assert p.returncode == 0 and "PASSED" in out, f"rank {r}: {out[-2500:]}"


@requires_gpu
def test_stress_multirank():
    """Randomized mixed-collective soak at world 2 on one device
    (STRESS_SECONDS env scales it; default 30 s)."""
    secs = os.environ.get("STRESS_SECONDS", "30")
    run_gpu_ranks("gpu_stress", 2, timeout=int(secs) + 150,
                  extra_env={"STRESS_SECONDS": secs})


@requires_gpu
def test_ddp_multirank():
    """Bucketed DDP (autograd hooks + non-blocking allreduce) at world 2."""
    run_gpu_ranks("gpu_ddp", 2)


@requires_gpu
def test_zero1_multirank():
    run_gpu_ranks("gpu_zero1", 2)


@requires_gpu
def test_seqpar_multirank():
    run_gpu_ranks("gpu_seqpar", 2)


@requires_gpu
@pytest.mark.parametrize("world", [2, 4])
def test_rs_overlap_multirank(world):
    run_gpu_ranks("gpu_rs_overlap", world)


@requires_gpu
def test_hybrid_rs_multirank():
    run_gpu_ranks("gpu_hybrid_rs", 8, timeout=300)
