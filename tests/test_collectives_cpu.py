"""Multi-process CPU (TCP transport) collective tests — the reference's
`mpiexec -n N ./mlsl_test` matrix re-done natively on 127.0.0.1."""
import pytest

from tests.mp import run_ranks


@pytest.mark.parametrize("world", [1, 2, 4])
def test_plumbing_allreduce(world):
    # driver config 1: mlsl_sample AllReduce COUNT=128 fp32
    run_ranks("plumbing_allreduce", world)


@pytest.mark.parametrize("world", [1, 2, 3, 4, 8])
def test_collectives_sweep(world):
    run_ranks("collectives_sweep", world)


def test_hybrid_grid():
    run_ranks("hybrid_grid", 4)


@pytest.mark.parametrize("world", [2])
def test_inline_progress_mode(world):
    run_ranks("collectives_sweep", world, extra_env={"MLSL_PROGRESS": "inline"})


@pytest.mark.parametrize("world", [2, 4])
def test_priority_concurrent(world):
    run_ranks("priority_and_inline", world,
              extra_env={"MLSL_MSG_PRIORITY": "1", "MLSL_MSG_PRIORITY_THRESHOLD": "4000"})


@pytest.mark.parametrize("world", [2, 3, 4, 8])
def test_chunked_channels(world):
    # exercise the chunk-over-channels fan-out (endpoint-parallelism analog)
    # across ALL ops — including the block-chunked RS/AG(v)/AlltoAll(v)
    # builders (reference endpoint split, src/comm_ep.cpp:598-736)
    run_ranks("collectives_sweep", world,
              extra_env={"MLSL_NUM_CHANNELS": "4", "MLSL_LARGE_MSG_SIZE_MB": "0"})


def test_rhd_algo():
    run_ranks("collectives_sweep", 4, extra_env={"MLSL_ALLREDUCE_ALGO": "rhd"})


def test_ring_algo():
    run_ranks("collectives_sweep", 3, extra_env={"MLSL_ALLREDUCE_ALGO": "ring"})


@pytest.mark.parametrize("world", [1, 2, 4, 8])
def test_quantized_allreduce(world):
    run_ranks("quantized_allreduce", world)


@pytest.mark.parametrize("world", [1, 2])
def test_compat_shim(world):
    run_ranks("compat_shim", world)


def test_configure_tenants():
    run_ranks("configure_tenants", 4)


@pytest.mark.parametrize("world", [2, 4, 8])
def test_rma_window(world):
    run_ranks("rma_window", world)


def test_fork_safety():
    run_ranks("fork_safety", 2)


def test_fault_peer_death():
    # rank 1 exits deliberately; rank 0 must fail fast, not hang
    import subprocess, sys, os
    from tests.mp import REPO, free_port
    port = free_port()
    procs = []
    for r in range(2):
        env = dict(os.environ, RANK=str(r), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MLSL_PORT=str(port),
                   MLSL_TRANSPORT="tcp", MLSL_TIMEOUT="10")
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "tests.workers", "fault_peer_death"],
            env=env, cwd=REPO, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, text=True))
    out0, _ = procs[0].communicate(timeout=60)
    procs[1].communicate(timeout=10)
    assert procs[0].returncode == 0, out0
    assert "expected failure" in out0


def test_bootstrap_accept_timeout():
    # rank 0 of a claimed world of 2, with no rank 1 ever dialing in:
    # Init must fail within MLSL_TIMEOUT (accept-side deadline), not hang.
    import subprocess, sys, os, time
    from tests.mp import REPO, free_port
    env = dict(os.environ, RANK="0", WORLD_SIZE="2",
               MASTER_ADDR="127.0.0.1", MLSL_PORT=str(free_port()),
               MLSL_TRANSPORT="tcp", MLSL_TIMEOUT="3")
    t0 = time.time()
    p = subprocess.run(
        [sys.executable, "-c",
         "import mlsl_amd as mx\n"
         "from mlsl_amd import MlslError\n"
         "try:\n"
         "    mx.init()\n"
         "except MlslError as e:\n"
         "    assert 'timeout' in str(e), e\n"
         "    print('OK timed out')\n"
         "else:\n"
         "    raise SystemExit('init unexpectedly succeeded')\n"],
        env=env, cwd=REPO, capture_output=True, text=True, timeout=30)
    assert p.returncode == 0, p.stdout + p.stderr
    assert "OK timed out" in p.stdout
    assert time.time() - t0 < 25


@pytest.mark.parametrize("world", [2])
def test_ddp_wrapper(world):
    run_ranks("ddp_wrapper", world, timeout=300)


@pytest.mark.parametrize("world", [2, 4])
def test_seqpar_reshard(world):
    run_ranks("seqpar_reshard", world)


def test_pointer_checker():
    run_ranks("pointer_checker", 2, extra_env={"MLSL_CHECK_POINTERS": "1"})


@pytest.mark.parametrize("world", [1, 2, 3])
def test_edge_cases(world):
    run_ranks("edge_cases", world)


@pytest.mark.parametrize("world", [2, 4])
def test_srlist_ring(world):
    run_ranks("srlist_ring", world)


@pytest.mark.parametrize("world", [2, 4])
def test_stress_random(world):
    run_ranks("stress_random", world, timeout=300)


@pytest.mark.parametrize("world", [2, 3])
def test_quant_plugin_dlopen(world):
    run_ranks("quant_plugin", world)


@pytest.mark.parametrize("world", [1, 2, 4])
def test_zero1_sharded_optimizer(world):
    run_ranks("zero1_sharded_opt", world)


@pytest.mark.parametrize("world", [2, 4])
def test_inplace_collectives(world):
    run_ranks("inplace_collectives", world)


@pytest.mark.parametrize("world", [2, 4])
def test_torch_distributed_backend(world):
    run_ranks("torch_backend", world)


@pytest.mark.parametrize("world", [2, 4])
def test_p2p_asymmetric(world):
    run_ranks("p2p_asymmetric", world)


@pytest.mark.parametrize("world", [2, 4])
def test_p2p_stress(world):
    run_ranks("p2p_stress", world, timeout=240)


def test_direct_algo():
    # one-shot exchange allreduce (full-mesh latency algorithm)
    run_ranks("collectives_sweep", 3,
              extra_env={"MLSL_ALLREDUCE_ALGO": "direct"})
