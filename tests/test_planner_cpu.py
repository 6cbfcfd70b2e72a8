"""Planner tests: the reference mlsl_test matrix (4 ranks x model_parts
{1,2,4} x dist_update {0,1}, tests/examples/mlsl_test/Makefile:59-107)
natively over the TCP transport, plus AlltoAll layout transitions."""
import pytest

from tests.mp import run_ranks


def test_grid_shapes():
    run_ranks("grid_shapes", 4)


@pytest.mark.parametrize("mp", [1, 2, 4])
@pytest.mark.parametrize("du", [0, 1])
def test_mlsl_net_matrix(mp, du):
    run_ranks("mlsl_net", 4, extra_env={"MP": str(mp), "DIST_UPDATE": str(du)})


def test_mlsl_net_world2():
    run_ranks("mlsl_net", 2, extra_env={"MP": "2", "DIST_UPDATE": "0"})


def test_mlsl_net_stats():
    run_ranks("mlsl_net", 4,
              extra_env={"MP": "2", "DIST_UPDATE": "1", "MLSL_STATS": "1"})


@pytest.mark.parametrize("world", [2, 4])
def test_alltoall_transition(world):
    run_ranks("alltoall_transition", world)


def test_mlsl_net_world1():
    run_ranks("mlsl_net", 1, extra_env={"MP": "1", "DIST_UPDATE": "0"})


def test_case2_allreduce():
    run_ranks("case2_allreduce", 4)


def test_case3_repartition():
    run_ranks("case3_repartition", 4)


@pytest.mark.parametrize("world", [2, 4])
def test_quantized_paramset(world):
    run_ranks("quantized_paramset", world)


def test_stats_log_dump():
    run_ranks("stats_log_dump", 2)
