"""Host-logic unit tests: collective schedule algebra via the C++ simulator."""
import os
import subprocess

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_schedule_selftest():
    exe = os.path.join(REPO, "build", "schedule_selftest")
    if not os.path.exists(exe):
        subprocess.run(["make", "selftest"], cwd=REPO, check=True,
                       capture_output=True, timeout=600)
    out = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "PASSED" in out.stdout


def test_schedule_selftest_asan():
    """Schedule algebra under ASan+UBSan (pure host C++, g++ build)."""
    exe = os.path.join(REPO, "build", "schedule_selftest_asan")
    subprocess.run(["make", "asan"], cwd=REPO, check=True,
                   capture_output=True, timeout=900)
    out = subprocess.run([exe], capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "PASSED" in out.stdout


def test_api_selftest_asan():
    """Full-stack ASan/UBSan: planner + engine + TCP mesh, world 4, two
    grid configs (device kernels stubbed; host paths fully sanitized)."""
    import os
    import subprocess
    from tests.mp import free_port
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    path = os.path.join(repo, "build", "api_selftest_asan")
    if not os.path.exists(path):
        subprocess.run(["make", "asan-api"], cwd=repo, check=True,
                       capture_output=True, timeout=900)
    for mp, du in ((2, 0), (2, 1)):
        port = free_port()
        procs = []
        for r in range(4):
            env = dict(os.environ, RANK=str(r), WORLD_SIZE="4",
                       MASTER_ADDR="127.0.0.1", MLSL_PORT=str(port),
                       MLSL_TRANSPORT="tcp", MP=str(mp), DIST_UPDATE=str(du),
                       ASAN_OPTIONS="detect_leaks=0")
            procs.append(subprocess.Popen([path], env=env, cwd=repo,
                                          stdout=subprocess.PIPE,
                                          stderr=subprocess.STDOUT, text=True))
        for r, p in enumerate(procs):
            out, _ = p.communicate(timeout=180)
            assert p.returncode == 0 and "PASSED" in out, f"rank {r}: {out[-2000:]}"


def test_submit_race_tsan():
    """Concurrent Start() from 4 application threads under TSan: regression
    test for the round-1 multi-producer submit race on the command ring
    (engine.hpp MpscRing). World 2 over TCP, per-thread groups."""
    import os
    import subprocess
    from tests.mp import free_port
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    path = os.path.join(repo, "build", "submit_race_tsan")
    if not os.path.exists(path):
        subprocess.run(["make", "tsan-submit"], cwd=repo, check=True,
                       capture_output=True, timeout=900)
    port = free_port()
    procs = []
    for r in range(2):
        env = dict(os.environ, RANK=str(r), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MLSL_PORT=str(port),
                   MLSL_TRANSPORT="tcp", THREADS="4", ITERS="30",
                   TSAN_OPTIONS="halt_on_error=1")
        procs.append(subprocess.Popen([path], env=env, cwd=repo,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    for r, p in enumerate(procs):
        out, _ = p.communicate(timeout=300)
        assert p.returncode == 0 and "PASSED" in out, f"rank {r}: {out[-3000:]}"
