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
