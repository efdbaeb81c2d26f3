"""Examples must stay runnable (CPU plumbing paths)."""

import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, *args, timeout=240):
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    return subprocess.run(
        [sys.executable, os.path.join(ROOT, "examples", script), *args],
        capture_output=True,
        text=True,
        timeout=timeout,
        cwd=ROOT,
        env=env,
    )


class TestExamples:
    def test_pi_estimation(self):
        proc = subprocess.run(
            [
                sys.executable,
                "-c",
                "import sys; sys.path.insert(0, %r); "
                "sys.path.insert(0, %r); "
                "from pi_estimation import main; "
                "pi = main(samples=20000, processes=2); "
                "assert 2.9 < pi < 3.4, pi"
                % (ROOT, os.path.join(ROOT, "examples")),
            ],
            capture_output=True,
            text=True,
            timeout=240,
            cwd=ROOT,
        )
        assert proc.returncode == 0, proc.stderr[-2000:]

    def test_poet_loop(self):
        proc = _run("poet_loop.py", "--pairs", "3", "--generations", "2")
        assert proc.returncode == 0, proc.stderr[-2000:]
        assert "archive best" in proc.stdout

    def test_ring_sgd_cpu(self):
        proc = _run("ring_sgd.py", "--world", "2")
        assert proc.returncode == 0, proc.stderr[-2000:]
        assert "done" in proc.stdout


class TestCli:
    def test_doctor(self):
        proc = subprocess.run(
            [sys.executable, "-m", "fiber_amd.cli", "doctor"],
            capture_output=True,
            text=True,
            timeout=120,
            cwd=ROOT,
        )
        assert proc.returncode == 0, proc.stdout + proc.stderr[-500:]
        assert "shm ring self-test" in proc.stdout
        assert "FAIL" not in proc.stdout

    def test_info(self):
        proc = subprocess.run(
            [sys.executable, "-m", "fiber_amd.cli", "info"],
            capture_output=True,
            text=True,
            timeout=120,
            cwd=ROOT,
        )
        assert proc.returncode == 0, proc.stderr[-2000:]
        assert "fiber_amd" in proc.stdout
        assert "config:" in proc.stdout

    def test_run_pins_devices(self):
        proc = subprocess.run(
            [
                sys.executable, "-m", "fiber_amd.cli", "run", "--gpu", "1",
                "--", sys.executable, "-c",
                "import os; print(os.environ.get('HIP_VISIBLE_DEVICES'))",
            ],
            capture_output=True,
            text=True,
            timeout=120,
            cwd=ROOT,
        )
        assert proc.returncode == 0, proc.stderr[-2000:]
        assert proc.stdout.strip() != "None"
