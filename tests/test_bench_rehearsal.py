"""Dress rehearsal of the driver's SCALE invocation (VERDICT r1 #3:
"when the driver gets an 8-GPU node, the curve exists on the first
try").  Runs the EXACT torchrun command shape the driver uses — nnodes
1, nproc-per-node N, master-addr 127.0.0.1 — against bench.py with
FAM_BENCH_CPU=1 (HIP kernels stubbed, gloo collectives), and validates
the emitted JSON contract."""

import json
import os
import socket
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class TestBenchRehearsal:
    def test_driver_invocation_world2(self):
        env = dict(os.environ)
        env["FAM_BENCH_CPU"] = "1"
        env["FAM_PG_TIMEOUT"] = "60"
        proc = subprocess.run(
            [
                sys.executable, "-m", "torch.distributed.run",
                "--nnodes=1", "--nproc-per-node", "2",
                "--master-addr", "127.0.0.1",
                "--master-port", str(_free_port()),
                os.path.join(ROOT, "bench.py"),
                "--gpus", "2", "--steps", "2", "--warmup", "1",
                "--pop-per-gpu", "8", "--horizon", "4",
            ],
            capture_output=True, text=True, timeout=420, cwd=ROOT, env=env,
        )
        assert proc.returncode == 0, proc.stderr[-3000:]
        # exactly ONE JSON line, from rank 0
        lines = [ln for ln in proc.stdout.splitlines()
                 if ln.startswith("{")]
        assert len(lines) == 1, proc.stdout
        d = json.loads(lines[0])
        assert d["metric"] == "es_rollouts_per_sec"
        assert d["n_gpus"] == 2
        assert d["steps"] == 2 and d["warmup"] == 1
        assert d["config"]["parallelism"] == "dp2"
        assert d["config"]["global_batch"] == 8 * 2 * 64
        assert d["value"] > 0 and d["ms_per_step"] > 0
        assert d["scaling"] == "weak"

    def test_single_rank_default_contract(self):
        env = dict(os.environ)
        env["FAM_BENCH_CPU"] = "1"
        proc = subprocess.run(
            [sys.executable, os.path.join(ROOT, "bench.py"),
             "--steps", "2", "--warmup", "1",
             "--pop-per-gpu", "8", "--horizon", "4"],
            capture_output=True, text=True, timeout=300, cwd=ROOT, env=env,
        )
        assert proc.returncode == 0, proc.stderr[-3000:]
        d = json.loads([ln for ln in proc.stdout.splitlines()
                        if ln.startswith("{")][0])
        assert d["n_gpus"] == 1
        assert d["higher_is_better"] is True
