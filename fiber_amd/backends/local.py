"""Single-node backend: jobs are subprocesses, each pinned to MI355X devices.

The reference's local/docker/kubernetes triad (``fiber/local_backend.py``,
``fiber/docker_backend.py``, ``fiber/kubernetes_backend.py``) collapses to
this one backend (SURVEY §1 "MI355X mapping").  GPU placement: a JobSpec
with ``gpu=k`` gets ``k`` device ordinals assigned round-robin from the
node's visible set, exported to the child as ``HIP_VISIBLE_DEVICES`` (and
``CUDA_VISIBLE_DEVICES`` for torch-on-ROCm, which aliases it).
"""

import os
import subprocess
import threading

from .. import config as fam_config
from ..core import Backend as BackendABC
from ..core import Job, ProcessStatus


def _visible_devices():
    conf = fam_config.get_object()
    if conf.devices:
        return [int(x) for x in conf.devices.split(",") if x.strip() != ""]
    env = os.environ.get("HIP_VISIBLE_DEVICES") or os.environ.get(
        "CUDA_VISIBLE_DEVICES"
    )
    if env:
        # The child's HIP_VISIBLE_DEVICES *replaces* (does not nest
        # within) this process's, so hand out the parent's actual
        # ordinals — returning range(len(...)) would pin children to
        # physical GPUs outside the parent's allocation.
        return [int(x) for x in env.split(",") if x.strip() != ""]
    try:
        import torch

        n = torch.cuda.device_count()
    except Exception:
        n = 0
    return list(range(n)) if n else [0]


class Backend(BackendABC):
    name = "local"

    def __init__(self):
        self._lock = threading.Lock()
        self._next_device = 0
        self._job_seq = 0

    def _assign_devices(self, count):
        devices = _visible_devices()
        with self._lock:
            picked = [
                devices[(self._next_device + i) % len(devices)]
                for i in range(count)
            ]
            self._next_device = (self._next_device + count) % len(devices)
        return picked

    def create_job(self, job_spec):
        env = dict(os.environ)
        env.update(job_spec.env)
        devices = job_spec.devices
        if devices is None and job_spec.gpu:
            devices = self._assign_devices(job_spec.gpu)
        if devices is not None:
            dev_str = ",".join(str(d) for d in devices)
            env["HIP_VISIBLE_DEVICES"] = dev_str
            env["CUDA_VISIBLE_DEVICES"] = dev_str
        # Keep dmabuf IPC mode for cross-process HIP tensor sharing.
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        conf = fam_config.get_object()
        log_path = None
        stdout = stderr = None
        if not conf.merge_output:
            # Capture job output so get_job_logs works (reference parity:
            # docker/k8s backends expose container/pod logs).
            with self._lock:
                self._job_seq += 1
                seq = self._job_seq
            log_path = os.path.join(
                conf.ipc_dir,
                "fam-job-%d-%d.log" % (os.getpid(), seq),
            )
            stdout = open(log_path, "ab")
            stderr = subprocess.STDOUT
        else:
            with self._lock:
                self._job_seq += 1
                seq = self._job_seq
        proc = subprocess.Popen(
            job_spec.command, env=env, stdout=stdout, stderr=stderr
        )
        if stdout is not None:
            stdout.close()
        jid = "local-%d-%d" % (proc.pid, seq)
        job = Job(proc, jid)
        job.devices = devices
        job.log_path = log_path
        return job

    def get_job_status(self, job):
        proc = job.data
        code = proc.poll()
        if code is None:
            return ProcessStatus.STARTED
        return ProcessStatus.STOPPED

    def get_job_exitcode(self, job):
        return job.data.poll()

    def get_job_logs(self, job):
        path = getattr(job, "log_path", None)
        if path and os.path.exists(path):
            with open(path, "rb") as fh:
                return fh.read().decode(errors="replace")
        return ""

    def cleanup_job(self, job):
        path = getattr(job, "log_path", None)
        if path:
            try:
                os.unlink(path)
            except OSError:
                pass

    def wait_for_job(self, job, timeout):
        try:
            job.data.wait(timeout=timeout)
        except subprocess.TimeoutExpired:
            return None
        return job.data.returncode

    def terminate_job(self, job):
        if job.data.poll() is None:
            job.data.terminate()

    def kill_job(self, job):
        if job.data.poll() is None:
            job.data.kill()

    def get_listen_addr(self):
        conf = fam_config.get_object()
        return conf.ipc_dir
