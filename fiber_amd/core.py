"""Job model + backend ABC.

Parity with reference ``fiber/core.py:28-113`` (JobSpec / Job / Backend),
reshaped for a single MI355X node: JobSpec carries a GPU *device list*
instead of a container image, and ``get_listen_addr`` returns an IPC
directory rather than a TCP interface.
"""


class ProcessStatus:
    INITIAL = "initial"
    STARTED = "started"
    STOPPED = "stopped"


class JobSpec:
    def __init__(
        self,
        command=None,
        name=None,
        cpu=None,
        gpu=None,
        mem=None,
        env=None,
        devices=None,
    ):
        self.command = command
        self.name = name
        self.cpu = cpu
        self.gpu = gpu
        self.mem = mem
        self.env = dict(env or {})
        # Explicit MI355X device ordinals to pin (HIP_VISIBLE_DEVICES).
        self.devices = devices

    def __repr__(self):
        return "JobSpec(name=%r, gpu=%r, devices=%r)" % (
            self.name,
            self.gpu,
            self.devices,
        )


class Job:
    def __init__(self, data, jid):
        self.data = data  # backend-specific handle (e.g. subprocess.Popen)
        self.jid = jid
        self.host = "127.0.0.1"


class Backend:
    """Abstract job backend (create/status/logs/wait/terminate)."""

    name = None

    def create_job(self, job_spec):
        raise NotImplementedError

    def get_job_status(self, job):
        raise NotImplementedError

    def get_job_logs(self, job):
        raise NotImplementedError

    def wait_for_job(self, job, timeout):
        raise NotImplementedError

    def terminate_job(self, job):
        raise NotImplementedError

    def get_listen_addr(self):
        raise NotImplementedError
