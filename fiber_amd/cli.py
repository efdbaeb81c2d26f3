"""``fam`` command-line tool.

Single-node reduction of the reference CLI (uber/fiber fiber/cli.py:338-473
``fiber run`` built+pushed docker images and launched k8s master pods):
here ``fam run`` launches a command as a backend job with MI355X device
pinning, ``fam bench`` drives the flagship benchmark, ``fam info`` prints
the node/config view.
"""

import os
import subprocess
import sys

import click


@click.group()
def main():
    """fiber_amd: MI355X-native worker-pool framework."""


@main.command(context_settings={"ignore_unknown_options": True})
@click.option("--gpu", default=0, help="GPUs to pin for the job")
@click.option("--cpu", default=1, help="CPU hint for the job")
@click.argument("command", nargs=-1, required=True)
def run(gpu, cpu, command):
    """Run COMMAND as a backend job (with HIP_VISIBLE_DEVICES pinning)."""
    from . import backend as fam_backend
    from .core import JobSpec

    backend = fam_backend.get_backend()
    spec = JobSpec(command=list(command), name="fam-run", cpu=cpu, gpu=gpu)
    job = backend.create_job(spec)
    code = backend.wait_for_job(job, None)
    sys.exit(code or 0)


@main.command()
@click.option("--gpus", default=1)
@click.option("--steps", default=20)
@click.option("--warmup", default=3)
@click.option("--model", default="mlp",
              type=click.Choice(["mlp", "conv"]))
@click.option("--mode", default="engine",
              type=click.Choice(["engine", "pool"]),
              help="engine: direct; pool: through the framework "
                   "(single-process)")
def bench(gpus, steps, warmup, model, mode):
    """Run the flagship ES benchmark (bench.py)."""
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = os.path.join(root, "bench.py")
    extra = ["--model", model, "--mode", mode]
    if gpus > 1 and mode == "engine":
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", str(gpus),
            "--master-addr", "127.0.0.1", script,
            "--gpus", str(gpus), "--steps", str(steps),
            "--warmup", str(warmup),
        ] + extra
    else:
        cmd = [sys.executable, script, "--gpus", str(gpus),
               "--steps", str(steps), "--warmup", str(warmup)] + extra
    sys.exit(subprocess.call(cmd))


@main.command()
def doctor():
    """Diagnose the node for fiber_amd readiness (devices, IPC mode,
    extensions, a live shm-ring self-test)."""
    import torch

    ok = True

    def check(label, good, detail=""):
        nonlocal ok
        mark = "ok " if good else "FAIL"
        if not good:
            ok = False
        click.echo("[%s] %-34s %s" % (mark, label, detail))

    check("python/torch", True,
          "%s / torch %s" % (sys.version.split()[0], torch.__version__))
    n = torch.cuda.device_count() if torch.cuda.is_available() else 0
    check("MI355X devices visible", True, str(n) or "0 (CPU-only mode)")
    if n:
        check("device 0", True, torch.cuda.get_device_name(0))
    ipc = os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY")
    check("HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC)", ipc == "0",
          repr(ipc) + ("" if ipc == "0"
                       else "  <- RCCL/tensor sharing will fail"))
    try:
        from . import _transport  # noqa: F401

        check("_transport extension", True, "loaded")
        from .transport import ShmRing, new_address

        name = new_address("fam-doctor")
        ring = ShmRing(name, True, 1 << 16, 5.0)
        ring.send(b"ping", 1.0)
        pong = ring.recv(1.0)
        ring.close()
        ring.unlink()
        check("shm ring self-test", pong == b"ping")
    except Exception as exc:  # noqa: BLE001
        check("_transport extension", False, repr(exc))
    try:
        from . import ops

        check("_ops (gfx950) extension", ops.ops_available(),
              "NPARAMS=%d" % ops.NPARAMS if ops.ops_available()
              else "not built — run `python setup.py build_ext --inplace`")
    except Exception as exc:  # noqa: BLE001
        check("_ops (gfx950) extension", False, repr(exc))
    import shutil

    free = shutil.disk_usage("/dev/shm").free if os.path.isdir(
        "/dev/shm") else 0
    check("/dev/shm free", free > (64 << 20), "%.1f GB" % (free / 2**30))
    sys.exit(0 if ok else 1)


@main.command()
def info():
    """Show node, device and config information."""
    import fiber_amd
    from . import config as fam_config

    click.echo("fiber_amd %s" % fiber_amd.__version__)
    click.echo("cpus: %d" % fiber_amd.cpu_count())
    try:
        import torch

        n = torch.cuda.device_count()
        click.echo("gpus: %d" % n)
        for i in range(n):
            click.echo("  cuda:%d %s" % (i, torch.cuda.get_device_name(i)))
    except Exception as exc:
        click.echo("gpus: unavailable (%s)" % exc)
    click.echo("config:")
    for key, value in sorted(fam_config.get_dict().items()):
        click.echo("  %s = %r" % (key, value))


if __name__ == "__main__":
    main()
