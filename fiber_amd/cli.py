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
def bench(gpus, steps, warmup):
    """Run the flagship ES benchmark (bench.py)."""
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = os.path.join(root, "bench.py")
    if gpus > 1:
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", str(gpus),
            "--master-addr", "127.0.0.1", script,
            "--gpus", str(gpus), "--steps", str(steps),
            "--warmup", str(warmup),
        ]
    else:
        cmd = [sys.executable, script, "--steps", str(steps),
               "--warmup", str(warmup)]
    sys.exit(subprocess.call(cmd))


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
