"""`colossalai_amd` CLI (reference: colossalai/cli — `colossalai run` / `check`).

`colossalai_amd run --nproc_per_node 8 train.py ...` wraps
torch.distributed.run on this node; `--hostfile` fans the same command out
over SSH for multi-node (one torchrun per host, rendezvous on MASTER_ADDR).
"""

import os
import socket
import subprocess
import sys

import click


@click.group()
def cli():
    pass


@cli.command(context_settings=dict(ignore_unknown_options=True))
@click.option("--nproc_per_node", "--nproc-per-node", type=int, default=1)
@click.option("--nnodes", type=int, default=1)
@click.option("--node_rank", type=int, default=0)
@click.option("--master_addr", type=str, default="127.0.0.1")
@click.option("--master_port", type=int, default=29500)
@click.option("--hostfile", type=str, default=None, help="one hostname per line; SSH fan-out")
@click.option("--ssh-port", type=int, default=22)
@click.argument("script", nargs=1)
@click.argument("script_args", nargs=-1, type=click.UNPROCESSED)
def run(nproc_per_node, nnodes, node_rank, master_addr, master_port, hostfile, ssh_port, script, script_args):
    """Launch a distributed training script (one process per GPU over RCCL)."""
    if hostfile:
        with open(hostfile) as f:
            hosts = [h.strip() for h in f if h.strip() and not h.startswith("#")]
        master_addr = hosts[0] if master_addr == "127.0.0.1" else master_addr
        procs = []
        for rank, host in enumerate(hosts):
            cmd = (
                f"cd {os.getcwd()} && {sys.executable} -m torch.distributed.run "
                f"--nnodes {len(hosts)} --node_rank {rank} --nproc-per-node {nproc_per_node} "
                f"--master-addr {master_addr} --master-port {master_port} {script} {' '.join(script_args)}"
            )
            if host in ("localhost", "127.0.0.1", socket.gethostname()):
                procs.append(subprocess.Popen(cmd, shell=True))
            else:
                procs.append(subprocess.Popen(["ssh", "-p", str(ssh_port), host, cmd]))
        rc = 0
        for p in procs:
            rc |= p.wait()
        sys.exit(rc)

    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes", str(nnodes),
        "--node_rank", str(node_rank),
        "--nproc-per-node", str(nproc_per_node),
        "--master-addr", master_addr,
        "--master-port", str(master_port),
        script, *script_args,
    ]
    os.execv(sys.executable, cmd)


@cli.command()
@click.option("-i", "--installation", is_flag=True, default=True)
def check(installation):
    """Environment sanity check (ROCm, RCCL, kernels)."""
    import torch

    print(f"torch: {torch.__version__}")
    print(f"hip: {torch.version.hip}")
    print(f"gpus visible: {torch.cuda.device_count() if torch.cuda.is_available() else 0}")
    try:
        from colossalai_amd import _C  # noqa: F401

        print("colossalai_amd._C (gfx950 HIP kernels): OK")
    except ImportError as e:
        print(f"colossalai_amd._C: MISSING ({e}) — run `python setup.py build_ext --inplace`")
    import colossalai_amd

    print(f"colossalai_amd: {colossalai_amd.__version__}")


if __name__ == "__main__":
    cli()
