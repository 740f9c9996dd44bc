"""Batch exec / copy / train launcher over the fabric.

Reference: /root/reference/python/dglrun/tools/launch.py — cmd types
exec_batch, copy_batch, copy_batch_container, train. The MI355X-native train
differs structurally from the reference: there are NO separate graph-server
or sampler processes (sampling is a HIP kernel, the feature store is GPU HBM
reached over RCCL alltoallv), so "train" is one torchrun per worker pod with
one rank per GPU (slots), rendezvous at worker 0.
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional

from .fabric import Fabric, get_fabric
from .hostfile import HostEntry, parse_hostfile


def exec_batch(hosts: List[HostEntry], command: str,
               fabric: Optional[Fabric] = None,
               env: Optional[Dict[str, str]] = None):
    fabric = fabric or get_fabric()
    errors = []

    def run(h):
        try:
            fabric.exec(h.pod, command, env=env)
        except Exception as e:  # noqa: BLE001
            errors.append((h.pod, e))

    threads = [threading.Thread(target=run, args=(h,), daemon=True) for h in hosts]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    if errors:
        raise RuntimeError(f"exec_batch failures: {errors}")


def copy_batch(hosts: List[HostEntry], local_path: str, remote_path: str,
               fabric: Optional[Fabric] = None, container: Optional[str] = None):
    fabric = fabric or get_fabric()
    for h in hosts:
        fabric.copy(local_path, h.pod, remote_path, container=container)


def train(
    hosts: List[HostEntry],
    script: str,
    script_args: str = "",
    master_port: int = 29400,
    fabric: Optional[Fabric] = None,
    env: Optional[Dict[str, str]] = None,
    num_trainers: Optional[int] = None,
):
    """Launch one torchrun per worker pod; ranks per pod = slots (or the
    dglrun --num-trainers override, reference launch.py:135-152); rendezvous
    at worker 0's pod IP (the operator's hostfile carries the IPs)."""
    fabric = fabric or get_fabric()
    assert hosts, "no workers in hostfile"
    master = hosts[0].ip
    nnodes = len(hosts)
    procs = []
    for i, h in enumerate(hosts):
        nproc = num_trainers if num_trainers else h.slots
        cmd = (
            f"python -m torch.distributed.run --nnodes={nnodes} "
            f"--node-rank={i} --nproc-per-node={nproc} "
            f"--master-addr={master} --master-port={master_port} "
            f"{script} {script_args}"
        )
        procs.append(fabric.exec(h.pod, cmd, env=env, block=False))
    rcs = [p.wait() for p in procs]
    if any(rc != 0 for rc in rcs):
        raise RuntimeError(f"training failed on some nodes: rcs={rcs}")


def main(argv=None):
    import argparse

    p = argparse.ArgumentParser()
    p.add_argument("--cmd_type", required=True,
                   choices=["exec_batch", "copy_batch", "copy_batch_container",
                            "train"])
    p.add_argument("--hostfile", default="/etc/dgl/hostfile")
    p.add_argument("--command", default="")
    p.add_argument("--source", default="")
    p.add_argument("--target", default="")
    p.add_argument("--container", default=None)
    p.add_argument("--master-port", type=int, default=29400)
    p.add_argument("--script", default="")
    p.add_argument("--script-args", default="")
    args = p.parse_args(argv)
    with open(args.hostfile) as f:
        hosts = parse_hostfile(f.read())
    if args.cmd_type == "exec_batch":
        exec_batch(hosts, args.command)
    elif args.cmd_type in ("copy_batch", "copy_batch_container"):
        copy_batch(hosts, args.source, args.target, container=args.container)
    elif args.cmd_type == "train":
        train(hosts, args.script, args.script_args, args.master_port)


if __name__ == "__main__":
    main()
