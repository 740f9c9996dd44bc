"""Batch exec / copy / train launcher over the fabric.

Reference: /root/reference/python/dglrun/tools/launch.py — cmd types
exec_batch, copy_batch, copy_batch_container, train. The MI355X-native train
differs structurally from the reference: there are NO separate graph-server
or sampler processes (sampling is a HIP kernel, the feature store is GPU HBM
reached over RCCL alltoallv), so "train" is one torchrun per worker pod with
one rank per GPU (slots), rendezvous at worker 0.
"""
from __future__ import annotations

import os
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

    # accepts BOTH this repo's spellings and the reference launch.py
    # invocation contract (python/dglrun/tools/launch.py:158-188 +
    # the call sites in exec/dglrun:202-234 and exec/dglkerun:190-233):
    # the command is POSITIONAL there, hosts come from --ip_config, copies
    # use --source_file_paths/--target_dir (+ --container), and train
    # carries --num_trainers/--num_samplers/--num_servers/--num_parts.
    p = argparse.ArgumentParser()
    p.add_argument("--cmd_type", required=True,
                   choices=["exec_batch", "copy_batch", "copy_batch_container",
                            "train"])
    p.add_argument("--hostfile", "--ip_config", "--ip-config",
                   default=os.environ.get("DGL_OPERATOR_HOSTFILE_PATH",
                                          "/etc/dgl/hostfile"))
    p.add_argument("--command", default="")
    p.add_argument("--source", "--source_file_paths", default="")
    p.add_argument("--target", "--target_dir", default="")
    p.add_argument("--container", default=None)
    p.add_argument("--master-port", type=int, default=29400)
    p.add_argument("--script", default="")
    p.add_argument("--script-args", default="")
    p.add_argument("--workspace", default=None)  # accepted (reference)
    p.add_argument("--num_trainers", "--num-trainers", type=int, default=None)
    p.add_argument("--num_samplers", "--num-samplers", type=int, default=0)
    p.add_argument("--num_servers", "--num-servers", type=int, default=0)
    p.add_argument("--num_parts", "--num-parts", type=int, default=None)
    p.add_argument("--part_config", "--part-config", default=None)
    p.add_argument("positional_command", nargs="?", default="",
                   help="reference style: the command/script as the last "
                        "positional argument")
    args = p.parse_args(argv)
    with open(args.hostfile) as f:
        hosts = parse_hostfile(f.read())
    command = args.command or args.positional_command
    if args.cmd_type == "exec_batch":
        exec_batch(hosts, command)
    elif args.cmd_type in ("copy_batch", "copy_batch_container"):
        copy_batch(hosts, args.source or command, args.target,
                   container=args.container)
    elif args.cmd_type == "train":
        # reference style passes the whole train command positionally
        # ("DGLBACKEND=pytorch python train.py --flag ..."); the reference
        # launcher strips the interpreter before wrapping with
        # torch.distributed.launch (launch.py:135-152) — same here
        script, script_args = args.script, args.script_args
        if not script and command:
            import shlex

            toks = shlex.split(command)
            while toks and "=" in toks[0] and not toks[0].startswith("-"):
                toks.pop(0)  # leading ENV=val assignments
            if toks and os.path.basename(toks[0]) in ("python", "python3"):
                toks.pop(0)
            assert toks, f"empty train command: {command!r}"
            script, script_args = toks[0], " ".join(toks[1:])
        if args.num_parts is not None:
            total = (args.num_trainers or hosts[0].slots) * len(hosts)
            assert args.num_parts == total, (
                f"one partition per rank: --num_parts {args.num_parts} vs "
                f"{total} ranks")
        train(hosts, script, script_args, args.master_port,
              num_trainers=args.num_trainers)


if __name__ == "__main__":
    main()
