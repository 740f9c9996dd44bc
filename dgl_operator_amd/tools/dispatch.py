"""Partition dispatcher: runs on the launcher after the partitioner delivered
its output; rewrites per-part paths into the workers' workload dir and ships
each part to its worker over the fabric.

Reference: /root/reference/python/dglrun/tools/dispatch.py:52-91. The
reference enforces one partition per worker POD (launch.py:107-108) because
its slots only multiply trainer processes over a shared partition; here
slots = ranks = GPUs per pod and the invariant generalizes to ONE PARTITION
PER RANK: pod i receives the contiguous partition block
[sum(slots[:i]), sum(slots[:i+1])), and torchrun global rank r trains
part-r. With slots=1 everywhere this reduces exactly to the reference
behavior.
"""
from __future__ import annotations

import json
import os
from typing import List, Optional

from .fabric import Fabric, get_fabric
from .hostfile import HostEntry, parse_hostfile

WORKLOAD_DIR = "workload"


def dispatch_partitions(
    dataset_dir: str,
    graph_name: str,
    hosts: List[HostEntry],
    fabric: Optional[Fabric] = None,
    workspace: str = "/dgl_workspace",
) -> dict:
    """Rewrite graph.json paths for the worker side and copy part payloads."""
    fabric = fabric or get_fabric()
    meta_path = os.path.join(dataset_dir, f"{graph_name}.json")
    with open(meta_path) as f:
        meta = json.load(f)
    num_parts = meta["num_parts"]
    total_ranks = sum(h.slots for h in hosts)
    assert num_parts == total_ranks, (
        f"one partition per rank required: {num_parts} parts vs "
        f"{total_ranks} ranks ({len(hosts)} pods x slots)"
    )
    revised = dict(meta)
    for i in range(num_parts):
        entry = dict(meta[f"part-{i}"])
        for k, v in list(entry.items()):
            # relative to the revised json's own directory (workload/)
            entry[k] = os.path.join(f"part{i}", os.path.basename(v))
        revised[f"part-{i}"] = entry
    revised_path = os.path.join(dataset_dir, f"{graph_name}_revised.json")
    with open(revised_path, "w") as f:
        json.dump(revised, f, indent=2)

    part = 0
    for host in hosts:
        dst_dir = f"{workspace}/{WORKLOAD_DIR}"
        for _ in range(host.slots):
            part_src = os.path.join(dataset_dir, f"part{part}")
            fabric.exec(host.pod, f"mkdir -p {dst_dir}/part{part}")
            for fname in os.listdir(part_src):
                fabric.copy(
                    os.path.join(part_src, fname), host.pod,
                    f"{dst_dir}/part{part}/{fname}",
                )
            part += 1
        fabric.copy(revised_path, host.pod, f"{dst_dir}/{graph_name}.json")
    return revised


def main(argv=None):
    import argparse

    # accepts BOTH this repo's spellings and the reference dispatch.py
    # invocation contract (dglrun:182-188: --workspace --rel_data_path
    # --rel_workload_path --part_config --ip_config)
    p = argparse.ArgumentParser()
    p.add_argument("--dataset-dir", default=None)
    p.add_argument("--graph-name", default=None)
    p.add_argument("--hostfile", "--ip_config", "--ip-config",
                   default="/etc/dgl/hostfile")
    p.add_argument("--workspace", default="/dgl_workspace")
    p.add_argument("--part_config", "--part-config", default=None,
                   help="reference spelling: path to <dataset>/<graph>.json")
    p.add_argument("--rel_data_path", "--rel-data-path", default="dataset")
    p.add_argument("--rel_workload_path", "--rel-workload-path",
                   default=WORKLOAD_DIR)
    args = p.parse_args(argv)
    dataset_dir, graph_name = args.dataset_dir, args.graph_name
    if args.part_config:
        dataset_dir = dataset_dir or os.path.dirname(args.part_config)
        graph_name = graph_name or os.path.splitext(
            os.path.basename(args.part_config))[0]
    if dataset_dir is None:
        dataset_dir = os.path.join(args.workspace, args.rel_data_path)
    assert graph_name, "need --graph-name or --part_config"
    with open(args.hostfile) as f:
        hosts = parse_hostfile(f.read())
    dispatch_partitions(dataset_dir, graph_name, hosts,
                        workspace=args.workspace)


if __name__ == "__main__":
    main()
