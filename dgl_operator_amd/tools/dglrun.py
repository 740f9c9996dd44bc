"""dglrun — the 5-phase in-container workflow driver.

Reference: /root/reference/python/dglrun/exec/dglrun (bash, 238 LoC).
Branches on DGL_OPERATOR_PHASE_ENV exactly like the reference:
  Launcher_Workload -> run the training entry point only (Skip mode)
  Partitioner       -> Phase 1 partition + Phase 2 deliver to launcher
  (unset, launcher) -> Phase 3 dispatch + Phase 4 revise + Phase 5 train
Each phase echoes its wall-clock time like the reference does
(dglrun:117-238).
"""
from __future__ import annotations

import argparse
import os
import subprocess
import time
from contextlib import contextmanager

from .dispatch import dispatch_partitions
from .fabric import get_fabric
from .hostfile import parse_hostfile, revise_for_dgl
from . import launch as launch_mod


@contextmanager
def phase(name: str):
    t0 = time.time()
    print(f"[dglrun] ---- {name} start ----", flush=True)
    yield
    print(f"[dglrun] ---- {name} done in {time.time() - t0:.1f}s ----", flush=True)


def _run(cmd: str, env=None):
    full_env = dict(os.environ)
    full_env.update(env or {})
    rc = subprocess.call(cmd, shell=True, env=full_env)
    if rc != 0:
        raise SystemExit(f"[dglrun] command failed rc={rc}: {cmd}")


def build_parser():
    # flag surface parity with the reference CLI (python/dglrun/exec/dglrun
    # arg parser; its misspelled --worksapce is accepted as an alias)
    p = argparse.ArgumentParser(prog="dglrun")
    p.add_argument("-g", "--graph-name", default="graph")
    p.add_argument("--partition-entry-point", default="")
    p.add_argument("--partition-entry-args", default="")
    p.add_argument("--num-partitions", type=int, default=1)
    p.add_argument("--balance-train", action="store_true")
    p.add_argument("--balance-edges", action="store_true")
    p.add_argument("--train-entry-point", default="")
    p.add_argument("--train-entry-args", default="")
    p.add_argument("--num-epochs", type=int, default=None,
                   help="appended to the train args when set")
    p.add_argument("--batch-size", type=int, default=None,
                   help="appended to the train args when set")
    p.add_argument("--partition-config-path", default="",
                   help="override the part-config path handed to training")
    p.add_argument("--dispatch-entry-point", default="",
                   help="custom dispatch script (default: built-in dispatch)")
    p.add_argument("--num-servers", type=int, default=0,
                   help="accepted for reference-CLI parity; this framework "
                        "has no server processes")
    # flags consumed by examples/v1alpha1/GraphSAGE_dist.yaml verbatim
    # (reference dglrun:2-105): trainers become torchrun ranks per pod;
    # samplers/servers are collapsed into the trainer process (the GPU
    # sampler is a HIP kernel, the feature store is HBM + RCCL alltoallv)
    p.add_argument("--num-trainers", type=int, default=None,
                   help="trainer ranks per worker pod (default: hostfile "
                        "slots)")
    p.add_argument("--num-samplers", type=int, default=0,
                   help="accepted for reference-CLI parity; sampling runs "
                        "in-process on the GPU (no sampler processes)")
    p.add_argument("--num-workers", type=int, default=None,
                   help="expected worker-pod count; validated against the "
                        "hostfile when set")
    p.add_argument("--dataset-url", default="",
                   help="forwarded to the partition entry point "
                        "(--dataset-url); offline partitioners fall back "
                        "to synthetic data")
    p.add_argument("--launch-entry-point", default="",
                   help="custom launch script for Phase 5, invoked with the "
                        "reference launch.py contract (default: built-in)")
    p.add_argument("--revise-hostfile-entry-point", default="",
                   help="custom hostfile-revise script (default: built-in)")
    p.add_argument("--workspace", "--worksapce", default=os.environ.get(
        "WORKSPACE", "/dgl_workspace"))
    p.add_argument("--hostfile", default=os.environ.get(
        "DGL_OPERATOR_HOSTFILE_PATH", "/etc/dgl/hostfile"))
    p.add_argument("--leadfile", default=os.environ.get(
        "DGL_OPERATOR_LEADFILE_PATH", "/etc/dgl/leadfile"))
    p.add_argument("--master-port", type=int, default=29400)
    p.add_argument("--ignore-partition", action="store_true",
                   help="reuse an existing partition under workspace/dataset "
                        "(the reference dglkerun's PVC-reuse path)")
    return p


def run_partitioner(args):
    dataset = os.path.join(args.workspace, "dataset")
    os.makedirs(dataset, exist_ok=True)
    if args.ignore_partition and os.path.exists(
        os.path.join(dataset, f"{args.graph_name}.json")
    ):
        print("[dglrun] Phase 1/5 skipped (--ignore-partition)", flush=True)
    else:
        run_phase1(args, dataset)
    _deliver(args, dataset)


def run_phase1(args, dataset):
    with phase("Phase 1/5 partition"):
        extra = args.partition_entry_args
        if not args.balance_train:
            extra += " --no-balance-train"
        if not args.balance_edges:
            extra += " --no-balance-edges"
        if args.dataset_url:
            extra += f" --dataset-url {args.dataset_url}"
        _run(
            f"python {args.partition_entry_point} "
            f"--graph-name {args.graph_name} "
            f"--num-partitions {args.num_partitions} "
            f"--output {dataset} {extra}"
        )


def _deliver(args, dataset):
    with phase("Phase 2/5 deliver"):
        # the leadfile is populated by the operator once the launcher pod has
        # an IP; wait for it like the watcher-loop waits for pods
        leads = []
        deadline = time.time() + 120
        while time.time() < deadline:
            try:
                with open(args.leadfile) as f:
                    leads = parse_hostfile(f.read())
            except FileNotFoundError:
                leads = []
            if leads:
                break
            time.sleep(0.5)
        if not leads:
            raise SystemExit("[dglrun] no launcher entry in leadfile")
        fabric = get_fabric()
        for lead in leads:
            # copy into the launcher's still-running watcher-loop-partitioner
            # init container (reference trick, dgljob_controller.go:1129-1138)
            fabric.copy(dataset, lead.pod, f"{args.workspace}/dataset",
                        container="watcher-loop-partitioner")


def run_launcher(args):
    with open(args.hostfile) as f:
        hosts = parse_hostfile(f.read())
    if args.num_workers is not None and args.num_workers != len(hosts):
        raise SystemExit(
            f"[dglrun] --num-workers {args.num_workers} but hostfile has "
            f"{len(hosts)} workers")
    if args.num_trainers:
        # one partition per RANK: --num-trainers overrides ranks-per-pod, so
        # dispatch must ship partitions per the same count (keeping hostfile
        # slots for dispatch while torchrun launches num_trainers ranks
        # would desync partition ownership)
        for h in hosts:
            if h.slots != args.num_trainers:
                print(f"[dglrun] note: hostfile slots={h.slots} overridden "
                      f"by --num-trainers {args.num_trainers} for {h.pod}",
                      flush=True)
            h.slots = args.num_trainers
    dataset = os.path.join(args.workspace, "dataset")
    with phase("Phase 3/5 dispatch"):
        if args.dispatch_entry_point:
            # custom dispatch scripts are most likely written against the
            # reference contract (dglrun:182-188) — pass those spellings;
            # the built-in dispatch CLI accepts them too
            part_cfg = os.path.join(dataset, f"{args.graph_name}.json")
            _run(f"python {args.dispatch_entry_point} "
                 f"--workspace {args.workspace} "
                 f"--rel_data_path dataset --rel_workload_path workload "
                 f"--part_config {part_cfg} --ip_config {args.hostfile}")
        else:
            dispatch_partitions(dataset, args.graph_name, hosts,
                                workspace=args.workspace)
    with phase("Phase 4/5 revise hostfile"):
        if args.revise_hostfile_entry_point:
            # reference invocation contract (dglrun:205)
            launch_mod.exec_batch(
                hosts,
                f"python {args.revise_hostfile_entry_point} "
                f"--workspace {args.workspace} "
                f"--ip_config {args.hostfile} --framework DGL",
            )
        else:
            revised = revise_for_dgl(hosts)
            launch_mod.exec_batch(
                hosts,
                f"sh -c 'mkdir -p {args.workspace} && printf %s \"{revised}\" "
                f"> {args.workspace}/hostfile_revised'",
            )
    with phase("Phase 5/5 train"):
        part_cfg = (args.partition_config_path
                    or f"{args.workspace}/workload/{args.graph_name}.json")
        targs = args.train_entry_args
        if args.num_epochs is not None:
            targs += f" --num-epochs {args.num_epochs}"
        if args.batch_size is not None:
            targs += f" --batch-size {args.batch_size}"
        if args.launch_entry_point:
            # custom launcher, reference invocation contract
            # (exec/dglrun:220-234): train command passed positionally
            train_cmd = (f"python {args.train_entry_point} "
                         f"--graph-name {args.graph_name} "
                         f"--part-config {part_cfg} {targs}")
            lep_args = (f"--workspace {args.workspace} "
                        f"--ip_config {args.hostfile} --cmd_type train "
                        f"--master-port {args.master_port}")
            if args.num_trainers:
                lep_args += f" --num_trainers {args.num_trainers}"
            _run(f"python {args.launch_entry_point} {lep_args} "
                 f"'{train_cmd}'")
        else:
            launch_mod.train(
                hosts,
                args.train_entry_point,
                f"--graph-name {args.graph_name} "
                f"--part-config {part_cfg} {targs}",
                master_port=args.master_port,
                num_trainers=args.num_trainers,
            )


def main(argv=None):
    args = build_parser().parse_args(argv)
    mode = os.environ.get("DGL_OPERATOR_PHASE_ENV", "")
    if mode == "Launcher_Workload":
        with phase("workload (Skip mode)"):
            _run(f"python {args.train_entry_point} {args.train_entry_args}")
    elif mode == "Partitioner":
        run_partitioner(args)
    else:
        run_launcher(args)


if __name__ == "__main__":
    main()
