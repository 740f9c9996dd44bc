"""dglkerun — DGL-KE workflow driver (the reference's second CLI,
/root/reference/python/dglrun/exec/dglkerun, 342 LoC bash).

Phases mirror dglrun but for KE jobs: partition triples (or reuse a
pre-partitioned dataset via --ignore-partition/--partitioned-dataset-dir,
the reference's PVC-reuse path), deliver/dispatch, revise hostfile to the
DGL-KE ipconfig format, then dglke_dist_train equivalent (one torchrun rank
per GPU; sharded kvstore replaces the server processes).
"""
from __future__ import annotations

import argparse
import os
import time
from contextlib import contextmanager

from . import launch as launch_mod
from .hostfile import parse_hostfile, revise_for_dglke


@contextmanager
def phase(name: str):
    t0 = time.time()
    print(f"[dglkerun] ---- {name} start ----", flush=True)
    yield
    print(f"[dglkerun] ---- {name} done in {time.time() - t0:.1f}s ----",
          flush=True)


def build_parser():
    # flag surface parity with the reference CLI (python/dglrun/exec/dglkerun;
    # --worksapce is its misspelling, accepted as an alias). Flags tied to the
    # reference's server/client process architecture (--num-servers,
    # --num-client-procs, --mix-cpus-and-single-gpu) are accepted and ignored:
    # this framework runs one rank per GPU with a sharded kvstore.
    p = argparse.ArgumentParser(prog="dglkerun")
    p.add_argument("--model-name", "--model", default="ComplEx")
    p.add_argument("--dataset", default="synthetic")
    p.add_argument("--custom-dataset", default="")
    p.add_argument("--dataset-files", default="")
    p.add_argument("--dataset-format", default="")
    p.add_argument("--num-partitions", type=int, default=0)
    p.add_argument("--num-servers", type=int, default=0)
    p.add_argument("--num-client-procs", type=int, default=0)
    p.add_argument("--mix-cpus-and-single-gpu", action="store_true")
    p.add_argument("--partition-config-path", default="")
    p.add_argument("--pvc-partitioned-dir", default="",
                   help="alias of --partitioned-dataset-dir (PVC reuse)")
    p.add_argument("--dispatch-entry-point", default="")
    p.add_argument("--launch-entry-point", default="")
    p.add_argument("--revise-hostfile-entry-point", default="")
    p.add_argument("--hidden-dim", type=int, default=400)
    p.add_argument("--gamma", type=float, default=143.0)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--batch-size", type=int, default=1024)
    p.add_argument("--neg-sample-size", type=int, default=256)
    p.add_argument("--max-step", type=int, default=1000)
    p.add_argument("--regularization-coef", "--regularization_coef",
                   type=float, default=1e-9,
                   help="forwarded to the trainer "
                        "(reference dglkerun:301 fixed value)")
    p.add_argument("--test", action="store_true",
                   help="evaluate after training (reference dglkerun:300 "
                        "--test; maps to the trainer's --eval)")
    p.add_argument("--batch-size-eval", "--batch_size_eval",
                   type=int, default=1024)
    p.add_argument("--log-interval", type=int, default=100)
    p.add_argument("--save-path", default="ckpts")
    p.add_argument("--no-save-emb", action="store_true")
    p.add_argument("--ignore-partition", action="store_true")
    p.add_argument("--partitioned-dataset-dir", default="")
    p.add_argument("--workspace", "--worksapce", default=os.environ.get(
        "WORKSPACE", "/dgl_workspace"))
    p.add_argument("--hostfile", default=os.environ.get(
        "DGL_OPERATOR_HOSTFILE_PATH", "/etc/dgl/hostfile"))
    p.add_argument("--leadfile", default=os.environ.get(
        "DGL_OPERATOR_LEADFILE_PATH", "/etc/dgl/leadfile"))
    p.add_argument("--master-port", type=int, default=29401)
    p.add_argument("--num-triples", type=int, default=100_000)
    p.add_argument("--num-entities", type=int, default=10_000)
    p.add_argument("--num-relations", type=int, default=100)
    p.add_argument("--train-entry-point",
                   default="examples/dgl_ke/train_ke.py")
    return p


def _dataset_name(args) -> str:
    return args.custom_dataset or args.dataset


def run_partitioner(args):
    """Phases 1-2 on the partitioner pod (reference dglkerun:133-205):
    partition the KG (built-in synthetic or custom triple files), then
    deliver the partitioned dataset into the launcher's still-waiting
    watcher-loop-partitioner init container."""
    from .fabric import get_fabric
    from . import kg_partition

    dataset_root = os.path.join(args.workspace, "dataset")
    os.makedirs(dataset_root, exist_ok=True)
    name = _dataset_name(args)
    meta_json = os.path.join(dataset_root, name, f"{name}.json")
    if args.ignore_partition and os.path.exists(meta_json):
        print("[dglkerun] Phase 1/5 skipped (--ignore-partition)", flush=True)
    else:
        with phase("Phase 1/5 partition KG"):
            argv = ["--dataset", name, "-k", str(args.num_partitions or 1),
                    "--data-path", dataset_root]
            if args.dataset_files:
                argv += ["--data-files"] + args.dataset_files.split(",")
                if args.dataset_format:
                    argv += ["--format", args.dataset_format]
            else:
                argv += ["--num-entities", str(args.num_entities),
                         "--num-relations", str(args.num_relations),
                         "--num-triples", str(args.num_triples)]
            kg_partition.main(argv)
    with phase("Phase 2/5 deliver"):
        if args.partitioned_dataset_dir or args.pvc_partitioned_dir:
            # PVC path (reference dglkerun:190-205 scp branch): the shared
            # volume already holds the partition; nothing to copy
            print("[dglkerun] PVC-partitioned dir in use; skip delivery",
                  flush=True)
            return
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
            raise SystemExit("[dglkerun] no launcher entry in leadfile")
        fabric = get_fabric()
        for lead in leads:
            fabric.copy(os.path.join(dataset_root, name), lead.pod,
                        f"{args.workspace}/dataset/{name}",
                        container="watcher-loop-partitioner")


def main(argv=None):
    args = build_parser().parse_args(argv)
    mode = os.environ.get("DGL_OPERATOR_PHASE_ENV", "")
    if mode == "Launcher_Workload":
        with phase("workload (Skip mode)"):
            import subprocess as sp

            rc = sp.call(f"python {args.train_entry_point}", shell=True)
            if rc != 0:
                raise SystemExit(rc)
        return
    if mode == "Partitioner":
        run_partitioner(args)
        return
    with open(args.hostfile) as f:
        hosts = parse_hostfile(f.read())
    name = _dataset_name(args)
    dataset_dir = (args.partitioned_dataset_dir or args.pvc_partitioned_dir
                   or os.path.join(args.workspace, "dataset"))
    have_partition = os.path.exists(
        os.path.join(dataset_dir, name, f"{name}.json"))
    if have_partition:
        # Phase 3 (reference dglkerun:225-233): the WHOLE partitioned
        # dataset goes to every worker (each rank reads its own part file)
        with phase("Phase 3/5 dispatch (copy dataset to workers)"):
            launch_mod.copy_batch(
                hosts, os.path.join(dataset_dir, name),
                f"{args.workspace}/dataset/{name}",
            )
    with phase("Phase 4/5 revise hostfile (dglke format)"):
        revised = revise_for_dglke(hosts, num_servers=1)
        launch_mod.exec_batch(
            hosts,
            f"sh -c 'mkdir -p {args.workspace} && printf %s \"{revised}\" "
            f"> {args.workspace}/hostfile_revised'",
        )
    with phase("Phase 5/5 dglke train"):
        extra = (
            f"--model-name {args.model_name} --hidden-dim {args.hidden_dim} "
            f"--gamma {args.gamma} --lr {args.lr} "
            f"--batch-size {args.batch_size} "
            f"--neg-sample-size {args.neg_sample_size} "
            f"--max-step {args.max_step} --save-path {args.save_path} "
            f"--regularization-coef {args.regularization_coef} "
            f"--log-interval {args.log_interval} "
            f"--batch-size-eval {args.batch_size_eval}"
            + (" --eval" if args.test else "")
            + (" --no-save-emb" if args.no_save_emb else "")
        )
        if have_partition:
            # PVC mode: workers read the shared dir; copy mode: the Phase 3
            # copy landed under the per-pod workspace
            dp = (dataset_dir
                  if (args.partitioned_dataset_dir or args.pvc_partitioned_dir)
                  else f"{args.workspace}/dataset")
            extra += f" --data-path {dp} --dataset-name {name}"
        launch_mod.train(hosts, args.train_entry_point, extra,
                         master_port=args.master_port)


if __name__ == "__main__":
    main()
