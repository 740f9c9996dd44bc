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
    p.add_argument("--save-path", default="ckpts")
    p.add_argument("--no-save-emb", action="store_true")
    p.add_argument("--ignore-partition", action="store_true")
    p.add_argument("--partitioned-dataset-dir", default="")
    p.add_argument("--workspace", "--worksapce", default=os.environ.get(
        "WORKSPACE", "/dgl_workspace"))
    p.add_argument("--hostfile", default="/etc/dgl/hostfile")
    p.add_argument("--master-port", type=int, default=29401)
    p.add_argument("--train-entry-point",
                   default="examples/dgl_ke/train_ke.py")
    return p


def main(argv=None):
    args = build_parser().parse_args(argv)
    with open(args.hostfile) as f:
        hosts = parse_hostfile(f.read())
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
            f"--max-step {args.max_step} --save-path {args.save_path}"
            + (" --no-save-emb" if args.no_save_emb else "")
        )
        launch_mod.train(hosts, args.train_entry_point, extra,
                         master_port=args.master_port)


if __name__ == "__main__":
    main()
