"""Hostfile parsing + ipconfig revision.

Reference: /root/reference/python/dglrun/tools/revise_hostfile.py — the
operator writes `ip port podname slots=N` per worker; trainers need
`ip port` (DGL) or `ip port num_servers` (DGL-KE).
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import List


@dataclass
class HostEntry:
    ip: str
    port: int
    pod: str
    slots: int


def parse_hostfile(text: str) -> List[HostEntry]:
    """Hostfile lines are `ip port podname slots=N`; partfile/leadfile
    lines are the reference's 3-column `ip port podname`
    (dgljob_controller.go:1440-1469) — slots defaults to 1 there."""
    out = []
    for line in text.splitlines():
        parts = line.split()
        if len(parts) < 3:
            continue
        slots = 1
        if len(parts) >= 4 and "=" in parts[3]:
            slots = int(parts[3].split("=")[1])
        out.append(
            HostEntry(ip=parts[0], port=int(parts[1]), pod=parts[2],
                      slots=slots)
        )
    return out


def revise_for_dgl(entries: List[HostEntry]) -> str:
    return "".join(f"{e.ip} {e.port}\n" for e in entries)


def revise_for_dglke(entries: List[HostEntry], num_servers: int = 1) -> str:
    return "".join(f"{e.ip} {e.port} {num_servers}\n" for e in entries)


def main(argv=None):
    import argparse

    # accepts BOTH this repo's spellings and the reference
    # revise_hostfile.py contract (--workspace --ip_config --framework
    # DGL|DGLKE --num_servers; reference dglrun:205 / dglkerun:258-260)
    p = argparse.ArgumentParser(description="revise hostfile into ipconfig")
    p.add_argument("--hostfile", "--ip_config", "--ip-config",
                   default="/etc/dgl/hostfile")
    p.add_argument("--output", default=None)
    p.add_argument("--format", "--framework",
                   type=lambda s: s.lower(),
                   choices=["dgl", "dglke"], default="dgl")
    p.add_argument("--num-servers", "--num_servers", type=int, default=1)
    p.add_argument("--workspace", default=None)
    args = p.parse_args(argv)
    with open(args.hostfile) as f:
        entries = parse_hostfile(f.read())
    text = (
        revise_for_dgl(entries)
        if args.format == "dgl"
        else revise_for_dglke(entries, args.num_servers)
    )
    out = args.output or os.path.join(
        args.workspace or os.environ.get("WORKSPACE", "."),
        "hostfile_revised"
    )
    with open(out, "w") as f:
        f.write(text)
    print(f"wrote {out} ({len(entries)} hosts)")


if __name__ == "__main__":
    main()
