"""watcher-loop equivalent: block until the pods named in a hostfile-format
watch file reach Running (mode=ready) or Succeeded (mode=finished).

Reference: /root/reference/watcher-loop/app/server.go:38-120 +
controllers/controller.go:121-254 — pod names come from column 3 of the watch
file, entries ending in "launcher" are skipped, a 500 ms ticker exits when the
watched set drains. This implementation polls the Cluster abstraction (the
informer machinery is a k8s-specific optimization; the observable contract —
block-until-state — is identical).
"""
from __future__ import annotations

import time
from typing import Iterable, List, Optional

from .api import PodPhase
from .cluster import Cluster


def parse_watchfile(text: str) -> List[str]:
    """Hostfile format: `ip port podname slots=N`; returns pod names, skipping
    any *launcher entry (watcher-loop/app/server.go:108-120)."""
    names = []
    for line in text.splitlines():
        parts = line.split()
        if len(parts) < 3:
            continue
        name = parts[2]
        if name.endswith("launcher"):
            continue
        names.append(name)
    return names


def watch(
    cluster: Cluster,
    namespace: str,
    pod_names: Iterable[str],
    mode: str = "ready",
    poll_interval: float = 0.5,
    timeout: Optional[float] = None,
) -> bool:
    """Block until every pod satisfies the mode predicate. Returns True on
    success, False on timeout."""
    assert mode in ("ready", "finished")
    watched = set(pod_names)
    deadline = time.time() + timeout if timeout else None
    while watched:
        for name in list(watched):
            p = cluster.get_pod(namespace, name)
            if p is None:
                continue
            if mode == "ready" and p.is_real_running():
                watched.discard(name)
            elif mode == "finished" and p.phase == PodPhase.SUCCEEDED:
                watched.discard(name)
        if not watched:
            return True
        if deadline and time.time() > deadline:
            return False
        time.sleep(poll_interval)
    return True


def watch_file(cluster: Cluster, namespace: str, watchfile_text: str,
               mode: str, **kw) -> bool:
    return watch(cluster, namespace, parse_watchfile(watchfile_text), mode, **kw)


def _satisfied(pod, mode: str) -> bool:
    if mode == "ready":
        return pod.is_real_running()
    return pod.phase == PodPhase.SUCCEEDED


def watch_events(
    cluster: Cluster,
    namespace: str,
    pod_names: Iterable[str],
    mode: str = "ready",
    timeout: Optional[float] = None,
) -> bool:
    """Event-driven variant of :func:`watch` — the informer + workqueue
    shape of the reference watcher-loop
    (/root/reference/watcher-loop/controllers/controller.go:84-152):
    an initial list sweep (:121-132), then pod UPDATE events filtered on
    the watched names (:84-100) drain the set; returns when it is empty
    (:140-152). No steady-state polling of the apiserver.

    Requires the cluster to expose ``watch_pods(namespace)`` — a generator
    of Pod events (``None`` items are keep-alive heartbeats used for the
    timeout check). Falls back to :func:`watch` when it does not.
    """
    assert mode in ("ready", "finished")
    stream_fn = getattr(cluster, "watch_pods", None)
    if stream_fn is None:
        return watch(cluster, namespace, pod_names, mode, timeout=timeout)
    watched = set(pod_names)
    deadline = time.time() + timeout if timeout else None
    stream = stream_fn(namespace)
    try:
        # initial sweep (events may predate subscription)
        for name in list(watched):
            p = cluster.get_pod(namespace, name)
            if p is not None and _satisfied(p, mode):
                watched.discard(name)
        while watched:
            if deadline and time.time() > deadline:
                return False
            ev = next(stream, StopIteration)
            if ev is StopIteration:
                return not watched
            if ev is None:  # heartbeat
                continue
            if ev.name in watched and _satisfied(ev, mode):
                watched.discard(ev.name)
        return True
    finally:
        close = getattr(stream, "close", None)
        if close:
            close()


def main(argv=None):
    """CLI parity with the reference watcher-loop binary
    (/root/reference/watcher-loop/app/server.go:38-64): env NAMESPACE,
    WATCHERFILE (a path under /etc/dgl), WATCHERMODE=ready|finished."""
    import argparse
    import os
    import sys

    from .cluster import KubectlCluster

    p = argparse.ArgumentParser(prog="watcher-loop")
    p.add_argument("--namespace", default=os.environ.get("NAMESPACE", "default"))
    p.add_argument("--watcherfile",
                   default=os.environ.get("WATCHERFILE", "/etc/dgl/hostfile"))
    p.add_argument("--mode", default=os.environ.get("WATCHERMODE", "ready"))
    p.add_argument("--timeout", type=float, default=None)
    args = p.parse_args(argv)
    path = args.watcherfile
    if not os.path.isabs(path):
        path = os.path.join("/etc/dgl", path)
    with open(path) as f:
        names = parse_watchfile(f.read())
    print(f"[watcher-loop] waiting for {len(names)} pods to be {args.mode}",
          flush=True)
    # event-driven (informer-style) against the real apiserver; watch()
    # remains the fallback for clusters without a watch stream
    ok = watch_events(KubectlCluster(), args.namespace, names, args.mode,
                      timeout=args.timeout)
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
