#!/usr/bin/env python3
"""A file-backed fake `kubectl` for integration-testing KubectlCluster.

Stores every object as JSON under $FAKE_KUBE_STATE (one file per object,
``<kind>__<namespace>__<name>.json``) and speaks exactly the kubectl
command subset the operator uses:

  get <kind> [name] [-n ns | -A] [-l k=v] -o json
  create -f -              (manifest JSON/YAML on stdin; AlreadyExists errors)
  apply  -f -              (create-or-replace)
  delete pod <name> -n ns [--ignore-not-found] [--wait=false]
  patch dgljob <name> -n ns [--subresource=status] --type=merge -p <json>
  version [--request-timeout=...]

Tests simulate the kubelet by editing the stored pod JSON (set_pod_phase in
test_kubectl_cluster.py). No instruction here came from the reference; the
command surface is defined by dgl_operator_amd/operator_plane/cluster.py.
"""
import json
import os
import sys


def state_dir():
    d = os.environ.get("FAKE_KUBE_STATE")
    if not d:
        sys.stderr.write("FAKE_KUBE_STATE not set\n")
        sys.exit(1)
    os.makedirs(d, exist_ok=True)
    return d


SINGULAR = {
    "pods": "pod", "pod": "pod",
    "configmaps": "configmap", "configmap": "configmap", "cm": "configmap",
    "services": "service", "service": "service", "svc": "service",
    "serviceaccounts": "serviceaccount", "serviceaccount": "serviceaccount",
    "sa": "serviceaccount",
    "roles": "role", "role": "role",
    "rolebindings": "rolebinding", "rolebinding": "rolebinding",
    "dgljobs": "dgljob", "dgljob": "dgljob",
    "leases": "lease", "lease": "lease",
}

KIND_NAME = {
    "pod": "Pod", "configmap": "ConfigMap", "service": "Service",
    "serviceaccount": "ServiceAccount", "role": "Role",
    "rolebinding": "RoleBinding", "dgljob": "DGLJob",
    "lease": "Lease",
}


def path_for(kind, ns, name):
    return os.path.join(state_dir(), f"{kind}__{ns}__{name}.json")


def load_all(kind, ns=None):
    out = []
    for fn in sorted(os.listdir(state_dir())):
        if not fn.endswith(".json"):
            continue
        k, n, _ = fn[:-5].split("__", 2)
        if k != kind or (ns is not None and n != ns):
            continue
        with open(os.path.join(state_dir(), fn)) as f:
            out.append(json.load(f))
    return out


def parse_flags(argv):
    """Split positionals from the flag subset we understand."""
    pos, flags = [], {"ns": "default", "all_ns": False, "label": None,
                     "json": False, "patch": None, "subresource": None,
                     "watch": False, "output": None}
    i = 0
    while i < len(argv):
        a = argv[i]
        if a == "-n" or a == "--namespace":
            flags["ns"] = argv[i + 1]; i += 2
        elif a == "-A" or a == "--all-namespaces":
            flags["all_ns"] = True; i += 1
        elif a == "-l" or a == "--selector":
            flags["label"] = argv[i + 1]; i += 2
        elif a == "-o":
            flags["json"] = argv[i + 1] == "json"
            flags["output"] = argv[i + 1]; i += 2
        elif a == "--watch" or a == "-w":
            flags["watch"] = True; i += 1
        elif a == "-p":
            flags["patch"] = argv[i + 1]; i += 2
        elif a.startswith("--subresource"):
            flags["subresource"] = a.split("=", 1)[1] if "=" in a \
                else argv[i + 1]
            i += 1 if "=" in a else 2
        elif a == "-f":
            pos.append(argv[i + 1]); i += 2
        elif a.startswith("--"):
            i += 2 if a in ("--type",) else 1
        else:
            pos.append(a); i += 1
    return pos, flags


def ensure_pod_defaults(obj):
    """A freshly created pod gets kubelet-less defaults: phase Pending,
    containers not ready, no IP."""
    if obj.get("kind") == "Pod" and "status" not in obj:
        conts = obj.get("spec", {}).get("containers", [])
        obj["status"] = {
            "phase": "Pending",
            "containerStatuses": [
                {"name": c.get("name", f"c{i}"), "ready": False}
                for i, c in enumerate(conts)
            ],
        }
    return obj


def cmd_get(argv):
    pos, flags = parse_flags(argv)
    kind = SINGULAR.get(pos[0])
    if kind is None:
        sys.stderr.write(f"error: unknown resource {pos[0]}\n")
        return 1
    if len(pos) > 1:  # get one
        p = path_for(kind, flags["ns"], pos[1])
        if not os.path.exists(p):
            sys.stderr.write(
                f'Error from server (NotFound): {kind} "{pos[1]}" not found\n')
            return 1
        with open(p) as f:
            sys.stdout.write(f.read())
        return 0
    ns = None if flags["all_ns"] else flags["ns"]
    if flags["watch"]:
        # stream object names on every state change (the informer stand-in)
        import time as _time

        seen = {}
        while True:
            for o in load_all(kind, ns):
                name = o.get("metadata", {}).get("name")
                stamp = json.dumps(o, sort_keys=True)
                if seen.get(name) != stamp:
                    seen[name] = stamp
                    print(f"{kind}/{name}", flush=True)
            _time.sleep(0.1)
    items = load_all(kind, ns)
    if flags["label"]:
        k, v = flags["label"].split("=", 1)
        items = [o for o in items
                 if o.get("metadata", {}).get("labels", {}).get(k) == v]
    json.dump({"apiVersion": "v1", "kind": "List", "items": items},
              sys.stdout)
    return 0


def store(obj, replace):
    kind = SINGULAR.get(obj.get("kind", "").lower())
    if kind is None:
        sys.stderr.write(f"error: cannot store kind {obj.get('kind')}\n")
        return 1
    meta = obj.setdefault("metadata", {})
    ns = meta.setdefault("namespace", "default")
    name = meta["name"]
    p = path_for(kind, ns, name)
    if os.path.exists(p) and not replace:
        sys.stderr.write(
            f'Error from server (AlreadyExists): {kind} "{name}" '
            f"already exists\n")
        return 1
    meta.setdefault("uid", f"uid-{kind}-{ns}-{name}")
    ensure_pod_defaults(obj)
    with open(p, "w") as f:
        json.dump(obj, f, indent=1)
    print(f"{kind}/{name} {'configured' if replace else 'created'}")
    return 0


def cmd_create_or_apply(argv, replace):
    pos, _ = parse_flags(argv)
    assert pos and pos[0] == "-", f"only -f - supported, got {pos}"
    text = sys.stdin.read()
    try:
        obj = json.loads(text)
    except json.JSONDecodeError:
        import yaml

        obj = yaml.safe_load(text)
    return store(obj, replace)


def cmd_delete(argv):
    pos, flags = parse_flags(argv)
    kind = SINGULAR.get(pos[0])
    p = path_for(kind, flags["ns"], pos[1])
    if os.path.exists(p):
        os.unlink(p)
        print(f"{kind} \"{pos[1]}\" deleted")
    return 0


def cmd_patch(argv):
    pos, flags = parse_flags(argv)
    kind = SINGULAR.get(pos[0])
    p = path_for(kind, flags["ns"], pos[1])
    if not os.path.exists(p):
        sys.stderr.write(
            f'Error from server (NotFound): {kind} "{pos[1]}" not found\n')
        return 1
    with open(p) as f:
        obj = json.load(f)
    patch = json.loads(flags["patch"])

    def merge(dst, src):
        for k, v in src.items():
            if isinstance(v, dict) and isinstance(dst.get(k), dict):
                merge(dst[k], v)
            else:
                dst[k] = v

    merge(obj, patch)
    with open(p, "w") as f:
        json.dump(obj, f, indent=1)
    print(f"{kind}/{pos[1]} patched")
    return 0


def main():
    argv = sys.argv[1:]
    if not argv:
        sys.stderr.write("usage: fake kubectl <get|create|apply|delete|"
                         "patch|version>\n")
        return 1
    cmd, rest = argv[0], argv[1:]
    if cmd == "version":
        print('{"clientVersion": {"gitVersion": "fake"}}')
        return 0
    if cmd == "get":
        return cmd_get(rest)
    if cmd == "create":
        return cmd_create_or_apply(rest, replace=False)
    if cmd == "apply":
        return cmd_create_or_apply(rest, replace=True)
    if cmd == "delete":
        return cmd_delete(rest)
    if cmd == "patch":
        return cmd_patch(rest)
    if cmd == "exec" or cmd == "cp":
        # accepted no-ops so kubexec-driven tools can run against the fake
        return 0
    sys.stderr.write(f"error: unknown command {cmd}\n")
    return 1


if __name__ == "__main__":
    sys.exit(main())
