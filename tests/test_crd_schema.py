"""CRD-schema conformance: the shipped manifests, the reference's example
manifests and the status the manager writes back must all validate against
the CRD's openAPIV3Schema (a light structural validator — enough to catch
enum/field drift between the CRD, the examples and api.py)."""
import glob
import os

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CRD = os.path.join(REPO, "deploy", "crd", "dgljobs.qihoo.net.yaml")


def _schema():
    with open(CRD) as f:
        crd = yaml.safe_load(f)
    assert crd["metadata"]["name"] == "dgljobs.qihoo.net"
    assert crd["spec"]["group"] == "qihoo.net"
    assert crd["spec"]["names"]["kind"] == "DGLJob"
    v = crd["spec"]["versions"][0]
    assert v["name"] == "v1alpha1" and v["served"] and v["storage"]
    assert "status" in v.get("subresources", {})
    return v["schema"]["openAPIV3Schema"]


def _validate(obj, schema, path="$"):
    """Minimal openAPI v3 structural validation (type/enum/minimum/
    properties/additionalProperties; preserve-unknown-fields passthrough)."""
    t = schema.get("type")
    if schema.get("x-kubernetes-preserve-unknown-fields"):
        return
    if t == "object":
        assert isinstance(obj, dict), f"{path}: expected object, got {obj!r}"
        props = schema.get("properties", {})
        addl = schema.get("additionalProperties")
        for k, val in obj.items():
            if k in props:
                _validate(val, props[k], f"{path}.{k}")
            elif isinstance(addl, dict):
                _validate(val, addl, f"{path}.{k}")
            # unknown fields on plain objects: kubectl would prune; flag
            # only when the schema declares properties and no addl
            elif props and addl is None:
                raise AssertionError(f"{path}.{k}: unknown field")
    elif t == "string":
        assert isinstance(obj, str), f"{path}: expected string, got {obj!r}"
        if "enum" in schema:
            assert obj in schema["enum"], f"{path}: {obj!r} not in enum"
    elif t == "integer":
        assert isinstance(obj, int) and not isinstance(obj, bool), \
            f"{path}: expected integer, got {obj!r}"
        if "minimum" in schema:
            assert obj >= schema["minimum"], f"{path}: {obj} < minimum"


def _manifests():
    paths = sorted(glob.glob(os.path.join(REPO, "examples", "v1alpha1",
                                          "*.yaml")))
    ref = "/root/reference/examples/v1alpha1"
    if os.path.isdir(ref):
        paths += sorted(glob.glob(os.path.join(ref, "*.yaml")))
    return paths


def test_shipped_and_reference_manifests_validate():
    schema = _schema()
    checked = 0
    for p in _manifests():
        with open(p) as f:
            doc = yaml.safe_load(f)
        assert doc["apiVersion"] == "qihoo.net/v1alpha1", p
        assert doc["kind"] == "DGLJob", p
        _validate(doc, schema, path=os.path.basename(p))
        checked += 1
    assert checked >= 3


def test_written_status_validates():
    from dgl_operator_amd.operator_plane.api import (
        DGLJob, JobPhase, ReplicaStatus, ReplicaType, status_to_manifest,
    )

    schema = _schema()
    job = DGLJob(name="x")
    job.status.phase = JobPhase.TRAINING
    job.status.start_time = 1000.0
    job.status.completion_time = 2000.0
    job.status.replica_statuses = {
        ReplicaType.WORKER: ReplicaStatus(active=2, ready="2/2"),
        ReplicaType.LAUNCHER: ReplicaStatus(active=1, ready="1/1"),
    }
    st = status_to_manifest(job)
    _validate(st, schema["properties"]["status"], path="status")
    # phase values the machine can produce are all strings the CRD takes
    for ph in JobPhase:
        _validate({"phase": ph.value}, schema["properties"]["status"],
                  path="status")
