"""waxctl-equivalent deployment CLI (reference
docs/guide/deployment/waxctl.md): the generated Kubernetes resources
must implement bytewax_amd.run's in-cluster env-var contract."""

import io
import sys
from contextlib import redirect_stdout
from pathlib import Path

import pytest
import yaml

from bytewax_amd.waxctl import build_manifests, main

REPO = Path(__file__).resolve().parent.parent
SCRIPT = REPO / "examples" / "wordcount.py"


def _by_kind(manifests):
    return {m["kind"]: m for m in manifests}


def test_statefulset_contract():
    ms = build_manifests(
        SCRIPT, name="wc", namespace="prod", processes=3, workers=2,
        env=["KAFKA=broker:9092"], recovery=True,
    )
    kinds = _by_kind(ms)
    assert set(kinds) == {
        "Namespace", "ConfigMap", "Service", "StatefulSet"
    }
    sts = kinds["StatefulSet"]
    assert sts["spec"]["replicas"] == 3
    env = {
        e["name"]: e.get("value")
        for e in sts["spec"]["template"]["spec"]["containers"][0]["env"]
    }
    assert env["BYTEWAX_WORKERS_PER_PROCESS"] == "2"
    assert env["BYTEWAX_STATEFULSET_NAME"] == "wc"
    assert env["BYTEWAX_HOSTFILE_PATH"] == "/etc/bytewax/hostfile.txt"
    assert env["BYTEWAX_RECOVERY_DIRECTORY"] == "/var/recovery"
    assert env["KAFKA"] == "broker:9092"
    # POD_NAME comes from the downward API (ordinal -> process id).
    pod_env = [
        e
        for e in sts["spec"]["template"]["spec"]["containers"][0]["env"]
        if e["name"] == "BYTEWAX_POD_NAME"
    ][0]
    assert pod_env["valueFrom"]["fieldRef"]["fieldPath"] == "metadata.name"
    # Recovery persists through a volume claim per pod.
    assert sts["spec"]["volumeClaimTemplates"][0]["spec"]["resources"][
        "requests"
    ]["storage"] == "10Gi"
    # The hostfile lists one stable DNS address per process.
    hosts = kinds["ConfigMap"]["data"]["hostfile.txt"].splitlines()
    assert hosts == [
        f"wc-{i}.wc.prod.svc.cluster.local:9999" for i in range(3)
    ]


def test_job_mode_and_tar_guard(tmp_path):
    ms = build_manifests(SCRIPT, name="batch", job_mode=True, processes=2)
    job = _by_kind(ms)["Job"]
    assert job["spec"]["completions"] == 2
    assert job["spec"]["completionMode"] == "Indexed"
    tar = tmp_path / "bundle.tar"
    tar.write_bytes(b"notreally")
    with pytest.raises(ValueError, match="python-file-name"):
        build_manifests(tar, name="x")


def test_cli_dry_run_yaml_roundtrip():
    buf = io.StringIO()
    with redirect_stdout(buf):
        rc = main(
            [
                "dataflow", "deploy", str(SCRIPT), "--name", "wc",
                "-p", "2", "--dry-run",
            ]
        )
    assert rc == 0
    docs = [d for d in yaml.safe_load_all(buf.getvalue()) if d]
    assert {d["kind"] for d in docs} == {
        "ConfigMap", "Service", "StatefulSet"
    }
