"""`waxctl`-style deployment CLI: generate/apply Kubernetes resources
for a dataflow.

Role parity with the reference's `waxctl dataflow deploy|list|delete`
(reference docs/guide/deployment/waxctl.md; the reference ships a Go
binary — this is the same contract as a Python CLI).  The generated
resources target :mod:`bytewax_amd.run`'s in-cluster env-var contract:
``BYTEWAX_POD_NAME`` / ``BYTEWAX_STATEFULSET_NAME`` derive the process
id from the StatefulSet ordinal, ``BYTEWAX_HOSTFILE_PATH`` lists the
cluster addresses, and ``BYTEWAX_WORKERS_PER_PROCESS`` /
``BYTEWAX_RECOVERY_*`` map to `-w` / `-r` / `-s` / `-b`.

Usage:
    python -m bytewax_amd.waxctl dataflow deploy my_script.py \
        --name my-dataflow -p 4 -w 2 --dry-run
    python -m bytewax_amd.waxctl dataflow list
    python -m bytewax_amd.waxctl dataflow delete --name my-dataflow

Without ``--dry-run`` the manifests are applied with ``kubectl apply
-f -`` (kubectl must be on PATH and configured); ``list``/``delete``
shell out to kubectl likewise.
"""

import argparse
import base64
import json
import shutil
import subprocess
import sys
from pathlib import Path
from typing import Dict, List, Optional

__all__ = ["build_manifests", "main"]

_HOSTFILE_DIR = "/etc/bytewax"
_SCRIPT_DIR = "/var/bytewax"


def _labels(name: str) -> Dict[str, str]:
    return {
        "app.kubernetes.io/name": name,
        "app.kubernetes.io/managed-by": "waxctl",
    }


def build_manifests(
    script_path: Path,
    *,
    name: str = "bytewax",
    namespace: Optional[str] = None,
    processes: int = 1,
    workers: int = 1,
    env: Optional[List[str]] = None,
    image: str = "bytewax-amd/bytewax-amd",
    tag: str = "latest",
    image_pull_policy: str = "Always",
    job_mode: bool = False,
    keep_alive: bool = False,
    recovery: bool = False,
    recovery_parts: int = 1,
    recovery_size: str = "10Gi",
    recovery_storageclass: Optional[str] = None,
    recovery_snapshot_interval: int = 30,
    recovery_backup_interval: int = 0,
    python_file_name: Optional[str] = None,
    requirements_file_name: Optional[str] = None,
    create_namespace: bool = True,
) -> List[dict]:
    """Build the Kubernetes resource dicts for a dataflow deployment.

    The script travels in a ConfigMap (binaryData, like waxctl's
    tarball mount); a headless Service gives the StatefulSet pods
    stable DNS names that the generated hostfile lists, one
    `host:port` per process.
    """
    script_path = Path(script_path)
    payload = script_path.read_bytes()
    run_file = python_file_name or script_path.name
    if script_path.suffix == ".tar" and python_file_name is None:
        msg = "a .tar payload needs --python-file-name"
        raise ValueError(msg)

    meta = {"name": name, "labels": _labels(name)}
    if namespace:
        meta["namespace"] = namespace

    hostfile = "\n".join(
        f"{name}-{i}.{name}.{namespace or 'default'}"
        ".svc.cluster.local:9999"
        for i in range(processes)
    )
    configmap = {
        "apiVersion": "v1",
        "kind": "ConfigMap",
        "metadata": dict(meta),
        "binaryData": {run_file: base64.b64encode(payload).decode()},
        "data": {"hostfile.txt": hostfile},
    }
    if requirements_file_name:
        configmap["binaryData"][
            Path(requirements_file_name).name
        ] = base64.b64encode(
            Path(requirements_file_name).read_bytes()
        ).decode()

    service = {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": dict(meta),
        "spec": {
            "clusterIP": "None",
            "selector": _labels(name),
            "ports": [{"name": "worker", "port": 9999}],
        },
    }

    container_env = [
        {"name": "BYTEWAX_WORKERS_PER_PROCESS", "value": str(workers)},
        {"name": "BYTEWAX_STATEFULSET_NAME", "value": name},
        {
            "name": "BYTEWAX_POD_NAME",
            "valueFrom": {
                "fieldRef": {"fieldPath": "metadata.name"}
            },
        },
        {
            "name": "BYTEWAX_HOSTFILE_PATH",
            "value": f"{_HOSTFILE_DIR}/hostfile.txt",
        },
    ]
    if recovery:
        container_env += [
            {
                "name": "BYTEWAX_RECOVERY_DIRECTORY",
                "value": "/var/recovery",
            },
            {
                "name": "BYTEWAX_SNAPSHOT_INTERVAL",
                "value": str(recovery_snapshot_interval),
            },
            {
                "name": "BYTEWAX_RECOVERY_BACKUP_INTERVAL",
                "value": str(recovery_backup_interval),
            },
        ]
    for kv in env or []:
        k, _, v = kv.partition("=")
        container_env.append({"name": k, "value": v})

    flow_arg = f"{Path(run_file).stem}:flow"
    command = [
        "sh",
        "-c",
        f"cd {_SCRIPT_DIR} && "
        + (
            f"pip install -r {Path(requirements_file_name).name} && "
            if requirements_file_name
            else ""
        )
        + f"python -m bytewax_amd.run {flow_arg}"
        + ("; sleep infinity" if keep_alive else ""),
    ]

    volume_mounts = [
        {"name": "dataflow-files", "mountPath": _SCRIPT_DIR},
        {"name": "hostfile", "mountPath": _HOSTFILE_DIR},
    ]
    volumes = [
        {
            "name": "dataflow-files",
            "configMap": {"name": name},
        },
        {
            "name": "hostfile",
            "configMap": {
                "name": name,
                "items": [
                    {"key": "hostfile.txt", "path": "hostfile.txt"}
                ],
            },
        },
    ]
    container = {
        "name": "process",
        "image": f"{image}:{tag}",
        "imagePullPolicy": image_pull_policy,
        "command": command,
        "ports": [{"containerPort": 9999, "name": "worker"}],
        "env": container_env,
        "volumeMounts": volume_mounts,
    }
    if recovery:
        container["volumeMounts"] = volume_mounts + [
            {"name": "recovery", "mountPath": "/var/recovery"}
        ]

    pod_spec = {"containers": [container], "volumes": volumes}
    if recovery:
        # First boot initializes the recovery DB partitions
        # (idempotent: `python -m bytewax_amd.recovery` skips an
        # already-initialized directory).
        pod_spec["initContainers"] = [
            {
                "name": "init-recovery",
                "image": f"{image}:{tag}",
                "imagePullPolicy": image_pull_policy,
                "command": [
                    "sh",
                    "-c",
                    "python -m bytewax_amd.recovery /var/recovery "
                    f"{recovery_parts} || true",
                ],
                "volumeMounts": [
                    {"name": "recovery", "mountPath": "/var/recovery"}
                ],
            }
        ]
    template = {
        "metadata": {"labels": _labels(name)},
        "spec": pod_spec,
    }

    if job_mode:
        workload = {
            "apiVersion": "batch/v1",
            "kind": "Job",
            "metadata": dict(meta),
            "spec": {
                "completions": processes,
                "parallelism": processes,
                "completionMode": "Indexed",
                "template": {
                    "metadata": template["metadata"],
                    "spec": dict(pod_spec, restartPolicy="OnFailure"),
                },
            },
        }
    else:
        spec = {
            "serviceName": name,
            "replicas": processes,
            "podManagementPolicy": "Parallel",
            "selector": {"matchLabels": _labels(name)},
            "template": template,
        }
        if recovery:
            claim = {
                "metadata": {"name": "recovery"},
                "spec": {
                    "accessModes": ["ReadWriteOnce"],
                    "resources": {
                        "requests": {"storage": recovery_size}
                    },
                },
            }
            if recovery_storageclass:
                claim["spec"]["storageClassName"] = recovery_storageclass
            spec["volumeClaimTemplates"] = [claim]
        workload = {
            "apiVersion": "apps/v1",
            "kind": "StatefulSet",
            "metadata": dict(meta),
            "spec": spec,
        }

    out: List[dict] = []
    if namespace and create_namespace:
        out.append(
            {
                "apiVersion": "v1",
                "kind": "Namespace",
                "metadata": {"name": namespace},
            }
        )
    out += [configmap, service, workload]
    return out


def _dump(manifests: List[dict], fmt: str) -> str:
    if fmt == "json":
        return json.dumps(manifests, indent=2)
    import yaml

    return "---\n".join(
        yaml.safe_dump(m, sort_keys=False) for m in manifests
    )


def _kubectl(args: List[str], stdin: Optional[str] = None) -> int:
    if shutil.which("kubectl") is None:
        print(
            "kubectl not found on PATH; re-run with --dry-run to "
            "print the manifests",
            file=sys.stderr,
        )
        return 1
    res = subprocess.run(
        ["kubectl", *args],
        input=stdin.encode() if stdin is not None else None,
    )
    return res.returncode


def main(argv: Optional[List[str]] = None) -> int:
    p = argparse.ArgumentParser(
        prog="waxctl", description=__doc__.splitlines()[0]
    )
    sub = p.add_subparsers(dest="noun", required=True)
    df = sub.add_parser(
        "dataflow", aliases=["df"], help="manage dataflows"
    )
    verbs = df.add_subparsers(dest="verb", required=True)

    dep = verbs.add_parser("deploy", help="deploy a dataflow")
    dep.add_argument("path", type=Path)
    dep.add_argument("-N", "--name", default="bytewax")
    dep.add_argument("-n", "--namespace", default=None)
    dep.add_argument("-p", "--processes", type=int, default=1)
    dep.add_argument("-w", "--workers", type=int, default=1)
    dep.add_argument(
        "-e",
        "--environment-variables",
        action="append",
        default=[],
        metavar="KEY=VALUE",
    )
    dep.add_argument(
        "-i", "--image-repository", default="bytewax-amd/bytewax-amd"
    )
    dep.add_argument("-t", "--image-tag", default="latest")
    dep.add_argument(
        "-l",
        "--image-pull-policy",
        default="Always",
        choices=["Always", "IfNotPresent", "Never"],
    )
    dep.add_argument("--job-mode", action="store_true")
    dep.add_argument("--keep-alive", action="store_true")
    dep.add_argument("--recovery", action="store_true")
    dep.add_argument("--recovery-parts", type=int, default=1)
    dep.add_argument("--recovery-size", default="10Gi")
    dep.add_argument("--recovery-storageclass", default=None)
    dep.add_argument(
        "--recovery-snapshot-interval", type=int, default=30
    )
    dep.add_argument(
        "--recovery-backup-interval", type=int, default=0
    )
    dep.add_argument("-f", "--python-file-name", default=None)
    dep.add_argument("-r", "--requirements-file-name", default=None)
    dep.add_argument(
        "--create-namespace",
        action=argparse.BooleanOptionalAction,
        default=True,
    )
    dep.add_argument("--dry-run", action="store_true")
    dep.add_argument(
        "-o",
        "--output-format",
        default="yaml",
        choices=["yaml", "json"],
    )

    ls = verbs.add_parser("list", aliases=["ls"], help="list dataflows")
    ls.add_argument("-n", "--namespace", default=None)

    rm = verbs.add_parser(
        "delete", aliases=["rm"], help="delete a dataflow"
    )
    rm.add_argument("-N", "--name", required=True)
    rm.add_argument("-n", "--namespace", default=None)
    rm.add_argument("--yes", action="store_true")

    args = p.parse_args(argv)

    if args.verb in ("deploy",):
        manifests = build_manifests(
            args.path,
            name=args.name,
            namespace=args.namespace,
            processes=args.processes,
            workers=args.workers,
            env=args.environment_variables,
            image=args.image_repository,
            tag=args.image_tag,
            image_pull_policy=args.image_pull_policy,
            job_mode=args.job_mode,
            keep_alive=args.keep_alive,
            recovery=args.recovery,
            recovery_parts=args.recovery_parts,
            recovery_size=args.recovery_size,
            recovery_storageclass=args.recovery_storageclass,
            recovery_snapshot_interval=args.recovery_snapshot_interval,
            recovery_backup_interval=args.recovery_backup_interval,
            python_file_name=args.python_file_name,
            requirements_file_name=args.requirements_file_name,
            create_namespace=args.create_namespace,
        )
        text = _dump(manifests, args.output_format)
        if args.dry_run:
            print(text)
            return 0
        return _kubectl(["apply", "-f", "-"], stdin=text)
    if args.verb in ("list", "ls"):
        sel = ["-l", "app.kubernetes.io/managed-by=waxctl"]
        ns = ["-n", args.namespace] if args.namespace else []
        return _kubectl(
            ["get", "statefulsets,jobs", *ns, *sel, "-o", "wide"]
        )
    if args.verb in ("delete", "rm"):
        if not args.yes:
            print("refusing without --yes", file=sys.stderr)
            return 1
        ns = ["-n", args.namespace] if args.namespace else []
        return _kubectl(
            [
                "delete",
                "statefulset,job,service,configmap",
                args.name,
                *ns,
                "--ignore-not-found",
            ]
        )
    return 2


if __name__ == "__main__":
    sys.exit(main())
