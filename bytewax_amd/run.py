"""Executing dataflows and the ``python -m bytewax_amd.run`` CLI.

Parity target: ``bytewax.run`` (reference pysrc/bytewax/run.py:30-355):
Flask-style import-string resolution (including factory calls),
argparse with env-var fallbacks, Kubernetes helpers, and dispatch to
single-process or cluster execution.
"""

import argparse
import ast
import inspect
import os
import sys
from datetime import timedelta
from pathlib import Path
from typing import Optional

from ._engine import cluster_main, run_main
from .dataflow import Dataflow
from .recovery import RecoveryConfig

__all__ = ["cli_main", "cluster_main", "run_main"]


def _locate_dataflow(module_name: str, dataflow_name: str) -> Dataflow:
    """Import a module and find the Dataflow instance or factory call.

    `dataflow_name` may be `"flow"` or a factory expression like
    `"build_flow()"` or `"build_flow('arg')"`.
    """
    try:
        __import__(module_name)
    except ImportError:
        if sys.exc_info()[2].tb_next:
            raise
        msg = (
            f"While importing {module_name!r}, an ImportError was raised:"
        )
        raise ImportError(msg)
    module = sys.modules[module_name]

    try:
        expr = ast.parse(dataflow_name.strip(), mode="eval").body
    except SyntaxError:
        msg = f"Failed to parse {dataflow_name!r} as attribute name or function call"
        raise SyntaxError(msg) from None

    if isinstance(expr, ast.Name):
        name = expr.id
        args: list = []
        kwargs: dict = {}
    elif isinstance(expr, ast.Call) and isinstance(expr.func, ast.Name):
        name = expr.func.id
        try:
            args = [ast.literal_eval(a) for a in expr.args]
            kwargs = {kw.arg: ast.literal_eval(kw.value) for kw in expr.keywords}
        except ValueError:
            msg = f"Failed to parse arguments as literal values: {dataflow_name!r}"
            raise ValueError(msg) from None
    else:
        msg = f"Failed to parse {dataflow_name!r} as attribute name or function call"
        raise ValueError(msg)

    try:
        attr = getattr(module, name)
    except AttributeError:
        msg = f"Failed to find attribute {name!r} in {module.__name__!r}"
        raise AttributeError(msg) from None

    if inspect.isfunction(attr) and isinstance(expr, ast.Name):
        if not args and not kwargs:
            sig = inspect.signature(attr)
            if any(
                p.default is inspect.Parameter.empty
                for p in sig.parameters.values()
            ):
                msg = (
                    f"Function {name!r} requires arguments; call it like "
                    f"{name}('arg') in the import string"
                )
                raise TypeError(msg)
        attr = attr()
    elif isinstance(expr, ast.Call):
        attr = attr(*args, **kwargs)

    if isinstance(attr, Dataflow):
        return attr
    msg = (
        f"A valid Dataflow was not obtained from '{module_name}:{dataflow_name}'; "
        f"got {type(attr)!r} instead"
    )
    raise TypeError(msg)


def _prepare_import(import_str: str):
    """Split `"module:flow"` / `"path/to/file.py:flow"` into a
    `(module_str, attrs_str)` pair (reference run.py:153-190
    contract: file paths become dotted module paths)."""
    module_str, _, flow_str = import_str.partition(":")
    if not flow_str:
        flow_str = "flow"
    path = Path(module_str)
    if path.suffix == ".py":
        if path.is_absolute():
            sys.path.insert(0, str(path.parent.resolve()))
            module_str = path.stem
        else:
            module_str = ".".join(path.with_suffix("").parts)
    return module_str, flow_str


class _EnvDefault(argparse.Action):
    """Use an env var as the default for an argparse argument."""

    def __init__(self, envvar, required=True, default=None, **kwargs):
        if envvar and envvar in os.environ:
            default = os.environ[envvar]
            required = False
        super().__init__(default=default, required=required, **kwargs)
        self.envvar = envvar

    def __call__(self, parser, namespace, values, option_string=None):
        setattr(namespace, self.dest, values)


def _parse_args(args=None):
    parser = argparse.ArgumentParser(
        prog="python -m bytewax_amd.run",
        description="Run a bytewax_amd dataflow",
    )
    parser.add_argument(
        "import_str",
        metavar="IMPORT_STR",
        help="Dataflow import string in the format "
        "<module_name>[:<dataflow_variable_or_factory>] e.g. "
        "'src.dataflow:flow' or 'src.dataflow:get_flow()'",
    )
    scaling = parser.add_argument_group("Scaling")
    scaling.add_argument(
        "-w",
        "--workers-per-process",
        type=int,
        default=None,
        action=_EnvDefault,
        envvar="BYTEWAX_WORKERS_PER_PROCESS",
        required=False,
        help="Number of workers for each process",
    )
    scaling.add_argument(
        "-i",
        "--process-id",
        type=int,
        default=None,
        action=_EnvDefault,
        envvar="BYTEWAX_PROCESS_ID",
        required=False,
        help="Process id",
    )
    scaling.add_argument(
        "-a",
        "--addresses",
        action=_EnvDefault,
        envvar="BYTEWAX_ADDRESSES",
        required=False,
        help="Addresses of other processes, separated by semicolon",
    )
    recovery = parser.add_argument_group("Recovery")
    recovery.add_argument(
        "-r",
        "--recovery-directory",
        type=Path,
        action=_EnvDefault,
        envvar="BYTEWAX_RECOVERY_DIRECTORY",
        required=False,
        help="Local directory containing pre-initialized recovery partitions",
    )
    recovery.add_argument(
        "-s",
        "--snapshot-interval",
        type=lambda v: timedelta(seconds=float(v)),
        action=_EnvDefault,
        envvar="BYTEWAX_SNAPSHOT_INTERVAL",
        required=False,
        help="System time duration in seconds to snapshot state for recovery",
    )
    recovery.add_argument(
        "-b",
        "--backup-interval",
        type=lambda v: timedelta(seconds=float(v)),
        action=_EnvDefault,
        envvar="BYTEWAX_RECOVERY_BACKUP_INTERVAL",
        required=False,
        help="System time duration in seconds to keep extra state snapshots around",
    )

    args = parser.parse_args(args)

    # Kubernetes helpers: derive process id from the pod name and
    # addresses from a hostfile.
    if args.process_id is None and "BYTEWAX_POD_NAME" in os.environ:
        pod = os.environ["BYTEWAX_POD_NAME"]
        ss = os.environ.get("BYTEWAX_STATEFULSET_NAME", "")
        if pod.startswith(ss + "-"):
            try:
                args.process_id = int(pod[len(ss) + 1 :])
            except ValueError:
                pass
    if args.addresses is None and "BYTEWAX_HOSTFILE_PATH" in os.environ:
        hostfile = Path(os.environ["BYTEWAX_HOSTFILE_PATH"])
        if hostfile.exists():
            args.addresses = ";".join(
                line for line in hostfile.read_text().splitlines() if line
            )

    if args.recovery_directory is not None and (
        args.snapshot_interval is None or args.backup_interval is None
    ):
        parser.error(
            "recovery requires -s/--snapshot-interval and -b/--backup-interval"
        )
    return args


def cli_main(
    flow: Dataflow,
    *,
    workers_per_process: Optional[int] = None,
    process_id: Optional[int] = None,
    addresses: Optional[list] = None,
    epoch_interval: Optional[timedelta] = None,
    recovery_config: Optional[RecoveryConfig] = None,
) -> None:
    """Run a dataflow, dispatching on the scaling arguments."""
    if addresses and len(addresses) > 1:
        if process_id is None:
            msg = "-i/--process-id required with multiple addresses"
            raise ValueError(msg)
        cluster_main(
            flow,
            addresses,
            process_id,
            epoch_interval=epoch_interval,
            recovery_config=recovery_config,
            worker_count_per_proc=workers_per_process or 1,
        )
    elif workers_per_process is not None and workers_per_process > 1:
        cluster_main(
            flow,
            [],
            0,
            epoch_interval=epoch_interval,
            recovery_config=recovery_config,
            worker_count_per_proc=workers_per_process,
        )
    else:
        run_main(
            flow,
            epoch_interval=epoch_interval,
            recovery_config=recovery_config,
        )


def _main() -> None:
    args = _parse_args()
    flow = _locate_dataflow(*_prepare_import(args.import_str))
    epoch_interval = args.snapshot_interval
    recovery_config = None
    if args.recovery_directory is not None:
        backup = args.backup_interval
        recovery_config = RecoveryConfig(
            args.recovery_directory, backup_interval=backup
        )
    addresses = args.addresses.split(";") if args.addresses else None
    cli_main(
        flow,
        workers_per_process=(
            int(args.workers_per_process)
            if args.workers_per_process is not None
            else None
        ),
        process_id=(
            int(args.process_id) if args.process_id is not None else None
        ),
        addresses=addresses,
        epoch_interval=epoch_interval,
        recovery_config=recovery_config,
    )


if __name__ == "__main__":
    _main()
