"""Shared fixtures.

Mirrors the reference's test strategy (reference pytests/conftest.py:
15-63): every operator test runs through multiple entry points —
1 worker in-thread, and in-process clusters of 1 and 2 worker threads —
plus a tmp-dir recovery fixture.

GPU tests are marked ``gpu`` and only run on a machine with an MI355X.
"""

from datetime import timedelta
from pathlib import Path

import pytest

from bytewax_amd.recovery import RecoveryConfig, init_db_dir
from bytewax_amd.testing import cluster_main, run_main


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X) to run"
    )


@pytest.fixture(
    params=[
        "run_main",
        "cluster_main-1thread",
        "cluster_main-2thread",
    ]
)
def entry_point_name(request):
    return request.param


def _wrapped_cluster_main1(*args, **kwargs):
    return cluster_main(*args, [], 0, worker_count_per_proc=1, **kwargs)


def _wrapped_cluster_main2(*args, **kwargs):
    return cluster_main(*args, [], 0, worker_count_per_proc=2, **kwargs)


@pytest.fixture
def entry_point(entry_point_name):
    if entry_point_name == "run_main":
        return run_main
    elif entry_point_name == "cluster_main-1thread":
        return _wrapped_cluster_main1
    elif entry_point_name == "cluster_main-2thread":
        return _wrapped_cluster_main2
    else:
        msg = "unknown entry point name"
        raise ValueError(msg)


@pytest.fixture
def recovery_config(tmp_path: Path):
    init_db_dir(tmp_path, 4)
    return RecoveryConfig(tmp_path)


@pytest.fixture
def now():
    from datetime import datetime, timezone

    return datetime(2024, 1, 1, tzinfo=timezone.utc)


ZERO_TD = timedelta(seconds=0)
