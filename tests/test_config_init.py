"""Constructor / config-resolution parity tests.

Models the reference's test_init coverage (reference
tests/ssh_test.py:46-69): explicit ctor args win, config fills the rest,
hardcoded defaults backstop everything.
"""

import os
from pathlib import Path

from covalent_ssh_plugin_amd import (
    _EXECUTOR_PLUGIN_DEFAULTS,
    EXECUTOR_PLUGIN_NAME,
    SSHExecutor,
)
from covalent_ssh_plugin_amd.compat import set_config


def test_plugin_metadata():
    assert EXECUTOR_PLUGIN_NAME == "SSHExecutor"
    # The ten reference config keys must all be present (reference
    # ssh.py:39-50).
    for key in [
        "username",
        "hostname",
        "ssh_key_file",
        "cache_dir",
        "python_path",
        "conda_env",
        "remote_cache",
        "run_local_on_ssh_fail",
        "remote_workdir",
        "create_unique_workdir",
    ]:
        assert key in _EXECUTOR_PLUGIN_DEFAULTS, key


def test_explicit_args_win():
    ex = SSHExecutor(
        username="alice",
        hostname="node0",
        ssh_key_file="/tmp/k",
        python_path="python3.11",
        conda_env="rocm",
        remote_workdir="wd",
        poll_freq=3,
        do_cleanup=False,
        retry_connect=False,
        max_connection_attempts=2,
        retry_wait_time=1,
        gpu_slots=4,
    )
    assert ex.username == "alice"
    assert ex.hostname == "node0"
    assert ex.ssh_key_file == "/tmp/k"
    assert ex.python_path == "python3.11"
    assert ex.conda_env == "rocm"
    assert ex.remote_workdir == "wd"
    assert ex.poll_freq == 3
    assert ex.do_cleanup is False
    assert ex.retry_connect is False
    assert ex.max_connection_attempts == 2
    assert ex.retry_wait_time == 1
    assert ex.gpu_slots == 4


def test_defaults_without_config():
    ex = SSHExecutor(username="u", hostname="h")
    assert ex.python_path == "python"
    assert ex.remote_cache == ".cache/covalent"
    assert ex.remote_workdir == "covalent-workdir"
    assert ex.create_unique_workdir is False
    assert ex.poll_freq == 15
    assert ex.do_cleanup is True
    assert ex.retry_connect is True
    assert ex.max_connection_attempts == 5
    assert ex.retry_wait_time == 5
    assert ex.gpu_slots == 8
    assert ex.slots_per_gpu == 1
    assert ex.hip_visible_devices_policy == "roundrobin"
    assert ex.batch_roundtrips is True


def test_config_fills_unset_args():
    set_config("executors.ssh.username", "configured-user")
    set_config("executors.ssh.python_path", "python3.10")
    try:
        ex = SSHExecutor(hostname="h")
        assert ex.username == "configured-user"
        assert ex.python_path == "python3.10"
        # explicit arg still wins over config
        ex2 = SSHExecutor(username="explicit", hostname="h")
        assert ex2.username == "explicit"
    finally:
        set_config("executors.ssh.username", "")
        set_config("executors.ssh.python_path", "python")


def test_key_file_expansion():
    ex = SSHExecutor(username="u", hostname="h", ssh_key_file="~/somekey")
    assert ex.ssh_key_file == os.path.join(os.path.expanduser("~"), "somekey")


def test_cache_dir_resolved(tmp_path):
    ex = SSHExecutor(username="u", hostname="h", cache_dir=str(tmp_path / "c"))
    assert Path(ex.cache_dir).is_absolute()
