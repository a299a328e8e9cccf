"""Connect/retry policy matrix — the reference's retry tests
(reference tests/ssh_test.py:193-257) rebuilt against the transport-pool
design: a fake transport factory fails N times, then succeeds."""

import asyncio

import pytest

from covalent_ssh_plugin_amd import SSHExecutor
from covalent_ssh_plugin_amd.transport import TransportConnectError
from covalent_ssh_plugin_amd.transport.base import CompletedCommand, Transport


class FlakyTransport(Transport):
    """Fails ``failures`` connect attempts, then connects."""

    def __init__(self, failures: int):
        self.failures = failures
        self.attempts = 0
        self._connected = False
        self.endpoint = "fake"

    async def connect(self):
        self.attempts += 1
        if self.attempts <= self.failures:
            raise TransportConnectError(f"refused (attempt {self.attempts})")
        self._connected = True

    @property
    def is_connected(self):
        return self._connected

    async def run(self, command, *, input_data=None, env=None, timeout=None):
        return CompletedCommand(0, b"", b"")

    async def put_files(self, files):
        pass

    async def get_file(self, remote_path, local_path):
        pass

    async def open_channel(self, command, env=None):
        raise NotImplementedError

    async def open_pipe(self, command, env=None):
        raise NotImplementedError

    async def close(self):
        self._connected = False


def _executor(**kw):
    kw.setdefault("username", "u")
    kw.setdefault("hostname", "h")
    kw.setdefault("ssh_key_file", "/dev/null")  # exists -> validation passes
    return SSHExecutor(**kw)


def _patch_factory(ex, transport):
    ex._make_transport = lambda: transport


def test_connect_immediate_success():
    ex = _executor(retry_wait_time=0)
    t = FlakyTransport(failures=0)
    _patch_factory(ex, t)
    got = asyncio.run(ex._client_connect())
    assert got is t
    assert t.attempts == 1


def test_connect_eventual_success():
    ex = _executor(retry_wait_time=0, max_connection_attempts=5)
    t = FlakyTransport(failures=3)
    _patch_factory(ex, t)
    got = asyncio.run(ex._client_connect())
    assert got is t
    assert t.attempts == 4


def test_connect_exhausted_returns_none():
    """Reference returns None after exhaustion (reference ssh.py:282)."""
    ex = _executor(retry_wait_time=0, max_connection_attempts=3)
    t = FlakyTransport(failures=99)
    _patch_factory(ex, t)
    assert asyncio.run(ex._client_connect()) is None
    assert t.attempts == 3


def test_connect_no_retry_raises_immediately():
    """retry_connect=False raises on first failure (reference ssh.py:271-273)."""
    ex = _executor(retry_connect=False, retry_wait_time=0)
    t = FlakyTransport(failures=99)
    _patch_factory(ex, t)
    with pytest.raises(TransportConnectError):
        asyncio.run(ex._client_connect())
    assert t.attempts == 1


def test_run_surfaces_connect_exhaustion_as_policy():
    """run() maps connect exhaustion to _on_ssh_fail: RuntimeError when
    fallback is off, local execution when it is on (reference
    ssh.py:499-501)."""
    ex = _executor(retry_wait_time=0, max_connection_attempts=2)
    _patch_factory(ex, FlakyTransport(failures=99))
    with pytest.raises(RuntimeError, match="Could not connect"):
        asyncio.run(ex.execute(lambda: 1, [], {}))

    ex2 = _executor(retry_wait_time=0, max_connection_attempts=2, run_local_on_ssh_fail=True)
    _patch_factory(ex2, FlakyTransport(failures=99))
    assert asyncio.run(ex2.execute(lambda x: x * 5, [5], {})) == 25


def test_missing_key_file_raises():
    """Credential validation parity (reference ssh.py:317-335)."""
    ex = _executor(ssh_key_file="/nonexistent/key/file")
    with pytest.raises(RuntimeError, match="no SSH key file"):
        asyncio.run(ex._validate_credentials())
    assert asyncio.run(ex._validate_credentials(raise_exception=False)) is False


def test_cancel_not_implemented():
    """Parity: cancellation unsupported (reference ssh.py:460-464)."""
    ex = _executor()
    with pytest.raises(NotImplementedError):
        asyncio.run(ex.cancel())
