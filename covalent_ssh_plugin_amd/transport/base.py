"""Transport abstraction for the MI355X SSH executor.

The reference plugin hard-wires asyncssh (reference ssh.py:263-268,
360-361, 451).  This build abstracts the remote channel behind a small
async interface so the same executor logic drives:

* :class:`~covalent_ssh_plugin_amd.transport.openssh.OpenSSHTransport` —
  the production transport, multiplexing every command and file transfer
  over one pooled OpenSSH ControlMaster connection per (host, user, key),
* :class:`~covalent_ssh_plugin_amd.transport.local.LocalTransport` — a
  loopback transport with identical semantics (spawns the same remote
  stub processes via bash) for offline CI and for benchmarking on boxes
  with no sshd.

All methods are coroutines; one transport instance may serve many
concurrent tasks (commands multiplex; there is no per-task connection).
"""

from __future__ import annotations

import dataclasses
from abc import ABC, abstractmethod
from typing import Optional, Sequence, Tuple


class TransportConnectError(ConnectionError):
    """Raised when the transport cannot establish its connection."""


class TransportCommandError(RuntimeError):
    """Raised when a command cannot be executed at the transport level.

    (A *remote command* failing with nonzero exit is NOT this error — that
    is reported through :class:`CompletedCommand.returncode` so executor
    policy code can decide what to do, matching the reference's use of
    ``SSHCompletedProcess.exit_status`` at reference ssh.py:383,553.)
    """


@dataclasses.dataclass
class CompletedCommand:
    """Result of one remote command (reference analog: SSHCompletedProcess)."""

    returncode: int
    stdout: bytes
    stderr: bytes

    @property
    def ok(self) -> bool:
        return self.returncode == 0

    def text_out(self) -> str:
        return self.stdout.decode(errors="replace")

    def text_err(self) -> str:
        return self.stderr.decode(errors="replace")


class Transport(ABC):
    """Async channel to the execution host."""

    #: human-readable endpoint, e.g. "user@host" or "local"
    endpoint: str = "?"

    @abstractmethod
    async def connect(self) -> None:
        """Establish the (master) connection.  Idempotent.

        Raises :class:`TransportConnectError` on failure.
        """

    @abstractmethod
    async def run(
        self,
        command: str,
        *,
        input_data: Optional[bytes] = None,
        env: Optional[dict] = None,
        timeout: Optional[float] = None,
    ) -> CompletedCommand:
        """Run ``command`` through the remote login shell.

        ``input_data`` is fed to the command's stdin (used by the fused
        dispatch path to stream the staged files as a tar archive in the
        same round trip as the execution command).  ``env`` entries are
        injected as ``K=V`` prefixes (GPU slot pinning uses this for
        ``HIP_VISIBLE_DEVICES``).
        """

    @abstractmethod
    async def put_files(self, files: Sequence[Tuple[str, str]]) -> None:
        """Upload ``[(local_path, remote_path), ...]`` in one batch.

        Remote paths may be relative (to the remote home) or absolute;
        parent directories are created.
        """

    @abstractmethod
    async def get_file(self, remote_path: str, local_path: str) -> None:
        """Download one remote file."""

    @abstractmethod
    async def close(self) -> None:
        """Tear down the connection.  Idempotent."""

    @property
    @abstractmethod
    def is_connected(self) -> bool: ...

    @abstractmethod
    async def open_channel(self, command: str, env: Optional[dict] = None):
        """Start a long-lived remote process and return a
        :class:`~covalent_ssh_plugin_amd.transport.channel.Channel` to
        its stdin/stdout (used for persistent workers)."""

    @abstractmethod
    async def open_pipe(self, command: str, env: Optional[dict] = None):
        """Start a remote command with piped stdin/stdout/stderr and
        return the raw :class:`asyncio.subprocess.Process`.

        Unlike :meth:`run` (which buffers the whole stdout in memory via
        ``communicate``), the caller streams both directions — the fused
        dispatch path uses this to spool multi-GiB results to disk with
        bounded dispatcher RSS."""

    # -- helpers shared by implementations ---------------------------------

    @staticmethod
    def _env_prefix(env: Optional[dict]) -> str:
        if not env:
            return ""
        import shlex

        parts = [f"{k}={shlex.quote(str(v))}" for k, v in env.items()]
        return "export " + " ".join(parts) + " && "


def make_tar_stream(files: Sequence[Tuple[str, str]]) -> Tuple[bytes, str]:
    """Pack local files into an in-memory tar whose member names are the
    remote paths, for single-round-trip extraction on the far side.

    All remote paths must be uniformly absolute or uniformly relative:
    absolute paths are stored with the leading ``/`` stripped and extracted
    with ``tar -C /``; relative paths extract relative to the remote $HOME
    (sshd's login cwd).  Returns ``(tar_bytes, extract_base)`` where
    ``extract_base`` is ``"/"`` or ``""`` (home).
    """
    import io
    import tarfile

    abs_flags = {remote.startswith("/") for _, remote in files}
    if len(abs_flags) > 1:
        raise ValueError("mixed absolute/relative remote paths in one batch")
    is_abs = abs_flags.pop() if abs_flags else False

    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w") as tf:
        for local, remote in files:
            name = remote.lstrip("/") if is_abs else remote
            tf.add(local, arcname=name)
    return buf.getvalue(), "/" if is_abs else ""


def make_tar_spool(files: Sequence[Tuple[str, str]], max_memory: int = 32 << 20):
    """Like :func:`make_tar_stream` but into a ``SpooledTemporaryFile``:
    archives beyond ``max_memory`` roll over to disk, so staging a
    multi-GiB function payload never holds a full tar copy in dispatcher
    RAM.  Returns ``(fileobj_at_pos0, extract_base)``;
    the caller closes the file object."""
    import tarfile
    import tempfile

    abs_flags = {remote.startswith("/") for _, remote in files}
    if len(abs_flags) > 1:
        raise ValueError("mixed absolute/relative remote paths in one batch")
    is_abs = abs_flags.pop() if abs_flags else False

    spool = tempfile.SpooledTemporaryFile(max_size=max_memory)
    with tarfile.open(fileobj=spool, mode="w") as tf:
        for local, remote in files:
            name = remote.lstrip("/") if is_abs else remote
            tf.add(local, arcname=name)
    spool.seek(0)
    return spool, "/" if is_abs else ""
