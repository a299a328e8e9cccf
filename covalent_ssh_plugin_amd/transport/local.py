"""Loopback transport: identical semantics to SSH, no network hop.

Commands run through a local ``bash -c`` subprocess (the same shell
behavior sshd gives a command), file "uploads" are filesystem copies, and
relative remote paths resolve against a configurable fake remote home
(default: the real ``$HOME``).  This is what offline CI tests and
``bench.py`` use — the whole executor pipeline (serialization, staging,
stub subprocess spawn, result pickup, cleanup, GPU slot injection) is
exercised for real; only the network round trip is absent.  No sshd
exists in the CI or MI355X images, so this is also the measured transport
for BASELINE configs (noted in the bench output).
"""

from __future__ import annotations

import asyncio
import os
import shutil
from pathlib import Path
from typing import Optional, Sequence, Tuple

from .base import CompletedCommand, Transport, TransportConnectError


class LocalTransport(Transport):
    def __init__(self, home: Optional[str] = None):
        # `home` stands in for the remote user's $HOME (sshd login cwd).
        self._home = Path(home or os.path.expanduser("~")).resolve()
        self._connected = False
        self.endpoint = f"local:{self._home}"

    @property
    def home(self) -> Path:
        return self._home

    def _resolve(self, remote_path: str) -> Path:
        p = Path(os.path.expanduser(remote_path))
        if not p.is_absolute():
            p = self._home / p
        return p

    async def connect(self) -> None:
        if not self._home.is_dir():
            raise TransportConnectError(f"local home {self._home} does not exist")
        self._connected = True

    @property
    def is_connected(self) -> bool:
        return self._connected

    async def run(
        self,
        command: str,
        *,
        input_data: Optional[bytes] = None,
        env: Optional[dict] = None,
        timeout: Optional[float] = None,
    ) -> CompletedCommand:
        if not self._connected:
            raise TransportConnectError("transport not connected")
        full_cmd = self._env_prefix(env) + command
        proc = await asyncio.create_subprocess_exec(
            "bash",
            "-c",
            full_cmd,
            cwd=str(self._home),
            stdin=asyncio.subprocess.PIPE if input_data is not None else asyncio.subprocess.DEVNULL,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
        )
        try:
            stdout, stderr = await asyncio.wait_for(
                proc.communicate(input=input_data), timeout=timeout
            )
        except asyncio.TimeoutError:
            proc.kill()
            await proc.wait()
            raise
        return CompletedCommand(proc.returncode, stdout, stderr)

    async def put_files(self, files: Sequence[Tuple[str, str]]) -> None:
        for local, remote in files:
            dst = self._resolve(remote)
            dst.parent.mkdir(parents=True, exist_ok=True)
            await asyncio.to_thread(shutil.copyfile, local, dst)

    async def get_file(self, remote_path: str, local_path: str) -> None:
        src = self._resolve(remote_path)
        Path(local_path).parent.mkdir(parents=True, exist_ok=True)
        await asyncio.to_thread(shutil.copyfile, src, local_path)

    async def open_channel(self, command: str, env: Optional[dict] = None):
        from .channel import Channel

        full_cmd = self._env_prefix(env) + command
        proc = await asyncio.create_subprocess_exec(
            "bash",
            "-c",
            full_cmd,
            cwd=str(self._home),
            stdin=asyncio.subprocess.PIPE,
            stdout=asyncio.subprocess.PIPE,
            stderr=None,
            limit=64 * 1024 * 1024,
        )
        return Channel(proc, label=f"local-worker")

    async def open_pipe(self, command: str, env: Optional[dict] = None):
        full_cmd = self._env_prefix(env) + command
        return await asyncio.create_subprocess_exec(
            "bash",
            "-c",
            full_cmd,
            cwd=str(self._home),
            stdin=asyncio.subprocess.PIPE,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
            limit=4 * 1024 * 1024,
        )

    async def close(self) -> None:
        self._connected = False
