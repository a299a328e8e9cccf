"""Long-lived framed channels over a transport.

A :class:`Channel` wraps one persistent remote process (a worker,
remote/worker_template.py) whose stdin/stdout carry 4-byte
length-prefixed frames (8-byte big-endian lengths: single tensor
buffers can exceed 4 GiB on a 288 GB HBM3E node).  Channels are how persistent workers are fed:
over SSH the process rides the pooled ControlMaster (no per-task
handshake), locally it is a plain subprocess — either way the remote
python + HIP runtime stay warm across electrons.
"""

from __future__ import annotations

import asyncio
import struct
from typing import List, Optional


class ChannelClosed(ConnectionError):
    """The remote worker process went away (EOF on its stdout)."""


class Channel:
    def __init__(self, proc: asyncio.subprocess.Process, label: str = "worker"):
        self._proc = proc
        self._label = label
        self._lock = asyncio.Lock()  # guards the write side
        self._inflight = 0  # requests sent, replies not yet resolved
        self.loop = asyncio.get_event_loop()

    @property
    def alive(self) -> bool:
        return self._proc.returncode is None

    @property
    def inflight(self) -> int:
        return self._inflight

    async def send_frame(self, payload) -> None:
        """``payload`` is bytes or any buffer-protocol object (ctypes
        views of pinned/tensor memory go out without copies)."""
        if not self.alive:
            raise ChannelClosed(f"{self._label}: process exited")
        view = memoryview(payload).cast("B")
        if view.nbytes <= 64 * 1024:
            # hot path (small control/request frames): one buffered write
            # instead of two halves the syscall/transport work per frame
            self._proc.stdin.write(struct.pack(">Q", view.nbytes) + view.tobytes())
        else:
            # large frames (tensor buffers): never copy
            self._proc.stdin.write(struct.pack(">Q", view.nbytes))
            self._proc.stdin.write(view)
        await self._proc.stdin.drain()

    async def recv_frame(self, timeout: Optional[float] = None) -> bytes:
        async def _read() -> bytes:
            header = await self._proc.stdout.readexactly(8)
            (length,) = struct.unpack(">Q", header)
            if length == 0:
                return b""
            if length <= (4 << 20):
                return await self._proc.stdout.readexactly(length)
            # large frame (tensor buffer): assemble into ONE preallocated
            # bytearray instead of readexactly's chunk-list + join + a
            # later bytearray copy — tensors reconstruct zero-copy on it
            buf = bytearray(length)
            view = memoryview(buf)
            pos = 0
            while pos < length:
                # read(remaining): takes the reader's ENTIRE internal
                # buffer in one pop.  Small fixed-size reads here are
                # quadratic — StreamReader deletes consumed bytes from
                # the front of its (up to 2*limit = 128 MiB) buffer, an
                # O(buffer) memmove per call.
                chunk = await self._proc.stdout.read(length - pos)
                if not chunk:
                    raise asyncio.IncompleteReadError(bytes(view[:pos]), length)
                view[pos : pos + len(chunk)] = chunk
                pos += len(chunk)
            return buf

        try:
            if timeout is None:
                return await _read()
            return await asyncio.wait_for(_read(), timeout=timeout)
        except (asyncio.IncompleteReadError, ConnectionResetError) as e:
            raise ChannelClosed(f"{self._label}: EOF mid-frame") from e

    async def request(self, payload: bytes, timeout: Optional[float] = None) -> bytes:
        """Serialized request/response round trip (single response frame)."""
        async with self._lock:
            await self.send_frame(payload)
            return await self.recv_frame(timeout=timeout)

    def transaction(self):
        """Async context manager holding the channel's request lock for a
        multi-frame exchange (e.g. a response followed by raw
        tensor-buffer frames)."""
        return _Transaction(self)

    # -- pipelined exchange --------------------------------------------
    #
    # The worker protocol is strictly ordered: responses come back in
    # request order.  ``exchange`` therefore only needs the write lock
    # while SENDING its request frames; replies are consumed by a single
    # reader pump that resolves waiter futures FIFO.  Over a real SSH
    # link this removes the one-RTT-per-electron serialization the
    # plain transaction() exchange pays.

    def _ensure_pump(self) -> None:
        if getattr(self, "_pump_task", None) is None or self._pump_task.done():
            self._waiters: "asyncio.Queue" = asyncio.Queue()
            self._pump_task = asyncio.get_running_loop().create_task(self._pump())

    def _fail_waiters(self, first=None) -> None:
        if first is not None and not first.done():
            first.set_exception(ChannelClosed(f"{self._label}: channel down"))
        waiters = getattr(self, "_waiters", None)
        while waiters is not None and not waiters.empty():
            w, _ = waiters.get_nowait()
            if not w.done():
                try:
                    w.set_exception(ChannelClosed(f"{self._label}: channel down"))
                except RuntimeError:
                    pass  # waiter's loop already gone

    async def _pump(self) -> None:
        fut = None
        try:
            while True:
                fut, reply_frames = await self._waiters.get()
                main = await self.recv_frame()
                extra_count = reply_frames(main)
                while extra_count is None:
                    # control frame (e.g. the worker's A1 execution ack)
                    # consumed by the callback: the real reply follows
                    main = await self.recv_frame()
                    extra_count = reply_frames(main)
                extras = [await self.recv_frame() for _ in range(extra_count)]
                if not fut.done():
                    fut.set_result((main, extras))
                fut = None
        except asyncio.CancelledError:
            # channel killed/closed: fail whatever is still waiting
            self._fail_waiters(fut)
            raise
        except Exception as e:  # noqa: BLE001 - fail this + later waiters
            if fut is not None and not fut.done():
                fut.set_exception(
                    e
                    if isinstance(e, ChannelClosed)
                    else ChannelClosed(f"{self._label}: reader failed: {e!r}")
                )
            self._fail_waiters()

    async def exchange(
        self,
        request_frames,
        reply_frames,
        timeout: Optional[float] = None,
    ):
        """Pipelined request/response.

        ``request_frames``: iterable of payloads sent back-to-back under
        the write lock.  ``reply_frames``: callable(main_frame) -> number
        of extra raw frames to read for this reply, or ``None`` to mean
        "that was a control frame; keep reading".  Returns
        ``(main_frame, [extra_frames])``.  Multiple callers may have
        requests in flight concurrently; replies resolve in order.
        """
        self._ensure_pump()
        fut = asyncio.get_running_loop().create_future()
        self._inflight += 1
        try:
            async with self._lock:
                self._waiters.put_nowait((fut, reply_frames))
                for frame in request_frames:
                    await self.send_frame(frame)
            if timeout is None:
                return await fut
            return await asyncio.wait_for(fut, timeout=timeout)
        finally:
            self._inflight -= 1

    def kill(self) -> None:
        """Synchronous hard-kill (for reaping workers whose event loop is
        already gone)."""
        task = getattr(self, "_pump_task", None)
        if task is not None and not task.done():
            try:
                task.cancel()
            except RuntimeError:
                pass  # task's loop already closed
        if self.alive:
            try:
                self._proc.kill()
            except ProcessLookupError:
                pass

    async def close(self) -> None:
        task = getattr(self, "_pump_task", None)
        if task is not None and not task.done():
            task.cancel()
        if self.loop is not asyncio.get_event_loop() or self.loop.is_closed():
            # channel belongs to another (likely closed) loop: its pipe
            # transports cannot be driven from here — hard-kill instead
            self.kill()
            return
        if self.alive:
            try:
                # zero-length frame = orderly shutdown
                self._proc.stdin.write(struct.pack(">Q", 0))
                await self._proc.stdin.drain()
                self._proc.stdin.close()
            except (ConnectionResetError, BrokenPipeError, RuntimeError):
                pass
            try:
                await asyncio.wait_for(self._proc.wait(), timeout=5)
            except asyncio.TimeoutError:
                self._proc.kill()
                await self._proc.wait()


class _Transaction:
    def __init__(self, channel: "Channel"):
        self._channel = channel

    async def __aenter__(self) -> "Channel":
        await self._channel._lock.acquire()
        return self._channel

    async def __aexit__(self, *exc) -> None:
        self._channel._lock.release()


async def open_subprocess_channel(argv: List[str], label: str) -> Channel:
    proc = await asyncio.create_subprocess_exec(
        *argv,
        stdin=asyncio.subprocess.PIPE,
        stdout=asyncio.subprocess.PIPE,
        stderr=None,  # worker/user stderr passes through to our logs
        limit=64 * 1024 * 1024,
    )
    return Channel(proc, label=label)
