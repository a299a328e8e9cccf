"""Multi-node fan-out: one executor per 8×MI355X node, least-loaded
dispatch across them.

The reference (and the base ``SSHExecutor``) targets ONE remote host.
``SSHClusterExecutor`` composes per-node executors — each with its own
pooled transport, GPU slot table and persistent workers — and places
every electron on the node with the fewest tasks in flight (ties broken
round-robin).  All per-node mechanics (slot pinning, warm workers,
pinned staging, retries) apply unchanged.

Usage::

    cluster = SSHClusterExecutor(
        hosts=["alice@node0", "alice@node1"],
        gpu_slots=8,
        persistent_workers=True,
    )
    result = await cluster.execute(fn, args, kwargs)
"""

from __future__ import annotations

import asyncio
import re
import time
from typing import Any, Callable, Dict, List, Optional, Union

from .compat import app_log
from .ssh import SSHConnectError, SSHExecutor

_HOST_RE = re.compile(r"^(?:(?P<user>[^@]+)@)?(?P<host>[^:@]+)(?::(?P<port>\d+))?$")


def _parse_host(spec: Union[str, dict]) -> dict:
    if isinstance(spec, dict):
        return dict(spec)
    m = _HOST_RE.match(spec)
    if not m:
        raise ValueError(f"bad host spec {spec!r} (want [user@]host[:port])")
    out: dict = {"hostname": m.group("host")}
    if m.group("user"):
        out["username"] = m.group("user")
    if m.group("port"):
        out["ssh_port"] = int(m.group("port"))
    return out


class SSHClusterExecutor:
    """Least-loaded fan-out over a list of MI355X nodes.

    With ``failover=True`` (default), a node that fails BEFORE task
    execution could start — connect exhaustion or environment-check
    failure, surfaced as :class:`SSHConnectError` — is marked unhealthy
    for ``failover_cooldown`` seconds and the electron transparently
    retries on the next-least-loaded node.  Failures after execution may
    have begun are never failed over (a non-idempotent task could run
    twice); they surface to the caller.
    """

    def __init__(
        self,
        hosts: List[Union[str, dict]],
        *,
        failover: bool = True,
        failover_cooldown: float = 30.0,
        **common_kwargs: Any,
    ):
        if not hosts:
            raise ValueError("hosts must be non-empty")
        self.failover = failover
        self.failover_cooldown = failover_cooldown
        self.executors: List[SSHExecutor] = []
        for spec in hosts:
            kwargs = dict(common_kwargs)
            kwargs.update(_parse_host(spec))
            self.executors.append(SSHExecutor(**kwargs))
        self._inflight = [0] * len(self.executors)
        self._unhealthy_until = [0.0] * len(self.executors)
        self._rr = 0

    def _pick(self, exclude: Optional[set] = None) -> Optional[int]:
        exclude = exclude or set()
        now = time.monotonic()
        healthy = [
            i
            for i in range(len(self.executors))
            if i not in exclude and self._unhealthy_until[i] <= now
        ]
        if not healthy:
            # every node cooling down / tried: fall back to anything
            # not yet tried this task rather than refusing outright
            healthy = [i for i in range(len(self.executors)) if i not in exclude]
        if not healthy:
            return None
        low = min(self._inflight[i] for i in healthy)
        candidates = [i for i in healthy if self._inflight[i] == low]
        idx = candidates[self._rr % len(candidates)]
        self._rr += 1
        return idx

    async def run(
        self,
        function: Callable,
        args: list,
        kwargs: dict,
        task_metadata: Optional[dict] = None,
    ) -> Any:
        tried: set = set()
        last_error: Optional[Exception] = None
        attempts = len(self.executors) if self.failover else 1
        for _ in range(attempts):
            idx = self._pick(exclude=tried)
            if idx is None:
                break
            tried.add(idx)
            self._inflight[idx] += 1
            try:
                return await self.executors[idx].run(
                    function, args, kwargs, task_metadata
                )
            except SSHConnectError as e:
                # pre-execution failure: node is unreachable/unfit —
                # cool it down and (maybe) fail over
                last_error = e
                self._unhealthy_until[idx] = (
                    time.monotonic() + self.failover_cooldown
                )
                if not self.failover:
                    raise
                app_log.warning(
                    "node %s failed pre-execution (%s); failing over",
                    self.executors[idx].hostname or idx,
                    e,
                )
            finally:
                self._inflight[idx] -= 1
        assert last_error is not None
        raise last_error

    async def execute(
        self,
        function: Callable,
        args: Optional[list] = None,
        kwargs: Optional[dict] = None,
        dispatch_id: str = "dispatch",
        node_id: int = 0,
    ) -> Any:
        return await self.run(
            function,
            list(args or []),
            dict(kwargs or {}),
            {"dispatch_id": dispatch_id, "node_id": node_id},
        )

    async def prewarm(self) -> int:
        counts = await asyncio.gather(*[ex.prewarm() for ex in self.executors])
        return sum(counts)

    def stats(self) -> Dict[str, dict]:
        return {
            ex.hostname or f"node{i}": ex.stats()
            for i, ex in enumerate(self.executors)
        }

    @property
    def capacity(self) -> int:
        return sum(ex.gpu_slots * ex.slots_per_gpu for ex in self.executors)
