"""Module-level keyed transport pool.

The reference opens and closes one SSH connection per task (reference
ssh.py:497, 585-587) and leaks it when the task raised (ssh.py:581-583 —
bug noted in SURVEY.md §3.1).  Here, all executor instances that target
the same (hostname, username, key, port) share ONE pooled transport; the
pool also hosts the per-endpoint one-time environment checks (python
version, conda env existence) so they are paid once per endpoint, not
once per task (hoisted out of reference ssh.py:508-524).

The pool is keyed module-level state on purpose: multiple SSHExecutor
instances pointing at the same host must share connection + GPU-slot
accounting (SURVEY.md §7 "Hard parts").
"""

from __future__ import annotations

import asyncio
import time
from typing import Callable, Dict, Optional, Tuple

from .base import Transport

PoolKey = Tuple[str, ...]

#: how long a FAILED check result stays cached (seconds).  Successful
#: checks are cached for the life of the process (the hoisting win);
#: failures must heal: a transient remote hiccup (conda glitch, disk
#: full during mkdir) may be gone on the next attempt (ADVICE r1:
#: "failed environment checks are cached forever").
NEGATIVE_CHECK_TTL = 60.0

_pool: Dict[PoolKey, Transport] = {}
_pool_lock: Optional[asyncio.Lock] = None
_pool_lock_loop = None
# One-time environment check results, keyed by (pool_key, check_name):
# (value, monotonic_timestamp, ok)
_env_checks: Dict[Tuple[PoolKey, str], Tuple[object, float, bool]] = {}


def _lock() -> asyncio.Lock:
    # Created lazily so the lock binds to the RUNNING loop; recreated if
    # a new asyncio.run() started a fresh loop (the old loop is closed,
    # so it can have no live waiters).
    global _pool_lock, _pool_lock_loop
    loop = asyncio.get_running_loop()
    if _pool_lock is None or _pool_lock_loop is not loop:
        _pool_lock = asyncio.Lock()
        _pool_lock_loop = loop
    return _pool_lock


_check_locks: Dict[Tuple[PoolKey, object], Tuple[asyncio.Lock, object]] = {}


def check_lock(key: PoolKey) -> asyncio.Lock:
    """Loop-aware per-endpoint lock serializing one-time environment
    checks (a cold-start fan of N electrons must run them once, not N
    times)."""
    loop = asyncio.get_running_loop()
    entry = _check_locks.get(key)
    if entry is None or entry[1] is not loop:
        entry = (asyncio.Lock(), loop)
        _check_locks[key] = entry
    return entry[0]


async def get_transport(key: PoolKey, factory: Callable[[], Transport]) -> Transport:
    """Return the pooled, connected transport for ``key``, creating it
    with ``factory`` (and connecting it) on first use.

    If the pooled transport has dropped its connection, reconnect it —
    and on a successful RE-connect, drop any cached FAILED environment
    checks for the endpoint: the failure may have died with the old
    connection (e.g. the remote host was rebooted/fixed).
    """
    async with _lock():
        transport = _pool.get(key)
        if transport is None:
            transport = factory()
            _pool[key] = transport
    if not transport.is_connected:
        reconnecting = bool(getattr(transport, "_csp_ever_connected", False))
        await transport.connect()
        if reconnecting:
            invalidate_failed_checks(key)
    transport._csp_ever_connected = True
    return transport


def cached_check(key: PoolKey, name: str):
    """Cached check value, or None if absent / an expired failure."""
    entry = _env_checks.get((key, name))
    if entry is None:
        return None
    value, ts, ok = entry
    if not ok and (time.monotonic() - ts) > NEGATIVE_CHECK_TTL:
        _env_checks.pop((key, name), None)
        return None
    return value


def store_check(key: PoolKey, name: str, value: object, ok: bool = True) -> None:
    """Cache a check result.  ``ok=False`` marks it as a failure, which
    expires after NEGATIVE_CHECK_TTL and is dropped on reconnect."""
    _env_checks[(key, name)] = (value, time.monotonic(), ok)


def invalidate_failed_checks(key: PoolKey) -> None:
    """Drop every cached FAILED check for ``key`` (called on transport
    reconnect; a fresh connection deserves a fresh verdict)."""
    stale = [k for k, (_v, _ts, ok) in _env_checks.items() if k[0] == key and not ok]
    for k in stale:
        _env_checks.pop(k, None)


async def close_all() -> None:
    async with _lock():
        transports = list(_pool.values())
        _pool.clear()
        _env_checks.clear()
        _check_locks.clear()
    for t in transports:
        try:
            await t.close()
        except Exception:
            pass


def reset() -> None:
    """Synchronous test hook: forget pooled state without closing."""
    _pool.clear()
    _env_checks.clear()
    _check_locks.clear()
