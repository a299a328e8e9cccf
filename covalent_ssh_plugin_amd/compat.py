"""Covalent host-framework compatibility layer.

The plugin consumes four things from the covalent host package (reference:
/root/reference/covalent_ssh_plugin/ssh.py:29-32): ``Result``, ``logger``,
``get_config`` and the ``RemoteExecutor`` base class.  When a real covalent
install is present we use it directly, so the plugin remains a drop-in
executor for covalent workflows.  When covalent is absent (CI containers,
the MI355X bench boxes — no network, no covalent wheel) we provide a
minimal, contract-faithful fallback so the executor works standalone.

The fallback ``RemoteExecutor`` reproduces the template contract that the
reference implements (abstract async methods ``_validate_credentials``,
``_upload_task``, ``submit_task``, ``get_status``, ``_poll_task``,
``query_result``, ``cancel``; ctor args ``poll_freq``/``remote_cache`` —
see SURVEY.md §1-L1 and reference ssh.py:98, CHANGELOG.md:110-118).
"""

from __future__ import annotations

import logging
import os
from typing import Any, Callable, Dict, Optional

__all__ = [
    "RemoteExecutor",
    "get_config",
    "set_config",
    "update_config_defaults",
    "app_log",
    "COVALENT_AVAILABLE",
]

try:  # pragma: no cover - exercised only when covalent is installed
    from covalent._shared_files import logger as _cov_logger
    from covalent._shared_files.config import get_config as _cov_get_config
    from covalent._shared_files.config import set_config as _cov_set_config
    from covalent.executor.executor_plugins.remote_executor import (
        RemoteExecutor as _CovRemoteExecutor,
    )

    COVALENT_AVAILABLE = True
except Exception:  # ModuleNotFoundError and any partial-install breakage
    COVALENT_AVAILABLE = False
    _cov_logger = None
    _cov_get_config = None
    _cov_set_config = None
    _CovRemoteExecutor = None


# ---------------------------------------------------------------------------
# Logger
# ---------------------------------------------------------------------------

if COVALENT_AVAILABLE:
    app_log = _cov_logger.app_log
else:
    app_log = logging.getLogger("covalent_ssh_plugin_amd")
    if os.environ.get("CSP_AMD_DEBUG"):
        logging.basicConfig(level=logging.DEBUG)


# ---------------------------------------------------------------------------
# Config store
# ---------------------------------------------------------------------------

# Flat dotted-key config store used when covalent's TOML config manager is
# absent.  Populated with the plugin defaults at import time (see ssh.py) the
# same way covalent merges _EXECUTOR_PLUGIN_DEFAULTS under [executors.ssh].
_fallback_config: Dict[str, Any] = {}


def update_config_defaults(prefix: str, defaults: Dict[str, Any]) -> None:
    """Merge ``defaults`` under ``prefix.`` without clobbering user values."""
    for key, value in defaults.items():
        _fallback_config.setdefault(f"{prefix}.{key}", value)


def set_config(key_or_dict, value: Any = None) -> None:
    if COVALENT_AVAILABLE:  # pragma: no cover
        if isinstance(key_or_dict, dict):
            _cov_set_config(key_or_dict)
        else:
            _cov_set_config(key_or_dict, value)
        return
    if isinstance(key_or_dict, dict):
        _fallback_config.update(key_or_dict)
    else:
        _fallback_config[key_or_dict] = value


def get_config(key: str) -> Any:
    """Dotted-key config lookup (e.g. ``executors.ssh.username``).

    With covalent installed, defer to its config manager (which raises
    KeyError for unknown keys, the gotcha noted in SURVEY.md §3.3).  The
    fallback raises KeyError for unknown keys too, so executor code paths
    behave identically in both environments.
    """
    if COVALENT_AVAILABLE:  # pragma: no cover
        return _cov_get_config(key)
    return _fallback_config[key]


# ---------------------------------------------------------------------------
# RemoteExecutor base
# ---------------------------------------------------------------------------

if COVALENT_AVAILABLE:  # pragma: no cover
    RemoteExecutor = _CovRemoteExecutor
else:

    class RemoteExecutor:
        """Minimal stand-in for covalent's RemoteExecutor template.

        Mirrors the observable contract the reference plugin relies on:
        the constructor stores ``poll_freq`` and ``remote_cache`` and the
        dispatcher awaits ``run(function, args, kwargs, task_metadata)``.
        """

        def __init__(
            self,
            poll_freq: int = 15,
            remote_cache: str = "",
            *args: Any,
            **kwargs: Any,
        ) -> None:
            self.poll_freq = poll_freq
            self.remote_cache = remote_cache

        # --- template methods the concrete executor must implement -------
        async def _validate_credentials(self, *a: Any, **kw: Any) -> bool:
            raise NotImplementedError

        async def _upload_task(self, *a: Any, **kw: Any) -> None:
            raise NotImplementedError

        async def submit_task(self, *a: Any, **kw: Any) -> Any:
            raise NotImplementedError

        async def get_status(self, *a: Any, **kw: Any) -> Any:
            raise NotImplementedError

        async def _poll_task(self, *a: Any, **kw: Any) -> Any:
            raise NotImplementedError

        async def query_result(self, *a: Any, **kw: Any) -> Any:
            raise NotImplementedError

        async def cancel(self, *a: Any, **kw: Any) -> None:
            raise NotImplementedError

        async def run(
            self,
            function: Callable,
            args: list,
            kwargs: dict,
            task_metadata: Optional[dict] = None,
        ) -> Any:
            raise NotImplementedError

        # Convenience used by tests / direct (covalent-less) invocation.
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
