"""MI355X-native Covalent SSH executor plugin.

Public surface mirrors the reference package
(/root/reference/covalent_ssh_plugin/__init__.py:17): the plugin exports
``SSHExecutor``; covalent discovers it through the
``covalent.executor.executor_plugins`` entry point (setup.py).
"""

from .cluster import SSHClusterExecutor
from .ssh import _EXECUTOR_PLUGIN_DEFAULTS, EXECUTOR_PLUGIN_NAME, SSHExecutor

__all__ = [
    "SSHExecutor",
    "SSHClusterExecutor",
    "EXECUTOR_PLUGIN_NAME",
    "_EXECUTOR_PLUGIN_DEFAULTS",
]

__version__ = "0.1.0"
