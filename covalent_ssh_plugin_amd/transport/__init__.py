from .base import (
    CompletedCommand,
    Transport,
    TransportCommandError,
    TransportConnectError,
    make_tar_stream,
)
from .local import LocalTransport
from .openssh import OpenSSHTransport

__all__ = [
    "CompletedCommand",
    "Transport",
    "TransportCommandError",
    "TransportConnectError",
    "make_tar_stream",
    "LocalTransport",
    "OpenSSHTransport",
]
