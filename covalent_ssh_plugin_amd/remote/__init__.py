from .stub import render_stub  # noqa: F401
