from .rccl_smoke import make_rccl_smoke_electron  # noqa: F401
