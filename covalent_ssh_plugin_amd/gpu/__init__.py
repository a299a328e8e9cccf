from . import slots  # noqa: F401
from .probe import GpuLibError, local_gpu_lib_path  # noqa: F401
