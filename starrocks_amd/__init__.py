"""starrocks_amd — MI355X-native execution engine for the StarRocks BE hot path.

The product path is the HIP C-ABI library (csrc/gpue.hip -> libgpue.so),
declared in include/gpue.h. This package is the host-side mirror of the
reference's pipeline::Operator surface for that path plus the
shared synthetic-data generator. It has NO CPU compute fallback: engine use
without the HIP library or an AMD GPU raises.
"""

from .engine import Engine, GpueError, lib_path  # noqa: F401
from . import gen  # noqa: F401
