"""Kernel-level operations.

The CDNA4 kernels live in ``csrc/swarmq_module.hip`` (compiled to the
in-tree ``_swarmq`` extension by ``build_ext.py``); their Python-facing
surface is :class:`swarmdb_amd.runtime.gpu_engine.GpuEngine` and the raw
``_swarmq.DeviceQueue`` binding. Kernel inventory and geometry:
docs/ARCHITECTURE.md §Kernels.
"""

from ..runtime.engine import (  # noqa: F401
    BROADCAST,
    NO_BITMAP,
    REC_DTYPE,
    VIS_ALL,
    VIS_BITMAP,
    VIS_GROUP,
)
