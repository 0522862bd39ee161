"""MI355X HIP/CDNA4 fused RL kernels.

This package owns the native compute lane replacing the reference's external
DI-hpc CUDA wheel (opendilab/DI-engine ding/hpc_rl/wrapper.py:61-72 dispatch
table). Kernels live in ``csrc/*.hip`` and are built in-tree for gfx950 by
``python setup_ops.py`` (driven from __graft_entry__.build()).

Dispatch policy (ding/ops/dispatch.py): rl_utils entry points call
``dispatch.use_hip(x)`` — True iff the extension is importable AND x is on a
HIP device AND DI_ENGINE_DISABLE_HIP is unset. On a GPU box with the
extension missing the dispatch raises, never silently falls back.
"""
from . import dispatch
from .dispatch import is_available, hip_ops
