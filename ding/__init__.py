"""MI355X-native decision-intelligence engine.

Capability parity target: opendilab/DI-engine v0.5.3 (``ding`` package).
This is a from-scratch build for AMD Instinct MI355X (gfx950):

* PyTorch-ROCm compute path; hot RL math ops are hand-written HIP/CDNA4
  kernels (``ding.ops``) dispatched transparently from ``ding.rl_utils``.
* Multi-GPU training is one process per GPU over RCCL/xGMI
  (``torch.distributed`` backend "nccl" == RCCL on ROCm) with bucketed
  async all-reduce overlapping backward (``ding.utils.dist_helper``).
* The async Task/Middleware runtime (``ding.framework``) and env managers
  run on CPU processes; cross-process transport is a stdlib TCP event bus.

Environment flags (parity with reference ding/__init__.py:10-12):
  DI_ENGINE_DISABLE_HIP=1  -- force the pure PyTorch op lane even on GPU.
"""
import os

__TITLE__ = "DI-engine-MI355X"
__VERSION__ = "v0.1.0"
__version__ = __VERSION__

enable_hip = os.environ.get("DI_ENGINE_DISABLE_HIP", "0") not in ("1", "true", "True")
