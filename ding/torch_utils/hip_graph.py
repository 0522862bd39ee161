"""hipGraph capture for launch-bound training steps.

MI355X note: rocprof on the Atari PPO bench showed ~570 kernel dispatches per
2.5 ms minibatch (conv fwd/bwd + optimizer) — the step is launch-bound, not
kernel-bound. `torch.cuda.CUDAGraph` on ROCm records the whole step into ONE
hipGraph that replays with a single `hipGraphLaunch`, removing per-kernel
launch latency and host-side dispatch work.

Capture rules honored here:
- warmup iterations run on a side stream first (MIOpen find / cuDNN-benchmark
  style autotuning and cold allocations must not happen inside capture);
- all tensors the step reads are STATIC buffers — callers pass fresh data and
  we ``copy_`` it in before replay;
- the step function must be sync-free (no ``.item()``/``.cpu()``) — return
  0-dim GPU tensors and read them after replay.
"""
from typing import Callable, Dict, Optional

import torch


class GraphedStep:
    """Capture ``step_fn(static_inputs) -> dict[str, Tensor]`` into a hipGraph
    keyed by input shapes; re-captures transparently if shapes change.

    ``step_fn`` may run model forward, loss, ``backward()`` and (single-GPU)
    ``optimizer.step()``; everything it launches lands in the graph.
    """

    def __init__(self, step_fn: Callable[[Dict[str, torch.Tensor]], Dict[str, torch.Tensor]], warmup: int = 3):
        self._fn = step_fn
        self._warmup = warmup
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        self._key = None
        self._static_in: Optional[Dict[str, torch.Tensor]] = None
        self._static_out: Optional[Dict[str, torch.Tensor]] = None

    @staticmethod
    def _shape_key(inputs: Dict[str, torch.Tensor]):
        return tuple(
            (k, tuple(v.shape), v.dtype) for k, v in sorted(inputs.items()) if isinstance(v, torch.Tensor)
        )

    def __call__(self, inputs: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        key = self._shape_key(inputs)
        if self._graph is None or key != self._key:
            self._capture(inputs, key)
        else:
            for k, v in inputs.items():
                if isinstance(v, torch.Tensor):
                    self._static_in[k].copy_(v, non_blocking=True)
        self._graph.replay()
        return self._static_out

    def _capture(self, inputs: Dict[str, torch.Tensor], key) -> None:
        self._key = key
        self._static_in = {
            k: (v.clone() if isinstance(v, torch.Tensor) else v) for k, v in inputs.items()
        }
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(self._warmup):
                self._fn(self._static_in)
        torch.cuda.current_stream().wait_stream(side)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._static_out = self._fn(self._static_in)
