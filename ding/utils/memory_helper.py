"""Layer-wise parameter/activation memory profiling.

Parity: reference ding/utils/memory_helper.py:249 (SimpleMemoryProfiler with
activation hooking). HIP note: device memory queried through
torch.cuda.memory_allocated (HIP allocator on ROCm).
"""
import os
from typing import Any, Dict, Optional

import torch
import torch.nn as nn


class SimpleMemoryProfiler:

    def __init__(self, model: nn.Module, optimizer=None, log_folder: str = "./memory_profile", total_steps: int = 1):
        self._model = model
        self._optimizer = optimizer
        self._log_folder = log_folder
        self._total_steps = total_steps
        self._step = 0
        self._activation_sizes: Dict[str, int] = {}
        self._hooks = []
        os.makedirs(log_folder, exist_ok=True)
        self._record_param_memory()
        self._register_hooks()

    def _record_param_memory(self):
        lines = []
        total = 0
        for name, p in self._model.named_parameters():
            bytes_ = p.numel() * p.element_size()
            total += bytes_
            lines.append(f"{name}: {bytes_ / 1024:.1f} KiB {tuple(p.shape)}")
        lines.append(f"TOTAL PARAMS: {total / 1024 / 1024:.2f} MiB")
        with open(os.path.join(self._log_folder, "params.txt"), "w") as f:
            f.write("\n".join(lines))

    def _register_hooks(self):
        def make_hook(name):
            def hook(module, inputs, output):
                if isinstance(output, torch.Tensor):
                    self._activation_sizes[name] = output.numel() * output.element_size()
            return hook

        for name, module in self._model.named_modules():
            if len(list(module.children())) == 0:
                self._hooks.append(module.register_forward_hook(make_hook(name)))

    def step(self):
        self._step += 1
        if self._step >= self._total_steps:
            self.dump()
            self.remove()

    def dump(self):
        lines = [f"{k}: {v / 1024:.1f} KiB" for k, v in self._activation_sizes.items()]
        lines.append(f"TOTAL ACTIVATIONS: {sum(self._activation_sizes.values()) / 1024 / 1024:.2f} MiB")
        if torch.cuda.is_available():
            lines.append(f"hip allocated: {torch.cuda.memory_allocated() / 1024 / 1024:.2f} MiB")
            lines.append(f"hip reserved: {torch.cuda.memory_reserved() / 1024 / 1024:.2f} MiB")
        with open(os.path.join(self._log_folder, "activations.txt"), "w") as f:
            f.write("\n".join(lines))

    def remove(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
