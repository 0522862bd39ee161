"""Small aux helpers: maze value-iteration BFS, render, k8s/slurm launch
stubs with real manifest/script generation.

Parity: reference ding/utils/bfs_helper.py, render_helper.py,
k8s_helper.py:118, slurm_helper.py, orchestrator_launcher.py:7.
"""
import os
import threading
from typing import Any, List, Optional, Tuple

import numpy as np
import torch


# ----------------------------------------------------------------- bfs maze
def get_vi_sequence(env, observation: np.ndarray) -> Tuple[np.ndarray, List]:
    """Value-iteration over a maze env exposing ``maze`` walls grid; returns
    (value sequence [N, H, W], bfs order). Used by procedure cloning."""
    maze = env.maze if hasattr(env, 'maze') else observation
    H, W = maze.shape[:2]
    target = getattr(env, 'target_location', (H - 1, W - 1))
    values = np.full((H, W), -np.inf, dtype=np.float32)
    values[target[0], target[1]] = 0.0
    seq = [values.copy()]
    frontier = [tuple(target)]
    order = [tuple(target)]
    while frontier:
        nxt = []
        for (y, x) in frontier:
            for dy, dx in ((1, 0), (-1, 0), (0, 1), (0, -1)):
                ny, nx_ = y + dy, x + dx
                if 0 <= ny < H and 0 <= nx_ < W and values[ny, nx_] == -np.inf:
                    if hasattr(env, 'maze') and env.maze[ny, nx_] != 0:
                        continue  # wall
                    values[ny, nx_] = values[y, x] - 1
                    nxt.append((ny, nx_))
                    order.append((ny, nx_))
        if nxt:
            seq.append(values.copy())
        frontier = nxt
    return np.stack(seq), order


# ------------------------------------------------------------------- render
def render(env, render_mode: str = 'rgb_array') -> Optional[np.ndarray]:
    """Best-effort frame grab for replay saving."""
    if hasattr(env, 'render'):
        try:
            return env.render(mode=render_mode)
        except TypeError:
            return env.render()
    return None


def fps(env_manager) -> int:
    return getattr(env_manager, '_fps', 30)


# ----------------------------------------------------------- cluster launch
DEFAULT_K8S_YAML = """apiVersion: batch/v1
kind: Job
metadata:
  name: {name}
spec:
  parallelism: {workers}
  template:
    spec:
      containers:
      - name: ding-worker
        image: {image}
        command: ["ditask", "--main", "{main}", "--parallel-workers", "1",
                  "--topology", "mesh", "--attach-to", "{attach_to}"]
        resources:
          limits:
            amd.com/gpu: {gpus}
      restartPolicy: OnFailure
"""


class K8sLauncher:
    """Generate a Job manifest for a ditask fleet (kubectl not available
    offline; manifest generation is the testable surface)."""

    def __init__(self, cfg: Optional[dict] = None):
        self.cfg = cfg or {}

    def create_manifest(self, name: str, main: str, workers: int = 1, image: str = 'ding:latest', gpus: int = 1,
                        attach_to: str = '', output_path: Optional[str] = None) -> str:
        manifest = DEFAULT_K8S_YAML.format(
            name=name, workers=workers, image=image, main=main, gpus=gpus, attach_to=attach_to
        )
        if output_path:
            with open(output_path, 'w') as f:
                f.write(manifest)
        return manifest

    def launch(self, *args, **kwargs):
        raise RuntimeError("no kubectl in the offline image; apply the generated manifest on a real cluster")


SLURM_TEMPLATE = """#!/bin/bash
#SBATCH --job-name={name}
#SBATCH --nodes={nodes}
#SBATCH --ntasks-per-node={tasks_per_node}
#SBATCH --gpus-per-task={gpus}
export MASTER_ADDR=$(scontrol show hostnames $SLURM_JOB_NODELIST | head -n1)
export MASTER_PORT={port}
srun {command}
"""


def generate_slurm_script(name: str, command: str, nodes: int = 1, tasks_per_node: int = 8, gpus: int = 1,
                          port: int = 29500, output_path: Optional[str] = None) -> str:
    script = SLURM_TEMPLATE.format(
        name=name, nodes=nodes, tasks_per_node=tasks_per_node, gpus=gpus, port=port, command=command
    )
    if output_path:
        with open(output_path, 'w') as f:
            f.write(script)
    return script


def find_free_port(host: str = '127.0.0.1') -> int:
    import socket
    with socket.socket() as s:
        s.bind((host, 0))
        return s.getsockname()[1]


def node_to_partition(node: str) -> str:
    return node.split('-')[0] if '-' in node else node


def node_to_host(node: str) -> str:
    return node


def get_ip() -> str:
    """Best-effort non-loopback IP (reference utils/system_helper.py:10)."""
    import socket
    try:
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.connect(('10.255.255.255', 1))
        ip = s.getsockname()[0]
        s.close()
        return ip
    except Exception:
        return '127.0.0.1'


def get_pid() -> int:
    import os
    return os.getpid()


class PropagatingThread(threading.Thread):
    """Thread that re-raises its exception in join() (reference
    utils/system_helper.py:40)."""

    def run(self):
        self._exc = None
        try:
            self._ret = self._target(*self._args, **self._kwargs)
        except Exception as e:
            self._exc = e

    def join(self, timeout=None):
        super().join(timeout)
        if getattr(self, '_exc', None) is not None:
            raise self._exc
        return getattr(self, '_ret', None)


def deprecated(since: str, removed_in: str, up_to: str = None):
    """Mark an API deprecated; warns on call (reference utils/deprecation.py)."""
    import functools
    import warnings

    def deco(fn):

        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            msg = f'{fn.__name__} is deprecated since {since}, will be removed in {removed_in}'
            if up_to:
                msg += f'; use {up_to} instead'
            warnings.warn(msg, DeprecationWarning, stacklevel=2)
            return fn(*args, **kwargs)

        return wrapper

    return deco
