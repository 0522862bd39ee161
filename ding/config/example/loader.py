"""Lookup for the per-(algorithm, env) preset configs.

``get_example_config('DQN', 'LunarLander-v2')`` imports
``ding.config.example.DQN.gym_lunarlander_v2`` and returns its ``cfg``
EasyDict (deep-copied). The filename convention matches the reference:
lowercase env id with ``-``/``.`` collapsed, prefixed ``gym_``.
"""
import copy
import importlib
import os
import pkgutil
from typing import List, Optional

from ding.utils import EasyDict

_ROOT = os.path.dirname(os.path.abspath(__file__))


def _module_name(env_id: str) -> str:
    return 'gym_' + env_id.replace('-', '_').replace('.', '_').lower()


def get_example_config(algo: str, env_id: str) -> Optional[EasyDict]:
    """Return the tuned preset for (algo, env) or None if not shipped."""
    mod_name = f'ding.config.example.{algo}.{_module_name(env_id)}'
    try:
        mod = importlib.import_module(mod_name)
    except ImportError:
        return None
    return copy.deepcopy(mod.cfg)


def list_example_configs(algo: str = None) -> List[str]:
    """Enumerate shipped presets as 'ALGO/module' strings."""
    out = []
    for algo_dir in sorted(os.listdir(_ROOT)):
        full = os.path.join(_ROOT, algo_dir)
        if not os.path.isdir(full) or algo_dir.startswith('_'):
            continue
        if algo is not None and algo_dir != algo:
            continue
        for m in pkgutil.iter_modules([full]):
            out.append(f'{algo_dir}/{m.name}')
    return out
