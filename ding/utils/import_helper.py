"""Dynamic module import. Parity: reference ding/utils/import_helper.py."""
import importlib
from typing import List, Optional

from .default_helper import one_time_warning


def try_import_ceph():
    one_time_warning("ceph is not supported offline")
    return None


def try_import_redis():
    one_time_warning("redis client is not installed offline")
    return None


def import_module(modules: Optional[List[str]] = None) -> None:
    if modules is None:
        return
    for name in modules:
        importlib.import_module(name)
