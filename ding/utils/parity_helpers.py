"""Small reference-surface helpers kept in one module: iteration/dict
utilities, singleton metaclass, cloudpickle wrapper, DDP config scaling and
optional-dependency import stubs.

Parity: reference ding/utils/default_helper.py (iter_mapping, override,
default_get, list_split, make_key_as_identifier, remove_illegal_item),
design_helper.py (SingletonMetaclass), dict_helper.py
(convert_easy_dict_to_dict), compression_helper.py (CloudPickleWrapper),
pytorch_ddp_dist_helper.py (dist_mode:147, to_ddp_config:274),
import_helper.py (try_import_mc/link/rediscluster), k8s_helper.py port
constants, render_helper.py (render_env, get_env_fps).
"""
import math
import pickle
from functools import wraps
from typing import Any, Callable, Iterable, List, Mapping, Optional, Tuple

from .edict import EasyDict

DEFAULT_K8S_COLLECTOR_PORT = 22270
DEFAULT_K8S_LEARNER_PORT = 22271
DEFAULT_K8S_AGGREGATOR_SLAVE_PORT = 22272
DEFAULT_K8S_COORDINATOR_PORT = 22273


def iter_mapping(iter_: Iterable, mapping: Callable[[Any], Any]) -> List[Any]:
    """Apply ``mapping`` over an iterable, returning a list."""
    return [mapping(e) for e in iter_]


def override(cls: type) -> Callable:
    """Decorator asserting the decorated function overrides a method of ``cls``."""

    def decorator(method: Callable) -> Callable:
        assert method.__name__ in dir(cls), f"{method.__name__} does not override any method of {cls.__name__}"
        return method

    return decorator


def default_get(data: Mapping, name: str, default_value: Any = None, default_fn: Optional[Callable] = None,
                judge_fn: Optional[Callable] = None) -> Any:
    """dict.get with an optional lazy default and a validity predicate on hits."""
    if name in data:
        value = data[name]
        if judge_fn is None or judge_fn(value):
            return value
        raise RuntimeError(f"key '{name}' present but value fails judge_fn: {value}")
    return default_fn() if default_fn is not None else default_value


def list_split(data: list, step: int) -> Tuple[List[list], list]:
    """Split into step-sized chunks; the ragged tail is returned separately."""
    if len(data) < step:
        return [], data
    n = len(data) // step
    chunks = [data[i * step:(i + 1) * step] for i in range(n)]
    return chunks, data[n * step:]


def make_key_as_identifier(data: dict) -> dict:
    """Rewrite keys into valid python identifiers (for namedtuple/logging)."""

    def fix(k: str) -> str:
        out = ''.join(c if (c.isalnum() or c == '_') else '_' for c in str(k))
        return '_' + out if (out and out[0].isdigit()) else out

    return {fix(k): v for k, v in data.items()}


def remove_illegal_item(data: dict) -> dict:
    """Drop values that cannot be serialized to plain logs (callables, modules)."""
    import types
    bad = (types.FunctionType, types.ModuleType, types.MethodType)
    return {k: v for k, v in data.items() if not isinstance(v, bad)}


class SingletonMetaclass(type):
    """Classes with this metaclass construct at most one instance; it is also
    exposed as ``cls.instance``."""

    instances: dict = {}

    def __call__(cls, *args, **kwargs):
        if cls not in SingletonMetaclass.instances:
            SingletonMetaclass.instances[cls] = super().__call__(*args, **kwargs)
            cls.instance = SingletonMetaclass.instances[cls]
        return SingletonMetaclass.instances[cls]


def convert_easy_dict_to_dict(data) -> dict:
    """Recursively convert EasyDict (and nested) into plain dicts."""
    if isinstance(data, dict):
        return {k: convert_easy_dict_to_dict(v) for k, v in data.items()}
    if isinstance(data, (list, tuple)):
        return type(data)(convert_easy_dict_to_dict(v) for v in data)
    return data


class CloudPickleWrapper:
    """Wraps an object so multiprocessing can move lambdas/closures: pickles
    via cloudpickle when available, stdlib pickle otherwise."""

    def __init__(self, data: Any) -> None:
        self.data = data

    def __getstate__(self):
        try:
            import cloudpickle
            return cloudpickle.dumps(self.data)
        except ImportError:
            return pickle.dumps(self.data)

    def __setstate__(self, state):
        self.data = pickle.loads(state)


def dist_mode(func: Callable) -> Callable:
    """Run the wrapped function inside a dist_init/dist_finalize bracket."""
    from .dist_helper import dist_init, dist_finalize

    @wraps(func)
    def wrapper(*args, **kwargs):
        dist_init()
        try:
            return func(*args, **kwargs)
        finally:
            dist_finalize()

    return wrapper


def to_ddp_config(cfg: EasyDict) -> EasyDict:
    """Divide per-job batch/collect sizes by world size so the global batch
    stays what the single-process config named."""
    from .dist_helper import get_world_size
    w = max(1, get_world_size())
    pol = cfg.policy
    if 'batch_size' in pol:
        pol.batch_size = int(math.ceil(pol.batch_size / w))
    if 'learn' in pol and 'batch_size' in pol.learn:
        pol.learn.batch_size = int(math.ceil(pol.learn.batch_size / w))
    if 'collect' in pol and 'n_sample' in pol.collect:
        pol.collect.n_sample = int(math.ceil(pol.collect.n_sample / w))
    if 'collect' in pol and 'n_episode' in pol.collect:
        pol.collect.n_episode = int(math.ceil(pol.collect.n_episode / w))
    return cfg


def _unavailable(name: str):

    def _fail(*args, **kwargs):
        raise ImportError(f"optional dependency '{name}' is not installed in this image")

    return _fail


def try_import_mc():
    """memcache is not shipped in the MI355X image."""
    return None


def try_import_link():
    """linklink (SenseTime internal) never applies here: torch.distributed
    over RCCL is the one distributed backend."""
    return None


def try_import_rediscluster():
    try:
        import rediscluster
        return rediscluster
    except ImportError:
        return None


def render_env(env, render_mode: Optional[str] = 'rgb_array'):
    """Render a frame from a ding env (HWC ndarray)."""
    from .misc_helpers import render
    return render(env, render_mode)


def get_env_fps(env) -> int:
    """Best-effort FPS from env metadata; 30 when unspecified."""
    meta = getattr(env, 'metadata', None) or {}
    for key in ('render_fps', 'video.frames_per_second'):
        if key in meta:
            return int(meta[key])
    return 30


def find_free_port_slurm(node: str) -> int:
    """Free port on a slurm node, derived from the job id when present so
    concurrent array tasks do not collide; falls back to an OS-chosen port."""
    import os
    job = os.environ.get('SLURM_JOB_ID')
    if job is not None:
        return 15000 + int(job) % 20000
    from .misc_helpers import find_free_port
    return find_free_port()
