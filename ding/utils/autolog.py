"""Time-windowed metric logging (autolog).

Parity: reference ding/utils/autolog/ (LoggedModel:34, LoggedValue,
TimeMode) — rolling windows of values with range_values/mean/max queries
used by TickMonitor in BaseLearner and buffer throughput monitors.
"""
import time
from enum import Enum
from typing import Any, Callable, Dict, List, Optional, Tuple


class TimeMode(Enum):
    ABSOLUTE = 0
    RELATIVE_LIFECYCLE = 1
    RELATIVE_CURRENT_TIME = 2


class NaturalTime:

    def time(self) -> float:
        return time.time()


class TickTime:

    def __init__(self, init: int = 0):
        self._tick = init

    def step(self, delta: int = 1) -> int:
        self._tick += delta
        return self._tick

    def time(self) -> int:
        return self._tick


class TimeProxy:

    def __init__(self, time_obj):
        self._time = time_obj
        self._frozen = None

    def freeze(self):
        self._frozen = self._time.time()

    def unfreeze(self):
        self._frozen = None

    def time(self):
        return self._frozen if self._frozen is not None else self._time.time()


class LoggedValue:
    """Descriptor recording (timestamp, value) history on set."""

    def __init__(self, type_: type = object):
        self._type = type_

    def __set_name__(self, owner, name):
        self._name = name

    def __get__(self, instance, owner):
        if instance is None:
            return self
        return instance._logged_data.get(self._name, (None, None))[-1][1] \
            if instance._logged_data.get(self._name) else None

    def __set__(self, instance, value):
        if not isinstance(value, self._type):
            raise TypeError(f"{self._name} expects {self._type}, got {type(value)}")
        now = instance._time.time()
        instance._logged_data.setdefault(self._name, []).append((now, value))
        instance._prune(self._name)


class LoggedModel:
    """Base for monitors: keeps per-property (t, v) history within ``expire``
    of current time; register_attribute_value adds reduction functions."""

    def __init__(self, time_obj, expire: float):
        self._time = time_obj
        self.expire = expire
        self._logged_data: Dict[str, List[Tuple[float, Any]]] = {}
        self._methods: Dict[str, Dict[str, Callable]] = {}

    def _prune(self, name: str):
        now = self._time.time()
        hist = self._logged_data.get(name, [])
        self._logged_data[name] = [(t, v) for t, v in hist if now - t <= self.expire]

    def range_values(self, name: str) -> List[Tuple[float, Any]]:
        self._prune(name)
        return list(self._logged_data.get(name, []))

    def register_attribute_value(self, attribute_name: str, property_name: str, value: Callable):
        self._methods.setdefault(property_name, {})[attribute_name] = value

    def get_property_attribute(self, property_name: str) -> List[str]:
        return list(self._methods.get(property_name, {}).keys())

    def __getattr__(self, key: str):
        if key.startswith('_'):
            raise AttributeError(key)
        methods = object.__getattribute__(self, '_methods') if '_methods' in self.__dict__ else {}
        if key in methods:
            class _Proxy:
                def __init__(self, fns):
                    self._fns = fns

                def __getattr__(self, attr):
                    return self._fns[attr]

            return _Proxy(methods[key])
        raise AttributeError(key)

    # convenience reductions over a property's history
    def _values(self, name: str) -> List[float]:
        return [v for _, v in self.range_values(name)]

    def avg(self, name: str) -> float:
        vals = self._values(name)
        return sum(vals) / len(vals) if vals else 0.0

    def max(self, name: str) -> float:
        vals = self._values(name)
        return max(vals) if vals else 0.0

    def sum(self, name: str) -> float:
        return sum(self._values(name))


class TickMonitor(LoggedModel):
    """Learner monitor: train-time/data-time/forward-time windows.

    Parity: reference worker/learner/base_learner.py:510.
    """
    train_time = LoggedValue(float)
    data_time = LoggedValue(float)
    forward_time = LoggedValue(float)
    backward_time = LoggedValue(float)

    def __init__(self, time_obj=None, expire: float = 10):
        super().__init__(time_obj or TickTime(), expire)
