"""Attribute-access dict (drop-in for the external ``easydict`` package the
reference depends on; that package is not available offline).

Behaviour matched to easydict.EasyDict as used throughout DI-engine configs:
recursive conversion of nested dicts/lists, attribute get/set, and plain-dict
compatibility (isinstance(cfg, dict) holds).
"""
from typing import Any


class EasyDict(dict):

    def __init__(self, d: Any = None, **kwargs):
        super().__init__()
        if d is None:
            d = {}
        if kwargs:
            d = dict(d, **kwargs)
        for k, v in d.items():
            self[k] = v

    @staticmethod
    def _convert(value: Any) -> Any:
        if isinstance(value, EasyDict):
            return value
        if isinstance(value, dict):
            return EasyDict(value)
        if isinstance(value, (list, tuple)):
            conv = [EasyDict._convert(x) for x in value]
            return type(value)(conv) if isinstance(value, tuple) else conv
        return value

    def __setitem__(self, key: str, value: Any) -> None:
        super().__setitem__(key, EasyDict._convert(value))

    def __setattr__(self, key: str, value: Any) -> None:
        self[key] = value

    def __getattr__(self, key: str) -> Any:
        try:
            return self[key]
        except KeyError:
            raise AttributeError(key)

    def __delattr__(self, key: str) -> None:
        try:
            del self[key]
        except KeyError:
            raise AttributeError(key)

    def update(self, other=None, **kwargs):  # keep conversion on update
        other = dict(other or {}, **kwargs)
        for k, v in other.items():
            self[k] = v

    def __deepcopy__(self, memo):
        import copy
        out = EasyDict()
        memo[id(self)] = out
        for k, v in self.items():
            dict.__setitem__(out, copy.deepcopy(k, memo), copy.deepcopy(v, memo))
        return out
