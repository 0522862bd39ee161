"""Config-validation DSL: typed, composable loaders.

Parity: reference ding/utils/loader/ (Loader combinators: is_type, to_type,
interval, enum, item, dict_, collection, check_only, optional, or_/and_).
"""
from typing import Any, Callable, Iterable, Optional


class LoaderError(ValueError):
    pass


class Loader:
    """A validating transform: call it with a value, get the (possibly
    converted) value back or raise LoaderError. Compose with | (or), &
    (and), >> (pipe)."""

    def __init__(self, fn: Callable[[Any], Any], name: str = "loader"):
        self._fn = fn
        self.name = name

    def __call__(self, value: Any) -> Any:
        return self._fn(value)

    def load(self, value: Any) -> Any:
        return self(value)

    def check(self, value: Any) -> bool:
        try:
            self(value)
            return True
        except Exception:
            return False

    def __or__(self, other: 'Loader') -> 'Loader':
        other = to_loader(other)

        def _or(value):
            try:
                return self(value)
            except Exception:
                return other(value)

        return Loader(_or, f"({self.name}|{other.name})")

    def __and__(self, other: 'Loader') -> 'Loader':
        other = to_loader(other)

        def _and(value):
            self(value)
            return other(value)

        return Loader(_and, f"({self.name}&{other.name})")

    def __rshift__(self, other: 'Loader') -> 'Loader':
        other = to_loader(other)
        return Loader(lambda v: other(self(v)), f"({self.name}>>{other.name})")


def to_loader(x) -> Loader:
    if isinstance(x, Loader):
        return x
    if isinstance(x, type):
        return is_type(x)
    if callable(x):
        return Loader(x, getattr(x, '__name__', 'fn'))
    # plain value: equality check
    return Loader(lambda v: v if v == x else _raise(f"expected {x!r}, got {v!r}"), f"eq({x!r})")


def _raise(msg: str):
    raise LoaderError(msg)


def is_type(t: type) -> Loader:
    return Loader(
        lambda v: v if isinstance(v, t) else _raise(f"expected {t.__name__}, got {type(v).__name__}"),
        f"is_type({t.__name__})"
    )


def to_type(t: type) -> Loader:
    return Loader(lambda v: t(v), f"to_type({t.__name__})")


def interval(lo=None, hi=None, left_ok: bool = True, right_ok: bool = True) -> Loader:

    def _check(v):
        if lo is not None and (v < lo or (not left_ok and v == lo)):
            _raise(f"{v} below interval min {lo}")
        if hi is not None and (v > hi or (not right_ok and v == hi)):
            _raise(f"{v} above interval max {hi}")
        return v

    return Loader(_check, f"interval({lo},{hi})")


def enum(*values, case_sensitive: bool = True) -> Loader:

    def _check(v):
        if case_sensitive:
            ok = v in values
        else:
            ok = isinstance(v, str) and v.lower() in [str(x).lower() for x in values]
        if not ok:
            _raise(f"{v!r} not in {values}")
        return v

    return Loader(_check, f"enum{values}")


def item(key: str, loader=None) -> Loader:
    loader = to_loader(loader) if loader is not None else None

    def _check(d):
        if key not in d:
            _raise(f"missing key {key!r}")
        v = d[key]
        return loader(v) if loader else v

    return Loader(_check, f"item({key})")


def dict_(**key_loaders) -> Loader:
    loaders = {k: to_loader(l) for k, l in key_loaders.items()}

    def _check(d):
        return {k: l(d[k]) if k in d else _raise(f"missing key {k!r}") for k, l in loaders.items()}

    return Loader(_check, "dict")


def collection(elem_loader) -> Loader:
    elem_loader = to_loader(elem_loader)

    def _check(xs):
        if not isinstance(xs, (list, tuple)):
            _raise(f"expected collection, got {type(xs).__name__}")
        return type(xs)(elem_loader(x) for x in xs)

    return Loader(_check, "collection")


def optional(loader) -> Loader:
    loader = to_loader(loader)
    return Loader(lambda v: None if v is None else loader(v), f"optional({loader.name})")


def check_only(loader) -> Loader:
    loader = to_loader(loader)

    def _check(v):
        loader(v)
        return v

    return Loader(_check, f"check_only({loader.name})")
