"""Small utilities (Java API parity: Arms.java AutoCloseable helpers,
Pair.java, Preconditions.java, FieldUtils.java)."""
from contextlib import contextmanager
from typing import Any, Callable, Iterable, NamedTuple, Optional, TypeVar

T = TypeVar("T")


class Pair(NamedTuple):
    left: Any
    right: Any


def check_argument(cond: bool, msg: str = "invalid argument"):
    if not cond:
        raise ValueError(msg)


def check_state(cond: bool, msg: str = "invalid state"):
    if not cond:
        raise RuntimeError(msg)


def check_non_negative(v: int, name: str = "value") -> int:
    if v < 0:
        raise ValueError(f"{name} must be non-negative, got {v}")
    return v


def close_quietly(*objs):
    """Arms.closeQuietly: close all, swallow errors."""
    for o in objs:
        try:
            if o is not None and hasattr(o, "close"):
                o.close()
        except Exception:
            pass


@contextmanager
def closing_all(*objs):
    """Arms.withResource over several closeables."""
    try:
        yield objs if len(objs) > 1 else objs[0]
    finally:
        close_quietly(*objs)


def close_all_except(objs: Iterable[Any], keep: Any):
    for o in objs:
        if o is not keep:
            close_quietly(o)
