"""Builtin engram registry.

EngramTemplate.builtin / .image names resolve here to in-process
implementations (role replacement for the reference's container images —
SURVEY.md §2.7).
"""
from __future__ import annotations

import typing as _t

from .base import Engram

_REGISTRY: _t.Dict[str, _t.Callable[[], Engram]] = {}
_INSTANCES: _t.Dict[str, Engram] = {}


class UnknownEngram(KeyError):
    pass


def register(name: str, factory: _t.Callable[[], Engram]) -> None:
    _REGISTRY[name] = factory


def register_class(cls: _t.Type[Engram]) -> _t.Type[Engram]:
    """Class decorator: @register_class on an Engram subclass with .name."""
    if not cls.name:
        raise ValueError(f"{cls.__name__} has no registry name")
    register(cls.name, cls)
    return cls


def resolve(name: str) -> Engram:
    """Get (and cache) the implementation instance for a registry name."""
    inst = _INSTANCES.get(name)
    if inst is not None:
        return inst
    factory = _REGISTRY.get(name)
    if factory is None:
        _load_builtins()
        factory = _REGISTRY.get(name)
    if factory is None:
        raise UnknownEngram(
            f"no builtin engram implementation named {name!r} "
            f"(known: {sorted(_REGISTRY)})"
        )
    inst = factory()
    _INSTANCES[name] = inst
    return inst


def known() -> _t.List[str]:
    _load_builtins()
    return sorted(_REGISTRY)


def reset_instances() -> None:
    _INSTANCES.clear()


_loaded = False
_load_lock = __import__("threading").Lock()


def _load_builtins() -> None:
    """Thread-safe: worker slots resolve engrams concurrently, so the flag
    must only flip after every builtin module finished importing."""
    global _loaded
    if _loaded:
        return
    with _load_lock:
        if _loaded:
            return
        from . import filter_transform  # noqa: F401
        from . import llm_infer  # noqa: F401
        from . import embed  # noqa: F401
        from . import join  # noqa: F401
        from . import materialize  # noqa: F401

        _loaded = True
