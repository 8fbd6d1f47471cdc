"""Algorithm / evaluation registries.

Parity with the reference's decorator registry (sheeprl/utils/registry.py:97,
:104): every algorithm module registers its ``main(runtime, cfg)`` entrypoint
under a name, with a ``decoupled`` flag; evaluation entrypoints register
separately.  The CLI resolves names through these tables.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, Dict, Optional


@dataclass
class AlgoEntry:
    name: str
    module: str
    entrypoint: Callable
    decoupled: bool = False


algorithm_registry: Dict[str, AlgoEntry] = {}
evaluation_registry: Dict[str, Callable] = {}


def register_algorithm(name: Optional[str] = None, decoupled: bool = False) -> Callable:
    def deco(fn: Callable) -> Callable:
        key = name or fn.__module__.rsplit(".", 2)[-2]
        algorithm_registry[key] = AlgoEntry(
            name=key, module=fn.__module__, entrypoint=fn, decoupled=decoupled
        )
        return fn

    return deco


def register_evaluation(algorithms: Optional[list] = None) -> Callable:
    def deco(fn: Callable) -> Callable:
        keys = algorithms or [fn.__module__.rsplit(".", 2)[-2]]
        for key in keys:
            evaluation_registry[key] = fn
        return fn

    return deco
