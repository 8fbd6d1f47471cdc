"""Attribute-access dict used for all configuration objects.

The reference framework converts its composed config into a plain dot-accessible
dict early (sheeprl/cli.py:364, sheeprl/utils/utils.py:34) so the algorithm code
never depends on the config library.  We do the same: the YAML composer in
``sheeprl_amd.config`` emits :class:`DotDict` trees.
"""

from __future__ import annotations

import copy
from typing import Any, Iterable, Mapping


class DotDict(dict):
    """A dict whose items are also reachable as attributes, recursively."""

    def __init__(self, *args: Any, **kwargs: Any) -> None:
        super().__init__(*args, **kwargs)
        for k, v in list(self.items()):
            super().__setitem__(k, _wrap(v))

    # -- attribute protocol -------------------------------------------------
    def __getattr__(self, name: str) -> Any:
        try:
            return self[name]
        except KeyError as e:  # pragma: no cover - error path
            raise AttributeError(name) from e

    def __setattr__(self, name: str, value: Any) -> None:
        self[name] = value

    def __delattr__(self, name: str) -> None:
        try:
            del self[name]
        except KeyError as e:  # pragma: no cover - error path
            raise AttributeError(name) from e

    # -- item protocol ------------------------------------------------------
    def __setitem__(self, key: Any, value: Any) -> None:
        super().__setitem__(key, _wrap(value))

    def setdefault(self, key: Any, default: Any = None) -> Any:
        if key not in self:
            self[key] = default
        return self[key]

    def update(self, other: Mapping[str, Any] | Iterable = (), **kwargs: Any) -> None:  # type: ignore[override]
        items = other.items() if isinstance(other, Mapping) else other
        for k, v in items:
            self[k] = v
        for k, v in kwargs.items():
            self[k] = v

    def __deepcopy__(self, memo: dict) -> "DotDict":
        return DotDict({k: copy.deepcopy(v, memo) for k, v in self.items()})

    # -- helpers -------------------------------------------------------------
    def get_nested(self, dotted: str, default: Any = None) -> Any:
        node: Any = self
        for part in dotted.split("."):
            if not isinstance(node, Mapping) or part not in node:
                return default
            node = node[part]
        return node

    def set_nested(self, dotted: str, value: Any) -> None:
        parts = dotted.split(".")
        node: DotDict = self
        for part in parts[:-1]:
            nxt = node.get(part)
            if not isinstance(nxt, DotDict):
                nxt = DotDict()
                node[part] = nxt
            node = nxt
        node[parts[-1]] = value

    def to_plain(self) -> dict:
        """Recursively convert back to builtin dict/list (for YAML dump)."""
        return _unwrap(self)


def _wrap(value: Any) -> Any:
    if isinstance(value, DotDict):
        return value
    if isinstance(value, dict):
        return DotDict(value)
    if isinstance(value, (list, tuple)):
        t = type(value) if isinstance(value, list) else list
        return t(_wrap(v) for v in value)
    return value


def _unwrap(value: Any) -> Any:
    if isinstance(value, dict):
        return {k: _unwrap(v) for k, v in value.items()}
    if isinstance(value, list):
        return [_unwrap(v) for v in value]
    return value
