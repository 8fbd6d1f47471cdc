"""Minimal YAML config composition for sheeprl-amd.

Role parity with the reference's Hydra layer (sheeprl/configs + hydra_plugins,
SURVEY.md L9) without the Hydra dependency (not available in this image):

* config groups under ``sheeprl_amd/configs/<group>/<name>.yaml``
* a root ``config.yaml`` with a ``defaults`` list selecting one file per group
* experiment files (``exp/*.yaml``) merged at global level, which may override
  group selections through their own ``defaults`` list
* dotted command-line overrides ``a.b.c=value`` (applied last)
* ``${a.b.c}`` interpolation and ``${now:%fmt}`` timestamps
* external search paths via the ``SHEEPRL_AMD_SEARCH_PATH`` environment
  variable (colon-separated dirs), mirroring the reference's
  ``SHEEPRL_SEARCH_PATH`` hydra plugin (hydra_plugins/sheeprl_search_path.py:10).

Group files may declare their own ``defaults`` list whose entries are either a
plain string (a base file in the same group, e.g. ``dreamer_v3_S`` inheriting
``dreamer_v3_XL``) or a mapping ``{"override /<group>": name}`` /
``{"<group>": name}`` selecting another group's file (only meaningful in the
root and in exp files).
"""

from __future__ import annotations

import datetime
import importlib
import os
import re
from pathlib import Path
from typing import Any, Dict, List, Optional, Sequence

import yaml

from sheeprl_amd.utils.dotdict import DotDict

_PKG_CONFIG_DIR = Path(__file__).resolve().parent.parent / "configs"
_INTERP_RE = re.compile(r"\$\{([^${}]+)\}")


def _search_dirs() -> List[Path]:
    dirs: List[Path] = []
    env = os.environ.get("SHEEPRL_AMD_SEARCH_PATH", "")
    for part in env.split(":"):
        if part.strip():
            dirs.append(Path(part.strip()))
    dirs.append(_PKG_CONFIG_DIR)
    return dirs


def _find(relpath: str) -> Path:
    for base in _search_dirs():
        p = base / relpath
        if p.is_file():
            return p
    raise FileNotFoundError(
        f"config file '{relpath}' not found in search path "
        f"{[str(d) for d in _search_dirs()]} (set SHEEPRL_AMD_SEARCH_PATH to add dirs)"
    )


def load_yaml(path: str | Path) -> dict:
    with open(path) as f:
        data = yaml.safe_load(f)
    return data or {}


def merge(dst: dict, src: dict) -> dict:
    """Recursive dict merge; ``src`` wins. Lists are replaced, not merged."""
    for k, v in src.items():
        if isinstance(v, dict) and isinstance(dst.get(k), dict):
            merge(dst[k], v)
        else:
            dst[k] = v
    return dst


def _load_group_file(group: str, name: str, _seen: Optional[set] = None) -> dict:
    """Load ``<group>/<name>.yaml`` following same-group inheritance chains."""
    _seen = _seen or set()
    key = f"{group}/{name}"
    if key in _seen:
        raise ValueError(f"circular defaults chain at {key}")
    _seen.add(key)
    raw = load_yaml(_find(f"{group}/{name}.yaml"))
    defaults = raw.pop("defaults", [])
    out: dict = {}
    for entry in defaults:
        if isinstance(entry, str):
            if entry == "_self_":
                continue
            merge(out, _load_group_file(group, entry, _seen))
        elif isinstance(entry, dict):
            # cross-group selections inside a group file are ignored here;
            # they are handled by the exp/root pass.
            continue
    merge(out, raw)
    return out


def _parse_defaults(defaults: Sequence, group_sel: Dict[str, str], self_base: Optional[str], group: str) -> Optional[str]:
    """Update ``group_sel`` from a defaults list; return same-group base name."""
    base = self_base
    for entry in defaults:
        if isinstance(entry, str):
            if entry == "_self_":
                continue
            base = entry
        elif isinstance(entry, dict):
            for k, v in entry.items():
                k = k.strip()
                if k.startswith("override "):
                    k = k[len("override "):]
                k = k.lstrip("/")
                group_sel[k] = v
    return base


def _value_from_str(s: str) -> Any:
    try:
        return yaml.safe_load(s)
    except Exception:
        return s


def compose(overrides: Sequence[str] = (), root: str = "config.yaml") -> DotDict:
    """Compose the full config from the root file, group files and overrides."""
    root_raw = load_yaml(_find(root))
    root_defaults = root_raw.pop("defaults", [])

    group_sel: Dict[str, str] = {}
    _parse_defaults(root_defaults, group_sel, None, "")

    # Split CLI overrides into group selections vs dotted value overrides.
    dotted: List[tuple] = []
    for ov in overrides:
        if "=" not in ov:
            raise ValueError(f"override '{ov}' is not of the form key=value")
        k, v = ov.split("=", 1)
        k = k.strip()
        if k in group_sel or (("." not in k) and any((base / k).is_dir() for base in _search_dirs())):
            group_sel[k] = v.strip()
        else:
            dotted.append((k, _value_from_str(v)))

    # Experiment selection can override other groups.  Walk the exp
    # inheritance chain child -> parent, then apply group selections
    # PARENT-FIRST so a child's `override /algo: ...` beats its base exp's.
    exp_chain: List[dict] = []
    exp_defaults_chain: List[list] = []
    exp_name = group_sel.get("exp")
    if exp_name not in (None, "???"):
        seen: set = set()
        name: Optional[str] = exp_name
        while name is not None:
            if name in seen:
                raise ValueError(f"circular exp defaults at {name}")
            seen.add(name)
            raw = load_yaml(_find(f"exp/{name}.yaml"))
            defaults = raw.pop("defaults", [])
            exp_defaults_chain.append(defaults)
            exp_chain.append(raw)
            # find the same-group base without touching group_sel yet
            base = None
            for entry in defaults:
                if isinstance(entry, str) and entry != "_self_":
                    base = entry
            name = base
        exp_chain.reverse()
        exp_defaults_chain.reverse()
        for defaults in exp_defaults_chain:  # base first, child last (wins)
            _parse_defaults(defaults, group_sel, None, "exp")

    # Re-apply CLI group selections (they beat the exp's defaults).
    for ov in overrides:
        k, _, v = ov.partition("=")
        k = k.strip()
        if k != "exp" and k in group_sel and "." not in k and "=" in ov:
            if any((base / k).is_dir() for base in _search_dirs()):
                group_sel[k] = v.strip()

    cfg: dict = {}
    for group, name in group_sel.items():
        if group == "exp" or name in (None, "???"):
            continue
        cfg[group] = _load_group_file(group, name)
    merge(cfg, root_raw)  # root-level plain keys
    for body in exp_chain:  # exp bodies merge at global level
        merge(cfg, body)
    for k, v in dotted:
        _set_dotted(cfg, k, v)

    resolve(cfg)
    _check_required(cfg)
    return DotDict(cfg)


def _set_dotted(cfg: dict, dotted: str, value: Any) -> None:
    parts = dotted.split(".")
    node = cfg
    for p in parts[:-1]:
        if p not in node or not isinstance(node[p], dict):
            node[p] = {}
        node = node[p]
    node[parts[-1]] = value


# --------------------------------------------------------------------------
# interpolation
# --------------------------------------------------------------------------

def resolve(cfg: dict) -> dict:
    """Resolve ``${a.b.c}`` and ``${now:%fmt}`` in place (fixpoint iteration)."""
    now = datetime.datetime.now()

    def lookup(path: str) -> Any:
        if path.startswith("now:"):
            return now.strftime(path[len("now:"):])
        node: Any = cfg
        for part in path.split("."):
            if not isinstance(node, dict) or part not in node:
                raise KeyError(path)
            node = node[part]
        return node

    def resolve_value(v: Any) -> Any:
        if isinstance(v, str):
            m = _INTERP_RE.fullmatch(v.strip())
            if m:
                try:
                    return lookup(m.group(1))
                except KeyError:
                    return v

            def sub(mm: re.Match) -> str:
                try:
                    return str(lookup(mm.group(1)))
                except KeyError:
                    return mm.group(0)

            return _INTERP_RE.sub(sub, v)
        return v

    def walk(node: Any) -> tuple:
        changed = False
        if isinstance(node, dict):
            for k, v in node.items():
                if isinstance(v, (dict, list)):
                    c, _ = walk(v)
                    changed |= c
                else:
                    nv = resolve_value(v)
                    if nv is not v and nv != v:
                        node[k] = nv
                        changed = True
        elif isinstance(node, list):
            for i, v in enumerate(node):
                if isinstance(v, (dict, list)):
                    c, _ = walk(v)
                    changed |= c
                else:
                    nv = resolve_value(v)
                    if nv is not v and nv != v:
                        node[i] = nv
                        changed = True
        return changed, node

    for _ in range(10):
        changed, _ = walk(cfg)
        if not changed:
            break
    return cfg


def _check_required(cfg: dict, prefix: str = "") -> None:
    missing: List[str] = []

    def walk(node: Any, path: str) -> None:
        if isinstance(node, dict):
            for k, v in node.items():
                walk(v, f"{path}.{k}" if path else str(k))
        elif node == "???":
            missing.append(path)

    walk(cfg, prefix)
    if missing:
        raise ValueError(f"missing required config values: {missing}")


def save_config(cfg: DotDict | dict, path: str | Path) -> None:
    plain = cfg.to_plain() if isinstance(cfg, DotDict) else cfg
    Path(path).parent.mkdir(parents=True, exist_ok=True)
    with open(path, "w") as f:
        yaml.safe_dump(plain, f, sort_keys=False)


def instantiate(spec: Any, *args: Any, **extra: Any) -> Any:
    """Build an object from a ``{target: dotted.path, **kwargs}`` mapping."""
    if spec is None:
        return None
    if isinstance(spec, str):
        module, _, attr = spec.rpartition(".")
        return getattr(importlib.import_module(module), attr)
    spec = dict(spec)
    target = spec.pop("target")
    module, _, attr = target.rpartition(".")
    fn = getattr(importlib.import_module(module), attr)
    kw = {k: v for k, v in spec.items()}
    kw.update(extra)
    return fn(*args, **kw)
