"""CLI / orchestration.

Parity with sheeprl/cli.py: ``run`` (:358) composes the
config, applies resume logic (:23-57), validates it (:271-345), builds the
Runtime (the Fabric replacement) and launches the registered algorithm
entrypoint; ``evaluation`` (:369) restores a checkpoint on a single device and
dispatches the registered evaluator; ``available_agents`` lists the registry.

Usage::

    python -m sheeprl_amd exp=ppo env.num_envs=4 runtime.devices=2
    python -m sheeprl_amd eval checkpoint_path=... [overrides]
    python -m sheeprl_amd agents
"""

from __future__ import annotations

import copy
import os
import pathlib
import sys
import warnings
from typing import List, Optional

import torch

from sheeprl_amd.config import compose
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.callback import CheckpointCallback
from sheeprl_amd.utils.dotdict import DotDict
from sheeprl_amd.utils.registry import algorithm_registry, evaluation_registry
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import print_config, seed_everything


def _import_algorithms() -> None:
    """Populate the registry (parity: sheeprl/__init__.py:18-47)."""
    import sheeprl_amd.algos  # noqa: F401  (imports register all algorithms)


def resume_from_checkpoint(cfg: DotDict) -> DotDict:
    """Merge the run config stored next to the checkpoint
    (parity: cli.py:23-57 — env/algo must match; new ``total_steps`` and
    ``learning_starts`` are kept)."""
    ckpt_path = pathlib.Path(cfg.checkpoint.resume_from)
    old_cfg_path = ckpt_path.parent.parent / "config.yaml"
    if not old_cfg_path.exists():
        raise RuntimeError(f"no config.yaml found next to checkpoint: {old_cfg_path}")
    import yaml

    with open(old_cfg_path) as f:
        old_cfg = DotDict(yaml.safe_load(f))
    if old_cfg.env.id != cfg.env.id:
        raise RuntimeError(
            f"environment mismatch on resume: checkpoint was trained on '{old_cfg.env.id}', requested '{cfg.env.id}'"
        )
    if old_cfg.algo.name != cfg.algo.name:
        raise RuntimeError(
            f"algorithm mismatch on resume: checkpoint is '{old_cfg.algo.name}', requested '{cfg.algo.name}'"
        )
    merged = copy.deepcopy(old_cfg)
    merged.algo.total_steps = cfg.algo.total_steps
    if "learning_starts" in cfg.algo:
        merged.algo.learning_starts = cfg.algo.learning_starts
    merged.checkpoint = cfg.checkpoint
    merged.runtime = cfg.runtime
    merged.metric = cfg.metric
    return merged


def check_configs(cfg: DotDict) -> None:
    """Config validation (parity: cli.py:271-345)."""
    _import_algorithms()
    name = cfg.algo.name
    if name not in algorithm_registry:
        raise ValueError(f"algorithm '{name}' is not registered; known: {sorted(algorithm_registry)}")
    entry = algorithm_registry[name]
    strategy = cfg.runtime.get("strategy", "auto")
    if strategy not in ("auto", "ddp"):
        raise ValueError(f"runtime.strategy must be 'auto' or 'ddp', got '{strategy}'")
    if entry.decoupled and int(cfg.runtime.devices) < 2:
        raise RuntimeError(
            f"algorithm '{name}' is decoupled (player + trainers) and needs runtime.devices >= 2, "
            f"got {cfg.runtime.devices}"
        )
    if cfg.algo.get("learning_starts", 0) < 0:
        raise ValueError("algo.learning_starts must be >= 0")
    if cfg.env.get("action_repeat", 1) < 1:
        raise ValueError("env.action_repeat must be >= 1")


def _build_runtime(cfg: DotDict) -> Runtime:
    callbacks = [CheckpointCallback(keep_last=cfg.checkpoint.get("keep_last"))]
    return Runtime(
        devices=cfg.runtime.devices,
        accelerator=cfg.runtime.accelerator,
        precision=cfg.runtime.precision,
        strategy=cfg.runtime.get("strategy", "auto"),
        callbacks=callbacks,
        bucket_cap_mb=cfg.runtime.get("bucket_cap_mb", 64),
    )


class _ReproducibleEntry:
    """Picklable launch target: re-resolves the entrypoint from the registry
    in the child process (multiprocessing 'spawn' cannot pickle closures) and
    applies the reproducibility settings (parity: cli.py:187-198)."""

    def __init__(self, algo_name: str) -> None:
        self.algo_name = algo_name

    def __call__(self, runtime: Runtime, cfg: DotDict):
        _import_algorithms()
        _import_external_modules(cfg)
        fn = algorithm_registry[self.algo_name].entrypoint
        seed_everything(cfg.seed + runtime.global_rank)
        torch.set_num_threads(int(cfg.get("num_threads", 1)))
        if cfg.get("torch_use_deterministic_algorithms", False):
            os.environ.setdefault("CUBLAS_WORKSPACE_CONFIG", cfg.get("cublas_workspace_config") or ":4096:8")
            torch.use_deterministic_algorithms(True, warn_only=True)
        torch.backends.cudnn.benchmark = cfg.get("torch_backends_cudnn_benchmark", True)
        torch.backends.cudnn.deterministic = cfg.get("torch_backends_cudnn_deterministic", False)
        fmm = cfg.get("float32_matmul_precision", "high")
        if fmm:
            torch.set_float32_matmul_precision(fmm)
        import torch.distributions as _td

        _td.Distribution.set_default_validate_args(bool(cfg.get("distribution", {}).get("validate_args", False)))
        return fn(runtime, cfg)


def _import_external_modules(cfg: DotDict) -> None:
    """Import user modules listed in ``import_modules`` so out-of-tree
    ``@register_algorithm`` entrypoints are registered (the reference's
    external-algorithm howto flow; combine with SHEEPRL_AMD_SEARCH_PATH for
    the configs)."""
    import importlib

    for mod in cfg.get("import_modules", []) or []:
        importlib.import_module(mod)


def run_algorithm(cfg: DotDict) -> None:
    """Registry lookup + launch (parity: cli.py:60-199)."""
    _import_algorithms()
    _import_external_modules(cfg)
    if cfg.metric.get("log_level", 1) <= 0 or cfg.metric.get("disable_timer", False):
        timer.disabled = True
    runtime = _build_runtime(cfg)
    runtime.launch(_ReproducibleEntry(cfg.algo.name), cfg)


def run(args: Optional[List[str]] = None) -> None:
    argv = list(sys.argv[1:] if args is None else args)
    cfg = compose(argv)
    _import_algorithms()
    _import_external_modules(cfg)
    if cfg.checkpoint.get("resume_from"):
        cfg = resume_from_checkpoint(cfg)
    check_configs(cfg)
    if cfg.get("print_config", True):
        print_config(cfg)
    run_algorithm(cfg)


def eval_algorithm(cfg: DotDict) -> None:
    """Load ckpt on a single-device runtime and dispatch the evaluator
    (parity: cli.py:202-268)."""
    _import_algorithms()
    _import_external_modules(cfg)
    name = cfg.algo.name
    if name not in evaluation_registry:
        raise ValueError(f"no evaluation registered for '{name}'")
    evaluate = evaluation_registry[name]
    runtime = Runtime(devices=1, accelerator=cfg.runtime.accelerator, precision=cfg.runtime.precision)

    def entry(rt: Runtime, cfg: DotDict) -> None:
        state = rt.load(cfg.checkpoint_path)
        with torch.no_grad():
            evaluate(rt, cfg, state)

    runtime.launch(entry, cfg)


def evaluation(args: Optional[List[str]] = None) -> None:
    argv = list(sys.argv[1:] if args is None else args)
    kv = dict(a.split("=", 1) for a in argv if "=" in a)
    ckpt = kv.pop("checkpoint_path", None)
    if ckpt is None:
        raise ValueError("evaluation requires checkpoint_path=<path to .ckpt>")
    ckpt_path = pathlib.Path(ckpt)
    run_cfg_path = ckpt_path.parent.parent / "config.yaml"
    if not run_cfg_path.exists():
        raise RuntimeError(f"no config.yaml next to checkpoint: {run_cfg_path}")
    import yaml

    with open(run_cfg_path) as f:
        cfg = DotDict(yaml.safe_load(f))
    for k, v in kv.items():
        cfg.set_nested(k, yaml.safe_load(v))
    cfg.runtime.devices = 1
    cfg.checkpoint_path = str(ckpt_path)
    eval_algorithm(cfg)


def registration(args: Optional[List[str]] = None) -> None:
    """Register a checkpoint's models into the local model registry
    (parity: cli.py:408-450; local backend — MLflow is not in this image)."""
    argv = list(sys.argv[1:] if args is None else args)
    kv = dict(a.split("=", 1) for a in argv if "=" in a)
    ckpt = kv.pop("checkpoint_path", None)
    if ckpt is None:
        raise ValueError("registration requires checkpoint_path=<path to .ckpt>")
    ckpt_path = pathlib.Path(ckpt)
    run_cfg_path = ckpt_path.parent.parent / "config.yaml"
    if not run_cfg_path.exists():
        raise RuntimeError(f"no config.yaml next to checkpoint: {run_cfg_path}")
    import yaml

    from sheeprl_amd.utils.model_manager import register_models_from_checkpoint

    with open(run_cfg_path) as f:
        cfg = DotDict(yaml.safe_load(f))
    registry_dir = kv.get("registry_dir", "models_registry")
    versions = register_models_from_checkpoint(cfg, str(ckpt_path), registry_dir)
    for name, v in versions.items():
        print(f"registered {name} -> v{v}")


def available_agents() -> None:
    """Print the registered algorithms (parity: available_agents.py:7)."""
    _import_algorithms()
    try:
        from rich.console import Console
        from rich.table import Table

        table = Table(title="sheeprl-amd agents")
        table.add_column("Algorithm")
        table.add_column("Module")
        table.add_column("Decoupled")
        for name, entry in sorted(algorithm_registry.items()):
            table.add_row(name, entry.module, str(entry.decoupled))
        Console().print(table)
    except Exception:
        for name, entry in sorted(algorithm_registry.items()):
            print(f"{name:24s} {entry.module} decoupled={entry.decoupled}")


_USAGE = """usage: python -m sheeprl_amd exp=<experiment> [group=value] [a.b.c=value ...]
       python -m sheeprl_amd eval checkpoint_path=<ckpt> [overrides...]
       python -m sheeprl_amd register checkpoint_path=<ckpt> [overrides...]
       python -m sheeprl_amd agents

Experiments live in sheeprl_amd/configs/exp/ (e.g. exp=dreamer_v3, exp=ppo);
run `python -m sheeprl_amd agents` to list the registered algorithms."""


def main() -> None:
    warnings.filterwarnings("ignore", category=UserWarning, module="torch.distributed")
    argv = sys.argv[1:]
    if not argv or argv[0] in ("-h", "--help", "help"):
        print(_USAGE)
        return
    if argv[0] == "eval":
        evaluation(argv[1:])
    elif argv[0] == "register":
        registration(argv[1:])
    elif argv[0] == "agents":
        available_agents()
    else:
        try:
            run(argv)
        except ValueError as e:
            if "missing required config values" in str(e):
                raise SystemExit(f"{e}\n\n{_USAGE}") from None
            raise


if __name__ == "__main__":
    main()
