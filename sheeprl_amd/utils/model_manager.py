"""Model registry.

Parity surface with sheeprl/utils/mlflow.py (``MlflowModelManager`` :75,
``register_model`` :384, ``register_model_from_checkpoint`` :330): versioned
model registration with descriptions and a changelog, plus transition/delete
operations.  MLflow is not in this image, so the backing store is a local
directory tree::

    models_registry/<model_name>/v<N>/model.pt      # state_dict
    models_registry/<model_name>/v<N>/meta.yaml     # description, tags, source
    models_registry/<model_name>/changelog.md
"""

from __future__ import annotations

import datetime
import shutil
from pathlib import Path
from typing import Any, Dict, Optional

import torch
import yaml


class ModelManager:
    def __init__(self, registry_dir: str | Path = "models_registry") -> None:
        self.registry_dir = Path(registry_dir)
        self.registry_dir.mkdir(parents=True, exist_ok=True)

    # -- helpers ------------------------------------------------------------
    def _model_dir(self, name: str) -> Path:
        return self.registry_dir / name

    def _versions(self, name: str) -> list:
        d = self._model_dir(name)
        if not d.exists():
            return []
        return sorted(
            (int(p.name[1:]) for p in d.iterdir() if p.is_dir() and p.name.startswith("v")),
        )

    def get_latest_version(self, name: str) -> Optional[int]:
        vs = self._versions(name)
        return vs[-1] if vs else None

    def _log_changelog(self, name: str, text: str) -> None:
        with open(self._model_dir(name) / "changelog.md", "a") as f:
            f.write(f"- {datetime.datetime.now().isoformat(timespec='seconds')}: {text}\n")

    # -- operations ----------------------------------------------------------
    def register_model(
        self,
        name: str,
        state_dict: Dict[str, torch.Tensor],
        description: str = "",
        tags: Optional[Dict[str, Any]] = None,
        source: str = "",
    ) -> int:
        version = (self.get_latest_version(name) or 0) + 1
        vdir = self._model_dir(name) / f"v{version}"
        vdir.mkdir(parents=True, exist_ok=True)
        torch.save(state_dict, vdir / "model.pt", pickle_protocol=4)
        with open(vdir / "meta.yaml", "w") as f:
            yaml.safe_dump(
                {
                    "name": name,
                    "version": version,
                    "description": description,
                    "tags": tags or {},
                    "source": source,
                    "stage": "None",
                    "registered_at": datetime.datetime.now().isoformat(timespec="seconds"),
                },
                f,
            )
        self._log_changelog(name, f"registered version {version} ({description or 'no description'})")
        return version

    def load_model(self, name: str, version: Optional[int] = None) -> Dict[str, torch.Tensor]:
        version = version or self.get_latest_version(name)
        if version is None:
            raise FileNotFoundError(f"no registered versions for model '{name}'")
        return torch.load(self._model_dir(name) / f"v{version}" / "model.pt", map_location="cpu", weights_only=False)

    def get_model_info(self, name: str, version: Optional[int] = None) -> Dict[str, Any]:
        version = version or self.get_latest_version(name)
        with open(self._model_dir(name) / f"v{version}" / "meta.yaml") as f:
            return yaml.safe_load(f)

    def transition_model(self, name: str, version: Optional[int] = None, stage: str = "staging") -> None:
        version = version or self.get_latest_version(name)
        meta_path = self._model_dir(name) / f"v{version}" / "meta.yaml"
        meta = yaml.safe_load(open(meta_path))
        meta["stage"] = stage
        with open(meta_path, "w") as f:
            yaml.safe_dump(meta, f)
        self._log_changelog(name, f"version {version} -> stage '{stage}'")

    def delete_model(self, name: str, version: Optional[int] = None) -> None:
        if version is None:
            shutil.rmtree(self._model_dir(name), ignore_errors=True)
        else:
            shutil.rmtree(self._model_dir(name) / f"v{version}", ignore_errors=True)
            self._log_changelog(name, f"deleted version {version}")

    def registered_models(self) -> Dict[str, list]:
        return {d.name: self._versions(d.name) for d in self.registry_dir.iterdir() if d.is_dir()}


def register_models_from_checkpoint(
    cfg: Any, ckpt_path: str, registry_dir: str | Path = "models_registry"
) -> Dict[str, int]:
    """Register every model the algorithm declares in MODELS_TO_REGISTER
    (parity: sheeprl/utils/mlflow.py:330 + cli registration :408-450)."""
    import importlib

    state = torch.load(ckpt_path, map_location="cpu", weights_only=False)
    algo_name = cfg.algo.name
    base = algo_name.replace("_exploration", "").replace("_finetuning", "").replace("_decoupled", "")
    to_register = None
    # MODELS_TO_REGISTER lives either in the algo's utils module or in the
    # entrypoint module itself (e.g. a2c/a2c.py, p2e_dv3/p2e_dv3_exploration.py)
    for mod_name in (
        f"sheeprl_amd.algos.{base}.utils",
        f"sheeprl_amd.algos.{base}.{algo_name}",
        f"sheeprl_amd.algos.{base}.{base}",
    ):
        try:
            mod = importlib.import_module(mod_name)
        except ImportError:
            continue
        to_register = getattr(mod, "MODELS_TO_REGISTER", None)
        if to_register:
            break
    if not to_register:
        to_register = {k for k in state if isinstance(state[k], dict) and any("weight" in kk for kk in state[k])}
    manager = ModelManager(registry_dir)
    versions = {}
    for model_name in to_register:
        if model_name in state and isinstance(state[model_name], dict):
            versions[model_name] = manager.register_model(
                f"{algo_name}_{cfg.env.id}_{model_name}",
                state[model_name],
                description=f"{model_name} from {ckpt_path}",
                tags={"algo": algo_name, "env": cfg.env.id},
                source=str(ckpt_path),
            )
    return versions
