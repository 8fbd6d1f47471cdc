"""Checkpoint callback.

Parity with sheeprl/utils/callback.py:14-148:
* ``on_checkpoint_coupled`` — gathers per-rank replay buffers to rank 0 (gloo
  object gather in the reference, callback.py:42-51) and saves the full state.
* ``on_checkpoint_player`` / ``on_checkpoint_trainer`` — decoupled variants.
* buffer-consistency trick (:87-120): before saving, mark the slot *before*
  the write head truncated so resumed sampling never stitches across the
  discontinuity; restored afterwards.
* ``keep_last`` pruning (:144-148).
"""

from __future__ import annotations

import os
from pathlib import Path
from typing import Any, Dict, Optional

import numpy as np


class CheckpointCallback:
    def __init__(self, keep_last: Optional[int] = None) -> None:
        self.keep_last = keep_last

    # -- helpers -----------------------------------------------------------
    def _experiment_dir(self, ckpt_path: str) -> Path:
        return Path(ckpt_path).parent

    def _prune(self, ckpt_dir: Path) -> None:
        if not self.keep_last:
            return
        ckpts = sorted(Path(ckpt_dir).glob("ckpt_*.ckpt"), key=os.path.getmtime)
        for old in ckpts[: -self.keep_last]:
            try:
                old.unlink()
            except OSError:
                pass

    @staticmethod
    def _ckpt_rb(rb: Any) -> Dict[str, Any]:
        """Temporarily mark the entry before the write head truncated
        (reference callback.py:87-120); returns restore info."""
        restore: Dict[str, Any] = {}
        from sheeprl_amd.data.buffers import EnvIndependentReplayBuffer, EpisodeBuffer, ReplayBuffer

        if isinstance(rb, ReplayBuffer):
            if "truncated" in rb and len(rb) > 0:
                pos = (rb._pos - 1) % rb.buffer_size
                arr = rb._buf["truncated"]
                restore = {"kind": "rb", "pos": pos, "old": np.asarray(arr[pos]).copy()}
                arr[pos] = np.ones_like(np.asarray(arr[pos]))
        elif isinstance(rb, EnvIndependentReplayBuffer):
            restore = {"kind": "env_independent", "subs": [CheckpointCallback._ckpt_rb(b) for b in rb.buffer]}
        elif isinstance(rb, EpisodeBuffer):
            restore = {"kind": "episode", "open": [o for o in rb._open]}
            rb._open = [None] * rb.n_envs  # drop open episodes from the snapshot
        return restore

    @staticmethod
    def _restore_rb(rb: Any, restore: Dict[str, Any]) -> None:
        if not restore:
            return
        kind = restore.get("kind")
        if kind == "rb":
            rb._buf["truncated"][restore["pos"]] = restore["old"]
        elif kind == "env_independent":
            for b, r in zip(rb.buffer, restore["subs"]):
                CheckpointCallback._restore_rb(b, r)
        elif kind == "episode":
            rb._open = restore["open"]

    # -- hooks --------------------------------------------------------------
    def on_checkpoint_coupled(
        self,
        runtime: Any,
        ckpt_path: str,
        state: Dict[str, Any],
        replay_buffer: Any = None,
    ) -> None:
        if replay_buffer is not None:
            restore = self._ckpt_rb(replay_buffer)
            gathered = runtime.gather_object(replay_buffer.state_dict(), dst=0)
            if runtime.global_rank == 0:
                state = dict(state)
                state["rb"] = gathered
            self._restore_rb(replay_buffer, restore)
        runtime.save(ckpt_path, state)
        if runtime.global_rank == 0:
            self._prune(self._experiment_dir(ckpt_path))

    def on_checkpoint_player(
        self,
        runtime: Any,
        player_trainer_collective: Any,
        ckpt_path: str,
        replay_buffer: Any = None,
    ) -> None:
        # trainers broadcast the state object to the player (reference :58)
        payload = [None]
        runtime.broadcast_object_list(payload, src=1, group=player_trainer_collective)
        state = payload[0] or {}
        if replay_buffer is not None:
            restore = self._ckpt_rb(replay_buffer)
            state["rb"] = [replay_buffer.state_dict()]
            self._restore_rb(replay_buffer, restore)
        runtime.save(ckpt_path, state)
        if runtime.global_rank == 0:
            self._prune(self._experiment_dir(ckpt_path))

    def on_checkpoint_trainer(
        self, runtime: Any, player_trainer_collective: Any, ckpt_path: str, state: Dict[str, Any]
    ) -> None:
        if runtime.global_rank != 1:
            return  # only the lead trainer ships state to the player
        import torch
        from torch import nn

        payload: Dict[str, Any] = {}
        for k, v in state.items():
            if isinstance(v, (nn.Module, torch.optim.Optimizer)):
                payload[k] = v.state_dict()
            elif hasattr(v, "state_dict") and not isinstance(v, dict):
                payload[k] = v.state_dict()
            else:
                payload[k] = v
        runtime.broadcast_object_list([payload], src=1, group=player_trainer_collective)
