"""PPO helpers (parity: sheeprl/algos/ppo/utils.py)."""

from __future__ import annotations

from typing import Any, Dict

import numpy as np
import torch
from torch import Tensor


AGGREGATOR_KEYS = {
    "Rewards/rew_avg",
    "Game/ep_len_avg",
    "Loss/value_loss",
    "Loss/policy_loss",
    "Loss/entropy_loss",
}
MODELS_TO_REGISTER = {"agent"}


def prepare_obs(obs: Dict[str, np.ndarray], cfg: Any, device: torch.device) -> Dict[str, Tensor]:
    """numpy env obs -> torch device tensors; uint8 images stay uint8 (the
    encoder normalizes on device with the fused kernel)."""
    out: Dict[str, Tensor] = {}
    for k in set(list(cfg.algo.cnn_keys.encoder or []) + list(cfg.algo.mlp_keys.encoder or [])):
        v = torch.as_tensor(np.ascontiguousarray(obs[k]))
        if k in (cfg.algo.cnn_keys.encoder or []):
            out[k] = v.to(device)
        else:
            out[k] = v.float().to(device)
    return out


@torch.no_grad()
def test(player: Any, env_fn: Any, cfg: Any, log_dir: str, device: torch.device, aggregator=None) -> float:
    """One greedy episode; returns cumulative reward
    (parity: ppo/utils.py test())."""
    env = env_fn()
    obs, _ = env.reset(seed=cfg.seed)
    done = False
    cum_reward = 0.0
    while not done:
        batched = {k: np.expand_dims(v, 0) for k, v in obs.items()}
        t_obs = prepare_obs(batched, cfg, device)
        actions, _, _ = player.get_actions(t_obs, greedy=True)
        a = actions.cpu().numpy().reshape(-1)
        if not player.actor.is_continuous:
            a = a[0] if a.shape[0] == 1 else a
        obs, reward, term, trunc, _ = env.step(a)
        cum_reward += float(reward)
        done = bool(term or trunc)
    env.close()
    return cum_reward
