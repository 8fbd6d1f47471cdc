"""SAC helpers (parity: sheeprl/algos/sac/utils.py)."""

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
    "Loss/alpha_loss",
}
MODELS_TO_REGISTER = {"agent"}


def prepare_obs(obs: Dict[str, np.ndarray], cfg: Any, device: torch.device) -> Tensor:
    """Concat the configured mlp keys into one flat float tensor."""
    keys = list(cfg.algo.mlp_keys.encoder or [])
    arrs = [np.asarray(obs[k], dtype=np.float32).reshape(np.asarray(obs[k]).shape[0], -1) for k in keys]
    return torch.as_tensor(np.concatenate(arrs, axis=-1), device=device)


@torch.no_grad()
def test(player: Any, env_fn: Any, cfg: Any, device: torch.device) -> float:
    env = env_fn()
    obs, _ = env.reset(seed=cfg.seed)
    done = False
    cum_reward = 0.0
    while not done:
        batched = {k: np.expand_dims(np.asarray(v), 0) for k, v in obs.items()}
        t_obs = prepare_obs(batched, cfg, device)
        action = player.get_actions(t_obs, greedy=True).cpu().numpy().reshape(-1)
        obs, reward, term, trunc, _ = env.step(action)
        cum_reward += float(reward)
        done = bool(term or trunc)
    env.close()
    return cum_reward
