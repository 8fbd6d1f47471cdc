"""DroQ (parity: sheeprl/algos/droq — DROQCritic with dropout+LayerNorm
droq/agent.py:20, DROQAgent :63, train droq.py:31: high replay-ratio critic
updates with dropout-regularized Q ensemble, delayed actor update).

Reuses the SAC training loop with DroQ critics: the loop's replay-ratio
scheduler provides the G critic updates per env step and
``algo.actor.update_freq`` delays the actor to once per G updates."""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import numpy as np
import torch
from torch import Tensor, nn

from sheeprl_amd.algos.sac.agent import SACActor, SACAgent, SACPlayer
from sheeprl_amd.algos.sac.sac import main as sac_main
from sheeprl_amd.algos.sac.utils import test
from sheeprl_amd.envs import make_env, spaces
from sheeprl_amd.models import LayerNorm
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation


class DROQCritic(nn.Module):
    """Q net with Dropout + LayerNorm + ReLU blocks (reference droq/agent.py:20)."""

    def __init__(self, input_dim: int, hidden_size: int = 256, dropout: float = 0.01) -> None:
        super().__init__()
        self.model = nn.Sequential(
            nn.Linear(input_dim, hidden_size),
            nn.Dropout(dropout),
            LayerNorm(hidden_size),
            nn.ReLU(),
            nn.Linear(hidden_size, hidden_size),
            nn.Dropout(dropout),
            LayerNorm(hidden_size),
            nn.ReLU(),
            nn.Linear(hidden_size, 1),
        )

    def forward(self, obs: Tensor, action: Tensor) -> Tensor:
        return self.model(torch.cat([obs, action], dim=-1))


MODELS_TO_REGISTER = {"agent"}


def build_agent(
    runtime: Runtime,
    cfg: Any,
    obs_space: spaces.Dict,
    action_space: spaces.Box,
    agent_state: Optional[Dict[str, Tensor]] = None,
) -> Tuple[SACAgent, SACPlayer]:
    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    obs_dim = sum(int(np.prod(obs_space[k].shape)) for k in mlp_keys)
    act_dim = int(np.prod(action_space.shape))
    actor = SACActor(
        obs_dim,
        act_dim,
        hidden_size=cfg.algo.actor.hidden_size,
        action_low=action_space.low,
        action_high=action_space.high,
    )
    critics = [
        DROQCritic(obs_dim + act_dim, cfg.algo.critic.hidden_size, dropout=cfg.algo.critic.dropout)
        for _ in range(cfg.algo.critic.n)
    ]
    agent = SACAgent(
        actor,
        critics,
        target_entropy=-act_dim,
        alpha=cfg.algo.alpha.alpha,
        tau=cfg.algo.tau,
        device=runtime.device,
    )
    if agent_state:
        agent.load_state_dict(agent_state)
    agent = runtime.setup_module(agent)
    player = SACPlayer(agent.actor)
    return agent, player


@register_algorithm(name="droq")
def main(runtime: Runtime, cfg: Any) -> None:
    # swap the agent builder the SAC loop uses, then run it unchanged
    import sheeprl_amd.algos.sac.sac as sac_mod

    orig = sac_mod.build_agent
    sac_mod.build_agent = build_agent
    try:
        sac_main.__wrapped__(runtime, cfg) if hasattr(sac_main, "__wrapped__") else sac_main(runtime, cfg)
    finally:
        sac_mod.build_agent = orig


@register_evaluation(algorithms=["droq"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    env_fn = make_env(cfg, cfg.seed, 0)
    env = env_fn()
    obs_space, action_space = env.observation_space, env.action_space
    env.close()
    _, player = build_agent(runtime, cfg, obs_space, action_space, state["agent"])
    reward = test(player, env_fn, cfg, runtime.device)
    runtime.print(f"Test/cumulative_reward: {reward}")
    return reward
