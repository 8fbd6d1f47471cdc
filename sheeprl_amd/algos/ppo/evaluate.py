"""PPO evaluation entrypoint (parity: sheeprl/algos/ppo/evaluate.py)."""

from __future__ import annotations

from typing import Any, Dict

from sheeprl_amd.algos.ppo.agent import build_agent
from sheeprl_amd.algos.ppo.utils import test
from sheeprl_amd.envs import make_env
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.registry import register_evaluation


@register_evaluation(algorithms=["ppo", "ppo_decoupled"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    env_fn = make_env(cfg, cfg.seed, 0)
    env = env_fn()
    obs_space = env.observation_space
    action_space = env.action_space
    env.close()
    _, player = build_agent(runtime, obs_space, action_space, cfg, state["agent"])
    reward = test(player, env_fn, cfg, ".", runtime.device)
    runtime.print(f"Test/cumulative_reward: {reward}")
    return reward
