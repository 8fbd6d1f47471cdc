"""Plan2Explore (Dreamer-V3 base) agent.

Parity surface: sheeprl/algos/p2e_dv3/agent.py:27-120 — the DV3 world model
plus an ensemble of next-stochastic-state predictors (disagreement =
intrinsic reward), a task actor/critic and an exploration actor with a DICT
of critics (intrinsic/extrinsic, weighted), each with its own EMA target and
Moments normalizer.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, Optional, Sequence

import numpy as np
import torch
from torch import nn

from sheeprl_amd.algos.dreamer_v3.agent import (
    Actor,
    PlayerDV3,
    build_agent as dv3_build_agent,
    init_weights,
    uniform_init_weights,
)
from sheeprl_amd.algos.dreamer_v3.utils import Moments
from sheeprl_amd.envs import spaces
from sheeprl_amd.models import MLP
from sheeprl_amd.parallel import Runtime


def build_agent(
    runtime: Runtime,
    actions_dim: Sequence[int],
    is_continuous: bool,
    cfg: Any,
    obs_space: spaces.Dict,
    world_model_state=None,
    ensembles_state=None,
    actor_task_state=None,
    critic_task_state=None,
    target_critic_task_state=None,
    actor_exploration_state=None,
    critics_exploration_state: Optional[Dict[str, Any]] = None,
):
    # task agent = plain DV3 agent
    world_model, actor_task, critic_task, target_critic_task, player = dv3_build_agent(
        runtime, actions_dim, is_continuous, cfg, obs_space,
        world_model_state, actor_task_state, critic_task_state, target_critic_task_state,
    )

    wm_cfg = cfg.algo.world_model
    stoch_state_size = wm_cfg.stochastic_size * wm_cfg.discrete_size
    latent_state_size = stoch_state_size + wm_cfg.recurrent_model.recurrent_state_size

    # exploration actor (same family as the task actor)
    actor_exploration = Actor(
        latent_state_size=latent_state_size,
        actions_dim=actions_dim,
        is_continuous=is_continuous,
        distribution=cfg.distribution.get("type", "auto"),
        init_std=cfg.algo.actor.init_std,
        min_std=cfg.algo.actor.min_std,
        max_std=cfg.algo.actor.get("max_std", 1.0),
        dense_units=cfg.algo.actor.dense_units,
        mlp_layers=cfg.algo.actor.mlp_layers,
        unimix=cfg.algo.unimix,
        action_clip=cfg.algo.actor.get("action_clip", 1.0),
    )
    actor_exploration.apply(init_weights)
    if cfg.algo.hafner_initialization:
        actor_exploration.mlp_heads.apply(uniform_init_weights(1.0))
    if actor_exploration_state:
        actor_exploration.load_state_dict(actor_exploration_state)
    actor_exploration = runtime.setup_module(actor_exploration)

    # exploration critics: dict of {name: {module, target, moments, weight, reward_type}}
    critics_exploration: Dict[str, Dict[str, Any]] = {}
    for name, spec in cfg.algo.critics_exploration.items():
        critic = MLP(
            latent_state_size,
            cfg.algo.critic.bins,
            [cfg.algo.critic.dense_units] * cfg.algo.critic.mlp_layers,
            activation="silu",
            layer_norm=True,
            layer_norm_eps=1e-3,
        )
        critic.apply(init_weights)
        if cfg.algo.hafner_initialization:
            critic.model[-1].apply(uniform_init_weights(0.0))
        st = (critics_exploration_state or {}).get(name)
        if st:
            critic.load_state_dict(st["module"])
        critic = runtime.setup_module(critic)
        target = copy.deepcopy(critic)
        if st and "target" in st:
            target.load_state_dict(st["target"])
        target = runtime.setup_module(target, sync=False)
        for p in target.parameters():
            p.requires_grad_(False)
        moments = Moments(
            cfg.algo.actor.moments.decay,
            cfg.algo.actor.moments.max,
            cfg.algo.actor.moments.percentile.low,
            cfg.algo.actor.moments.percentile.high,
        ).to(runtime.device)
        if st and "moments" in st:
            moments.load_state_dict(st["moments"])
        critics_exploration[name] = {
            "module": critic,
            "target_module": target,
            "moments": moments,
            "weight": spec.weight,
            "reward_type": spec.reward_type,
        }

    # disagreement ensembles: predict the next stochastic state from
    # (latent, action) (p2e_dv3_exploration.py:205-231)
    ens_cfg = cfg.algo.ensembles
    ensembles = nn.ModuleList(
        [
            MLP(
                latent_state_size + int(np.sum(actions_dim)),
                stoch_state_size,
                [ens_cfg.dense_units] * ens_cfg.mlp_layers,
                activation="silu",
                layer_norm=True,
                layer_norm_eps=1e-3,
            )
            for _ in range(ens_cfg.n)
        ]
    )
    for i, ens in enumerate(ensembles):
        # decorrelate the ensemble members' inits
        torch.manual_seed(cfg.seed + runtime.global_rank * 1000 + i)
        ens.apply(init_weights)
    if ensembles_state:
        ensembles.load_state_dict(ensembles_state)
    ensembles = runtime.setup_module(ensembles)

    player.actor_type = cfg.algo.player.actor_type
    if cfg.algo.player.actor_type == "exploration":
        player.actor = actor_exploration
    return (
        world_model,
        ensembles,
        actor_task,
        critic_task,
        target_critic_task,
        actor_exploration,
        critics_exploration,
        player,
    )
