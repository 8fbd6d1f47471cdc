"""Dreamer-V2 agent (parity surface: sheeprl/algos/dreamer_v2/agent.py —
RSSM :301, WorldModel :707 (shared by DV3's import, dreamer_v3/agent.py:24),
PlayerDV2 :735).

Identical machinery to DV3 with V2's choices: ELU activations, no LayerNorm
in the dense/conv stacks (the GRU cell keeps its LN), discrete 32x32 latents
WITHOUT unimix, Normal(.,1) reward/value heads (size-1 outputs), optional
continue model, hard target-critic copies.  Reuses the DV3 building blocks
(sheeprl_amd/algos/dreamer_v3/agent.py) parameterized for ELU/no-LN.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, Optional, Sequence, Tuple

import numpy as np
import torch
from torch import Tensor, nn

from sheeprl_amd.algos.dreamer_v3.agent import (
    Actor,
    CNNDecoder,
    CNNEncoder,
    MLPDecoder,
    MLPEncoder,
    PlayerDV3,
    RecurrentModel,
    RSSM,
    WorldModel,
)
from sheeprl_amd.envs import spaces
from sheeprl_amd.models import MLP, MultiDecoder, MultiEncoder
from sheeprl_amd.parallel import Runtime

PlayerDV2 = PlayerDV3  # same inference player (parity: dreamer_v2/agent.py:735)


def build_agent(
    runtime: Runtime,
    actions_dim: Sequence[int],
    is_continuous: bool,
    cfg: Any,
    obs_space: spaces.Dict,
    world_model_state: Optional[Dict[str, Tensor]] = None,
    actor_state: Optional[Dict[str, Tensor]] = None,
    critic_state: Optional[Dict[str, Tensor]] = None,
    target_critic_state: Optional[Dict[str, Tensor]] = None,
) -> Tuple[WorldModel, Actor, MLP, nn.Module, PlayerDV3]:
    wm_cfg = cfg.algo.world_model
    act = cfg.algo.get("dense_act", "elu")
    ln = bool(cfg.algo.get("layer_norm", False))
    eps = 1e-3

    recurrent_state_size = wm_cfg.recurrent_model.recurrent_state_size
    stochastic_size = wm_cfg.stochastic_size * wm_cfg.discrete_size
    latent_state_size = stochastic_size + recurrent_state_size
    cnn_keys = list(cfg.algo.cnn_keys.encoder or [])
    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    stages = int(np.log2(cfg.env.screen_size) - np.log2(4))

    cnn_encoder = (
        CNNEncoder(
            cnn_keys,
            [int(np.prod(obs_space[k].shape[:-2])) for k in cnn_keys],
            tuple(obs_space[cnn_keys[0]].shape[-2:]),
            wm_cfg.encoder.cnn_channels_multiplier,
            eps,
            stages,
            activation=act,
            layer_norm=ln,
        )
        if cnn_keys
        else None
    )
    mlp_encoder = (
        MLPEncoder(
            mlp_keys,
            [int(obs_space[k].shape[0]) for k in mlp_keys],
            wm_cfg.encoder.mlp_layers,
            wm_cfg.encoder.dense_units,
            eps,
            symlog_inputs=False,
            activation=act,
            layer_norm=ln,
        )
        if mlp_keys
        else None
    )
    encoder = MultiEncoder(cnn_encoder, mlp_encoder)

    recurrent_model = RecurrentModel(
        int(sum(actions_dim) + stochastic_size),
        recurrent_state_size,
        wm_cfg.recurrent_model.dense_units,
        eps,
        activation=act,
        layer_norm=True,  # DV2 keeps LN inside the recurrent model
    )
    representation_model = MLP(
        encoder.output_dim + recurrent_state_size,
        stochastic_size,
        [wm_cfg.representation_model.hidden_size],
        activation=act,
        layer_norm=ln,
        layer_norm_eps=eps,
    )
    transition_model = MLP(
        recurrent_state_size,
        stochastic_size,
        [wm_cfg.transition_model.hidden_size],
        activation=act,
        layer_norm=ln,
        layer_norm_eps=eps,
    )
    rssm = RSSM(
        recurrent_model,
        representation_model,
        transition_model,
        discrete=wm_cfg.discrete_size,
        unimix=0.0,  # DV2 has no uniform mixing
        learnable_initial_recurrent_state=wm_cfg.get("learnable_initial_recurrent_state", False),
    )

    cnn_decoder = (
        CNNDecoder(
            cnn_keys,
            [int(np.prod(obs_space[k].shape[:-2])) for k in cnn_keys],
            wm_cfg.observation_model.cnn_channels_multiplier,
            latent_state_size,
            cnn_encoder.output_dim,
            tuple(obs_space[cnn_keys[0]].shape[-2:]),
            eps,
            stages,
            activation=act,
            layer_norm=ln,
        )
        if cnn_keys
        else None
    )
    mlp_decoder = (
        MLPDecoder(
            mlp_keys,
            [int(obs_space[k].shape[0]) for k in mlp_keys],
            latent_state_size,
            wm_cfg.observation_model.mlp_layers,
            wm_cfg.observation_model.dense_units,
            eps,
            activation=act,
            layer_norm=ln,
        )
        if mlp_keys
        else None
    )
    observation_model = MultiDecoder(cnn_decoder, mlp_decoder)

    reward_model = MLP(
        latent_state_size,
        1,
        [wm_cfg.reward_model.dense_units] * wm_cfg.reward_model.mlp_layers,
        activation=act,
        layer_norm=ln,
        layer_norm_eps=eps,
    )
    continue_model = MLP(
        latent_state_size,
        1,
        [wm_cfg.discount_model.dense_units] * wm_cfg.discount_model.mlp_layers,
        activation=act,
        layer_norm=ln,
        layer_norm_eps=eps,
    )
    world_model = WorldModel(encoder, rssm, observation_model, reward_model, continue_model)

    actor = Actor(
        latent_state_size=latent_state_size,
        actions_dim=actions_dim,
        is_continuous=is_continuous,
        distribution=cfg.distribution.get("type", "auto"),
        init_std=cfg.algo.actor.init_std,
        min_std=cfg.algo.actor.min_std,
        max_std=cfg.algo.actor.get("max_std", 1.0),
        dense_units=cfg.algo.actor.dense_units,
        mlp_layers=cfg.algo.actor.mlp_layers,
        layer_norm_eps=eps,
        unimix=0.0,
        action_clip=cfg.algo.actor.get("action_clip", 1.0),
        activation=act,
        layer_norm=ln,
    )
    critic = MLP(
        latent_state_size,
        1,
        [cfg.algo.critic.dense_units] * cfg.algo.critic.mlp_layers,
        activation=act,
        layer_norm=ln,
        layer_norm_eps=eps,
    )

    if world_model_state:
        world_model.load_state_dict(world_model_state)
    if actor_state:
        actor.load_state_dict(actor_state)
    if critic_state:
        critic.load_state_dict(critic_state)

    world_model = runtime.setup_module(world_model)
    actor = runtime.setup_module(actor)
    critic = runtime.setup_module(critic)
    target_critic = copy.deepcopy(critic)
    if target_critic_state:
        target_critic.load_state_dict(target_critic_state)
    target_critic = runtime.setup_module(target_critic, sync=False)
    for p in target_critic.parameters():
        p.requires_grad_(False)

    player = PlayerDV2(
        world_model.encoder,
        world_model.rssm,
        actor,
        actions_dim,
        cfg.env.num_envs,
        wm_cfg.stochastic_size,
        recurrent_state_size,
        runtime.device,
        discrete_size=wm_cfg.discrete_size,
    )
    return world_model, actor, critic, target_critic, player
