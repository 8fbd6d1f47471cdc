"""Dreamer-V1: continuous (Normal) stochastic state world model.

Parity surface: sheeprl/algos/dreamer_v1 — RSSM dreamer_v1/agent.py:64,
WorldModel :192, PlayerDV1 :219; losses dreamer_v1/loss.py (critic :9,
actor :27 — pure dynamics backprop, reconstruction :41 with free-nats KL
between Normal posterior/prior); compute_stochastic_state
dreamer_v1/utils.py:80-108 (mean/softplus-std reparameterized sample).

Reuses the DV3 building blocks (encoder/decoder/recurrent cell) in their
ELU/no-LN configuration; the stochastic state is a 30-dim diagonal Normal.
"""

from __future__ import annotations

import os
from typing import Any, Dict, Optional, Sequence

import numpy as np
import torch
import torch.distributions as td
import torch.nn.functional as F
from torch import Tensor, nn

from sheeprl_amd import ops
from sheeprl_amd.algos.dreamer_v3.agent import (
    Actor,
    CNNDecoder,
    CNNEncoder,
    MLPDecoder,
    MLPEncoder,
    RecurrentModel,
)
from sheeprl_amd.config import save_config
from sheeprl_amd.data import EnvIndependentReplayBuffer, SequentialReplayBuffer
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.models import MLP, MultiDecoder, MultiEncoder
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import Ratio

AGGREGATOR_KEYS = {
    "Rewards/rew_avg",
    "Game/ep_len_avg",
    "Loss/world_model_loss",
    "Loss/value_loss",
    "Loss/policy_loss",
    "Loss/observation_loss",
    "Loss/reward_loss",
    "Loss/state_loss",
    "State/kl",
}
MODELS_TO_REGISTER = {"world_model", "actor", "critic"}



from sheeprl_amd.algos.dreamer_v2.dreamer_v2 import _FastUnitNormal


def _unit_scale(x):
    """Scale=1 as a device tensor: td.Normal(x, 1) materializes the python
    scalar with a pageable H2D copy, which is illegal inside hipGraph capture."""
    return torch.ones((), device=x.device, dtype=x.dtype)

def compute_stochastic_state(info: Tensor, min_std: float = 0.1):
    """(mean, std), sample — parity: dreamer_v1/utils.py:80-108."""
    mean, std = torch.chunk(info.float(), 2, -1)
    std = F.softplus(std) + min_std
    dist = td.Independent(td.Normal(mean, std), 1)
    return (mean, std), dist.rsample()


class RSSMV1(nn.Module):
    def __init__(self, recurrent_model, representation_model, transition_model, min_std: float = 0.1):
        super().__init__()
        self.recurrent_model = recurrent_model
        self.representation_model = representation_model
        self.transition_model = transition_model
        self.min_std = min_std

    @property
    def _dtype(self):
        return next(self.recurrent_model.parameters()).dtype

    def dynamic(self, stochastic_state, recurrent_state, action, embedded_obs):
        dt = self._dtype
        recurrent_state = self.recurrent_model(
            torch.cat((stochastic_state.to(dt), action.to(dt)), -1), recurrent_state.to(dt)
        )
        prior_state_mean_std, prior = self._transition(recurrent_state)
        posterior_mean_std, posterior = self._representation(recurrent_state, embedded_obs)
        return recurrent_state, posterior, prior, posterior_mean_std, prior_state_mean_std

    def _representation(self, recurrent_state, embedded_obs):
        dt = self._dtype
        info = self.representation_model(torch.cat((recurrent_state.to(dt), embedded_obs.to(dt)), -1))
        (mean, std), sample = compute_stochastic_state(info, self.min_std)
        return (mean, std), sample.to(dt)

    def _transition(self, recurrent_state):
        info = self.transition_model(recurrent_state.to(self._dtype))
        (mean, std), sample = compute_stochastic_state(info, self.min_std)
        return (mean, std), sample.to(self._dtype)

    def imagination(self, stochastic_state, recurrent_state, actions):
        dt = self._dtype
        recurrent_state = self.recurrent_model(
            torch.cat((stochastic_state.to(dt), actions.to(dt)), -1), recurrent_state.to(dt)
        )
        _, imagined_state = self._transition(recurrent_state)
        return imagined_state, recurrent_state


class WorldModelV1(nn.Module):
    def __init__(self, encoder, rssm, observation_model, reward_model, continue_model=None):
        super().__init__()
        self.encoder = encoder
        self.rssm = rssm
        self.observation_model = observation_model
        self.reward_model = reward_model
        self.continue_model = continue_model


def add_exploration_noise(actions, amount, is_continuous):
    """Exploration noise on sampled actions (reference dreamer_v1/agent.py
    :558-574): Gaussian-perturb-and-clip for continuous actions,
    epsilon-random one-hot swap for discrete heads."""
    if amount <= 0:
        return actions
    if is_continuous:
        a = torch.cat(list(actions), -1)
        a = torch.clip(torch.normal(a, amount), -1, 1)
        return (a,)
    out = []
    for act in actions:
        rand_idx = torch.randint(0, act.shape[-1], act.shape[:-1], device=act.device)
        rand_onehot = torch.nn.functional.one_hot(rand_idx, act.shape[-1]).to(act.dtype)
        swap = (torch.rand(act.shape[:1], device=act.device) < amount).view(-1, *([1] * (act.dim() - 1)))
        out.append(torch.where(swap, rand_onehot, act))
    return tuple(out)


class PlayerDV1(nn.Module):
    def __init__(self, encoder, rssm, actor, actions_dim, num_envs, stochastic_size, recurrent_state_size, device,
                 expl_amount: float = 0.0, expl_min: float = 0.0, expl_decay: float = 0.0):
        super().__init__()
        self.encoder = encoder
        self.rssm = rssm
        self.actor = actor
        self.actions_dim = list(actions_dim)
        self.num_envs = num_envs
        self.stochastic_size = stochastic_size
        self.recurrent_state_size = recurrent_state_size
        self.device = device
        self.expl_amount = expl_amount
        self.expl_min = expl_min
        self.expl_decay = expl_decay

    def _expl(self, step: int) -> float:
        amount = self.expl_amount
        if self.expl_decay:
            amount *= 0.5 ** (float(step) / self.expl_decay)
        return max(amount, self.expl_min)

    @torch.no_grad()
    def get_exploration_actions(self, obs, step: int = 0, mask=None):
        actions = self.get_actions(obs, greedy=False, mask=mask)
        amount = self._expl(step)
        if amount > 0:
            actions = add_exploration_noise(actions, amount, self.actor.is_continuous)
            self.actions = torch.cat(list(actions), -1).to(self.actions.dtype)
        return actions

    @torch.no_grad()
    def init_states(self, reset_envs: Optional[Sequence[int]] = None) -> None:
        dtype = next(self.rssm.parameters()).dtype
        if reset_envs is None or len(reset_envs) == 0:
            self.actions = torch.zeros(1, self.num_envs, int(np.sum(self.actions_dim)), device=self.device, dtype=dtype)
            self.recurrent_state = torch.zeros(1, self.num_envs, self.recurrent_state_size, device=self.device,
                                               dtype=dtype)
            self.stochastic_state = torch.zeros(1, self.num_envs, self.stochastic_size, device=self.device,
                                                dtype=dtype)
        else:
            self.actions[:, reset_envs] = 0.0
            self.recurrent_state[:, reset_envs] = 0.0
            self.stochastic_state[:, reset_envs] = 0.0

    @torch.no_grad()
    def get_actions(self, obs, greedy: bool = False, mask=None):
        embedded = self.encoder(obs)
        self.recurrent_state = self.rssm.recurrent_model(
            torch.cat((self.stochastic_state, self.actions), -1), self.recurrent_state
        )
        _, self.stochastic_state = self.rssm._representation(self.recurrent_state, embedded)
        actions, _ = self.actor(torch.cat((self.stochastic_state, self.recurrent_state), -1), greedy, mask)
        self.actions = torch.cat(actions, -1).to(self.stochastic_state.dtype)
        return actions


def build_agent(
    runtime: Runtime,
    actions_dim: Sequence[int],
    is_continuous: bool,
    cfg: Any,
    obs_space: spaces.Dict,
    world_model_state=None,
    actor_state=None,
    critic_state=None,
):
    wm_cfg = cfg.algo.world_model
    act = cfg.algo.get("dense_act", "elu")
    eps = 1e-3
    stochastic_size = wm_cfg.stochastic_size
    recurrent_state_size = wm_cfg.recurrent_model.recurrent_state_size
    latent_state_size = stochastic_size + recurrent_state_size
    cnn_keys = list(cfg.algo.cnn_keys.encoder or [])
    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    stages = int(np.log2(cfg.env.screen_size) - np.log2(4))

    cnn_encoder = (
        CNNEncoder(cnn_keys, [int(np.prod(obs_space[k].shape[:-2])) for k in cnn_keys],
                   tuple(obs_space[cnn_keys[0]].shape[-2:]), wm_cfg.encoder.cnn_channels_multiplier, eps, stages,
                   activation=act, layer_norm=False)
        if cnn_keys else None
    )
    mlp_encoder = (
        MLPEncoder(mlp_keys, [int(obs_space[k].shape[0]) for k in mlp_keys], wm_cfg.encoder.mlp_layers,
                   wm_cfg.encoder.dense_units, eps, symlog_inputs=False, activation=act, layer_norm=False)
        if mlp_keys else None
    )
    encoder = MultiEncoder(cnn_encoder, mlp_encoder)

    recurrent_model = RecurrentModel(
        int(sum(actions_dim)) + stochastic_size, recurrent_state_size, wm_cfg.recurrent_model.dense_units,
        eps, activation=act, layer_norm=True,
    )
    representation_model = MLP(
        encoder.output_dim + recurrent_state_size, stochastic_size * 2,
        [wm_cfg.representation_model.hidden_size], activation=act, layer_norm=False,
    )
    transition_model = MLP(
        recurrent_state_size, stochastic_size * 2, [wm_cfg.transition_model.hidden_size],
        activation=act, layer_norm=False,
    )
    rssm = RSSMV1(recurrent_model, representation_model, transition_model, min_std=wm_cfg.min_std)

    cnn_decoder = (
        CNNDecoder(cnn_keys, [int(np.prod(obs_space[k].shape[:-2])) for k in cnn_keys],
                   wm_cfg.observation_model.cnn_channels_multiplier, latent_state_size, cnn_encoder.output_dim,
                   tuple(obs_space[cnn_keys[0]].shape[-2:]), eps, stages, activation=act, layer_norm=False)
        if cnn_keys else None
    )
    mlp_decoder = (
        MLPDecoder(mlp_keys, [int(obs_space[k].shape[0]) for k in mlp_keys], latent_state_size,
                   wm_cfg.observation_model.mlp_layers, wm_cfg.observation_model.dense_units, eps,
                   activation=act, layer_norm=False)
        if mlp_keys else None
    )
    observation_model = MultiDecoder(cnn_decoder, mlp_decoder)
    reward_model = MLP(latent_state_size, 1, [wm_cfg.reward_model.dense_units] * wm_cfg.reward_model.mlp_layers,
                       activation=act, layer_norm=False)
    continue_model = None
    if cfg.algo.world_model.get("use_continues", False):
        dm = cfg.algo.world_model.get("discount_model", {})
        continue_model = MLP(latent_state_size, 1,
                             [dm.get("dense_units", 400)] * dm.get("mlp_layers", 4),
                             activation=act, layer_norm=False)
    world_model = WorldModelV1(encoder, rssm, observation_model, reward_model, continue_model)

    actor = Actor(
        latent_state_size=latent_state_size,
        actions_dim=actions_dim,
        is_continuous=is_continuous,
        distribution="tanh_normal" if is_continuous else "discrete",
        init_std=cfg.algo.actor.init_std,
        min_std=cfg.algo.actor.min_std,
        dense_units=cfg.algo.actor.dense_units,
        mlp_layers=cfg.algo.actor.mlp_layers,
        layer_norm_eps=eps,
        unimix=0.0,
        action_clip=1.0,
        activation=act,
        layer_norm=False,
    )
    critic = MLP(latent_state_size, 1, [cfg.algo.critic.dense_units] * cfg.algo.critic.mlp_layers,
                 activation=act, layer_norm=False)

    if world_model_state:
        world_model.load_state_dict(world_model_state)
    if actor_state:
        actor.load_state_dict(actor_state)
    if critic_state:
        critic.load_state_dict(critic_state)

    world_model = runtime.setup_module(world_model)
    actor = runtime.setup_module(actor)
    critic = runtime.setup_module(critic)
    player = PlayerDV1(world_model.encoder, world_model.rssm, actor, actions_dim, cfg.env.num_envs,
                       stochastic_size, recurrent_state_size, runtime.device,
                       expl_amount=cfg.algo.actor.get("expl_amount", 0.0),
                       expl_min=cfg.algo.actor.get("expl_min", 0.0),
                       expl_decay=cfg.algo.actor.get("expl_decay", 0.0))
    return world_model, actor, critic, player


def train(
    runtime: Runtime,
    world_model,
    actor,
    critic,
    world_optimizer,
    actor_optimizer,
    critic_optimizer,
    data: Dict[str, torch.Tensor],
    aggregator,
    cfg: Any,
) -> None:
    """Parity: dreamer_v1/dreamer_v1.py train :37 — dynamic learning, pure
    dynamics-backprop actor, Normal-regression critic."""
    batch_size = cfg.algo.per_rank_batch_size
    sequence_length = cfg.algo.per_rank_sequence_length
    recurrent_state_size = cfg.algo.world_model.recurrent_model.recurrent_state_size
    stochastic_size = cfg.algo.world_model.stochastic_size
    device = runtime.device
    dtype = runtime.param_dtype

    batch_obs = {k: ops.normalize_obs(data[k]).to(dtype) for k in cfg.algo.cnn_keys.encoder}
    batch_obs.update({k: data[k].to(dtype) for k in cfg.algo.mlp_keys.encoder})
    batch_actions = torch.cat((torch.zeros_like(data["actions"][:1]), data["actions"][:-1]), dim=0)

    recurrent_state = torch.zeros(1, batch_size, recurrent_state_size, device=device, dtype=dtype)
    stochastic_state = torch.zeros(1, batch_size, stochastic_size, device=device, dtype=dtype)
    recurrent_states = torch.empty(sequence_length, batch_size, recurrent_state_size, device=device, dtype=dtype)
    stochastic_states = torch.empty(sequence_length, batch_size, stochastic_size, device=device, dtype=dtype)
    post_means = torch.empty(sequence_length, batch_size, stochastic_size, device=device)
    post_stds = torch.empty(sequence_length, batch_size, stochastic_size, device=device)
    prior_means = torch.empty(sequence_length, batch_size, stochastic_size, device=device)
    prior_stds = torch.empty(sequence_length, batch_size, stochastic_size, device=device)

    embedded_obs = world_model.encoder(batch_obs)
    for i in range(sequence_length):
        recurrent_state, stochastic_state, _, post_ms, prior_ms = world_model.rssm.dynamic(
            stochastic_state, recurrent_state, batch_actions[i : i + 1], embedded_obs[i : i + 1]
        )
        recurrent_states[i] = recurrent_state
        stochastic_states[i] = stochastic_state
        post_means[i], post_stds[i] = post_ms[0], post_ms[1]
        prior_means[i], prior_stds[i] = prior_ms[0], prior_ms[1]
    latent_states = torch.cat((stochastic_states, recurrent_states), -1)

    decoded = world_model.observation_model(latent_states)
    po = {k: _FastUnitNormal(v.float(), len(v.shape[2:])) for k, v in decoded.items()}
    _rm_out = world_model.reward_model(latent_states).float()

    pr = _FastUnitNormal(_rm_out, 1)

    posteriors_dist = td.Independent(td.Normal(post_means, post_stds), 1)
    priors_dist = td.Independent(td.Normal(prior_means, prior_stds), 1)

    observation_loss = -sum(po[k].log_prob(batch_obs[k].float()).mean() for k in po)
    reward_loss = -pr.log_prob(data["rewards"]).mean()
    kl = td.kl_divergence(posteriors_dist, priors_dist).mean()
    state_loss = torch.clamp(kl, min=cfg.algo.world_model.kl_free_nats)
    continue_loss = torch.zeros((), device=device)
    if cfg.algo.world_model.get("use_continues", False) and world_model.continue_model is not None:
        qc_out = world_model.continue_model(latent_states).float()
        qc = td.Independent(td.Bernoulli(logits=qc_out), 1)
        continues_targets = (1 - data["terminated"]) * cfg.algo.gamma
        continue_loss = cfg.algo.world_model.get("continue_scale_factor", 1.0) * -qc.log_prob(
            continues_targets
        ).mean()
    rec_loss = (
        cfg.algo.world_model.kl_regularizer * state_loss + observation_loss + reward_loss + continue_loss
    )

    world_optimizer.zero_grad(set_to_none=True)
    runtime.backward(rec_loss)
    if cfg.algo.world_model.clip_gradients and cfg.algo.world_model.clip_gradients > 0:
        runtime.clip_gradients(world_model, world_optimizer, cfg.algo.world_model.clip_gradients)
    world_optimizer.step()

    # behaviour learning: imagination + dynamics-backprop actor
    horizon = cfg.algo.horizon
    flat = batch_size * sequence_length
    imagined_state = stochastic_states.detach().reshape(1, -1, stochastic_size)
    recurrent_state = recurrent_states.detach().reshape(1, -1, recurrent_state_size)
    imagined_latent_states = torch.empty(horizon, flat, stochastic_size + recurrent_state_size,
                                         device=device, dtype=dtype)
    for i in range(horizon):
        actions = torch.cat(actor(torch.cat((imagined_state, recurrent_state), -1))[0], dim=-1).to(dtype)
        imagined_state, recurrent_state = world_model.rssm.imagination(imagined_state, recurrent_state, actions)
        imagined_latent_states[i] = torch.cat((imagined_state, recurrent_state), -1)

    predicted_values = critic(imagined_latent_states).float()
    predicted_rewards = world_model.reward_model(imagined_latent_states).float()
    if cfg.algo.world_model.get("use_continues", False) and world_model.continue_model is not None:
        continues = torch.sigmoid(world_model.continue_model(imagined_latent_states).float()) * cfg.algo.gamma
    else:
        continues = torch.ones_like(predicted_rewards) * cfg.algo.gamma
    # λ-values with v_{t+1} alignment and bootstrap = last value
    next_values = torch.cat((predicted_values[1:], predicted_values[-1:]), dim=0)
    lambda_values = ops.lambda_values(predicted_rewards, next_values, continues, cfg.algo.lmbda)
    with torch.no_grad():
        discount = torch.cumprod(torch.cat((torch.ones_like(continues[:1]), continues[:-1]), 0), 0)

    actor_optimizer.zero_grad(set_to_none=True)
    policy_loss = -torch.mean(discount * lambda_values)
    runtime.backward(policy_loss)
    if cfg.algo.actor.clip_gradients and cfg.algo.actor.clip_gradients > 0:
        runtime.clip_gradients(actor, actor_optimizer, cfg.algo.actor.clip_gradients)
    actor_optimizer.step()

    _qv_out = critic(imagined_latent_states.detach()).float()


    qv = _FastUnitNormal(_qv_out, 1)
    critic_optimizer.zero_grad(set_to_none=True)
    value_loss = -torch.mean(discount[..., 0] * qv.log_prob(lambda_values.detach()))
    runtime.backward(value_loss)
    if cfg.algo.critic.clip_gradients and cfg.algo.critic.clip_gradients > 0:
        runtime.clip_gradients(critic, critic_optimizer, cfg.algo.critic.clip_gradients)
    critic_optimizer.step()

    if aggregator and not MetricAggregator.disabled:
        aggregator.update("Loss/world_model_loss", rec_loss.detach())
        aggregator.update("Loss/observation_loss", observation_loss.detach())
        aggregator.update("Loss/reward_loss", reward_loss.detach())
        aggregator.update("Loss/state_loss", state_loss.detach())
        aggregator.update("State/kl", kl.detach())
        aggregator.update("Loss/policy_loss", policy_loss.detach())
        aggregator.update("Loss/value_loss", value_loss.detach())


def _capture_train_step(
    runtime, world_model, actor, critic,
    world_optimizer, actor_optimizer, critic_optimizer, example_batch, cfg,
):
    """hipGraph-capture the DV1 gradient step; None on capture failure."""
    from sheeprl_amd.parallel.graphs import CUDAGraphStep
    from sheeprl_amd.utils.metric import MetricAggregator

    def train_fn(batch):
        was_disabled = MetricAggregator.disabled
        MetricAggregator.disabled = True
        try:
            train(runtime, world_model, actor, critic, world_optimizer, actor_optimizer,
                  critic_optimizer, batch, None, cfg)
        finally:
            MetricAggregator.disabled = was_disabled

    try:
        step = CUDAGraphStep(train_fn, example_batch, warmup=2)
        runtime.print("[dreamer_v1] gradient step captured in a hipGraph")
        return step
    except Exception as e:  # noqa: BLE001
        runtime.print(f"[dreamer_v1] hipGraph capture failed ({e}); eager training")
        return None


@register_algorithm(name="dreamer_v1")
def main(runtime: Runtime, cfg: Any) -> None:
    device = runtime.device
    log_dir = get_log_dir(runtime, cfg.root_dir, cfg.run_name)
    logger = get_logger(runtime, cfg, log_dir)
    runtime.logger = logger
    if runtime.is_global_zero:
        save_config(cfg, os.path.join(log_dir, "config.yaml"))

    envs = vectorize_env(cfg, cfg.seed, runtime.global_rank)
    obs_space = envs.single_observation_space
    action_space = envs.single_action_space
    is_continuous = isinstance(action_space, spaces.Box)
    is_multidiscrete = isinstance(action_space, spaces.MultiDiscrete)
    actions_dim = tuple(
        action_space.shape if is_continuous else (action_space.nvec.tolist() if is_multidiscrete else [action_space.n])
    )
    cnn_keys = list(cfg.algo.cnn_keys.encoder or [])
    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    obs_keys = cnn_keys + mlp_keys

    state: Dict[str, Any] = {}
    if cfg.checkpoint.resume_from:
        state = runtime.load(cfg.checkpoint.resume_from)

    world_model, actor, critic, player = build_agent(
        runtime, actions_dim, is_continuous, cfg, obs_space,
        state.get("world_model"), state.get("actor"), state.get("critic"),
    )
    world_optimizer = make_optimizer(world_model.parameters(), cfg.algo.world_model.optimizer)
    actor_optimizer = make_optimizer(actor.parameters(), cfg.algo.actor.optimizer)
    critic_optimizer = make_optimizer(critic.parameters(), cfg.algo.critic.optimizer)

    aggregator = MetricAggregator({k: "mean" for k in AGGREGATOR_KEYS})
    rb = EnvIndependentReplayBuffer(
        max(int(cfg.buffer.size), 1), n_envs=cfg.env.num_envs, obs_keys=obs_keys,
        buffer_cls=SequentialReplayBuffer,
    )

    world_size = runtime.world_size
    num_envs = cfg.env.num_envs
    policy_steps_per_iter = int(num_envs * world_size)
    total_iters = int(cfg.algo.total_steps // policy_steps_per_iter) if not cfg.dry_run else 1
    learning_starts = cfg.algo.learning_starts // policy_steps_per_iter if not cfg.dry_run else 0
    # replay-ratio accounting starts AFTER the prefill (parity:
    # sheeprl dreamer_v3.py:661, sac.py:301 — the reference subtracts the
    # prefill policy steps before asking Ratio how many grad steps are owed,
    # otherwise the first train iteration pays a learning_starts-sized backlog)
    prefill_steps = max(learning_starts - 1, 0) * policy_steps_per_iter
    policy_step = int(state.get("policy_step", 0))
    last_log = 0
    graphed_step = None
    _n_train_calls = 0
    want_graphs = (
        (runtime.device.type == "cuda" or os.environ.get("SHEEPRL_AMD_FORCE_GRAPHS") == "1")
        and cfg.algo.get("hip_graphs", True)
        and os.environ.get("SHEEPRL_AMD_NO_GRAPHS", "0") != "1"
    )
    last_checkpoint = 0
    ratio = Ratio(cfg.algo.replay_ratio, pretrain_steps=cfg.algo.per_rank_pretrain_steps)

    from sheeprl_amd.algos.dreamer_v3.utils import prepare_obs, test

    step_data: Dict[str, np.ndarray] = {}
    obs, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)
    for k in obs_keys:
        step_data[k] = np.asarray(obs[k])[np.newaxis]
    step_data["rewards"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["terminated"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["truncated"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    player.init_states()

    for iter_num in range(1, total_iters + 1):
        policy_step += policy_steps_per_iter
        with torch.inference_mode():
            with timer("Time/env_interaction_time"):
                if iter_num <= learning_starts and cfg.checkpoint.resume_from is None:
                    real_actions = actions = np.stack([envs.single_action_space.sample() for _ in range(num_envs)])
                    if not is_continuous:
                        actions = np.concatenate(
                            [
                                F.one_hot(torch.as_tensor(a).long(), d).numpy()
                                for a, d in zip(np.asarray(actions).reshape(num_envs, -1).T, actions_dim)
                            ],
                            axis=-1,
                        ).reshape(num_envs, -1)
                else:
                    torch_obs = prepare_obs(runtime, obs, cnn_keys=cnn_keys, num_envs=num_envs)
                    acts = player.get_exploration_actions(torch_obs, step=policy_step)
                    actions = torch.cat(acts, -1).view(num_envs, -1).float().cpu().numpy()
                    if is_continuous:
                        real_actions = actions
                    else:
                        real_actions = (
                            torch.stack([a.argmax(dim=-1) for a in acts], dim=-1).view(num_envs, -1).cpu().numpy()
                        )
                        if real_actions.shape[-1] == 1:
                            real_actions = real_actions[..., 0]
                step_data["actions"] = np.asarray(actions, np.float32).reshape(1, num_envs, -1)
                rb.add(step_data)
                next_obs, rewards, terminated, truncated, infos = envs.step(real_actions)
                dones = np.logical_or(terminated, truncated)

            for ep in infos.get("episode", []):
                if ep is not None:
                    aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                    aggregator.update("Game/ep_len_avg", float(ep["l"][0]))
            for k in obs_keys:
                step_data[k] = np.asarray(next_obs[k])[np.newaxis]
            obs = next_obs
            step_data["rewards"] = np.asarray(rewards, np.float32).reshape(1, num_envs, 1)
            step_data["terminated"] = np.asarray(terminated, np.float32).reshape(1, num_envs, 1)
            step_data["truncated"] = np.asarray(truncated, np.float32).reshape(1, num_envs, 1)
            dones_idxes = np.nonzero(dones)[0].tolist()
            if dones_idxes:
                # player states are inference tensors (created under the
                # action-selection inference_mode); reset them under it too
                with torch.inference_mode():
                    player.init_states(dones_idxes)

        rb_ready = any(len(b) >= cfg.algo.per_rank_sequence_length for b in rb.buffer)
        if iter_num >= learning_starts and rb_ready:
            n_steps = ratio((policy_step - prefill_steps) / world_size)
            if n_steps > 0:
                with timer("Time/train_time"):
                    for _ in range(n_steps):
                        sample = rb.sample_tensors(
                            cfg.algo.per_rank_batch_size,
                            sequence_length=cfg.algo.per_rank_sequence_length,
                            n_samples=1,
                            device=device,
                        )
                        batch = {k: v[0].to(device) for k, v in sample.items()}
                        _n_train_calls += 1
                        if graphed_step is not None and _n_train_calls % 64 != 0:  # every 64th step runs eager to feed metrics (~14x a replay; DV3 feeds metrics from the capture's static buffers instead)
                            graphed_step(batch)
                        else:
                            train(runtime, world_model, actor, critic, world_optimizer, actor_optimizer,
                                  critic_optimizer, batch, aggregator, cfg)
                        if graphed_step is None and want_graphs and _n_train_calls >= 3:
                            graphed_step = _capture_train_step(
                                runtime, world_model, actor, critic,
                                world_optimizer, actor_optimizer, critic_optimizer, batch, cfg,
                            )
                            if graphed_step is None:
                                want_graphs = False

        if policy_step - last_log >= cfg.metric.log_every or iter_num == total_iters or cfg.dry_run:
            runtime.log_dict(aggregator.compute(), policy_step)
            aggregator.reset()
            timer.reset()
            last_log = policy_step

        if (
            cfg.checkpoint.every > 0
            and policy_step - last_checkpoint >= cfg.checkpoint.every
            or cfg.dry_run
            or (iter_num == total_iters and cfg.checkpoint.save_last)
        ):
            last_checkpoint = policy_step
            ckpt_path = os.path.join(log_dir, "checkpoint", f"ckpt_{policy_step}_{runtime.global_rank}.ckpt")
            runtime.call(
                "on_checkpoint_coupled",
                ckpt_path=ckpt_path,
                state={
                    "world_model": world_model,
                    "actor": actor,
                    "critic": critic,
                    "world_optimizer": world_optimizer,
                    "actor_optimizer": actor_optimizer,
                    "critic_optimizer": critic_optimizer,
                    "ratio": ratio,
                    "policy_step": policy_step,
                    "batch_size": cfg.algo.per_rank_batch_size * world_size,
                },
                replay_buffer=rb if cfg.buffer.get("checkpoint", False) else None,
            )

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        reward = test(player, runtime, make_env(cfg, cfg.seed, 0), cfg, log_dir)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


@register_evaluation(algorithms=["dreamer_v1"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    from sheeprl_amd.algos.dreamer_v3.utils import test

    env_fn = make_env(cfg, cfg.seed, 0)
    env = env_fn()
    obs_space, action_space = env.observation_space, env.action_space
    env.close()
    is_continuous = isinstance(action_space, spaces.Box)
    is_multidiscrete = isinstance(action_space, spaces.MultiDiscrete)
    actions_dim = tuple(
        action_space.shape if is_continuous else (action_space.nvec.tolist() if is_multidiscrete else [action_space.n])
    )
    _, _, _, player = build_agent(
        runtime, actions_dim, is_continuous, cfg, obs_space,
        state["world_model"], state["actor"], state["critic"],
    )
    reward = test(player, runtime, env_fn, cfg)
    runtime.print(f"Test/cumulative_reward: {reward}")
    return reward
