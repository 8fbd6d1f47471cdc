"""Dreamer-V3 agent: encoder/decoder, RSSM, actor, critic, player.

Parity surface: sheeprl/algos/dreamer_v3/agent.py — CNNEncoder :42,
MLPEncoder :102, CNNDecoder :154, MLPDecoder :230, RecurrentModel :281,
RSSM :344 (dynamic :398, _representation :458, _transition :474,
imagination :487), PlayerDV3 :596, Actor :694, build_agent :935 (Hafner init
application :1168-1180, player weight tying :1229-1235).

MI355X design: every Linear+LN+SiLU triple is the fused DenseBlock; the GRU
cell's post-GEMM math is the fused ``ops.gru_gates`` kernel; the stochastic
state sampling stays in fp32 for categorical stability under bf16-true.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.distributions as td
import torch.nn.functional as F
from torch import Tensor, nn

from sheeprl_amd import ops
from sheeprl_amd.distributions import OneHotCategoricalST, unimix_logits
from sheeprl_amd.envs import spaces
from sheeprl_amd.models import (
    FastLinear,
    CNN,
    DeCNN,
    DenseBlock,
    LayerNormGRUCell,
    MLP,
    MultiDecoder,
    MultiEncoder,
    cnn_forward,
)
from sheeprl_amd.parallel import Runtime


# ---------------------------------------------------------------------------
# init (Hafner) — parity: dreamer_v3/utils.py:143-187
# ---------------------------------------------------------------------------

def init_weights(m: nn.Module) -> None:
    if isinstance(m, nn.Linear):
        denoms = (m.in_features + m.out_features) / 2.0
        std = np.sqrt(1.0 / denoms) / 0.87962566103423978
        nn.init.trunc_normal_(m.weight.data, mean=0.0, std=std, a=-2.0 * std, b=2.0 * std)
        if m.bias is not None:
            m.bias.data.fill_(0.0)
    elif isinstance(m, (nn.Conv2d, nn.ConvTranspose2d)):
        space = m.kernel_size[0] * m.kernel_size[1]
        in_num = space * m.in_channels
        out_num = space * m.out_channels
        denoms = (in_num + out_num) / 2.0
        std = np.sqrt(1.0 / denoms) / 0.87962566103423978
        nn.init.trunc_normal_(m.weight.data, mean=0.0, std=std, a=-2.0 * std, b=2.0 * std)
        if m.bias is not None:
            m.bias.data.fill_(0.0)


def uniform_init_weights(given_scale: float):
    def f(m: nn.Module) -> None:
        if isinstance(m, nn.Linear):
            denoms = (m.in_features + m.out_features) / 2.0
            limit = np.sqrt(3 * given_scale / denoms)
            nn.init.uniform_(m.weight.data, a=-limit, b=limit)
            if m.bias is not None:
                m.bias.data.fill_(0.0)
        elif isinstance(m, (nn.Conv2d, nn.ConvTranspose2d)):
            space = m.kernel_size[0] * m.kernel_size[1]
            denoms = (space * m.in_channels + space * m.out_channels) / 2.0
            limit = np.sqrt(3 * given_scale / denoms)
            nn.init.uniform_(m.weight.data, a=-limit, b=limit)
            if m.bias is not None:
                m.bias.data.fill_(0.0)

    return f


def compute_stochastic_state(logits: Tensor, discrete: int = 32, sample: bool = True) -> Tensor:
    """One-hot straight-through sample of the [*, stoch, discrete] categorical
    (parity: dreamer_v2/utils.py:44)."""
    logits = logits.view(*logits.shape[:-1], -1, discrete)
    dist = td.Independent(OneHotCategoricalST(logits=logits.float()), 1)
    out = dist.rsample() if sample else dist.mode
    return out.to(logits.dtype) if out.dtype != logits.dtype else out


# ---------------------------------------------------------------------------
# encoders / decoders
# ---------------------------------------------------------------------------

class CNNEncoder(nn.Module):
    def __init__(
        self,
        keys: Sequence[str],
        input_channels: Sequence[int],
        image_size: Tuple[int, int],
        channels_multiplier: int,
        layer_norm_eps: float = 1e-3,
        stages: int = 4,
        activation: str = "silu",
        layer_norm: bool = True,
    ) -> None:
        super().__init__()
        self.keys = list(keys)
        self.input_dim = (sum(input_channels), *image_size)
        self.model = nn.Sequential(
            CNN(
                in_channels=self.input_dim[0],
                hidden_channels=[(2**i) * channels_multiplier for i in range(stages)],
                kernel_sizes=[4] * stages,
                strides=[2] * stages,
                paddings=[1] * stages,
                activation=activation,
                layer_norm=layer_norm,
                layer_norm_eps=layer_norm_eps,
            ),
            nn.Flatten(-3, -1),
        )
        with torch.no_grad():
            self.output_dim = self.model(torch.zeros(1, *self.input_dim)).shape[-1]

    def forward(self, obs: Dict[str, Tensor]) -> Tensor:
        x = torch.cat([obs[k] for k in self.keys], dim=-3)
        return cnn_forward(self.model, x, x.shape[-3:], flatten=False)


class MLPEncoder(nn.Module):
    def __init__(
        self,
        keys: Sequence[str],
        input_dims: Sequence[int],
        mlp_layers: int = 4,
        dense_units: int = 512,
        layer_norm_eps: float = 1e-3,
        symlog_inputs: bool = True,
        activation: str = "silu",
        layer_norm: bool = True,
    ) -> None:
        super().__init__()
        self.keys = list(keys)
        self.input_dim = sum(input_dims)
        self.model = MLP(
            self.input_dim,
            None,
            [dense_units] * mlp_layers,
            activation=activation,
            layer_norm=layer_norm,
            layer_norm_eps=layer_norm_eps,
        )
        self.output_dim = dense_units
        self.symlog_inputs = symlog_inputs

    def forward(self, obs: Dict[str, Tensor]) -> Tensor:
        x = torch.cat([ops.symlog(obs[k]) if self.symlog_inputs else obs[k] for k in self.keys], -1)
        return self.model(x)


class CNNDecoder(nn.Module):
    def __init__(
        self,
        keys: Sequence[str],
        output_channels: Sequence[int],
        channels_multiplier: int,
        latent_state_size: int,
        cnn_encoder_output_dim: int,
        image_size: Tuple[int, int],
        layer_norm_eps: float = 1e-3,
        stages: int = 4,
        activation: str = "silu",
        layer_norm: bool = True,
    ) -> None:
        super().__init__()
        self.keys = list(keys)
        self.output_channels = list(output_channels)
        self.output_dim = (sum(output_channels), *image_size)
        self.model = nn.Sequential(
            FastLinear(latent_state_size, cnn_encoder_output_dim),
            nn.Unflatten(1, (-1, 4, 4)),
            DeCNN(
                in_channels=(2 ** (stages - 1)) * channels_multiplier,
                hidden_channels=[(2**i) * channels_multiplier for i in reversed(range(stages - 1))]
                + [self.output_dim[0]],
                kernel_sizes=[4] * stages,
                strides=[2] * stages,
                paddings=[1] * stages,
                activation=activation,
                layer_norm=layer_norm,
                layer_norm_eps=layer_norm_eps,
                last_layer_plain=True,
            ),
        )

    def forward(self, latent_states: Tensor) -> Dict[str, Tensor]:
        out = cnn_forward(self.model, latent_states, (latent_states.shape[-1],), flatten=False)
        return {k: rec for k, rec in zip(self.keys, torch.split(out, self.output_channels, -3))}


class MLPDecoder(nn.Module):
    def __init__(
        self,
        keys: Sequence[str],
        output_dims: Sequence[int],
        latent_state_size: int,
        mlp_layers: int = 4,
        dense_units: int = 512,
        layer_norm_eps: float = 1e-3,
        activation: str = "silu",
        layer_norm: bool = True,
    ) -> None:
        super().__init__()
        self.keys = list(keys)
        self.model = MLP(
            latent_state_size,
            None,
            [dense_units] * mlp_layers,
            activation=activation,
            layer_norm=layer_norm,
            layer_norm_eps=layer_norm_eps,
        )
        self.heads = nn.ModuleList([FastLinear(dense_units, dim) for dim in output_dims])

    def forward(self, latent_states: Tensor) -> Dict[str, Tensor]:
        x = self.model(latent_states)
        return {k: h(x) for k, h in zip(self.keys, self.heads)}


# ---------------------------------------------------------------------------
# RSSM
# ---------------------------------------------------------------------------

class RecurrentModel(nn.Module):
    def __init__(
        self,
        input_size: int,
        recurrent_state_size: int,
        dense_units: int,
        layer_norm_eps: float = 1e-3,
        activation: str = "silu",
        layer_norm: bool = True,
    ) -> None:
        super().__init__()
        self.mlp = DenseBlock(
            input_size, dense_units, bias=not layer_norm, layer_norm=layer_norm,
            layer_norm_eps=layer_norm_eps, activation=activation
        )
        self.rnn = LayerNormGRUCell(dense_units, recurrent_state_size, bias=False, layer_norm=True,
                                    layer_norm_eps=layer_norm_eps)
        self.recurrent_state_size = recurrent_state_size

    def forward(self, input: Tensor, recurrent_state: Tensor) -> Tensor:
        return self.rnn(self.mlp(input), recurrent_state)


class RSSM(nn.Module):
    def __init__(
        self,
        recurrent_model: RecurrentModel,
        representation_model: MLP,
        transition_model: MLP,
        discrete: int = 32,
        unimix: float = 0.01,
        learnable_initial_recurrent_state: bool = True,
    ) -> None:
        super().__init__()
        self.recurrent_model = recurrent_model
        self.representation_model = representation_model
        self.transition_model = transition_model
        self.discrete = discrete
        self.unimix = unimix
        init = torch.zeros(recurrent_model.recurrent_state_size, dtype=torch.float32)
        if learnable_initial_recurrent_state:
            self.initial_recurrent_state = nn.Parameter(init)
        else:
            self.register_buffer("initial_recurrent_state", init)

    @property
    def _dtype(self) -> torch.dtype:
        return next(self.recurrent_model.parameters()).dtype

    def get_initial_states(self, batch_shape: Sequence[int]) -> Tuple[Tensor, Tensor]:
        initial_recurrent_state = torch.tanh(self.initial_recurrent_state).expand(*batch_shape, -1)
        initial_posterior = self._transition(initial_recurrent_state, sample_state=False)[1]
        return initial_recurrent_state, initial_posterior

    def dynamic(
        self, posterior: Tensor, recurrent_state: Tensor, action: Tensor, embedded_obs: Tensor, is_first: Tensor
    ) -> Tuple[Tensor, Tensor, Tensor, Tensor, Tensor]:
        dt = self._dtype
        is_first = is_first.to(dt)
        action = (1 - is_first) * action.to(dt)
        initial_recurrent_state, initial_posterior = self.get_initial_states(recurrent_state.shape[:2])
        recurrent_state = (1 - is_first) * recurrent_state.to(dt) + is_first * initial_recurrent_state.to(dt)
        posterior = posterior.view(*posterior.shape[:-2], -1).to(dt)
        posterior = (1 - is_first) * posterior + is_first * initial_posterior.view_as(posterior).to(dt)
        recurrent_state = self.recurrent_model(torch.cat((posterior, action), -1), recurrent_state)
        prior_logits, prior = self._transition(recurrent_state)
        posterior_logits, posterior = self._representation(recurrent_state, embedded_obs)
        return recurrent_state, posterior, prior, posterior_logits, prior_logits

    def dynamic_posterior(
        self,
        posterior: Tensor,
        recurrent_state: Tensor,
        action: Tensor,
        embedded_obs: Tensor,
        is_first: Tensor,
        initial_states: Optional[Tuple[Tensor, Tensor]] = None,
    ) -> Tuple[Tensor, Tensor, Tensor]:
        """Scan step WITHOUT the transition head.

        In training the prior is only consumed by the KL loss (the recurrence
        feeds back the posterior), so the transition MLP runs once batched
        over all T afterwards (``transition_logits``) instead of per step.
        ``initial_states`` lets the caller hoist the (parameter-dependent but
        step-invariant) initial-state computation out of the scan — together
        this removes two MLP+categorical-head evaluations per scan step.
        """
        dt = self._dtype
        is_first = is_first.to(dt)
        if initial_states is None:
            initial_recurrent_state, initial_posterior = self.get_initial_states(recurrent_state.shape[:2])
        else:
            initial_recurrent_state, initial_posterior = initial_states
        # fused per-row reset masking (one kernel per tensor instead of ~4)
        action = ops.masked_lerp(action.to(dt), None, is_first)
        recurrent_state = ops.masked_lerp(recurrent_state.to(dt), initial_recurrent_state.to(dt), is_first)
        posterior = posterior.view(*posterior.shape[:-2], -1).to(dt)
        posterior = ops.masked_lerp(
            posterior, initial_posterior.reshape(*posterior.shape[:-1], -1).to(dt), is_first
        )
        recurrent_state = self.recurrent_model(torch.cat((posterior, action), -1), recurrent_state)
        posterior_logits, posterior = self._representation(recurrent_state, embedded_obs)
        return recurrent_state, posterior, posterior_logits

    def transition_logits(self, recurrent_states: Tensor) -> Tensor:
        """Batched prior logits over a whole [T, B, H] stack."""
        raw = self.transition_model(recurrent_states.to(self._dtype))
        m, _ = self._stoch_head(raw, sample=False)
        return m

    def _stoch_head(self, raw_logits: Tensor, sample: bool) -> Tuple[Tensor, Tensor]:
        """Fused unimix + log-probs + one-hot-ST sampling (ops.categorical_st).

        Returns (mixed log-prob logits [*, S*K] fp32, sample [*, S, K])."""
        shape = raw_logits.shape
        raw4 = raw_logits.view(*shape[:-1], -1, self.discrete)
        m, onehot = ops.categorical_st(raw4, self.unimix, sample=sample)
        return m.reshape(*shape), onehot.to(self._dtype)

    def _uniform_mix(self, logits: Tensor) -> Tensor:
        dim = logits.dim()
        if dim == 3:
            logits = logits.view(*logits.shape[:-1], -1, self.discrete)
        elif dim != 4:
            raise RuntimeError(f"expected 3D or 4D logits, got {dim}D")
        logits = unimix_logits(logits.float(), self.unimix)
        return logits.view(*logits.shape[:-2], -1)

    def _representation(self, recurrent_state: Tensor, embedded_obs: Tensor) -> Tuple[Tensor, Tensor]:
        dt = self._dtype
        raw = self.representation_model(torch.cat((recurrent_state.to(dt), embedded_obs.to(dt)), -1))
        return self._stoch_head(raw, sample=True)

    def _transition(self, recurrent_out: Tensor, sample_state: bool = True) -> Tuple[Tensor, Tensor]:
        raw = self.transition_model(recurrent_out.to(self._dtype))
        return self._stoch_head(raw, sample=sample_state)

    def imagination(self, prior: Tensor, recurrent_state: Tensor, actions: Tensor) -> Tuple[Tensor, Tensor]:
        dt = self._dtype
        recurrent_state = self.recurrent_model(torch.cat((prior.to(dt), actions.to(dt)), -1), recurrent_state.to(dt))
        _, imagined_prior = self._transition(recurrent_state)
        return imagined_prior, recurrent_state


class DecoupledRSSM(RSSM):
    """Posterior computed for all timesteps in parallel from embeddings only
    (parity: dreamer_v3/agent.py:501-593) — the sequence-parallel seam: only
    the cheap gate recurrence stays sequential."""

    def dynamic(  # type: ignore[override]
        self, posterior: Tensor, recurrent_state: Tensor, action: Tensor, is_first: Tensor
    ) -> Tuple[Tensor, Tensor, Tensor]:
        dt = self._dtype
        is_first = is_first.to(dt)
        action = (1 - is_first) * action.to(dt)
        initial_recurrent_state, initial_posterior = self.get_initial_states(recurrent_state.shape[:2])
        recurrent_state = (1 - is_first) * recurrent_state.to(dt) + is_first * initial_recurrent_state.to(dt)
        posterior = posterior.view(*posterior.shape[:-2], -1).to(dt)
        posterior = (1 - is_first) * posterior + is_first * initial_posterior.view_as(posterior).to(dt)
        recurrent_state = self.recurrent_model(torch.cat((posterior, action), -1), recurrent_state)
        prior_logits, _ = self._transition(recurrent_state)
        return recurrent_state, None, prior_logits

    def _representation(self, embedded_obs: Tensor) -> Tuple[Tensor, Tensor]:  # type: ignore[override]
        raw = self.representation_model(embedded_obs.to(self._dtype))
        return self._stoch_head(raw, sample=True)


class WorldModel(nn.Module):
    def __init__(
        self,
        encoder: MultiEncoder,
        rssm: RSSM,
        observation_model: MultiDecoder,
        reward_model: MLP,
        continue_model: MLP,
    ) -> None:
        super().__init__()
        self.encoder = encoder
        self.rssm = rssm
        self.observation_model = observation_model
        self.reward_model = reward_model
        self.continue_model = continue_model


# ---------------------------------------------------------------------------
# actor
# ---------------------------------------------------------------------------

class Actor(nn.Module):
    def __init__(
        self,
        latent_state_size: int,
        actions_dim: Sequence[int],
        is_continuous: bool,
        distribution: str = "auto",
        init_std: float = 2.0,
        min_std: float = 0.1,
        max_std: float = 1.0,
        dense_units: int = 1024,
        mlp_layers: int = 5,
        layer_norm_eps: float = 1e-3,
        unimix: float = 0.01,
        action_clip: float = 1.0,
        activation: str = "silu",
        layer_norm: bool = True,
    ) -> None:
        super().__init__()
        self.distribution = distribution.lower()
        if self.distribution not in ("auto", "normal", "tanh_normal", "discrete", "scaled_normal"):
            raise ValueError(f"unknown actor distribution '{distribution}'")
        if self.distribution == "auto":
            self.distribution = "scaled_normal" if is_continuous else "discrete"
        self.model = MLP(
            latent_state_size,
            None,
            [dense_units] * mlp_layers,
            activation=activation,
            layer_norm=layer_norm,
            layer_norm_eps=layer_norm_eps,
        )
        if is_continuous:
            self.mlp_heads = nn.ModuleList([FastLinear(dense_units, int(sum(actions_dim)) * 2)])
        else:
            self.mlp_heads = nn.ModuleList([FastLinear(dense_units, d) for d in actions_dim])
        self.actions_dim = list(actions_dim)
        self.is_continuous = is_continuous
        self.init_std = init_std
        self.min_std = min_std
        self.max_std = max_std
        self._unimix = unimix
        self._action_clip = action_clip

    def forward(
        self, state: Tensor, greedy: bool = False, mask: Optional[Dict[str, Tensor]] = None
    ) -> Tuple[Tuple[Tensor, ...], Tuple[td.Distribution, ...]]:
        out = self.model(state)
        pre_dist = [head(out) for head in self.mlp_heads]
        if self.is_continuous:
            mean, std = torch.chunk(pre_dist[0].float(), 2, -1)
            if self.distribution == "tanh_normal":
                mean = 5 * torch.tanh(mean / 5)
                std = F.softplus(std + self.init_std) + self.min_std
                dist = td.Independent(
                    td.TransformedDistribution(td.Normal(mean, std), td.TanhTransform()), 1
                )
            elif self.distribution == "normal":
                dist = td.Independent(td.Normal(mean, std), 1)
            else:  # scaled_normal
                std = (self.max_std - self.min_std) * torch.sigmoid(std + self.init_std) + self.min_std
                dist = td.Independent(td.Normal(torch.tanh(mean), std), 1)
            if not greedy:
                actions = dist.rsample()
            else:
                sample = dist.sample((100,))
                log_prob = dist.log_prob(sample)
                actions = sample[log_prob.argmax(0)].view(1, 1, -1)
            if self._action_clip > 0.0:
                clip = torch.full_like(actions, self._action_clip)
                actions = actions * (clip / torch.maximum(clip, torch.abs(actions))).detach()
            return (actions,), (dist,)
        actions_list: List[Tensor] = []
        dists: List[Any] = []
        for logits in pre_dist:
            # fused unimix + sample + ST (ops.categorical_st); the dist wraps
            # the normalized log-probs for log_prob/entropy in the actor loss
            m, onehot = ops.categorical_st(logits, self._unimix, sample=not greedy)
            from sheeprl_amd.distributions import LogProbCategorical

            dists.append(LogProbCategorical(m))
            actions_list.append(onehot)
        return tuple(actions_list), tuple(dists)


class MinedojoActor(Actor):
    """Action-masked actor for MineDojo (reference dreamer_v3/agent.py:848-933).

    Head 0 (action type) is masked by ``mask["mask_action_type"]``; the
    craft-argument head is masked per sample where the sampled action type is
    15 (craft); the equip/place/destroy-argument head where the type is 16/17
    (equip/place, ``mask_equip_place``) or 18 (destroy, ``mask_destroy``).
    Masks apply AFTER the unimix transform, as -inf logits (probability 0),
    vectorized over [T, B] instead of the reference's python loops.
    """

    def forward(
        self, state: Tensor, greedy: bool = False, mask: Optional[Dict[str, Tensor]] = None
    ) -> Tuple[Tuple[Tensor, ...], Tuple[Any, ...]]:
        out = self.model(state)
        actions_list: List[Tensor] = []
        dists: List[Any] = []
        functional_action: Optional[Tensor] = None
        neg_inf = -torch.inf
        for i, head in enumerate(self.mlp_heads):
            logits = head(out).float()
            K = logits.shape[-1]
            mixed = torch.log((1 - self._unimix) * torch.softmax(logits, -1) + self._unimix / K)
            if mask is not None:
                if i == 0:
                    mixed = torch.where(mask["mask_action_type"].expand_as(mixed), mixed, neg_inf)
                elif i == 1:
                    sel = (functional_action == 15).unsqueeze(-1)  # craft
                    blocked = sel & ~mask["mask_craft_smelt"].expand_as(mixed)
                    mixed = torch.where(blocked, neg_inf, mixed)
                elif i == 2:
                    sel_ep = ((functional_action == 16) | (functional_action == 17)).unsqueeze(-1)
                    sel_d = (functional_action == 18).unsqueeze(-1)
                    blocked = (sel_ep & ~mask["mask_equip_place"].expand_as(mixed)) | (
                        sel_d & ~mask["mask_destroy"].expand_as(mixed)
                    )
                    mixed = torch.where(blocked, neg_inf, mixed)
            dist = OneHotCategoricalST(logits=mixed)
            dists.append(dist)
            actions_list.append(dist.mode if greedy else dist.rsample())
            if functional_action is None:
                functional_action = actions_list[0].argmax(dim=-1)
        return tuple(actions_list), tuple(dists)


# ---------------------------------------------------------------------------
# player
# ---------------------------------------------------------------------------

class PlayerDV3(nn.Module):
    def __init__(
        self,
        encoder: MultiEncoder,
        rssm: RSSM,
        actor: Actor,
        actions_dim: Sequence[int],
        num_envs: int,
        stochastic_size: int,
        recurrent_state_size: int,
        device: torch.device,
        discrete_size: int = 32,
        actor_type: Optional[str] = None,
    ) -> None:
        super().__init__()
        self.encoder = encoder
        self.rssm = rssm
        self.actor = actor
        self.actions_dim = list(actions_dim)
        self.num_envs = num_envs
        self.stochastic_size = stochastic_size
        self.recurrent_state_size = recurrent_state_size
        self.device = device
        self.discrete_size = discrete_size
        self.actor_type = actor_type
        self.decoupled_rssm = isinstance(rssm, DecoupledRSSM)

    @torch.no_grad()
    def set_exploration(self, amount: float = 0.0, minimum: float = 0.0, decay: float = 0.0) -> None:
        """Configure DV1/DV2-style exploration noise for
        :meth:`get_exploration_actions` (reference dreamer_v2/agent.py:663)."""
        self._expl_amount = amount
        self._expl_min = minimum
        self._expl_decay = decay

    def _expl(self, step: int) -> float:
        amount = getattr(self, "_expl_amount", 0.0)
        if getattr(self, "_expl_decay", 0.0):
            amount *= 0.5 ** (float(step) / self._expl_decay)
        return max(amount, getattr(self, "_expl_min", 0.0))

    @torch.no_grad()
    def get_exploration_actions(self, obs, step: int = 0, mask=None):
        from sheeprl_amd.algos.dreamer_v1.dreamer_v1 import add_exploration_noise

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
            self.recurrent_state, stochastic_state = self.rssm.get_initial_states((1, self.num_envs))
            self.recurrent_state = self.recurrent_state.contiguous().to(dtype)
            self.stochastic_state = stochastic_state.reshape(1, self.num_envs, -1).to(dtype)
        else:
            self.actions[:, reset_envs] = 0.0
            rec, stoch = self.rssm.get_initial_states((1, len(reset_envs)))
            self.recurrent_state[:, reset_envs] = rec.to(self.recurrent_state.dtype)
            self.stochastic_state[:, reset_envs] = stoch.reshape(1, len(reset_envs), -1).to(self.stochastic_state.dtype)

    @torch.no_grad()
    def get_actions(
        self, obs: Dict[str, Tensor], greedy: bool = False, mask: Optional[Dict[str, Tensor]] = None
    ) -> Sequence[Tensor]:
        embedded_obs = self.encoder(obs)
        self.recurrent_state = self.rssm.recurrent_model(
            torch.cat((self.stochastic_state, self.actions), -1), self.recurrent_state
        )
        if self.decoupled_rssm:
            _, stoch = self.rssm._representation(embedded_obs)
        else:
            _, stoch = self.rssm._representation(self.recurrent_state, embedded_obs)
        self.stochastic_state = stoch.view(*stoch.shape[:-2], self.stochastic_size * self.discrete_size)
        actions, _ = self.actor(torch.cat((self.stochastic_state, self.recurrent_state), -1), greedy, mask)
        self.actions = torch.cat(actions, -1).to(self.stochastic_state.dtype)
        return actions


# ---------------------------------------------------------------------------
# build
# ---------------------------------------------------------------------------

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
    actor_cfg = cfg.algo.actor
    critic_cfg = cfg.algo.critic
    eps = 1e-3

    recurrent_state_size = wm_cfg.recurrent_model.recurrent_state_size
    stochastic_size = wm_cfg.stochastic_size * wm_cfg.discrete_size
    latent_state_size = stochastic_size + recurrent_state_size
    cnn_keys = list(cfg.algo.cnn_keys.encoder or [])
    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    stages = int(np.log2(cfg.env.screen_size) - np.log2(4))

    cnn_encoder = (
        CNNEncoder(
            keys=cnn_keys,
            input_channels=[int(np.prod(obs_space[k].shape[:-2])) for k in cnn_keys],
            image_size=tuple(obs_space[cnn_keys[0]].shape[-2:]),
            channels_multiplier=wm_cfg.encoder.cnn_channels_multiplier,
            layer_norm_eps=eps,
            stages=stages,
        )
        if cnn_keys
        else None
    )
    mlp_encoder = (
        MLPEncoder(
            keys=mlp_keys,
            input_dims=[int(obs_space[k].shape[0]) for k in mlp_keys],
            mlp_layers=wm_cfg.encoder.mlp_layers,
            dense_units=wm_cfg.encoder.dense_units,
            layer_norm_eps=eps,
        )
        if mlp_keys
        else None
    )
    encoder = MultiEncoder(cnn_encoder, mlp_encoder)

    recurrent_model = RecurrentModel(
        input_size=int(sum(actions_dim) + stochastic_size),
        recurrent_state_size=recurrent_state_size,
        dense_units=wm_cfg.recurrent_model.dense_units,
        layer_norm_eps=eps,
    )
    repr_in = encoder.output_dim + (0 if wm_cfg.decoupled_rssm else recurrent_state_size)
    representation_model = MLP(
        repr_in,
        stochastic_size,
        [wm_cfg.representation_model.hidden_size],
        activation="silu",
        layer_norm=True,
        layer_norm_eps=eps,
    )
    transition_model = MLP(
        recurrent_state_size,
        stochastic_size,
        [wm_cfg.transition_model.hidden_size],
        activation="silu",
        layer_norm=True,
        layer_norm_eps=eps,
    )
    rssm_cls = DecoupledRSSM if wm_cfg.decoupled_rssm else RSSM
    recurrent_model.apply(init_weights)
    representation_model.apply(init_weights)
    transition_model.apply(init_weights)
    rssm = rssm_cls(
        recurrent_model=recurrent_model,
        representation_model=representation_model,
        transition_model=transition_model,
        discrete=wm_cfg.discrete_size,
        unimix=cfg.algo.unimix,
        learnable_initial_recurrent_state=wm_cfg.learnable_initial_recurrent_state,
    )

    cnn_decoder = (
        CNNDecoder(
            keys=list(cfg.algo.cnn_keys.decoder or cnn_keys),
            output_channels=[int(np.prod(obs_space[k].shape[:-2])) for k in cnn_keys],
            channels_multiplier=wm_cfg.observation_model.cnn_channels_multiplier,
            latent_state_size=latent_state_size,
            cnn_encoder_output_dim=cnn_encoder.output_dim,
            image_size=tuple(obs_space[cnn_keys[0]].shape[-2:]),
            layer_norm_eps=eps,
            stages=stages,
        )
        if cnn_keys
        else None
    )
    mlp_decoder = (
        MLPDecoder(
            keys=list(cfg.algo.mlp_keys.decoder or mlp_keys),
            output_dims=[int(obs_space[k].shape[0]) for k in mlp_keys],
            latent_state_size=latent_state_size,
            mlp_layers=wm_cfg.observation_model.mlp_layers,
            dense_units=wm_cfg.observation_model.dense_units,
            layer_norm_eps=eps,
        )
        if mlp_keys
        else None
    )
    observation_model = MultiDecoder(cnn_decoder, mlp_decoder)

    reward_model = MLP(
        latent_state_size,
        wm_cfg.reward_model.bins,
        [wm_cfg.reward_model.dense_units] * wm_cfg.reward_model.mlp_layers,
        activation="silu",
        layer_norm=True,
        layer_norm_eps=eps,
    )
    continue_model = MLP(
        latent_state_size,
        1,
        [wm_cfg.discount_model.dense_units] * wm_cfg.discount_model.mlp_layers,
        activation="silu",
        layer_norm=True,
        layer_norm_eps=eps,
    )
    world_model = WorldModel(encoder, rssm, observation_model, reward_model, continue_model)

    actor_cls = MinedojoActor if str(cfg.env.id).startswith("minedojo") else Actor
    actor = actor_cls(
        latent_state_size=latent_state_size,
        actions_dim=actions_dim,
        is_continuous=is_continuous,
        distribution=cfg.distribution.get("type", "auto"),
        init_std=actor_cfg.init_std,
        min_std=actor_cfg.min_std,
        max_std=actor_cfg.max_std,
        dense_units=actor_cfg.dense_units,
        mlp_layers=actor_cfg.mlp_layers,
        layer_norm_eps=eps,
        unimix=cfg.algo.unimix,
        action_clip=actor_cfg.action_clip,
    )
    critic = MLP(
        latent_state_size,
        critic_cfg.bins,
        [critic_cfg.dense_units] * critic_cfg.mlp_layers,
        activation="silu",
        layer_norm=True,
        layer_norm_eps=eps,
    )

    encoder.apply(init_weights)
    observation_model.apply(init_weights)
    reward_model.apply(init_weights)
    continue_model.apply(init_weights)
    actor.apply(init_weights)
    critic.apply(init_weights)

    if cfg.algo.hafner_initialization:
        actor.mlp_heads.apply(uniform_init_weights(1.0))
        critic.model[-1].apply(uniform_init_weights(0.0))
        rssm.transition_model.model[-1].apply(uniform_init_weights(1.0))
        rssm.representation_model.model[-1].apply(uniform_init_weights(1.0))
        world_model.reward_model.model[-1].apply(uniform_init_weights(0.0))
        world_model.continue_model.model[-1].apply(uniform_init_weights(1.0))
        if mlp_decoder is not None:
            mlp_decoder.heads.apply(uniform_init_weights(1.0))
        if cnn_decoder is not None:
            cnn_decoder.model[-1].model[-1].apply(uniform_init_weights(1.0))

    if world_model_state:
        world_model.load_state_dict(world_model_state)
    if actor_state:
        actor.load_state_dict(actor_state)
    if critic_state:
        critic.load_state_dict(critic_state)

    # one GradSync per training module: few, large buckets for the xGMI links
    world_model = runtime.setup_module(world_model)
    actor = runtime.setup_module(actor)
    critic = runtime.setup_module(critic)

    target_critic = copy.deepcopy(critic)
    if target_critic_state:
        target_critic.load_state_dict(target_critic_state)
    target_critic = runtime.setup_module(target_critic, sync=False)
    for p in target_critic.parameters():
        p.requires_grad_(False)

    player = PlayerDV3(
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
