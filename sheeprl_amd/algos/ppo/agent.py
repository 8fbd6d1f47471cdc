"""PPO agent (parity surface: sheeprl/algos/ppo/agent.py — PPOAgent :91,
PPOActor :72, PPOPlayer :242, build_agent :325 with player weight-tying
:363-368).

Supports dict observations (cnn + mlp keys) and discrete / multi-discrete /
continuous action spaces.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch
import torch.distributions as td
from torch import Tensor, nn

from sheeprl_amd import ops
from sheeprl_amd.envs import spaces
from sheeprl_amd.models import MLP, MultiEncoder, NatureCNN, cnn_forward, get_activation
from sheeprl_amd.parallel import Runtime


class PixelEncoder(nn.Module):
    """Stacks the cnn keys channel-wise and encodes with NatureCNN
    (parity: ppo/agent.py:20-36)."""

    def __init__(self, keys: Sequence[str], obs_space: spaces.Dict, features_dim: int = 512) -> None:
        super().__init__()
        self.keys = list(keys)
        in_ch = sum(obs_space[k].shape[0] for k in self.keys)
        size = obs_space[self.keys[0]].shape[1]
        self.cnn = NatureCNN(in_ch, features_dim, screen_size=size)
        self.input_dim = (in_ch, size, size)
        self.output_dim = features_dim

    def forward(self, obs: Dict[str, Tensor]) -> Tensor:
        x = torch.cat([obs[k] for k in self.keys], dim=-3)
        if x.dtype == torch.uint8:
            x = ops.normalize_obs(x)
        # the fused obs_norm emits fp32; bf16-true modules need the cast
        x = x.to(self.cnn.conv[0].weight.dtype)
        return cnn_forward(self.cnn, x, self.input_dim)


class VectorEncoder(nn.Module):
    def __init__(
        self,
        keys: Sequence[str],
        obs_space: spaces.Dict,
        features_dim: Optional[int],
        dense_units: int,
        mlp_layers: int,
        dense_act: Any,
        layer_norm: bool,
    ) -> None:
        super().__init__()
        self.keys = list(keys)
        in_dim = sum(int(torch.tensor(obs_space[k].shape).prod()) for k in self.keys)
        if mlp_layers > 0 and features_dim:
            self.model: nn.Module = MLP(
                in_dim,
                None,
                [dense_units] * (mlp_layers - 1) + [features_dim],
                activation=dense_act,
                layer_norm=layer_norm,
            )
            self.output_dim = features_dim
        else:
            self.model = nn.Identity()
            self.output_dim = in_dim

    def forward(self, obs: Dict[str, Tensor]) -> Tensor:
        x = torch.cat([obs[k].float() for k in self.keys], dim=-1)
        return self.model(x)


class PPOActor(nn.Module):
    """Action heads (parity: ppo/agent.py:72-89)."""

    def __init__(self, features_dim: int, action_space: spaces.Space, cfg_actor: Any) -> None:
        super().__init__()
        act = get_activation(cfg_actor.get("dense_act", "tanh"))
        units = cfg_actor.get("dense_units", 64)
        layers = cfg_actor.get("mlp_layers", 2)
        self.torso = (
            MLP(features_dim, None, [units] * layers, activation=act, layer_norm=cfg_actor.get("layer_norm", False))
            if layers > 0
            else nn.Identity()
        )
        torso_out = units if layers > 0 else features_dim
        self.is_continuous = isinstance(action_space, spaces.Box)
        self.is_multidiscrete = isinstance(action_space, spaces.MultiDiscrete)
        if self.is_continuous:
            dim = int(torch.tensor(action_space.shape).prod())
            self.head = nn.Linear(torso_out, dim * 2)  # mean + log_std
            self.action_dims: List[int] = [dim]
        elif self.is_multidiscrete:
            self.action_dims = [int(n) for n in action_space.nvec]
            self.head = nn.Linear(torso_out, sum(self.action_dims))
        else:
            self.action_dims = [action_space.n]
            self.head = nn.Linear(torso_out, action_space.n)

    def forward(self, features: Tensor) -> Tensor:
        return self.head(self.torso(features))

    def distributions(self, logits: Tensor) -> List[td.Distribution]:
        if self.is_continuous:
            mean, log_std = logits.chunk(2, dim=-1)
            std = log_std.clamp(-20, 2).exp()
            return [td.Independent(td.Normal(mean, std), 1)]
        outs = []
        start = 0
        for n in self.action_dims:
            outs.append(td.Categorical(logits=logits[..., start : start + n].float()))
            start += n
        return outs


class PPOAgent(nn.Module):
    """Encoder + actor + critic (parity: ppo/agent.py:91-240)."""

    def __init__(self, obs_space: spaces.Dict, action_space: spaces.Space, cfg_algo: Any) -> None:
        super().__init__()
        cnn_keys = list(cfg_algo.cnn_keys.encoder or [])
        mlp_keys = list(cfg_algo.mlp_keys.encoder or [])
        cnn_enc = (
            PixelEncoder(cnn_keys, obs_space, cfg_algo.encoder.get("cnn_features_dim", 512)) if cnn_keys else None
        )
        mlp_enc = (
            VectorEncoder(
                mlp_keys,
                obs_space,
                cfg_algo.encoder.get("mlp_features_dim"),
                cfg_algo.encoder.get("dense_units", 64),
                cfg_algo.encoder.get("mlp_layers", 1),
                cfg_algo.encoder.get("dense_act", "tanh"),
                cfg_algo.encoder.get("layer_norm", False),
            )
            if mlp_keys
            else None
        )
        self.feature_extractor = MultiEncoder(cnn_enc, mlp_enc)
        self.actor = PPOActor(self.feature_extractor.output_dim, action_space, cfg_algo.actor)
        crit = cfg_algo.critic
        layers = crit.get("mlp_layers", 2)
        self.critic = MLP(
            self.feature_extractor.output_dim,
            1,
            [crit.get("dense_units", 64)] * layers,
            activation=crit.get("dense_act", "tanh"),
            layer_norm=crit.get("layer_norm", False),
        )
        self.is_continuous = self.actor.is_continuous

    def forward(
        self, obs: Dict[str, Tensor], actions: Optional[Tensor] = None
    ) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
        """Returns (actions, log_prob, entropy, value)."""
        feats = self.feature_extractor(obs)
        logits = self.actor(feats)
        value = self.critic(feats)
        dists = self.actor.distributions(logits)
        if self.is_continuous:
            dist = dists[0]
            if actions is None:
                actions = dist.rsample()
            logp = dist.log_prob(actions).unsqueeze(-1)
            ent = dist.entropy().unsqueeze(-1)
            return actions, logp, ent, value
        if actions is None:
            acts = [d.sample() for d in dists]
        else:
            acts = list(actions.unbind(-1))
        logp = sum(d.log_prob(a) for d, a in zip(dists, acts)).unsqueeze(-1)
        ent = sum(d.entropy() for d in dists).unsqueeze(-1)
        return torch.stack(acts, dim=-1), logp, ent, value

    def get_values(self, obs: Dict[str, Tensor]) -> Tensor:
        return self.critic(self.feature_extractor(obs))


class PPOPlayer(nn.Module):
    """Single-device inference wrapper whose parameters are tied to the
    trained agent's (parity: ppo/agent.py:242-322, tying at :363-368)."""

    def __init__(self, feature_extractor: nn.Module, actor: PPOActor, critic: nn.Module) -> None:
        super().__init__()
        self.feature_extractor = feature_extractor
        self.actor = actor
        self.critic = critic

    @torch.no_grad()
    def get_actions(self, obs: Dict[str, Tensor], greedy: bool = False) -> Tuple[Tensor, Tensor, Tensor]:
        feats = self.feature_extractor(obs)
        logits = self.actor(feats)
        value = self.critic(feats)
        dists = self.actor.distributions(logits)
        if self.actor.is_continuous:
            d = dists[0]
            a = d.mean if greedy else d.rsample()
            return a, d.log_prob(a).unsqueeze(-1), value
        acts = [(torch.argmax(d.logits, dim=-1) if greedy else d.sample()) for d in dists]
        logp = sum(d.log_prob(a) for d, a in zip(dists, acts)).unsqueeze(-1)
        return torch.stack(acts, dim=-1), logp, value

    @torch.no_grad()
    def get_values(self, obs: Dict[str, Tensor]) -> Tensor:
        return self.critic(self.feature_extractor(obs))


def build_agent(
    runtime: Runtime,
    obs_space: spaces.Dict,
    action_space: spaces.Space,
    cfg: Any,
    agent_state: Optional[Dict[str, Tensor]] = None,
) -> Tuple[PPOAgent, PPOPlayer]:
    agent = PPOAgent(obs_space, action_space, cfg.algo)
    if agent_state is not None:
        agent.load_state_dict(agent_state)
    agent = runtime.setup_module(agent)
    player = PPOPlayer(agent.feature_extractor, agent.actor, agent.critic)
    return agent, player
