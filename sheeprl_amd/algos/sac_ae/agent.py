"""SAC-AE agent (parity surface: sheeprl/algos/sac_ae/agent.py — CNNEncoder
:26 / CNNDecoder :153 (SAC-AE paper conv stack, feature dim with
LayerNorm+tanh), SACAEAgent :321 with separate encoder/critic EMA taus).

Pixel (+vector) observations; actor consumes DETACHED encoder features, the
critic trains the encoder; the autoencoder adds reconstruction + latent-L2."""

from __future__ import annotations

import copy
from typing import Any, Dict, Optional, Sequence, Tuple

import numpy as np
import torch
from torch import Tensor, nn

from sheeprl_amd import ops
from sheeprl_amd.algos.sac.agent import SACActor
from sheeprl_amd.envs import spaces
from sheeprl_amd.models import MLP
from sheeprl_amd.parallel import Runtime


class SACAEEncoder(nn.Module):
    def __init__(self, keys: Sequence[str], obs_space: spaces.Dict, features_dim: int = 50, screen_size: int = 64):
        super().__init__()
        self.cnn_keys = [k for k in keys if len(obs_space[k].shape) == 3]
        self.mlp_keys = [k for k in keys if len(obs_space[k].shape) == 1]
        in_ch = sum(obs_space[k].shape[0] for k in self.cnn_keys)
        self.conv = nn.Sequential(
            nn.Conv2d(in_ch, 32, 3, stride=2), nn.ReLU(),
            nn.Conv2d(32, 32, 3, stride=1), nn.ReLU(),
            nn.Conv2d(32, 32, 3, stride=1), nn.ReLU(),
            nn.Conv2d(32, 32, 3, stride=1), nn.ReLU(),
            nn.Flatten(),
        )
        with torch.no_grad():
            n_flat = self.conv(torch.zeros(1, in_ch, screen_size, screen_size)).shape[1]
        self.conv_out_shape = None
        mlp_dim = sum(int(np.prod(obs_space[k].shape)) for k in self.mlp_keys)
        self.fc = nn.Linear(n_flat + mlp_dim, features_dim)
        self.ln = nn.LayerNorm(features_dim)
        self.output_dim = features_dim
        self._n_flat = n_flat
        self._in_ch = in_ch
        self._screen = screen_size

    def forward(self, obs: Dict[str, Tensor], detach: bool = False) -> Tensor:
        x = torch.cat([obs[k] for k in self.cnn_keys], dim=-3)
        x = ops.normalize_obs(x) if x.dtype == torch.uint8 else x
        h = self.conv(x)
        if self.mlp_keys:
            h = torch.cat([h] + [obs[k].float().flatten(1) for k in self.mlp_keys], dim=-1)
        if detach:
            h = h.detach()
        z = torch.tanh(self.ln(self.fc(h)))
        return z


class SACAEDecoder(nn.Module):
    def __init__(self, keys: Sequence[str], obs_space: spaces.Dict, features_dim: int = 50, screen_size: int = 64):
        super().__init__()
        self.cnn_keys = [k for k in keys if len(obs_space[k].shape) == 3]
        out_ch = sum(obs_space[k].shape[0] for k in self.cnn_keys)
        self.out_channels = [obs_space[k].shape[0] for k in self.cnn_keys]
        # inverse of the encoder conv stack
        conv_size = (screen_size - 3) // 2 + 1 - 6  # k3s2 then 3x k3s1 (64 -> 31 -> 25)
        self._conv_size = conv_size
        self.fc = nn.Linear(features_dim, 32 * conv_size * conv_size)
        self.deconv = nn.Sequential(
            nn.ConvTranspose2d(32, 32, 3, stride=1), nn.ReLU(),
            nn.ConvTranspose2d(32, 32, 3, stride=1), nn.ReLU(),
            nn.ConvTranspose2d(32, 32, 3, stride=1), nn.ReLU(),
            nn.ConvTranspose2d(32, out_ch, 3, stride=2, output_padding=1),
        )

    def forward(self, z: Tensor) -> Dict[str, Tensor]:
        h = self.fc(z).view(-1, 32, self._conv_size, self._conv_size)
        rec = self.deconv(h)
        return {k: r for k, r in zip(self.cnn_keys, torch.split(rec, self.out_channels, dim=-3))}


class SACAEQFunction(nn.Module):
    def __init__(self, features_dim: int, action_dim: int, hidden_size: int = 1024):
        super().__init__()
        self.model = MLP(features_dim + action_dim, 1, [hidden_size, hidden_size], activation="relu")

    def forward(self, z: Tensor, a: Tensor) -> Tensor:
        return self.model(torch.cat([z, a], dim=-1))


class SACAEAgent(nn.Module):
    def __init__(
        self,
        encoder: SACAEEncoder,
        decoder: SACAEDecoder,
        actor: SACActor,
        critics: Sequence[SACAEQFunction],
        target_entropy: float,
        alpha: float = 0.1,
        encoder_tau: float = 0.05,
        critic_tau: float = 0.01,
        device: torch.device = torch.device("cpu"),
    ) -> None:
        super().__init__()
        self.encoder = encoder
        self.decoder = decoder
        self.actor = actor
        self.qfs = nn.ModuleList(critics)
        self.encoder_target = copy.deepcopy(encoder)
        self.qfs_target = copy.deepcopy(self.qfs)
        for p in list(self.encoder_target.parameters()) + list(self.qfs_target.parameters()):
            p.requires_grad_(False)
        self.log_alpha = nn.Parameter(torch.tensor(float(np.log(alpha)), device=device))
        self.target_entropy = target_entropy
        self._encoder_tau = encoder_tau
        self._critic_tau = critic_tau

    @property
    def alpha(self) -> Tensor:
        return self.log_alpha.exp()

    def get_q_values(self, z: Tensor, a: Tensor) -> Tensor:
        return torch.cat([q(z, a) for q in self.qfs], dim=-1)

    @torch.no_grad()
    def get_target_q_values(self, z: Tensor, a: Tensor) -> Tensor:
        return torch.cat([q(z, a) for q in self.qfs_target], dim=-1)

    @torch.no_grad()
    def target_ema(self) -> None:
        ops.ema_update_(list(self.qfs_target.parameters()), list(self.qfs.parameters()), self._critic_tau)
        ops.ema_update_(list(self.encoder_target.parameters()), list(self.encoder.parameters()), self._encoder_tau)


class SACAEPlayer(nn.Module):
    def __init__(self, encoder: SACAEEncoder, actor: SACActor) -> None:
        super().__init__()
        self.encoder = encoder
        self.actor = actor

    @torch.no_grad()
    def get_actions(self, obs: Dict[str, Tensor], greedy: bool = False) -> Tensor:
        z = self.encoder(obs)
        if greedy:
            return self.actor.get_greedy_actions(z)
        return self.actor(z)[0]


def build_agent(
    runtime: Runtime,
    cfg: Any,
    obs_space: spaces.Dict,
    action_space: spaces.Box,
    agent_state: Optional[Dict[str, Tensor]] = None,
) -> Tuple[SACAEAgent, SACAEPlayer]:
    keys = list(cfg.algo.cnn_keys.encoder or []) + list(cfg.algo.mlp_keys.encoder or [])
    act_dim = int(np.prod(action_space.shape))
    screen = obs_space[list(cfg.algo.cnn_keys.encoder)[0]].shape[-1]
    encoder = SACAEEncoder(keys, obs_space, cfg.algo.encoder.features_dim, screen)
    decoder = SACAEDecoder(list(cfg.algo.cnn_keys.encoder), obs_space, cfg.algo.encoder.features_dim, screen)
    actor = SACActor(
        cfg.algo.encoder.features_dim,
        act_dim,
        hidden_size=cfg.algo.actor.hidden_size,
        action_low=action_space.low,
        action_high=action_space.high,
    )
    critics = [
        SACAEQFunction(cfg.algo.encoder.features_dim, act_dim, cfg.algo.critic.hidden_size)
        for _ in range(cfg.algo.critic.n)
    ]
    agent = SACAEAgent(
        encoder,
        decoder,
        actor,
        critics,
        target_entropy=-act_dim,
        alpha=cfg.algo.alpha.alpha,
        encoder_tau=cfg.algo.encoder.tau,
        critic_tau=cfg.algo.critic.tau,
        device=runtime.device,
    )
    if agent_state:
        agent.load_state_dict(agent_state)
    agent = runtime.setup_module(agent)
    player = SACAEPlayer(agent.encoder, agent.actor)
    return agent, player
