"""SAC agent (parity surface: sheeprl/algos/sac/agent.py — SACCritic :20
(ensemble), SACActor :57 (tanh-normal, log-prob correction :123-142),
SACAgent :145 (log_alpha :164, target EMA :265), SACPlayer :270)."""

from __future__ import annotations

import copy
from typing import Any, Dict, Optional, Sequence, Tuple

import numpy as np
import torch
from torch import Tensor, nn

from sheeprl_amd import ops
from sheeprl_amd.distributions import TanhNormal
from sheeprl_amd.envs import spaces
from sheeprl_amd.models import MLP
from sheeprl_amd.parallel import Runtime

LOG_STD_MIN, LOG_STD_MAX = -5.0, 2.0


class SACCritic(nn.Module):
    def __init__(self, observation_dim: int, hidden_size: int = 256, num_critics: int = 1) -> None:
        super().__init__()
        self.model = nn.ModuleList(
            [MLP(observation_dim, 1, [hidden_size, hidden_size], activation="relu") for _ in range(num_critics)]
        )

    def forward(self, obs: Tensor, action: Tensor) -> Tensor:
        x = torch.cat([obs, action], dim=-1)
        return torch.cat([m(x) for m in self.model], dim=-1)


class SACActor(nn.Module):
    def __init__(
        self,
        observation_dim: int,
        action_dim: int,
        distribution: str = "auto",
        hidden_size: int = 256,
        action_low: Optional[np.ndarray] = None,
        action_high: Optional[np.ndarray] = None,
    ) -> None:
        super().__init__()
        self.model = MLP(observation_dim, None, [hidden_size, hidden_size], activation="relu")
        self.fc_mean = nn.Linear(hidden_size, action_dim)
        self.fc_logstd = nn.Linear(hidden_size, action_dim)
        low = np.asarray(action_low, dtype=np.float32) if action_low is not None else -np.ones(action_dim, np.float32)
        high = np.asarray(action_high, dtype=np.float32) if action_high is not None else np.ones(action_dim, np.float32)
        self.register_buffer("action_scale", torch.tensor((high - low) / 2.0))
        self.register_buffer("action_bias", torch.tensor((high + low) / 2.0))

    def _dist(self, obs: Tensor) -> TanhNormal:
        x = self.model(obs)
        mean = self.fc_mean(x).float()
        log_std = self.fc_logstd(x).float()
        std = log_std.clamp(LOG_STD_MIN, LOG_STD_MAX).exp()
        return TanhNormal(mean, std)

    def forward(self, obs: Tensor) -> Tuple[Tensor, Tensor]:
        """Returns (squashed action in env range, summed log-prob).

        log pi(a) = logN(x) - sum log(scale * (1 - tanh(x)^2)); the tanh term
        is inside TanhNormal.rsample_with_log_prob, leaving the scale term.
        """
        x = self.model(obs)
        mean = self.fc_mean(x).float()
        log_std = self.fc_logstd(x).float()
        if mean.is_cuda and ops.use_hip(mean):
            # fused sample: clamp/exp + reparam + tanh + rescale + summed
            # log-prob in ONE kernel each way (§2.8 item 13); philox noise
            # stays a torch op so the step remains hipGraph-capturable
            eps = torch.randn_like(mean)
            return ops.tanh_normal_sample(
                mean, log_std, eps, self.action_scale.float(), self.action_bias.float(), obs,
                LOG_STD_MIN, LOG_STD_MAX,
            )
        dist = TanhNormal(mean, log_std.clamp(LOG_STD_MIN, LOG_STD_MAX).exp())
        y, logp = dist.rsample_with_log_prob()
        # sampling/log-prob math runs in fp32 for stability; the action feeds
        # the (possibly bf16) critics in the module dtype
        action = (y * self.action_scale + self.action_bias).to(obs.dtype)
        log_prob = (logp - torch.log(self.action_scale.float()).expand_as(logp)).sum(-1, keepdim=True)
        return action, log_prob

    def get_greedy_actions(self, obs: Tensor) -> Tensor:
        dist = self._dist(obs)
        return dist.mode * self.action_scale + self.action_bias


class SACAgent(nn.Module):
    def __init__(
        self,
        actor: SACActor,
        critics: Sequence[SACCritic],
        target_entropy: float,
        alpha: float = 1.0,
        tau: float = 0.005,
        device: torch.device = torch.device("cpu"),
    ) -> None:
        super().__init__()
        self.actor = actor
        self.qfs = nn.ModuleList(critics)
        self.qfs_target = copy.deepcopy(self.qfs)
        for p in self.qfs_target.parameters():
            p.requires_grad_(False)
        self.log_alpha = nn.Parameter(torch.tensor(float(np.log(alpha)), device=device))
        self.target_entropy = target_entropy
        self._tau = tau

    @property
    def alpha(self) -> Tensor:
        return self.log_alpha.exp()

    def get_q_values(self, obs: Tensor, action: Tensor) -> Tensor:
        return torch.cat([qf(obs, action) for qf in self.qfs], dim=-1)

    @torch.no_grad()
    def get_target_q_values(self, obs: Tensor, action: Tensor) -> Tensor:
        return torch.cat([qf(obs, action) for qf in self.qfs_target], dim=-1)

    @torch.no_grad()
    def qfs_target_ema(self, tau: Optional[float] = None) -> None:
        ops.ema_update_(
            list(self.qfs_target.parameters()), list(self.qfs.parameters()), self._tau if tau is None else tau
        )


class SACPlayer(nn.Module):
    def __init__(self, actor: SACActor) -> None:
        super().__init__()
        self.actor = actor

    @torch.no_grad()
    def get_actions(self, obs: Tensor, greedy: bool = False) -> Tensor:
        if greedy:
            return self.actor.get_greedy_actions(obs)
        return self.actor(obs)[0]


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
        SACCritic(obs_dim + act_dim, cfg.algo.critic.hidden_size, num_critics=1)
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
