"""Recurrent PPO agent (parity surface: sheeprl/algos/ppo_recurrent/agent.py —
RecurrentModel :18-80 (pre-MLP -> LSTM -> post-MLP), RecurrentPPOAgent :83).

The LSTM input is cat(encoded features, previous action); hidden states are
reset at episode starts via the is_first mask (replaces the reference's
packed-sequence handling with an explicit masked scan)."""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.distributions as td
from torch import Tensor, nn

from sheeprl_amd.algos.ppo.agent import PixelEncoder, VectorEncoder
from sheeprl_amd.envs import spaces
from sheeprl_amd.models import MLP, MultiEncoder
from sheeprl_amd.parallel import Runtime


class RecurrentModel(nn.Module):
    def __init__(
        self,
        input_size: int,
        lstm_hidden_size: int,
        dense_units: int,
        layer_norm: bool = True,
        pre_rnn_mlp: bool = False,
        post_rnn_mlp: bool = False,
    ) -> None:
        super().__init__()
        # pre/post MLPs around the LSTM are OFF by default, matching the
        # reference's rnn.pre_rnn_mlp.apply / post_rnn_mlp.apply defaults
        self.pre_mlp = (
            MLP(input_size, None, [dense_units], activation="relu", layer_norm=layer_norm)
            if pre_rnn_mlp
            else nn.Identity()
        )
        lstm_in = dense_units if pre_rnn_mlp else input_size
        self.lstm = nn.LSTM(lstm_in, lstm_hidden_size, batch_first=False)
        self.post_mlp = (
            MLP(lstm_hidden_size, None, [lstm_hidden_size], activation="relu", layer_norm=layer_norm)
            if post_rnn_mlp
            else nn.Identity()
        )
        self.output_dim = lstm_hidden_size
        self.hidden_size = lstm_hidden_size

    def forward(self, x: Tensor, states: Tuple[Tensor, Tensor]) -> Tuple[Tensor, Tuple[Tensor, Tensor]]:
        """x: [T, B, F] -> ([T, B, H], states)."""
        x = self.pre_mlp(x)
        out, states = self.lstm(x, states)
        shape = out.shape
        return self.post_mlp(out.reshape(-1, shape[-1])).view(*shape[:-1], -1), states

    def _post(self, out: Tensor) -> Tensor:
        shape = out.shape
        return self.post_mlp(out.reshape(-1, shape[-1])).view(*shape[:-1], -1)

    def masked_scan(
        self, x: Tensor, is_first: Tensor, states: Tuple[Tensor, Tensor]
    ) -> Tuple[Tensor, Tuple[Tensor, Tensor]]:
        """Step the LSTM over T with per-step state reset where is_first=1.

        On HIP the whole scan runs through the fused kernels (one batched
        input GEMM + one gates launch per step, hand-written backward) —
        SURVEY.md §2.8 item 12."""
        x = self.pre_mlp(x)
        from sheeprl_amd.ops.lstm import lstm_scan, lstm_scan_applicable

        if lstm_scan_applicable(self.lstm, x):
            out, states = lstm_scan(x, is_first, states, self.lstm)
            return self._post(out), states
        outs: List[Tensor] = []
        h, c = states
        for t in range(x.shape[0]):
            mask = (1.0 - is_first[t]).view(1, -1, 1).to(x.dtype)
            h = h * mask
            c = c * mask
            out, (h, c) = self.lstm(x[t : t + 1], (h.contiguous(), c.contiguous()))
            outs.append(out)
        out = torch.cat(outs, dim=0)
        return self._post(out), (h, c)


class RecurrentPPOAgent(nn.Module):
    def __init__(
        self,
        obs_space: spaces.Dict,
        action_space: spaces.Space,
        cfg_algo: Any,
        num_envs: int = 1,
    ) -> None:
        super().__init__()
        cnn_keys = list(cfg_algo.cnn_keys.encoder or [])
        mlp_keys = list(cfg_algo.mlp_keys.encoder or [])
        cnn_enc = PixelEncoder(cnn_keys, obs_space, cfg_algo.encoder.get("cnn_features_dim", 512)) if cnn_keys else None
        mlp_enc = (
            VectorEncoder(
                mlp_keys,
                obs_space,
                cfg_algo.encoder.get("mlp_features_dim", 64),
                cfg_algo.encoder.get("dense_units", 64),
                cfg_algo.encoder.get("mlp_layers", 1),
                cfg_algo.encoder.get("dense_act", "relu"),
                cfg_algo.encoder.get("layer_norm", False),
            )
            if mlp_keys
            else None
        )
        self.feature_extractor = MultiEncoder(cnn_enc, mlp_enc)
        self.is_continuous = isinstance(action_space, spaces.Box)
        self.is_multidiscrete = isinstance(action_space, spaces.MultiDiscrete)
        if self.is_continuous:
            self.actions_dim = [int(np.prod(action_space.shape))]
        elif self.is_multidiscrete:
            self.actions_dim = [int(n) for n in action_space.nvec]
        else:
            self.actions_dim = [action_space.n]
        act_input = int(np.sum(self.actions_dim)) if not self.is_continuous else self.actions_dim[0]
        self.rnn = RecurrentModel(
            self.feature_extractor.output_dim + act_input,
            cfg_algo.rnn.lstm.hidden_size,
            cfg_algo.rnn.get("dense_units", 64),
            layer_norm=cfg_algo.rnn.get("layer_norm", True),
            pre_rnn_mlp=cfg_algo.rnn.get("pre_rnn_mlp", {}).get("apply", False),
            post_rnn_mlp=cfg_algo.rnn.get("post_rnn_mlp", {}).get("apply", False),
        )
        units = cfg_algo.actor.dense_units
        self.actor_torso = MLP(self.rnn.output_dim, None, [units] * cfg_algo.actor.mlp_layers, activation="relu")
        if self.is_continuous:
            self.actor_head = nn.Linear(units, self.actions_dim[0] * 2)
        else:
            self.actor_head = nn.Linear(units, int(np.sum(self.actions_dim)))
        cunits = cfg_algo.critic.dense_units
        self.critic = MLP(self.rnn.output_dim, 1, [cunits] * cfg_algo.critic.mlp_layers, activation="relu")
        self.num_envs = num_envs

    def initial_states(self, batch: int, device, dtype=torch.float32) -> Tuple[Tensor, Tensor]:
        h = torch.zeros(1, batch, self.rnn.hidden_size, device=device, dtype=dtype)
        return h, torch.zeros_like(h)

    def _dists(self, logits: Tensor) -> List[td.Distribution]:
        if self.is_continuous:
            mean, log_std = logits.chunk(2, -1)
            return [td.Independent(td.Normal(mean.float(), log_std.clamp(-20, 2).exp().float()), 1)]
        out = []
        start = 0
        for n in self.actions_dim:
            out.append(td.Categorical(logits=logits[..., start : start + n].float()))
            start += n
        return out

    def encode(self, obs: Dict[str, Tensor], prev_actions: Tensor) -> Tensor:
        feats = self.feature_extractor(obs)
        return torch.cat((feats, prev_actions), dim=-1)

    def forward_sequence(
        self,
        obs: Dict[str, Tensor],
        prev_actions: Tensor,
        is_first: Tensor,
        states: Tuple[Tensor, Tensor],
        actions: Tensor,
    ) -> Tuple[Tensor, Tensor, Tensor]:
        """Training pass over [T, B, ...]; returns (logprob, entropy, values)."""
        x = self.encode(obs, prev_actions)
        out, _ = self.rnn.masked_scan(x, is_first, states)
        logits = self.actor_head(self.actor_torso(out))
        values = self.critic(out)
        dists = self._dists(logits)
        if self.is_continuous:
            logp = dists[0].log_prob(actions).unsqueeze(-1)
            ent = dists[0].entropy().unsqueeze(-1)
        else:
            acts = list(actions.unbind(-1))
            logp = sum(d.log_prob(a) for d, a in zip(dists, acts)).unsqueeze(-1)
            ent = sum(d.entropy() for d in dists).unsqueeze(-1)
        return logp, ent, values


class RecurrentPPOPlayer(nn.Module):
    def __init__(self, agent: RecurrentPPOAgent) -> None:
        super().__init__()
        self.agent = agent

    @torch.no_grad()
    def get_actions(
        self,
        obs: Dict[str, Tensor],
        prev_actions: Tensor,
        states: Tuple[Tensor, Tensor],
        greedy: bool = False,
    ):
        x = self.agent.encode(obs, prev_actions)  # [1, B, F]
        out, states = self.agent.rnn(x, states)
        logits = self.agent.actor_head(self.agent.actor_torso(out))
        values = self.agent.critic(out)
        dists = self.agent._dists(logits)
        if self.agent.is_continuous:
            a = dists[0].mean if greedy else dists[0].rsample()
            logp = dists[0].log_prob(a).unsqueeze(-1)
            return a, logp, values, states
        acts = [(torch.argmax(d.logits, -1) if greedy else d.sample()) for d in dists]
        logp = sum(d.log_prob(a) for d, a in zip(dists, acts)).unsqueeze(-1)
        return torch.stack(acts, -1), logp, values, states


def build_agent(
    runtime: Runtime,
    obs_space: spaces.Dict,
    action_space: spaces.Space,
    cfg: Any,
    agent_state: Optional[Dict[str, Tensor]] = None,
) -> Tuple[RecurrentPPOAgent, RecurrentPPOPlayer]:
    agent = RecurrentPPOAgent(obs_space, action_space, cfg.algo, num_envs=cfg.env.num_envs)
    if agent_state:
        agent.load_state_dict(agent_state)
    agent = runtime.setup_module(agent)
    player = RecurrentPPOPlayer(agent)
    return agent, player
