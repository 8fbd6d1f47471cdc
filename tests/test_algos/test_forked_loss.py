"""The stream-forked world-model loss (loss.py:reconstruction_loss_forked)
must be value- and gradient-identical to reconstruction_loss — the fork is a
scheduling hint, never a semantic change (parallel/streams.py).  On CPU the
Branches region is a no-op, so this exercises the restructured code path
itself.
"""

import copy

import torch
import torch.distributions as td
import torch.nn as nn

from sheeprl_amd.algos.dreamer_v3.loss import reconstruction_loss, reconstruction_loss_forked
from sheeprl_amd.distributions import (
    BernoulliSafeMode,
    MSEDistribution,
    SymlogDistribution,
    TwoHotEncodingDistribution,
)
from sheeprl_amd.parallel.streams import Branches

T, B = 5, 3
STOCH, DISC = 4, 6
H = 16
LATENT = STOCH * DISC + H


class _TinyWorldModel(nn.Module):
    def __init__(self):
        super().__init__()
        self.obs_head = nn.Linear(LATENT, 7)
        self.reward_head = nn.Linear(LATENT, 255)
        self.continue_head = nn.Linear(LATENT, 1)
        self.transition = nn.Linear(H, STOCH * DISC)
        self.rssm = self  # transition_logits lives on .rssm in the real model

    def observation_model(self, latents):
        return {"state": self.obs_head(latents)}

    def reward_model(self, latents):
        return self.reward_head(latents)

    def continue_model(self, latents):
        return self.continue_head(latents)

    def transition_logits(self, recurrent_states):
        return self.transition(recurrent_states)


def _inputs(seed=0):
    g = torch.Generator().manual_seed(seed)
    posteriors_logits = torch.randn(T, B, STOCH * DISC, generator=g, requires_grad=True)
    recurrent_states = torch.randn(T, B, H, generator=g, requires_grad=True)
    posteriors = torch.softmax(posteriors_logits.detach().view(T, B, STOCH, DISC), -1)
    latent_states = torch.cat((posteriors.view(T, B, -1), recurrent_states), -1)
    batch_obs = {"state": torch.randn(T, B, 7, generator=g)}
    rewards = torch.randn(T, B, 1, generator=g)
    continues_targets = torch.randint(0, 2, (T, B, 1), generator=g).float()
    return posteriors_logits, recurrent_states, latent_states, batch_obs, rewards, continues_targets


def test_forked_loss_matches_reference_path():
    torch.manual_seed(1)
    wm_a = _TinyWorldModel()
    wm_b = copy.deepcopy(wm_a)

    pl_a, rs_a, ls_a, obs, rew, cont = _inputs()
    pl_b = pl_a.detach().clone().requires_grad_(True)
    rs_b = rs_a.detach().clone().requires_grad_(True)
    ls_b = torch.cat((torch.softmax(pl_b.detach().view(T, B, STOCH, DISC), -1).view(T, B, -1), rs_b), -1)

    # ---- original path ----
    priors = wm_a.rssm.transition_logits(rs_a)
    recon = wm_a.observation_model(ls_a)
    po = {"state": SymlogDistribution(recon["state"].float(), dims=1)}
    pr = TwoHotEncodingDistribution(wm_a.reward_model(ls_a).float(), dims=1)
    pc = td.Independent(BernoulliSafeMode(logits=wm_a.continue_model(ls_a).float()), 1)
    out_a = reconstruction_loss(
        po,
        {k: v.float() for k, v in obs.items()},
        pr,
        rew,
        priors.view(T, B, STOCH, DISC),
        pl_a.view(T, B, STOCH, DISC),
        0.5, 0.1, 1.0, 1.0,
        pc, cont, 1.0,
    )
    out_a[0].backward()

    # ---- forked path (Branches no-op on CPU) ----
    out_b = reconstruction_loss_forked(
        Branches(enabled=False),
        wm_b,
        ls_b,
        rs_b,
        obs,
        rew,
        cont,
        pl_b,
        None,  # transition head computed inside the fork
        STOCH, DISC,
        [], ["state"],
        0.5, 0.1, 1.0, 1.0,
        1.0,
        False,
    )
    out_b[0].backward()

    for i, name in enumerate(["rec_loss", "kl", "state_loss", "reward_loss", "obs_loss", "cont_loss"]):
        torch.testing.assert_close(out_a[i], out_b[i], msg=f"{name} diverged")
    torch.testing.assert_close(pl_a.grad, pl_b.grad)
    torch.testing.assert_close(rs_a.grad, rs_b.grad)
    for (na, pa), (nb, pb) in zip(wm_a.named_parameters(), wm_b.named_parameters()):
        assert na == nb
        torch.testing.assert_close(pa.grad, pb.grad, msg=f"param grad {na} diverged")


def test_forked_loss_viewed_logits_returned():
    wm = _TinyWorldModel()
    pl, rs, ls, obs, rew, cont = _inputs(seed=3)
    out = reconstruction_loss_forked(
        Branches(enabled=False), wm, ls, rs, obs, rew, cont, pl, None,
        STOCH, DISC, [], ["state"], 0.5, 0.1, 1.0, 1.0, 1.0, False,
    )
    assert out[6].shape == (T, B, STOCH, DISC)  # priors viewed
    assert out[7].shape == (T, B, STOCH, DISC)  # posteriors viewed


def test_ppo_losses_eager_fallback_matches_components():
    """On CPU ppo_losses returns exactly the separate policy/value/entropy
    losses (the fused kernel is GPU-only)."""
    from sheeprl_amd.algos.ppo.loss import entropy_loss, policy_loss, ppo_losses, value_loss

    torch.manual_seed(5)
    N = 64
    lp_new = torch.randn(N, requires_grad=True)
    lp_old = torch.randn(N)
    adv = torch.randn(N)
    v_new = torch.randn(N, 1, requires_grad=True)
    v_old = torch.randn(N, 1)
    ret = torch.randn(N, 1)
    ent = torch.randn(N, requires_grad=True)
    for clip_vloss in (False, True):
        for reduction in ("mean", "sum"):
            pg, vl, el = ppo_losses(lp_new, lp_old, adv, v_new, v_old, ret, ent, 0.2, clip_vloss, reduction)
            torch.testing.assert_close(pg, policy_loss(lp_new, lp_old, adv, 0.2, reduction))
            torch.testing.assert_close(vl, value_loss(v_new, v_old, ret, 0.2, clip_vloss, reduction))
            torch.testing.assert_close(el, entropy_loss(ent, reduction))
