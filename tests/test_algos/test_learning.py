"""Learning-direction unit tests: the losses must push parameters the right
way on analytic problems (sign-convention regressions are otherwise invisible
to shape/smoke tests).  Reference parity: sheeprl/algos/sac/loss.py:10-28,
sheeprl/algos/ppo/loss.py:6-63."""

import torch

from sheeprl_amd.algos.ppo.loss import policy_loss as ppo_policy_loss
from sheeprl_amd.algos.sac.agent import SACActor, SACCritic
from sheeprl_amd.algos.sac.loss import critic_loss, entropy_loss, policy_loss


def test_sac_actor_climbs_analytic_q():
    torch.manual_seed(0)
    actor = SACActor(observation_dim=3, action_dim=1, hidden_size=32)
    opt = torch.optim.Adam(actor.parameters(), lr=1e-2)
    obs = torch.randn(64, 3)
    alpha = torch.tensor(0.01)
    for _ in range(300):
        a, logp = actor(obs)
        q = -(a - 0.7).pow(2)  # analytic critic, optimum at a=0.7
        loss = policy_loss(alpha, logp, q)
        opt.zero_grad()
        loss.backward()
        opt.step()
    greedy = actor.get_greedy_actions(obs)
    assert (greedy - 0.7).abs().mean() < 0.1


def test_sac_critic_fits_targets():
    torch.manual_seed(0)
    critic = SACCritic(observation_dim=4, hidden_size=32, num_critics=2)
    opt = torch.optim.Adam(critic.parameters(), lr=1e-2)
    obs = torch.randn(64, 3)
    act = torch.randn(64, 1)
    tgt = (obs.sum(-1, keepdim=True) + act).tanh()
    before = critic_loss(critic(obs, act), tgt, 2).item()
    for _ in range(200):
        loss = critic_loss(critic(obs, act), tgt, 2)
        opt.zero_grad()
        loss.backward()
        opt.step()
    after = critic_loss(critic(obs, act), tgt, 2).item()
    assert after < before * 0.05


def test_sac_alpha_moves_toward_target_entropy():
    log_alpha = torch.nn.Parameter(torch.tensor(0.0))
    opt = torch.optim.SGD([log_alpha], lr=0.1)
    # entropy below target (logp high) -> alpha must grow
    loss = entropy_loss(log_alpha, torch.full((8, 1), 2.0), target_entropy=-1.0)
    opt.zero_grad()
    loss.backward()
    opt.step()
    assert log_alpha.item() > 0.0
    # entropy above target (logp very negative) -> alpha must shrink
    log_alpha.data.zero_()
    loss = entropy_loss(log_alpha, torch.full((8, 1), -5.0), target_entropy=-1.0)
    opt.zero_grad()
    loss.backward()
    opt.step()
    assert log_alpha.item() < 0.0


def test_ppo_policy_loss_directions():
    # positive advantage -> gradient should raise the new logprob
    lp = torch.nn.Parameter(torch.tensor([0.0]))
    loss = ppo_policy_loss(lp, torch.tensor([0.0]), torch.tensor([1.0]), clip_coef=0.2)
    loss.backward()
    assert lp.grad.item() < 0  # minimizing moves logprob up
    # clipping: once ratio exceeds 1+clip, a positive advantage gives no grad
    lp2 = torch.nn.Parameter(torch.tensor([1.0]))  # ratio e^1 >> 1.2
    loss2 = ppo_policy_loss(lp2, torch.tensor([0.0]), torch.tensor([1.0]), clip_coef=0.2)
    loss2.backward()
    assert abs(lp2.grad.item()) < 1e-8
    # negative advantage with ratio below 1-clip is likewise clipped flat
    lp3 = torch.nn.Parameter(torch.tensor([-1.0]))
    loss3 = ppo_policy_loss(lp3, torch.tensor([0.0]), torch.tensor([-1.0]), clip_coef=0.2)
    loss3.backward()
    assert abs(lp3.grad.item()) < 1e-8


def test_dv3_kl_balancing_gradient_routing():
    """dyn term must push the PRIOR toward the (detached) posterior; repr term
    must push the POSTERIOR toward the (detached) prior; free-nats clamp kills
    gradients when KL is below the floor (dreamer_v3/loss.py:39-55)."""
    from sheeprl_amd.algos.dreamer_v3.loss import categorical_kl

    torch.manual_seed(0)
    post = torch.nn.Parameter(torch.randn(4, 8, 6))
    prior = torch.nn.Parameter(torch.randn(4, 8, 6))

    dyn = categorical_kl(post.detach(), prior)
    dyn.mean().backward()
    assert post.grad is None and prior.grad is not None and prior.grad.abs().sum() > 0
    g_prior = prior.grad.clone()
    prior.grad = None

    rep = categorical_kl(post, prior.detach())
    rep.mean().backward()
    assert prior.grad is None and post.grad is not None and post.grad.abs().sum() > 0

    # a gradient step on the prior reduces the dynamic KL
    with torch.no_grad():
        prior2 = prior - 5.0 * g_prior
    assert categorical_kl(post.detach(), prior2).mean() < dyn.mean()

    # free nats: identical distributions (KL=0 < floor) give zero gradient
    same = torch.nn.Parameter(torch.randn(4, 8, 6))
    kl = categorical_kl(same.detach(), same)
    clamped = torch.maximum(kl, torch.full_like(kl, 1.0))
    clamped.mean().backward()
    assert same.grad.abs().max() == 0
