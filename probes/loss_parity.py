"""Learning-quality parity: eager-fp32 module path vs graphed-bf16 fused path.

Trains two DV3-XS instances from the same seed on the same synthetic replay
stream and tracks each arm's world-model loss on a SHARED held-out batch,
evaluated with the same eager evaluator (module path, fp32 math) for both.
This guards the composition of every fused kernel + hipGraph against the
plain PyTorch path — unit tolerance tests cover each
kernel, this covers the whole gradient step.

    python probes/loss_parity.py [steps] [out.json]
"""

from __future__ import annotations

import json
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

OVERRIDES = [
    "algo=dreamer_v3_XS",
    "algo.per_rank_batch_size=8",
    "algo.per_rank_sequence_length=16",
    "buffer.size=512",
]


def build_arm(precision: str, fused: bool, seed: int = 11):
    os.environ["SHEEPRL_AMD_SCAN_IMPL"] = "v1"
    import bench

    cfg = bench._build(OVERRIDES + [
        f"runtime.precision={precision}",
        f"algo.fused_scan={'True' if fused else 'False'}",
        f"algo.fused_imagination={'True' if fused else 'False'}",
    ], "cuda")
    cfg.seed = seed
    runtime, envs, models, optims, moments, rb = bench._setup(cfg, 0, 1)
    n_act = int(envs.single_action_space.n)
    obs, step_data = bench._prefill(cfg, envs, rb, n_steps=96)
    envs.close()
    return cfg, runtime, models, optims, moments, rb, n_act


@torch.no_grad()
def eval_wm_loss(cfg, runtime, models, batch) -> float:
    """World-model loss on a fixed batch via the EAGER module path in fp32
    (identical evaluator for both arms)."""
    from sheeprl_amd.algos.dreamer_v3.loss import categorical_kl
    from sheeprl_amd import ops
    import torch.distributions as td

    world_model = models[0]
    device = runtime.device
    dtype = runtime.param_dtype
    bo = {k: ops.normalize_obs(batch[k]).to(dtype) for k in cfg.algo.cnn_keys.encoder}
    data = dict(batch)
    data["is_first"][0, :] = torch.ones_like(data["is_first"][0, :])
    batch_actions = torch.cat((torch.zeros_like(data["actions"][:1]), data["actions"][:-1]), dim=0)
    T, B = data["is_first"].shape[:2]
    H = cfg.algo.world_model.recurrent_model.recurrent_state_size
    S, K = cfg.algo.world_model.stochastic_size, cfg.algo.world_model.discrete_size
    embedded = world_model.encoder(bo)
    _ir, _ip = world_model.rssm.get_initial_states((1, B))
    init = (_ir.contiguous(), _ip.contiguous())
    posterior = torch.zeros(1, B, S, K, device=device, dtype=dtype)
    rec = torch.zeros(1, B, H, device=device, dtype=dtype)
    hs, zs, ms = [], [], []
    for i in range(T):
        rec, posterior, plogits = world_model.rssm.dynamic_posterior(
            posterior, rec, batch_actions[i : i + 1], embedded[i : i + 1],
            data["is_first"][i : i + 1], initial_states=init,
        )
        hs.append(rec[0])
        zs.append(posterior.reshape(B, S * K))
        ms.append(plogits[0])
    recurrent_states = torch.stack(hs)
    posteriors_logits = torch.stack(ms)
    priors_logits = world_model.rssm.transition_logits(recurrent_states)
    latent = torch.cat((torch.stack(zs), recurrent_states), -1)
    recon = world_model.observation_model(latent)
    obs_loss = sum(
        ((recon[k].float() - bo[k].float()) ** 2).sum(dim=tuple(range(2, recon[k].dim())))
        for k in cfg.algo.cnn_keys.decoder
    )
    from sheeprl_amd.distributions import TwoHotEncodingDistribution

    rl = world_model.reward_model(latent).float()
    reward_loss = -TwoHotEncodingDistribution(rl, dims=1).log_prob(data["rewards"].float())
    pl = priors_logits.float().view(T, B, S, K)
    ql = posteriors_logits.float().view(T, B, S, K)
    free = torch.full((T, B), cfg.algo.world_model.kl_free_nats, device=device)
    dyn = cfg.algo.world_model.kl_dynamic * torch.maximum(categorical_kl(ql.detach(), pl), free)
    rep = cfg.algo.world_model.kl_representation * torch.maximum(categorical_kl(ql, pl.detach()), free)
    cl = world_model.continue_model(latent).float()
    cont_loss = cfg.algo.world_model.continue_scale_factor * -td.Independent(
        td.Bernoulli(logits=cl), 1
    ).log_prob(1 - data["terminated"].float())
    total = (cfg.algo.world_model.kl_regularizer * (dyn + rep) + obs_loss + reward_loss + cont_loss).mean()
    return float(total.item())


def run(steps: int = 400, out_path: str = "profiles/loss_parity.json") -> dict:
    from sheeprl_amd.algos.dreamer_v3.dreamer_v3 import train, _capture_train_step
    from sheeprl_amd.utils.metric import MetricAggregator

    MetricAggregator.disabled = True
    arms = {}
    eval_batch = None
    for name, (precision, fused, graphs) in {
        "eager_fp32": ("fp32", False, False),
        "fused_bf16_graphed": ("bf16", True, True),
    }.items():
        cfg, runtime, models, optims, moments, rb, n_act = build_arm(precision, fused)
        torch.manual_seed(123)
        sample_g = torch.Generator()
        sample_g.manual_seed(99)
        wm, actor, critic, target_critic, player = models
        wo, ao, co = optims
        seq = cfg.algo.per_rank_sequence_length
        bsz = cfg.algo.per_rank_batch_size
        actions_dim = [n_act]
        if eval_batch is None:
            s = rb.sample_tensors(bsz, sequence_length=seq, n_samples=1, device=runtime.device)
            eval_batch = {k: v[0].clone() for k, v in s.items()}
        curve = []

        def train_fn(batch):
            train(runtime, wm, actor, critic, target_critic, wo, ao, co,
                  batch, None, cfg, False, actions_dim, moments)

        graphed = None
        losses_at = list(range(0, steps + 1, max(1, steps // 16)))
        from sheeprl_amd import ops as _o

        for it in range(steps + 1):
            if it in losses_at:
                curve.append(round(eval_wm_loss(cfg, runtime, models, eval_batch), 4))
            if it == steps:
                break
            s = rb.sample_tensors(bsz, sequence_length=seq, n_samples=1, device=runtime.device)
            batch = {k: v[0] for k, v in s.items()}
            _o.ema_update_(list(target_critic.parameters()), list(critic.parameters()), cfg.algo.critic.tau)
            if graphs and graphed is None and it == 3:
                graphed, _graph_metrics = _capture_train_step(
                    runtime, wm, actor, critic, target_critic, wo, ao, co,
                    batch, cfg, False, actions_dim, moments,
                )
            if graphed is not None:
                graphed(batch)
            else:
                train_fn(batch)
        arms[name] = curve
    first_key = "eager_fp32"
    a = arms[first_key]
    b = arms["fused_bf16_graphed"]
    tail = max(1, len(a) // 4)
    rel_final = abs(sum(a[-tail:]) / tail - sum(b[-tail:]) / tail) / max(abs(sum(a[-tail:]) / tail), 1e-6)
    result = {
        "steps": steps,
        "eval_points": len(a),
        "eager_fp32": a,
        "fused_bf16_graphed": b,
        "rel_final_divergence": round(rel_final, 4),
        "both_learning": a[-1] < a[0] and b[-1] < b[0],
    }
    Path(out_path).parent.mkdir(parents=True, exist_ok=True)
    with open(out_path, "w") as fh:
        json.dump(result, fh, indent=1)
    print(json.dumps(result))
    return result


if __name__ == "__main__":
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 400
    out = sys.argv[2] if len(sys.argv) > 2 else "profiles/loss_parity.json"
    run(steps, out)
