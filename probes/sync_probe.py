"""Find stream-sync / capture-unsafe ops in the DV3 train step."""
import os
import sys
import traceback
import warnings

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ["SHEEPRL_AMD_NO_GRAPHS"] = "1"

import torch

from bench import _build, _prefill, _setup
from sheeprl_amd.algos.dreamer_v3.dreamer_v3 import train
from sheeprl_amd.utils.metric import MetricAggregator

cfg = _build([], "cuda")
runtime, envs, models, optims, moments, rb = _setup(cfg, 0, 1)
world_model, actor, critic, target_critic, player = models
world_optimizer, actor_optimizer, critic_optimizer = optims
device = runtime.device
seq_len = cfg.algo.per_rank_sequence_length
obs, step_data = _prefill(cfg, envs, rb, n_steps=seq_len + 10)
aggregator = MetricAggregator({})
MetricAggregator.disabled = True
actions_dim = [envs.single_action_space.n]


def train_fn(batch):
    train(runtime, world_model, actor, critic, target_critic,
          world_optimizer, actor_optimizer, critic_optimizer,
          batch, aggregator, cfg, False, actions_dim, moments)


def get_batch():
    s = rb.sample_tensors(cfg.algo.per_rank_batch_size, sequence_length=seq_len, n_samples=1, device=device)
    return {k: v[0] for k, v in s.items()}


# 2 warmups
for _ in range(2):
    train_fn(get_batch())
torch.cuda.synchronize()

print("=== sync debug (warn) pass ===", flush=True)
torch.cuda.set_sync_debug_mode(1)
with warnings.catch_warnings(record=True) as ws:
    warnings.simplefilter("always")
    train_fn(get_batch())
torch.cuda.set_sync_debug_mode(0)
seen = set()
for w in ws:
    m = str(w.message)
    if m not in seen:
        seen.add(m)
        print("SYNC:", m, flush=True)
print(f"({len(ws)} sync events, {len(seen)} unique)")

print("=== capture attempt with traceback ===", flush=True)
static = get_batch()
side = torch.cuda.Stream()
side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    train_fn(static)
torch.cuda.current_stream().wait_stream(side)
torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
try:
    with torch.cuda.graph(g):
        train_fn(static)
    print("CAPTURE OK", flush=True)
    g.replay()
    torch.cuda.synchronize()
    print("REPLAY OK", flush=True)
except Exception:
    traceback.print_exc()
