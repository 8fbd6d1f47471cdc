"""2-rank REAL-RCCL proof on a single MI355X.

Launched under torchrun with --nproc-per-node 2 on a 1-GPU box: both ranks
share cuda:0 (RCCL supports multiple communicators per device in separate
processes).  Exercises exactly the first-contact risks of the driver's 8-GPU
scaling window:

1. GradSync flat-parameter broadcast over RCCL (device tensors);
2. bucketed async all-reduce in post-accumulate-grad hooks during backward,
   with a channels_last conv in the module;
3. the same step captured in a hipGraph (collective INSIDE the graph) and
   replayed 3x.

Each rank computes the expected world-averaged gradients locally (it knows
both ranks' deterministic inputs) and asserts the RCCL result matches.
Prints RCCL_2RANK_OK on success (checked by tests/test_gpu_rccl.py).
"""

from __future__ import annotations

import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

import torch
import torch.distributed as dist
from torch import nn


def make_model(device):
    torch.manual_seed(7)  # identical init on both ranks (broadcast re-checks)
    m = nn.Sequential(
        nn.Conv2d(3, 8, 3, padding=1),
        nn.SiLU(),
        nn.Flatten(),
        nn.Linear(8 * 8 * 8, 16),
    )
    return m.to(device).to(memory_format=torch.channels_last)


def rank_input(rank: int):
    g = torch.Generator().manual_seed(100 + rank)
    return torch.randn(4, 3, 8, 8, generator=g)


def expected_grads(device):
    """Average of both ranks' grads computed in-process (ground truth)."""
    m = make_model(device)
    grads = None
    for r in range(2):
        m.zero_grad()
        m(rank_input(r).to(device)).square().mean().backward()
        g = [p.grad.detach().clone() for p in m.parameters()]
        grads = g if grads is None else [a + b for a, b in zip(grads, g)]
    return [g / 2 for g in grads]


def main() -> None:
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    assert world == 2
    # one rank per device when the box has >= 2 (RCCL refuses co-located
    # ranks: "Duplicate GPU detected" — measured on ROCm 7.2 / NCCL 2.26)
    torch.cuda.set_device(rank % torch.cuda.device_count())
    device = torch.device("cuda", torch.cuda.current_device())
    dist.init_process_group("nccl", rank=rank, world_size=world)

    from sheeprl_amd.parallel.gradsync import GradSync

    model = make_model(device)
    # perturb rank-1 weights so broadcast_params is observable
    if rank == 1:
        with torch.no_grad():
            for p in model.parameters():
                p.add_(1.0)
    gs = GradSync(model, bucket_cap_mb=1)
    gs.broadcast_params(src=0)
    ref = make_model(device)
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.equal(p, q), "broadcast_params diverged from rank-0 init"
    print(f"[rank {rank}] param broadcast OK", flush=True)

    exp = expected_grads(device)
    x = rank_input(rank).to(device)

    # --- eager step: hooks launch bucketed all-reduce during backward
    model.zero_grad()
    model(x).square().mean().backward()
    gs.finalize()
    for p, e in zip(model.parameters(), exp):
        assert torch.allclose(p.grad, e, atol=1e-6), (
            f"eager RCCL grad mismatch: max err {(p.grad - e).abs().max().item()}"
        )
    print(f"[rank {rank}] eager bucketed all-reduce OK", flush=True)

    # --- hipGraph-captured step (collective inside the graph)
    static_x = x.clone()

    def step():
        model.zero_grad(set_to_none=False)
        model(static_x).square().mean().backward()
        gs.finalize()

    # warmup on a side stream (capture protocol)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            step()
    torch.cuda.current_stream().wait_stream(s)

    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        step()
    for it in range(3):
        graph.replay()
        torch.cuda.synchronize()
        for p, e in zip(model.parameters(), exp):
            assert torch.allclose(p.grad, e, atol=1e-6), (
                f"graphed RCCL grad mismatch at replay {it}: "
                f"max err {(p.grad - e).abs().max().item()}"
            )
    print(f"[rank {rank}] hipGraph-captured all-reduce OK (3 replays)", flush=True)

    dist.barrier()
    dist.destroy_process_group()
    if rank == 0:
        print("RCCL_2RANK_OK", flush=True)


if __name__ == "__main__":
    main()
