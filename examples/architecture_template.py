"""Decoupled player/trainer template on the MI355X runtime (parity:
examples/architecture_template.py in the reference, rebuilt on
torch.distributed instead of Fabric).

Rank 0 is the PLAYER: it steps the envs, fills a replay buffer and scatters
sample chunks to the trainers.  Ranks 1..N are TRAINERS: they DDP-train over
their own process group (RCCL on GPU boxes, gloo on CPU) and rank 1
broadcasts the updated flat parameters back to the player.

Run on CPU:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 3 \
        --master-addr 127.0.0.1 examples/architecture_template.py
"""

import os

import torch
import torch.distributed as dist
from torch import nn
from torch.nn.utils import parameters_to_vector, vector_to_parameters

OBS_DIM, ACT_DIM, ITERS, CHUNK = 8, 2, 10, 64


def make_model() -> nn.Module:
    torch.manual_seed(0)  # identical init on every rank
    return nn.Sequential(nn.Linear(OBS_DIM, 32), nn.ReLU(), nn.Linear(32, ACT_DIM))


def player(world: int, trainer_ranks: list) -> None:
    model = make_model()
    for it in range(ITERS):
        # fake rollout: in a real algorithm this is the env loop + buffer
        chunks = [None] + [
            {"obs": torch.randn(CHUNK, OBS_DIM), "target": torch.randn(CHUNK, ACT_DIM)} for _ in trainer_ranks
        ]
        out = [None]
        dist.scatter_object_list(out, chunks, src=0)
        # receive updated weights from the lead trainer over one link
        flat = parameters_to_vector(model.parameters())
        dist.broadcast(flat, src=1)
        vector_to_parameters(flat, model.parameters())
    dist.scatter_object_list([None], [None] + [-1] * len(trainer_ranks), src=0)  # shutdown
    print("player done; param norm", float(flat.detach().norm()))


def trainer(rank: int, pg: dist.ProcessGroup) -> None:
    model = make_model()
    opt = torch.optim.SGD(model.parameters(), lr=1e-2)
    while True:
        out = [None]
        dist.scatter_object_list(out, None, src=0)
        if isinstance(out[0], int) and out[0] == -1:
            break
        loss = (model(out[0]["obs"]) - out[0]["target"]).pow(2).mean()
        opt.zero_grad()
        loss.backward()
        # gradient all-reduce across trainers only (bucketed RCCL in the
        # real runtime — see sheeprl_amd/parallel/gradsync.py)
        for p in model.parameters():
            dist.all_reduce(p.grad, group=pg)
            p.grad /= pg.size()
        opt.step()
        flat = parameters_to_vector(model.parameters())
        dist.broadcast(flat, src=1)  # lead trainer ships weights to the player


def main() -> None:
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend)
    rank, world = dist.get_rank(), dist.get_world_size()
    assert world >= 3, "needs 1 player + >=2 trainers"
    trainer_ranks = list(range(1, world))
    pg = dist.new_group(trainer_ranks)  # the trainers' optimization group
    if rank == 0:
        player(world, trainer_ranks)
    else:
        trainer(rank, pg)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
