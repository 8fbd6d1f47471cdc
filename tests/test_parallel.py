"""GradSync correctness: bucketed async all-reduce must equal DDP-style
gradient averaging, including the unused-parameter case (gloo, 2 procs)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch import nn

from sheeprl_amd.parallel.gradsync import GradSync


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.a = nn.Linear(8, 8)
        self.b = nn.Linear(8, 4)
        self.unused = nn.Linear(3, 3)

    def forward(self, x):
        return self.b(torch.relu(self.a(x)))


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)
    net = Net()
    gs = GradSync(net, bucket_cap_mb=1)
    gs.broadcast_params(src=0)
    torch.manual_seed(100 + rank)
    x = torch.randn(16, 8)
    loss = net(x).pow(2).mean()
    loss.backward()
    gs.finalize()
    grads = {n: p.grad.clone() if p.grad is not None else None for n, p in net.named_parameters()}
    # reference: all-reduce raw per-rank grads computed independently
    torch.manual_seed(0)
    net2 = Net()
    for p2, p in zip(net2.parameters(), net.parameters()):
        p2.data.copy_(p.data)
    torch.manual_seed(100 + rank)
    loss2 = net2(torch.randn(16, 8)).pow(2).mean()
    loss2.backward()
    for n, p2 in net2.named_parameters():
        if p2.grad is None:
            continue
        g = p2.grad.clone()
        dist.all_reduce(g)
        g /= world
        assert torch.allclose(grads[n], g, atol=1e-6), f"mismatch on {n}"
    dist.destroy_process_group()
    q.put(rank)


@pytest.mark.timeout(120)
def test_gradsync_matches_ddp_average():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_worker, args=(r, 2, 29531, q)) for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(120)
    assert all(p.exitcode == 0 for p in ps)


def _collectives_worker(rank, world_size, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from sheeprl_amd.parallel import Runtime

        rt = Runtime(devices=world_size, accelerator="cpu")
        rt.global_rank = rank
        rt.world_size = world_size

        # object scatter with equal-size chunks (the decoupled rollout path)
        out = [None]
        chunks = [None] + [{"a": i + 1} for i in range(world_size - 1)] if rank == 0 else None
        rt.scatter_object_list(out, chunks, src=0)
        if rank > 0:
            assert out[0] == {"a": rank}

        # object broadcast
        payload = [{"x": 42}] if rank == 0 else [None]
        rt.broadcast_object_list(payload, src=0)
        assert payload[0] == {"x": 42}

        # tensor all_reduce through the wrapper (mean semantics)
        t = torch.tensor([float(rank + 1)])
        out_t = rt.all_reduce(t, op="sum")
        assert out_t.item() == sum(range(1, world_size + 1))
        assert rt.all_reduce(t, op="mean").item() == pytest.approx(1.5)

        # gather_object to rank 0
        gathered = rt.gather_object({"r": rank}, dst=0)
        if rank == 0:
            assert [g["r"] for g in gathered] == list(range(world_size))

        # rank-independent metric aggregation (all-gathered per-rank values)
        from sheeprl_amd.utils.metric import RankIndependentMetricAggregator

        agg = RankIndependentMetricAggregator({"m": "mean"})
        agg.update("m", float(rank + 1))
        per_rank = agg.compute()
        assert [d["m"] for d in per_rank] == [1.0, 2.0][:world_size]
    finally:
        dist.destroy_process_group()


def test_runtime_object_collectives_world2():
    mp.spawn(_collectives_worker, args=(2, 29581), nprocs=2, join=True)


class ConvNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.c1 = nn.Conv2d(3, 8, 3, padding=1)
        self.c2 = nn.Conv2d(8, 4, 3, padding=1)

    def forward(self, x):
        return self.c2(torch.relu(self.c1(x))).mean((-1, -2))


def _chlast_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        net = ConvNet().to(memory_format=torch.channels_last)
        gs = GradSync(net, bucket_cap_mb=1)
        gs.broadcast_params(src=0)
        torch.manual_seed(200 + rank)
        x = torch.randn(4, 3, 8, 8).to(memory_format=torch.channels_last)
        net(x).pow(2).mean().backward()
        gs.finalize()  # channels_last grads must flatten into the buckets
        torch.manual_seed(0)
        ref = ConvNet()
        for rp, p in zip(ref.parameters(), net.parameters()):
            rp.data.copy_(p.data)
        torch.manual_seed(200 + rank)
        ref(torch.randn(4, 3, 8, 8)).pow(2).mean().backward()
        for (n, p), rp in zip(net.named_parameters(), ref.parameters()):
            g = rp.grad.clone()
            dist.all_reduce(g)
            g /= world
            assert torch.allclose(p.grad, g, atol=1e-6), f"mismatch on {n}"
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_gradsync_channels_last_convs():
    """Conv grads under channels_last (the setup_module path on GPU boxes)
    must bucket correctly — view(-1) on permuted strides would raise."""
    mp.spawn(_chlast_worker, args=(2, 29591), nprocs=2, join=True)


def test_flat_params_roundtrip_channels_last():
    """params_to_flat/flat_to_params must survive channels_last conv weights
    and preserve the receiver's memory format (runtime.py helpers used for
    decoupled weight transport)."""
    from sheeprl_amd.parallel import flat_to_params, params_to_flat

    torch.manual_seed(0)
    src = ConvNet().to(memory_format=torch.channels_last)
    dst = ConvNet().to(memory_format=torch.channels_last)
    vec = params_to_flat(src.parameters())
    flat_to_params(vec, dst.parameters())
    for a, b in zip(src.parameters(), dst.parameters()):
        assert torch.equal(a, b)
    assert dst.c1.weight.is_contiguous(memory_format=torch.channels_last)


def _moments_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from sheeprl_amd.algos.dreamer_v3.utils import Moments
        from sheeprl_amd.parallel import Runtime

        rt = Runtime(devices=world, accelerator="cpu")
        rt.global_rank = rank
        rt.world_size = world
        m = Moments(decay=0.0)  # decay 0: buffers become this batch's quantiles
        x = torch.arange(10, dtype=torch.float32) + rank * 10  # rank0: 0..9, rank1: 10..19
        low, invscale = m(x, rt)
        # quantiles of the GLOBAL gathered sample 0..19, identical on all ranks
        full = torch.arange(20, dtype=torch.float32)
        assert torch.allclose(low, torch.quantile(full, 0.05), atol=1e-5)
        assert torch.allclose(low + invscale, torch.quantile(full, 0.95), atol=1e-5)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_moments_percentiles_are_global(tmp_path):
    """DV3 return normalization must use the ALL-GATHERED sample so every
    rank scales advantages identically (reference dreamer_v3/utils.py:56-63)."""
    mp.spawn(_moments_worker, args=(2, 29601), nprocs=2, join=True)
