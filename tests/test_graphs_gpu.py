"""hipGraph client-step fast path vs eager epoch (numerics, GPU only)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


class TinyNet(torch.nn.Module):
    """Dropout-free so graph vs eager is deterministic given same batches."""

    def __init__(self):
        super().__init__()
        self.fc1 = torch.nn.Linear(32, 64)
        self.fc2 = torch.nn.Linear(64, 5)

    def forward(self, x):
        return self.fc2(torch.relu(self.fc1(x)))


class TinyModel(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.net = TinyNet()

    def loss(self, batch):
        out = self.net(batch["x"])
        return torch.nn.functional.cross_entropy(out, batch["y"])


def _batches(n_batches=6, bs=16, seed=0):
    g = torch.Generator().manual_seed(seed)
    return [(torch.randn(bs, 32, generator=g).cuda(),
             torch.randint(0, 5, (bs,), generator=g).cuda())
            for _ in range(n_batches)]


def _run_eager(model, arena, batches, lr, max_norm):
    from msrflute_amd import ops
    stats = torch.zeros(2, device="cuda")
    loss_acc = torch.zeros((), device="cuda")
    for x, y in batches:
        arena.zero_grad()
        loss = model.loss({"x": x, "y": y})
        loss.backward()
        ops.clip_by_norm(arena.grad, max_norm)
        stats += ops.sum_sumsq(arena.grad)
        ops.sgd_step(arena.data, arena.grad, None, lr=lr)
        loss_acc += loss.detach()
    return stats, loss_acc


def test_graphed_epoch_matches_eager():
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.graphs import GraphCache

    torch.manual_seed(5)
    m1 = TinyModel().cuda()
    m2 = TinyModel().cuda()
    m2.load_state_dict(m1.state_dict())
    a1, a2 = ParameterArena(m1), ParameterArena(m2)
    batches = _batches()
    lr, max_norm = 0.05, 1.0

    s1, l1 = _run_eager(m1, a1, batches, lr, max_norm)

    cache = GraphCache(m2, a2, {"type": "sgd", "lr": lr}, max_norm)
    assert cache.supports()
    cache.set_lr(lr)
    g = cache.get(*batches[0])
    g.reset_client()
    for x, y in batches:
        g.run_batch(x, y)
    torch.cuda.synchronize()

    assert torch.allclose(a1.data, a2.data, rtol=1e-5, atol=1e-6)
    assert torch.allclose(s1, g.stats_acc, rtol=1e-4, atol=1e-5)
    assert torch.allclose(l1, g.loss_acc, rtol=1e-5, atol=1e-6)


def test_graph_replay_uses_updated_lr():
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.graphs import GraphCache

    torch.manual_seed(6)
    m = TinyModel().cuda()
    a = ParameterArena(m)
    cache = GraphCache(m, a, {"type": "sgd", "lr": 0.1}, None)
    x = torch.randn(8, 32).cuda()
    y = torch.randint(0, 5, (8,)).cuda()
    cache.set_lr(0.0)  # zero LR: replay must not move the weights
    g = cache.get(x, y)
    g.reset_client()
    w0 = a.data.clone()
    g.run_batch(x, y)
    torch.cuda.synchronize()
    assert torch.allclose(a.data, w0)
    cache.set_lr(0.5)
    g.run_batch(x, y)
    torch.cuda.synchronize()
    assert not torch.allclose(a.data, w0)


def test_whole_epoch_graph_matches_eager():
    """GraphedClientEpoch (one replay per epoch) vs the eager sequence on
    the same shuffle order, incl. a ragged tail (n % bs != 0)."""
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.graphs import GraphCache, epoch_graph_for
    from msrflute_amd import ops

    torch.manual_seed(11)
    m1 = TinyModel().cuda()
    m2 = TinyModel().cuda()
    m2.load_state_dict(m1.state_dict())
    a1, a2 = ParameterArena(m1), ParameterArena(m2)
    lr, max_norm, bs = 0.05, 1.0, 16
    n = 16 * 4 + 7  # ragged tail of 7
    xs = torch.randn(n, 32).cuda()
    ys = torch.randint(0, 5, (n,)).cuda()
    order = torch.randperm(n)

    # eager reference over the same order (full batches only)
    full = order[: (n // bs) * bs]
    batches = [(xs[full[i:i + bs].cuda()], ys[full[i:i + bs].cuda()])
               for i in range(0, len(full), bs)]
    s1, l1 = _run_eager(m1, a1, batches, lr, max_norm)

    cache = GraphCache(m2, a2, {"type": "sgd", "lr": lr}, max_norm)
    cache.set_lr(lr)
    eg = epoch_graph_for(cache, xs, ys, bs)
    assert eg is not None and eg.n_batches == 4
    eg.reset_client()
    eg.run_epoch(xs, ys, order)
    torch.cuda.synchronize()

    assert torch.allclose(a1.data, a2.data, rtol=1e-5, atol=1e-6)
    assert torch.allclose(s1, eg.stats_acc, rtol=1e-4, atol=1e-5)
    assert torch.allclose(l1, eg.loss_acc, rtol=1e-5, atol=1e-6)

    # second client replays with a different order and stays finite
    eg.reset_client()
    eg.run_epoch(xs, ys, torch.randperm(n))
    torch.cuda.synchronize()
    assert torch.isfinite(eg.loss_acc).item()
