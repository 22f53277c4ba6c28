"""ParameterArena invariants + fused optimizers vs torch.optim."""

import torch

from msrflute_amd.ops.arena import (ParameterArena, flatten_state_dict,
                                    unflatten_into_state_dict)
from msrflute_amd.ops.fused_optim import ArenaAdam, ArenaSGD


def small_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(torch.nn.Linear(10, 20), torch.nn.ReLU(),
                               torch.nn.Linear(20, 3))


def test_params_are_views_into_arena():
    m = small_model()
    before = [p.detach().clone() for p in m.parameters()]
    arena = ParameterArena(m)
    for p, b in zip(m.parameters(), before):
        assert torch.equal(p.detach(), b)
        assert p.data_ptr() >= arena.data.data_ptr()
    # writing the arena writes the params
    arena.data.zero_()
    for p in m.parameters():
        assert torch.all(p.detach() == 0)


def test_backward_accumulates_into_grad_arena():
    m = small_model()
    arena = ParameterArena(m)
    x = torch.randn(4, 10)
    m(x).sum().backward()
    assert arena.grad.abs().sum() > 0
    total = sum(float(p.grad.abs().sum()) for p in m.parameters())
    assert abs(float(arena.grad.abs().sum()) - total) < 1e-4


def test_load_state_dict_writes_through():
    m = small_model(0)
    other = small_model(1)
    arena = ParameterArena(m)
    m.load_state_dict(other.state_dict())
    for p, q in zip(m.parameters(), other.parameters()):
        assert torch.equal(p.detach(), q.detach())
    # views intact
    assert m[0].weight.data_ptr() >= arena.data.data_ptr()


def test_flatten_unflatten_state_dict_roundtrip():
    m = small_model()
    sd = m.state_dict()
    flat = flatten_state_dict(sd)
    sd2 = {k: torch.zeros_like(v) for k, v in sd.items()}
    unflatten_into_state_dict(flat, sd2)
    for k in sd:
        assert torch.equal(sd[k], sd2[k])


def _train_steps(model, opt, arena=None, steps=5, seed=42):
    torch.manual_seed(seed)
    for _ in range(steps):
        x = torch.randn(8, 10)
        if arena is not None:
            arena.zero_grad()
        else:
            opt.zero_grad()
        loss = model(x).pow(2).mean()
        loss.backward()
        opt.step()
    return torch.cat([p.detach().reshape(-1) for p in model.parameters()])


def test_arena_sgd_matches_torch_sgd():
    m1, m2 = small_model(7), small_model(7)
    a1 = ParameterArena(m1)
    opt1 = ArenaSGD(a1, lr=0.05, momentum=0.9, weight_decay=1e-4)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.05, momentum=0.9,
                           weight_decay=1e-4)
    w1 = _train_steps(m1, opt1, arena=a1)
    w2 = _train_steps(m2, opt2)
    assert torch.allclose(w1, w2, rtol=1e-5, atol=1e-7)


def test_arena_adam_matches_torch_adam():
    m1, m2 = small_model(9), small_model(9)
    a1 = ParameterArena(m1)
    opt1 = ArenaAdam(a1, lr=0.01)
    opt2 = torch.optim.Adam(m2.parameters(), lr=0.01)
    w1 = _train_steps(m1, opt1, arena=a1)
    w2 = _train_steps(m2, opt2)
    assert torch.allclose(w1, w2, rtol=1e-4, atol=1e-7)


def test_arena_optimizer_state_dict_roundtrip():
    m1 = small_model(11)
    a1 = ParameterArena(m1)
    opt1 = ArenaSGD(a1, lr=0.05, momentum=0.9)
    _train_steps(m1, opt1, arena=a1, steps=3)
    sd = opt1.state_dict()
    assert set(sd.keys()) == {"state", "param_groups"}

    m2 = small_model(11)
    a2 = ParameterArena(m2)
    opt2 = ArenaSGD(a2, lr=0.05, momentum=0.9)
    opt2.load_state_dict(sd)
    assert torch.allclose(opt1.momentum_buf, opt2.momentum_buf)

    # and it is loadable by a plain torch.optim.SGD (checkpoint parity)
    m3 = small_model(11)
    opt3 = torch.optim.SGD(m3.parameters(), lr=0.05, momentum=0.9)
    # torch requires params referenced in state to exist: warm it up
    m3(torch.randn(2, 10)).sum().backward()
    opt3.step()
    opt3.load_state_dict(sd)
    bufs = [opt3.state[p]["momentum_buffer"] for p in m3.parameters()]
    flat = torch.cat([b.reshape(-1) for b in bufs])
    assert torch.allclose(flat, opt1.momentum_buf)
