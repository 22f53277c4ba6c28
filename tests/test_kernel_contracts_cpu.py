"""CPU-checkable contracts the GPU kernels and the shared shuffle-order
discipline depend on (regressions here produced real GPU bugs this
round — wrong shuffle orders across training paths, and weight-layout
mismatches would silently corrupt the recurrences)."""

import torch

from msrflute_amd.ops.lstm import _pack_fwd, _pack_bwd


def test_pack_fwd_layout_mapping():
    """[4H, H] -> [H/4, 4H, 4] with element (kk, j, d) = W_hh[j, 4kk+d]
    (the forward kernels' per-step b128 weight read)."""
    H = 8
    w = torch.arange(4 * H * H, dtype=torch.float32).reshape(4 * H, H)
    p = _pack_fwd(w)
    assert p.shape == (H // 4, 4 * H, 4) and p.is_contiguous()
    for kk in range(H // 4):
        for j in (0, 3, 4 * H - 1):
            for d in range(4):
                assert p[kk, j, d] == w[j, 4 * kk + d]


def test_pack_bwd_layout_mapping():
    """[4H, H] -> [4H/4, H, 4] with element (jg, h, d) = W_hh[4jg+d, h]
    (the backward kernels' b128 form)."""
    H = 8
    w = torch.arange(4 * H * H, dtype=torch.float32).reshape(4 * H, H)
    p = _pack_bwd(w)
    assert p.shape == (H, H, 4) and p.is_contiguous()
    for jg in range(H):
        for h in (0, 1, H - 1):
            for d in range(4):
                assert p[jg, h, d] == w[4 * jg + d, h]


def test_unshuffled_loader_consumes_no_rng():
    """The graphed-path probe peeks a batch with shuffle off and assumes
    that consumes NO host RNG — otherwise the epoch's randperm would
    shift and the eager/graphed/fused/mega paths would train on
    different shuffle orders for the same (round, client) seed."""
    from msrflute_amd.models.generic_data import ArrayDataLoader
    blob = {"users": ["u0"], "num_samples": [7],
            "user_data": {"u0": {"x": [[float(i)] * 4 for i in range(7)]}},
            "user_data_label": {"u0": list(range(7))}}
    dl = ArrayDataLoader(blob, user_idx=0, mode="train",
                         args={"batch_size": 3})
    torch.manual_seed(1234)
    before = torch.get_rng_state()
    dl.shuffle = False
    batch = next(iter(dl.create_loader()), None)
    assert batch is not None
    assert torch.equal(before, torch.get_rng_state())
    # and the shuffled draw equals the seed's FIRST randperm
    dl.shuffle = True
    torch.manual_seed(77)
    order_direct = torch.randperm(7)
    torch.manual_seed(77)
    xs = torch.cat([b["x"] for b in dl.create_loader()])
    expect = torch.tensor(blob["user_data"]["u0"]["x"])[order_direct]
    assert torch.equal(xs.cpu(), expect)
