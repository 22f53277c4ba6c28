"""Fused LSTM sequence kernels vs torch nn.LSTM (GPU only): forward
outputs and ALL gradients (weights, biases, input) must match."""

import pytest
import torch

@pytest.mark.gpu
def test_fused_lstm_matches_nn_lstm():
    from msrflute_amd.ops.lstm import FusedLSTM

    torch.manual_seed(3)
    B, T, E, H, L = 4, 80, 8, 256, 2
    fused = FusedLSTM(E, H, num_layers=L).cuda()
    ref = torch.nn.LSTM(E, H, num_layers=L, batch_first=True).cuda()
    ref.load_state_dict(fused.state_dict())

    x1 = torch.randn(B, T, E, device="cuda", requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)

    out1, _ = fused(x1)
    out2, _ = ref(x2)
    assert torch.allclose(out1, out2, rtol=1e-4, atol=1e-5), \
        (out1 - out2).abs().max().item()

    g = torch.randn_like(out1)
    out1.backward(g)
    out2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, rtol=1e-3, atol=1e-5), \
        (x1.grad - x2.grad).abs().max().item()
    for (n1, p1), (n2, p2) in zip(fused.named_parameters(),
                                  ref.named_parameters()):
        assert n1 == n2
        assert torch.allclose(p1.grad, p2.grad, rtol=1e-3, atol=1e-4), \
            (n1, (p1.grad - p2.grad).abs().max().item())


def test_fused_lstm_cpu_fallback_is_exact():
    from msrflute_amd.ops.lstm import FusedLSTM
    torch.manual_seed(5)
    fused = FusedLSTM(8, 256, num_layers=2)
    ref = torch.nn.LSTM(8, 256, num_layers=2, batch_first=True)
    ref.load_state_dict(fused.state_dict())
    x = torch.randn(3, 12, 8)
    o1, _ = fused(x)
    o2, _ = ref(x)
    assert torch.allclose(o1, o2)
