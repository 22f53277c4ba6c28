"""Flat contiguous parameter arena.

The central MI355X-first data structure (SURVEY.md §7.1 divergence 1):
every model replica's parameters live in ONE contiguous device buffer and
each ``nn.Parameter``'s storage is a view into it.  Gradients likewise live
in a parallel flat buffer with ``p.grad`` bound to views.  Consequences:

* every hot op (pseudo-gradient, weighted accumulate, clip, DP noise,
  quantization, optimizer step) is a single fused kernel over one
  contiguous fp32 buffer instead of a per-tensor Python loop
  (reference hot loops: SURVEY.md §2.4);
* every collective (round-level all-reduce / broadcast) is one contiguous
  RCCL op — no per-tensor shape handshakes (reference C2/C5 in SURVEY §2.5);
* works identically on CPU (plain torch views) and GPU (HIP kernels).

The reference's equivalents were ``unroll_network``/``update_network``
copies (extensions/privacy/__init__.py:105-126) and per-tensor
send/accumulate loops (core/federated.py:112-124, strategies/utils.py:21-33).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch


class ParameterArena:
    """Rebinds a model's parameters (and optionally grads) into flat buffers."""

    def __init__(self, model: torch.nn.Module, bind_grads: bool = True,
                 dtype: torch.dtype = torch.float32):
        self.model = model
        params = [p for p in model.parameters()]
        self.names: List[str] = [n for n, _ in model.named_parameters()]
        self.shapes: List[torch.Size] = [p.shape for p in params]
        self.numels: List[int] = [p.numel() for p in params]
        self.total: int = sum(self.numels)
        device = params[0].device if params else torch.device("cpu")
        self.device = device
        self.dtype = dtype

        # offsets[i] is the start of param i; offsets[-1] == total
        offs = [0]
        for n in self.numels:
            offs.append(offs[-1] + n)
        self.offsets = offs
        # segment boundary tensor for segmented kernels (LAMB/LARS, per-layer quant)
        self.seg_offsets = torch.tensor(offs, dtype=torch.int64, device=device)

        self.data = torch.empty(self.total, dtype=dtype, device=device)
        for p, off, n in zip(params, offs, self.numels):
            self.data[off:off + n].copy_(p.detach().reshape(-1).to(dtype))
            # Rebind the parameter's storage to the arena view.
            p.data = self.data[off:off + n].view(p.shape)

        self.grad: Optional[torch.Tensor] = None
        if bind_grads:
            self.grad = torch.zeros(self.total, dtype=dtype, device=device)
            self._bind_grads()

    # -- gradient binding -------------------------------------------------
    def _bind_grads(self):
        for p, off, n, shape in zip(self.model.parameters(), self.offsets,
                                    self.numels, self.shapes):
            p.grad = self.grad[off:off + n].view(shape)

    def zero_grad(self):
        """Zero the flat grad buffer.  NOTE: never call
        ``optimizer.zero_grad(set_to_none=True)`` on an arena-bound model —
        it would unbind the views; use this instead."""
        if self.grad is not None:
            self.grad.zero_()
            self._bind_grads()  # cheap; restores any views torn off by user code

    # -- views ------------------------------------------------------------
    def segment_of(self, name: str) -> Tuple[int, int]:
        """(offset, numel) of the named parameter's arena segment."""
        i = self.names.index(name)
        return self.offsets[i], self.numels[i]

    def grad_segment(self, i: int) -> torch.Tensor:
        off, n = self.offsets[i], self.numels[i]
        return self.grad[off:off + n]

    def param_view(self, i: int) -> torch.Tensor:
        off, n = self.offsets[i], self.numels[i]
        return self.data[off:off + n].view(self.shapes[i])

    def clone_data(self) -> torch.Tensor:
        return self.data.clone()

    def copy_data_(self, src: torch.Tensor):
        self.data.copy_(src)

    def new_buffer(self, zero: bool = True) -> torch.Tensor:
        buf = torch.zeros_like(self.data) if zero else torch.empty_like(self.data)
        return buf

    def state(self) -> Tuple[torch.Tensor, torch.Tensor]:
        return self.data, self.grad


def flatten_state_dict(sd) -> torch.Tensor:
    """Flatten a state dict's tensors to one fp32 vector (FedLabels-style
    state-dict aggregation; reference fedlabels.py:84-92 concatenates dicts)."""
    parts = [v.detach().reshape(-1).float() for v in sd.values()
             if torch.is_tensor(v)]
    if not parts:
        return torch.zeros(0)
    return torch.cat(parts)


def unflatten_into_state_dict(flat: torch.Tensor, sd):
    """Inverse of flatten_state_dict: writes values back in iteration order."""
    off = 0
    for k, v in sd.items():
        if not torch.is_tensor(v):
            continue
        n = v.numel()
        v.copy_(flat[off:off + n].view(v.shape).to(v.dtype))
        off += n
    return sd
