"""nlg_gru task: next-word GRU LM with tied embedding/unembedding.

Reference: experiments/nlg_gru/model.py:11-133 (custom GRU2 cell iterated
per-step in Python).  Contract kept: ``loss`` masks padding (< 0), runs the
recurrence over ``x[:, :-1]`` and scores T positions — the hidden-state
stack includes h0, so position t predicts token t from tokens < t.
Accuracy rejects OOV (id 0) predictions unless ``OOV_correct``.

MI355X-native divergence: the input projection ``W_ih @ x_t`` for ALL
timesteps is one hipBLASLt GEMM over [B*T, E] up front; only the
hidden-hidden GEMM runs in the per-step loop.  The gate math per step is a
single fused elementwise op via ``ops.gru_gates`` (HIP kernel when built;
torch fallback otherwise) instead of 6 eager kernels.
"""

from typing import Tuple

import torch
from torch import Tensor, nn

from msrflute_amd.core.model import BaseModel
from msrflute_amd.ops import gru_gates
from msrflute_amd.utils import to_device


class GRUCellSeq(nn.Module):
    """GRU recurrence over a full sequence with precomputed input gates."""

    def __init__(self, input_size, hidden_size):
        super().__init__()
        self.hidden_size = hidden_size
        self.w_ih = nn.Linear(input_size, 3 * hidden_size, bias=True)
        self.w_hh = nn.Linear(hidden_size, 3 * hidden_size, bias=True)

    def forward(self, emb: Tensor) -> Tuple[Tensor, Tensor]:
        B, T, _ = emb.shape
        g_i = self.w_ih(emb)  # one GEMM for every step: [B, T, 3H]
        from msrflute_amd.ops.lstm import fused_gru_available, fused_gru_seq
        if T > 0 and fused_gru_available(self.hidden_size, emb):
            # whole recurrence = ONE kernel launch (csrc/gru_seq.hip)
            h_seq = fused_gru_seq(g_i, self.w_hh.weight, self.w_hh.bias)
            hiddens = torch.cat(
                [emb.new_zeros(B, 1, self.hidden_size), h_seq], dim=1)
            return hiddens, h_seq[:, -1]
        h = emb.new_zeros(B, self.hidden_size)
        hiddens = [h]
        for t in range(T):
            g_h = self.w_hh(h)
            h = gru_gates(g_i[:, t], g_h, h)
            hiddens.append(h)
        return torch.stack(hiddens, dim=1), h  # [B, T+1, H], h_T


class TiedEmbedding(nn.Module):
    def __init__(self, vocab_size, embedding_size):
        super().__init__()
        self.table = nn.Parameter(torch.zeros(vocab_size, embedding_size))
        self.unembedding_bias = nn.Parameter(torch.zeros(vocab_size))
        delta = (3 / embedding_size) ** 0.5
        nn.init.uniform_(self.table, -delta, delta)

    def embed(self, ids: Tensor) -> Tensor:
        return nn.functional.embedding(ids, self.table)

    def unembed(self, h: Tensor) -> Tensor:
        return h @ self.table.t() + self.unembedding_bias


class GRU(BaseModel):
    def __init__(self, model_config, **kwargs):
        super().__init__()
        self.vocab_size = model_config["vocab_size"]
        self.embedding = TiedEmbedding(self.vocab_size,
                                       model_config["embed_dim"])
        self.rnn = GRUCellSeq(model_config["embed_dim"],
                              model_config["hidden_dim"])
        self.squeeze = nn.Linear(model_config["hidden_dim"],
                                 model_config["embed_dim"], bias=False)
        self.OOV_correct = model_config.get("OOV_correct", False)
        p = float(model_config.get("dropout", 0.0) or 0.0)
        self.drop = nn.Dropout(p) if p > 0 else None

    def forward(self, x: Tensor) -> Tensor:
        hiddens, _ = self.rnn(self.embedding.embed(x))
        if self.drop is not None:
            hiddens = self.drop(hiddens)
        return self.embedding.unembed(self.squeeze(hiddens))

    def _masked_logits(self, input):
        x = to_device(input["x"] if isinstance(input, dict) else input).long()
        mask = (x >= 0).view(-1)
        x = x * (x >= 0).long()  # pads -> id 0 (ignored through the mask)
        logits = self.forward(x[:, :-1])  # [B, T, V] incl. h0 position
        targets = x.reshape(-1)[mask]
        preds = logits.reshape(-1, self.vocab_size)[mask]
        return preds, targets, x, logits

    def loss(self, input) -> Tensor:
        preds, targets, _, _ = self._masked_logits(input)
        return nn.functional.cross_entropy(preds, targets)

    def inference(self, input):
        preds, targets, x, logits = self._masked_logits(input)
        top = torch.argmax(preds, dim=1)
        if self.OOV_correct:
            acc = top.eq(targets).float().mean()
        else:
            acc = (top.eq(targets) & (top != 0)).float().mean()
        # 'output' is the full [B, T, V] logit tensor — the privacy leakage
        # metric indexes it per position (extensions/privacy/metrics.py)
        return {"output": logits, "acc": acc.item(), "batch_size": x.shape[0]}
