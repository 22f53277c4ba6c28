"""Fed-Shakespeare task: 2-layer char LSTM next-character model.

Benchmark task 4 (reference: experiments/nlp_rnn_fedshakespeare/model.py:12-67;
architecture from the FedML NLP RNN).  Vocab 90 with pad id 0; loss is
per-position CE with pad ignored; accuracy is masked next-char accuracy.
"""

import torch
from torch import nn
from torch.nn import functional as F

from msrflute_amd.core.model import BaseModel
from msrflute_amd.ops.lstm import FusedLSTM
from msrflute_amd.utils import to_device


class CharLSTM(nn.Module):
    def __init__(self, vocab_size=90, embedding_dim=8, hidden_size=256):
        super().__init__()
        self.embeddings = nn.Embedding(vocab_size, embedding_dim,
                                       padding_idx=0)
        # FusedLSTM: nn.LSTM param names/semantics, gfx950 sequence kernels
        # on GPU (MIOpen RNN is slow at FL batch sizes and capture-unsafe)
        self.lstm = FusedLSTM(embedding_dim, hidden_size, num_layers=2,
                              batch_first=True)
        self.fc = nn.Linear(hidden_size, vocab_size)

    def forward(self, x):
        out, _ = self.lstm(self.embeddings(x))
        # [B, T, V] -> [B, V, T] for per-position cross-entropy
        return self.fc(out).transpose(1, 2)


class RNN(BaseModel):
    def __init__(self, model_config):
        super().__init__()
        self.net = CharLSTM(
            vocab_size=model_config.get("vocab_size", 90),
            embedding_dim=model_config.get("embed_dim", 8),
            hidden_size=model_config.get("hidden_dim", 256))

    def loss(self, input):
        x = to_device(input["x"]).long()
        y = to_device(input["y"]).long()
        return F.cross_entropy(self.net(x), y, ignore_index=0)

    def inference(self, input):
        x = to_device(input["x"]).long()
        y = to_device(input["y"]).long()
        output = self.net(x)
        pred = torch.argmax(output, dim=1)
        mask = y != 0
        denom = mask.sum().clamp_min(1)
        acc = ((pred == y) & mask).sum().item() / denom.item()
        return {"output": output, "acc": acc, "batch_size": x.shape[0]}
