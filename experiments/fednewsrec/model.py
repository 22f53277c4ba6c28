"""fednewsrec task: attention-based news recommendation (FedNewsRec).

Reference: experiments/fednewsrec/fednewsrec_model.py (hand-rolled
multi-head attention via einsum, DocEncoder conv+attention+attentive
pooling, UserEncoder attention + GRU-tail fusion, frozen pretrained
embeddings) and model.py:47-129 (AUC/MRR/nDCG metrics).

MI355X-native divergences: multi-head attention uses
``F.scaled_dot_product_attention`` (fused flash-style kernel on ROCm)
instead of 4 einsum kernels; the embedding matrix is random-init frozen
(no GloVe download offline — pass ``embedding_path`` to a local ``.npy``
to load real vectors); metrics are computed in torch (no sklearn/nltk).
"""

import numpy as np
import torch
from torch import nn
from torch.nn import functional as F

from msrflute_amd.core.model import BaseModel
from msrflute_amd.utils import to_device

NPRATIO = 4


class AttentivePooling(nn.Module):
    """Additive attention pooling over a sequence (reference
    fednewsrec_model.py:12-42)."""

    def __init__(self, dim, hidden=200, dropout=0.2):
        super().__init__()
        self.drop = nn.Dropout(dropout)
        self.dense = nn.Linear(dim, hidden)
        self.dense2 = nn.Linear(hidden, 1)

    def forward(self, x):                     # [B, L, D] -> [B, D]
        x = self.drop(x)
        att = self.dense2(torch.tanh(self.dense(x))).squeeze(-1)
        att = torch.softmax(att, dim=1)
        return torch.einsum("bld,bl->bd", x, att)


class SelfAttention(nn.Module):
    """Multi-head self-attention via SDPA (reference Attention module,
    fednewsrec_model.py:44-106, was einsum-based)."""

    def __init__(self, dim, heads, head_dim):
        super().__init__()
        self.heads, self.head_dim = heads, head_dim
        out = heads * head_dim
        self.wq = nn.Linear(dim, out, bias=False)
        self.wk = nn.Linear(dim, out, bias=False)
        self.wv = nn.Linear(dim, out, bias=False)
        for w in (self.wq, self.wk, self.wv):
            nn.init.xavier_uniform_(w.weight, gain=np.sqrt(2))

    def forward(self, x):                     # [B, L, D] -> [B, L, H*Hd]
        B, L, _ = x.shape
        q = self.wq(x).view(B, L, self.heads, self.head_dim).transpose(1, 2)
        k = self.wk(x).view(B, L, self.heads, self.head_dim).transpose(1, 2)
        v = self.wv(x).view(B, L, self.heads, self.head_dim).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v)
        return o.transpose(1, 2).reshape(B, L, -1)


class DocEncoder(nn.Module):
    """Title encoder: conv over words -> self-attn -> attentive pooling
    (reference fednewsrec_model.py:128-197)."""

    def __init__(self, embed_dim=300, conv_dim=400, heads=20):
        super().__init__()
        self.drop1 = nn.Dropout(0.2)
        self.conv = nn.Conv1d(embed_dim, conv_dim, 3)
        self.drop2 = nn.Dropout(0.2)
        self.attn = SelfAttention(conv_dim, heads, conv_dim // heads)
        self.drop3 = nn.Dropout(0.2)
        self.pool = AttentivePooling(conv_dim)

    def forward(self, x):                     # [B, T, E] -> [B, D]
        h = self.conv(self.drop1(x).transpose(1, 2)).transpose(1, 2)
        h = self.drop2(F.relu(h))
        h = self.drop3(F.relu(self.attn(h)))
        return self.pool(h)


class UserEncoder(nn.Module):
    """User encoder: self-attn pooled vector fused with a GRU over the
    last 20 clicks via attentive pooling (reference
    fednewsrec_model.py:208-282)."""

    def __init__(self, dim=400, heads=20, tail=20):
        super().__init__()
        self.attn = SelfAttention(dim, heads, dim // heads)
        self.drop = nn.Dropout(0.2)
        self.pool_attn = AttentivePooling(dim)
        self.tail = tail
        self.gru = nn.GRU(dim, dim, batch_first=True)
        self.pool_fuse = AttentivePooling(dim)

    def forward(self, news_vecs):             # [B, H, D] -> [B, D]
        v2 = self.pool_attn(self.drop(self.attn(news_vecs)))
        g, _ = self.gru(news_vecs[:, -self.tail:, :])
        v1 = g[:, -1, :]
        return self.pool_fuse(torch.stack([v1, v2], dim=1))


class FedNewsRecNet(nn.Module):
    def __init__(self, embedding_matrix, freeze_embeddings=True):
        super().__init__()
        self.embedding = nn.Embedding.from_pretrained(
            torch.as_tensor(embedding_matrix, dtype=torch.float32),
            freeze=freeze_embeddings)
        self.doc_encoder = DocEncoder(self.embedding.embedding_dim)
        self.user_encoder = UserEncoder()

    def news_encoder(self, titles):           # [.., T] ids -> [.., D]
        shape = titles.shape
        flat = titles.reshape(-1, shape[-1])
        vecs = self.doc_encoder(self.embedding(flat))
        return vecs.reshape(*shape[:-1], -1)

    def forward(self, click, sample):
        # click: [B, H, T]; sample: [B, 1+npratio, T]
        user_vec = self.user_encoder(self.news_encoder(click))
        cand_vecs = self.news_encoder(sample)
        scores = torch.einsum("bkd,bd->bk", cand_vecs, user_vec)
        return scores, user_vec


def auc_score(labels, scores):
    order = np.argsort(-scores)
    ranked = labels[order]
    n_pos, n_neg = ranked.sum(), len(ranked) - ranked.sum()
    if n_pos == 0 or n_neg == 0:
        return 0.5
    # rank-sum AUC
    ranks = np.empty(len(scores))
    ranks[np.argsort(scores)] = np.arange(1, len(scores) + 1)
    return (ranks[labels > 0].sum() - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg)


def mrr_score(labels, scores):
    order = np.argsort(-scores)
    ranked = labels[order]
    rr = ranked / (np.arange(len(ranked)) + 1)
    return rr.sum() / max(ranked.sum(), 1)


def dcg_score(labels, scores, k):
    order = np.argsort(-scores)[:k]
    gains = (2 ** labels[order] - 1) / np.log2(np.arange(len(order)) + 2)
    return gains.sum()


def ndcg_score(labels, scores, k):
    ideal = dcg_score(labels, labels, k)
    return dcg_score(labels, scores, k) / ideal if ideal > 0 else 0.0


class FEDNEWS(BaseModel):
    def __init__(self, model_config):
        super().__init__()
        vocab = int(model_config.get("vocab_size", 5000))
        dim = int(model_config.get("embed_dim", 300))
        path = model_config.get("embedding_path")
        if path:
            matrix = np.load(path)
        else:
            rng = np.random.default_rng(1234)
            matrix = rng.standard_normal((vocab, dim)).astype(np.float32) * 0.1
            matrix[0] = 0.0
        self.net = FedNewsRecNet(matrix, freeze_embeddings=bool(
            model_config.get("freeze_embeddings", True)))

    def loss(self, input):
        if not self.net.training:
            return torch.tensor(0.0)
        click, sample = input["x"]
        label = to_device(input["y"]).long()
        scores, _ = self.net(to_device(click).long(), to_device(sample).long())
        return F.cross_entropy(scores, label)

    def inference(self, input):
        click, cands = input["x"]
        labels = input["y"]  # [B, K] binary relevance
        click = to_device(click).long()
        cands = to_device(cands).long()
        scores, _ = self.net(click, cands)
        scores = scores.detach().cpu().numpy()
        labels = labels.cpu().numpy()
        aucs, mrrs, n5, n10 = [], [], [], []
        for i in range(len(scores)):
            aucs.append(auc_score(labels[i], scores[i]))
            mrrs.append(mrr_score(labels[i], scores[i]))
            n5.append(ndcg_score(labels[i], scores[i], 5))
            n10.append(ndcg_score(labels[i], scores[i], 10))
        bs = len(scores)
        return {
            "output": 0.0, "acc": float(np.mean(aucs)), "batch_size": bs,
            "mrr": {"value": float(np.mean(mrrs)), "higher_is_better": True},
            "ndcg@5": {"value": float(np.mean(n5)), "higher_is_better": True},
            "ndcg@10": {"value": float(np.mean(n10)),
                        "higher_is_better": True},
        }

    def set_eval(self):
        self.eval()
        self.net.eval()

    def set_train(self):
        self.train()
        self.net.train()
