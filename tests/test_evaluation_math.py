"""Evaluation merge math + eval-client grouping (reference:
evaluation.py:159-216, metrics.py:43-73)."""

import torch

from msrflute_amd.core.evaluation import make_eval_clients
from msrflute_amd.core.metrics import Metrics


class _Ds:
    def __init__(self, counts):
        self.user_list = [f"u{i}" for i in range(len(counts))]
        self.num_samples = counts
        self.user_data = {u: {"x": [0] * c}
                          for u, c in zip(self.user_list, counts)}
        self.user_data_label = {u: [0] * c
                                for u, c in zip(self.user_list, counts)}


def test_make_eval_clients_balances_samples():
    ds = _Ds([50, 10, 40, 30, 20, 5, 45])

    cfg = {"server_config": {"type": "model_optimization",
                             "data_config": {"num_clients": 7}}}
    groups = list(make_eval_clients(ds, cfg, n_groups=3))
    covered = sorted(sum((g.client_id for g in groups), []))
    assert covered == list(range(7))  # exact cover, no dupes
    loads = [sum(ds.num_samples[i] for i in g.client_id) for g in groups]
    assert max(loads) <= 2 * min(loads) + max(ds.num_samples)


class _Model:
    def set_eval(self):
        pass

    def set_train(self):
        pass

    def loss(self, batch):
        return torch.tensor(float(batch["y"].float().mean()))

    def inference(self, batch):
        n = len(batch["y"])
        return {"output": 0.0, "acc": float(batch["y"].float().mean()),
                "batch_size": n,
                "f1": {"value": 0.5, "higher_is_better": True}}


def test_metrics_weighted_average():
    # batch sizes 2 and 8 with acc 1.0 / 0.0 -> weighted acc 0.2
    b1 = {"x": torch.zeros(2), "y": torch.ones(2)}
    b2 = {"x": torch.zeros(8), "y": torch.zeros(8)}
    m = Metrics()
    _, metrics = m.compute_metrics([b1, b2], _Model())
    assert abs(metrics["acc"]["value"] - 0.2) < 1e-6
    assert metrics["f1"]["value"] == 0.5
    assert metrics["f1"]["higher_is_better"] is True
