"""Coverage for remaining reference behaviors: DGA staleness simulation,
fall-back-to-best, BERT adapters, fednewsrec ranking metrics,
personalization alpha update, DeviceShardStore (CPU), schema rejection."""

import os
import subprocess
import sys

import numpy as np
import pytest
import torch
import yaml

from tests.test_tasks import REPO, _make_data
from tests.test_features import _base_cfg, _run_task


def test_dga_staleness_holds_gradients(tmp_path):
    """stale_prob=1 holds every gradient a round (reference dga.py:260-277)
    — training still completes and releases held grads next round."""
    data_dir = str(tmp_path / "data")
    _make_data("cv_lr_mnist", data_dir)
    cfg = _base_cfg(rounds=4)
    cfg["strategy"] = "DGA"
    cfg["server_config"].update(stale_prob=0.5, fast_aggregation=False)
    _run_task("cv_lr_mnist", cfg, tmp_path, data_dir)


def test_fall_back_to_best_model(tmp_path):
    data_dir = str(tmp_path / "data")
    _make_data("cv_lr_mnist", data_dir)
    cfg = _base_cfg(rounds=4)
    cfg["server_config"].update(fall_back_to_best_model=True, initial_val=True,
                                val_freq=2)
    _run_task("cv_lr_mnist", cfg, tmp_path, data_dir)


def test_bert_adapters_freeze_base():
    from importlib.machinery import SourceFileLoader
    mod = SourceFileLoader(
        "mlm_model", os.path.join(REPO, "experiments/mlm_bert/model.py")
    ).load_module()
    m = mod.BERT({"BERT": {"model": {
        "model_name": "t", "vocab_size": 100, "hidden_size": 32,
        "num_hidden_layers": 2, "num_attention_heads": 2,
        "intermediate_size": 64, "adapter": True, "adapter_dim": 8},
        "training": {"batch_size": 2, "label_smoothing_factor": 0.1,
                     "seed": 1}}})
    if torch.cuda.is_available():  # model.loss moves inputs via to_device
        m = m.cuda()
    trainable = [n for n, p in m.named_parameters() if p.requires_grad]
    frozen = [n for n, p in m.named_parameters() if not p.requires_grad]
    assert trainable and all("adapters" in n for n in trainable)
    assert any("encoder" in n for n in frozen)
    # adapters actually receive gradients through the hooks
    batch = {"input_ids": torch.randint(4, 100, (2, 8)),
             "attention_mask": torch.ones(2, 8, dtype=torch.int64),
             "labels": torch.randint(4, 100, (2, 8))}
    m.loss(batch).backward()
    g = [p.grad for n, p in m.named_parameters() if "adapters" in n]
    assert all(t is not None for t in g)
    assert any(t.abs().sum() > 0 for t in g)


def test_newsrec_ranking_metrics():
    from importlib.machinery import SourceFileLoader
    mod = SourceFileLoader(
        "fnr_model", os.path.join(REPO, "experiments/fednewsrec/model.py")
    ).load_module()
    labels = np.array([0, 1, 0, 0, 0])
    perfect = np.array([0.1, 0.9, 0.2, 0.0, 0.3])
    worst = np.array([0.9, 0.0, 0.8, 0.7, 0.6])
    assert mod.auc_score(labels, perfect) == 1.0
    assert mod.auc_score(labels, worst) == 0.0
    assert mod.mrr_score(labels, perfect) == 1.0
    assert abs(mod.mrr_score(labels, worst) - 1 / 5) < 1e-9
    assert mod.ndcg_score(labels, perfect, 5) == 1.0
    assert mod.ndcg_score(labels, worst, 5) < 0.5


def test_alpha_update_moves_toward_better_model():
    from msrflute_amd.utils.misc import alpha_update
    local = torch.nn.Linear(4, 1)
    glob = torch.nn.Linear(4, 1)
    with torch.no_grad():
        glob.weight.copy_(local.weight + 1.0)
        glob.bias.copy_(local.bias)
    # local grad points TOWARD global (negative dot with (local-global))
    local.weight.grad = torch.ones_like(local.weight)
    local.bias.grad = torch.zeros_like(local.bias)
    glob.weight.grad = torch.ones_like(glob.weight)
    glob.bias.grad = torch.zeros_like(glob.bias)
    a0 = 0.5
    a1 = alpha_update(local, glob, a0, lr=0.1)
    # grad_alpha = grad·(local-global) = 4*(-1) < 0  => alpha increases
    assert a1 > a0
    assert 0.0 <= a1 <= 1.0


def test_device_shard_store_cpu_views():
    from msrflute_amd.models.generic_data import DeviceShardStore
    from tools.create_data import make_femnist_blob

    blob = make_femnist_blob(n_users=4, samples_per_user=6, seed=0)

    class Ds:
        user_list = blob["users"]
        user_data = blob["user_data"]
        user_data_label = blob["user_data_label"]
    store = DeviceShardStore(Ds(), (28, 28), device="cpu")
    loader = store.loader_for("user00002", batch_size=4)
    xs = torch.as_tensor(np.asarray(blob["user_data"]["user00002"]["x"]))
    assert torch.allclose(loader.dataset.x, xs.float())
    assert len(loader.dataset) == 6
    batches = list(loader)
    assert sum(len(b["x"]) for b in batches) == 6


def test_schema_rejects_bad_optimizer():
    from msrflute_amd.config import FLUTEConfig
    with open(os.path.join(REPO, "configs", "cv_lr_mnist.yaml")) as f:
        cfg = yaml.safe_load(f)
    cfg["server_config"]["optimizer_config"]["type"] = "nonsense"
    with pytest.raises(Exception):
        FLUTEConfig.from_dict(cfg)


def test_scheduled_sampling_ramp():
    from msrflute_amd.utils.schedulers import ScheduledSamplingScheduler

    class M:
        pass
    m = M()
    s = ScheduledSamplingScheduler(m, ramp_start=2, ramp_stop=6,
                                   initial_rate=1.0, final_rate=0.0)
    rates = []
    for _ in range(9):
        s.step()
        rates.append(m.scheduled_sampling_rate)
    assert rates[0] == 1.0 and rates[1] == 1.0       # before ramp
    assert rates[3] < rates[2] <= 1.0                # ramping down
    assert rates[8] == 0.0 and m.scheduled_sampling is False
    # checkpoint round-trip excludes the model
    sd = s.state_dict()
    assert "model" not in sd
    s2 = ScheduledSamplingScheduler(m, 0, 1, 0, 0)
    s2.load_state_dict(sd)
    assert s2.iter == s.iter


def test_nbest_task_scheduler_stages():
    from msrflute_amd.utils.schedulers import NBestTaskScheduler
    s = NBestTaskScheduler(num_tasks=[1, 3, 5], iteration_per_task=[2, 4, 6])
    seen = []
    for _ in range(12):
        s.step()
        seen.append(s.current_num_tasks())
    # within each 6-iteration cycle: stages advance 1 -> 3 -> 5
    assert seen[0] == 1 and 3 in seen[:6] and 5 in seen[:6]
    assert s.no_label_updates() == 3  # 12 iters / 6 per cycle + 1


def test_val_loss_annealing_e2e(tmp_path):
    """Server optimizer LR decays on val-loss plateau (reference
    val_loss annealing, utils/utils.py:151-186 + trainer.py:139-155)."""
    data_dir = str(tmp_path / "data")
    _make_data("cv_lr_mnist", data_dir)
    cfg = _base_cfg(rounds=4)
    cfg["server_config"]["annealing_config"] = {
        "type": "val_loss", "gamma": 0.5, "step_interval": "epoch",
        "patience": 0, "step_size": 1}
    cfg["server_config"]["initial_val"] = True
    _run_task("cv_lr_mnist", cfg, tmp_path, data_dir)


def test_mlm_line_by_line_mode():
    """mlm_bert dataset line-by-line framing (reference dataset.py:70-82):
    one padded frame per utterance instead of concatenate-and-chunk."""
    from importlib.machinery import SourceFileLoader
    ds_mod = SourceFileLoader(
        "mlm_ds", os.path.join(REPO, "experiments/mlm_bert/dataloaders/"
                               "dataset.py")).load_module()
    blob = {"users": ["a"], "num_samples": [2],
            "user_data": {"a": {"x": [[5, 6, 7], [8, 9, 10, 11, 12]]}}}
    lbl = ds_mod.Dataset(blob, args={"max_seq_length": 4,
                                     "process_line_by_line": True},
                         user_idx=0)
    assert len(lbl) == 2                       # one frame per utterance
    ids0, attn0 = lbl[0]
    assert list(ids0) == [5, 6, 7, 0] and list(attn0) == [1, 1, 1, 0]
    ids1, _ = lbl[1]
    assert list(ids1) == [8, 9, 10, 11]        # truncated to max_seq_length

    grouped = ds_mod.Dataset(blob, args={"max_seq_length": 4}, user_idx=0)
    assert len(grouped) == 2                   # 8 tokens -> 2 frames of 4
    g0, _ = grouped[0]
    assert list(g0) == [5, 6, 7, 8]            # concatenated across utts


def test_quant_threshold_annealing_progresses(tmp_path):
    """quant_thresh multiplies by quant_anneal every round (reference
    server.py:295-298)."""
    import json
    data_dir = str(tmp_path / "data")
    _make_data("cv_lr_mnist", data_dir)
    cfg = _base_cfg(rounds=3)
    cfg["strategy"] = "DGA"
    cfg["client_config"].update(quant_thresh=0.5, quant_bits=4,
                                quant_anneal=0.5)
    out = _run_task("cv_lr_mnist", cfg, tmp_path, data_dir)
    path = os.path.join(out, "msrflute_amd", "log", "metrics_rank0.jsonl")
    with open(path) as f:
        vals = [json.loads(l)["value"] for l in f
                if json.loads(l)["key"] == "Quantization Thresh."]
    assert len(vals) == 3
    assert vals[0] == 0.25 and vals[1] == 0.125 and vals[2] == 0.0625


def test_client_range_sampling(tmp_path):
    """num_clients_per_iteration as a "min,max" random range (reference
    server.py:84-86, 284-291)."""
    import json
    data_dir = str(tmp_path / "data")
    _make_data("cv_lr_mnist", data_dir)
    cfg = _base_cfg(rounds=4)
    cfg["server_config"]["num_clients_per_iteration"] = "2,5"
    out = _run_task("cv_lr_mnist", cfg, tmp_path, data_dir)
    path = os.path.join(out, "msrflute_amd", "log", "metrics_rank0.jsonl")
    with open(path) as f:
        ns = [json.loads(l)["value"] for l in f
              if json.loads(l)["key"] == "Clients for round"]
    assert len(ns) == 4 and all(2 <= n <= 5 for n in ns)


def test_fedavg_freeze_layer_zeroes_segment():
    """freeze_layer zeroes that parameter's pseudo-gradient before upload
    (reference fedavg.py:81-86)."""
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.strategies.fedavg import FedAvg

    net = torch.nn.Sequential(torch.nn.Linear(4, 3), torch.nn.Linear(3, 2))
    arena = ParameterArena(net, bind_grads=True)
    arena.grad.fill_(1.0)
    cfg = {"model_config": {"freeze_layer": "1.weight"},
           "client_config": {}, "server_config": {}, "strategy": "FedAvg"}

    class Tr:
        num_samples = 7
    t = Tr()
    t.arena = arena
    strat = FedAvg("client", cfg)
    payload = strat.generate_client_payload(t)
    assert payload["weight"] == 7.0
    off, n = arena.segment_of("1.weight")
    assert payload["grad"][off:off + n].abs().sum() == 0
    other_off, other_n = arena.segment_of("0.weight")
    assert payload["grad"][other_off:other_off + other_n].abs().sum() > 0


def test_component_wise_lr(tmp_path):
    """trainer_config.updatable_names freezes/relearns components
    (reference utils/utils.py:725-751)."""
    data_dir = str(tmp_path / "data")
    _make_data("cv_lr_mnist", data_dir)
    cfg = _base_cfg(rounds=2)
    cfg["client_config"]["trainer_config"] = {
        "updatable_names": ["net.linear.weight"]}
    _run_task("cv_lr_mnist", cfg, tmp_path, data_dir)


def test_epoch_snapshots_and_best_acc_criterion(tmp_path):
    """model_backup_freq epoch{i} snapshots + best_model_criterion=acc
    (reference server.py:530-559, evaluation best tracking)."""
    data_dir = str(tmp_path / "data")
    _make_data("cv_lr_mnist", data_dir)
    cfg = _base_cfg(rounds=4)
    cfg["server_config"].update(model_backup_freq=2, initial_val=True,
                                best_model_criterion="acc")
    out = _run_task("cv_lr_mnist", cfg, tmp_path, data_dir)
    models = os.listdir(os.path.join(out, "msrflute_amd", "models"))
    assert any(m.startswith("epoch") for m in models), models
    assert "best_val_acc_model.tar" in models
