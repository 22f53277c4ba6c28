"""End-to-end subprocess tests (mirrors reference testing/test_e2e_trainer.py).

Runs the real entrypoint on tiny synthetic data: single process and a
2-process gloo group (works without a GPU), checks exit code, learning
progress, checkpoint format parity and resume.
"""

import json
import os
import subprocess
import sys

import pytest
import torch
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _write_config(tmp_path, **overrides):
    with open(os.path.join(REPO, "configs", "cv_lr_mnist.yaml")) as f:
        cfg = yaml.safe_load(f)
    cfg["server_config"]["max_iteration"] = overrides.pop("max_iteration", 4)
    cfg["server_config"]["val_freq"] = overrides.pop("val_freq", 2)
    cfg["server_config"]["rec_freq"] = overrides.pop("rec_freq", 4)
    cfg["server_config"]["initial_val"] = overrides.pop("initial_val", True)
    for k, v in overrides.items():
        cfg["server_config"][k] = v
    p = tmp_path / "cfg.yaml"
    with open(p, "w") as f:
        yaml.safe_dump(cfg, f)
    return str(p)


def _run(nproc, cfg, data_dir, out_dir, extra_env=None, port=29801):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    if extra_env:
        env.update(extra_env)
    if nproc == 1:
        cmd = [sys.executable, "e2e_trainer.py"]
    else:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
               "--master-port", str(port), "e2e_trainer.py"]
    cmd += ["-dataPath", data_dir, "-outputPath", out_dir,
            "-config", cfg, "-task", "cv_lr_mnist", "-backend", "gloo"]
    return subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                          text=True, timeout=600)


def _read_metrics(out_dir, rank=0):
    path = os.path.join(out_dir, "msrflute_amd", "log",
                        f"metrics_rank{rank}.jsonl")
    with open(path) as f:
        return [json.loads(l) for l in f]


def test_e2e_single_process(tmp_data_dir, tmp_path):
    out = str(tmp_path / "out1")
    cfg = _write_config(tmp_path, max_iteration=6)
    r = _run(1, cfg, tmp_data_dir, out)
    assert r.returncode == 0, r.stderr[-3000:]

    # learning: training loss decreases from first to last round
    metrics = _read_metrics(out)
    losses = [m["value"] for m in metrics if m["key"] == "Training loss"]
    assert len(losses) == 6
    assert losses[-1] < losses[0], losses

    # checkpoint format parity: .tar dict with the reference's keys/tokens
    models_dir = os.path.join(out, "msrflute_amd", "models")
    for token in ["latest", "best_val_loss", "best_val_acc", "best_test_acc"]:
        path = os.path.join(models_dir, f"{token}_model.tar")
        assert os.path.exists(path), token
    ckpt = torch.load(os.path.join(models_dir, "latest_model.tar"),
                      map_location="cpu", weights_only=False)
    assert set(ckpt.keys()) >= {"model_state_dict", "optimizer_state_dict",
                                "lr_scheduler_state_dict"}
    # status log parity
    with open(os.path.join(models_dir, "status_log.json")) as f:
        status = json.load(f)
    assert {"i", "best_val_loss", "best_val_acc", "weight"} <= set(status)


def test_e2e_two_process_gloo(tmp_data_dir, tmp_path):
    out = str(tmp_path / "out2")
    cfg = _write_config(tmp_path)
    r = _run(2, cfg, tmp_data_dir, out, port=29802)
    assert r.returncode == 0, r.stderr[-3000:]
    m0 = _read_metrics(out, 0)
    losses = [m["value"] for m in m0 if m["key"] == "Training loss"]
    assert len(losses) == 4


def test_single_vs_two_process_same_result(tmp_data_dir, tmp_path):
    """The symmetric runtime must give the same model no matter the world
    size (weighted sums are order/partition independent up to fp)."""
    cfg = _write_config(tmp_path, max_iteration=3, initial_val=False)
    out1, out2 = str(tmp_path / "w1"), str(tmp_path / "w2")
    r1 = _run(1, cfg, tmp_data_dir, out1)
    assert r1.returncode == 0, r1.stderr[-3000:]
    r2 = _run(2, cfg, tmp_data_dir, out2, port=29803)
    assert r2.returncode == 0, r2.stderr[-3000:]
    sd1 = torch.load(os.path.join(out1, "msrflute_amd", "models", "latest_model.tar"),
                     map_location="cpu", weights_only=False)["model_state_dict"]
    sd2 = torch.load(os.path.join(out2, "msrflute_amd", "models", "latest_model.tar"),
                     map_location="cpu", weights_only=False)["model_state_dict"]
    for k in sd1:
        assert torch.allclose(sd1[k], sd2[k], rtol=1e-4, atol=1e-6), k


def test_resume_from_checkpoint(tmp_data_dir, tmp_path):
    out = str(tmp_path / "resume")
    cfg1 = _write_config(tmp_path, max_iteration=2)
    r = _run(1, cfg1, tmp_data_dir, out)
    assert r.returncode == 0, r.stderr[-3000:]
    status = json.load(open(os.path.join(out, "msrflute_amd", "models",
                                         "status_log.json")))
    assert status["i"] == 2

    # resume to 4 rounds
    with open(cfg1) as f:
        cfg = yaml.safe_load(f)
    cfg["server_config"]["max_iteration"] = 4
    cfg["server_config"]["resume_from_checkpoint"] = True
    cfg2 = str(tmp_path / "cfg2.yaml")
    yaml.safe_dump(cfg, open(cfg2, "w"))
    r = _run(1, cfg2, tmp_data_dir, out)
    assert r.returncode == 0, r.stderr[-3000:]
    status = json.load(open(os.path.join(out, "msrflute_amd", "models",
                                         "status_log.json")))
    assert status["i"] == 4
    # resumed run must skip rounds 0-1: exactly 2 more Training loss records
    metrics = _read_metrics(out)
    losses = [m for m in metrics if m["key"] == "Training loss"]
    assert len(losses) == 4  # 2 from first run + 2 after resume


def test_single_vs_two_process_same_result_dga(tmp_data_dir, tmp_path):
    """World-size invariance must also hold for DGA's softmax weighting
    (weights derive from per-client losses which travel through the
    metadata gather — partition must not change the aggregate)."""
    cfg_path = _write_config(tmp_path, max_iteration=3, initial_val=False)
    with open(cfg_path) as f:
        cfg = yaml.safe_load(f)
    cfg["strategy"] = "DGA"
    cfg["server_config"]["aggregate_median"] = "softmax"
    cfg["server_config"]["softmax_beta"] = 1.0
    with open(cfg_path, "w") as f:
        yaml.safe_dump(cfg, f)
    out1, out2 = str(tmp_path / "d1"), str(tmp_path / "d2")
    r1 = _run(1, cfg_path, tmp_data_dir, out1)
    assert r1.returncode == 0, r1.stderr[-3000:]
    r2 = _run(2, cfg_path, tmp_data_dir, out2, port=29807)
    assert r2.returncode == 0, r2.stderr[-3000:]
    sd1 = torch.load(os.path.join(out1, "msrflute_amd", "models",
                                  "latest_model.tar"),
                     map_location="cpu", weights_only=False)["model_state_dict"]
    sd2 = torch.load(os.path.join(out2, "msrflute_amd", "models",
                                  "latest_model.tar"),
                     map_location="cpu", weights_only=False)["model_state_dict"]
    for k in sd1:
        assert torch.allclose(sd1[k], sd2[k], rtol=1e-4, atol=1e-6), k


def test_four_process_more_ranks_than_clients(tmp_data_dir, tmp_path):
    """world_size=4 with 3 clients/round: one rank is idle every round —
    the zero-row metadata gather and empty-partition paths must hold."""
    cfg = _write_config(tmp_path, max_iteration=3, initial_val=False,
                        num_clients_per_iteration=3)
    out = str(tmp_path / "w4")
    r = _run(4, cfg, tmp_data_dir, out, port=29809)
    assert r.returncode == 0, r.stderr[-3000:]
    m0 = _read_metrics(out, 0)
    losses = [m["value"] for m in m0 if m["key"] == "Training loss"]
    assert len(losses) == 3 and all(l > 0 for l in losses)
