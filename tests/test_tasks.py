"""Per-task end-to-end smoke tests: every shipped task trains a few FL
rounds on tiny synthetic data (CPU, world_size 1, in-process).

Mirrors the reference's task matrix (testing/test_e2e_trainer.py runs
nlg_gru / ecg_cnn / mlm_bert / classif_cnn; we cover all shipped tasks).
"""

import os
import subprocess
import sys

import pytest
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

# task -> (blob maker kwargs override, config knobs)
TASK_MATRIX = {
    "cv_lr_mnist": {},
    "cv_cnn_femnist": {},
    "cv_resnet_fedcifar100": {},
    "classif_cnn": {},
    "ecg_cnn": {},
    "nlp_rnn_fedshakespeare": {},
    "nlg_gru": {},
    "mlm_bert": {},
    "cv": {},
    "semisupervision": {},
    "fednewsrec": {},
}


def _make_data(task, data_dir, n_users=12, samples=8):
    from tools import create_data as cd
    fn, kw = cd.TASKS[task]
    kw = dict(kw)
    if "n_users" in kw:
        kw["n_users"] = n_users
    if "samples_per_user" in kw:
        kw["samples_per_user"] = samples
    if "utts_per_user" in kw:
        kw["utts_per_user"] = samples
    if "n_samples" in kw:
        kw["n_samples"] = 400
    train = fn(seed=0, **kw)
    kw_eval = dict(kw, n_users=3)
    if task == "fednewsrec":
        from tools.create_data import make_newsrec_eval_blob
        val = make_newsrec_eval_blob(seed=1, **kw_eval)
        test = make_newsrec_eval_blob(seed=2, **kw_eval)
    else:
        val = fn(seed=1, **kw_eval)
        test = fn(seed=2, **kw_eval)
    cd.save_blob(train, os.path.join(data_dir, task, "train_data.pt"))
    cd.save_blob(val, os.path.join(data_dir, task, "val_data.pt"))
    cd.save_blob(test, os.path.join(data_dir, task, "test_data.pt"))


def _shrink_config(task, tmp_path):
    with open(os.path.join(REPO, "configs", f"{task}.yaml")) as f:
        cfg = yaml.safe_load(f)
    sc = cfg["server_config"]
    sc["max_iteration"] = 2
    sc["num_clients_per_iteration"] = 3
    sc["val_freq"] = 2
    sc["rec_freq"] = 2
    sc["initial_val"] = False
    sc["initial_rec"] = False
    # tiny models where configurable, for CPU speed
    if task == "nlg_gru":
        cfg["model_config"].update(embed_dim=16, hidden_dim=32)
    if task == "ecg_cnn":
        cfg["model_config"].update(hid_size=16)
    if task == "cv_resnet_fedcifar100":
        cfg["model_config"].update(num_classes=100)
    if task == "semisupervision":
        cfg["model_config"].update(num_classes=10)
        cfg["client_config"]["semisupervision"].update(num_classes=10,
                                                       unl_bs=3, bs=3)
        cfg["client_config"]["data_config"]["train"][
            "num_labeled_per_user"] = 4
    if task == "fednewsrec":
        cfg["model_config"].update(embed_dim=32)  # vocab must match the blob
    if task == "cv":
        for sec in (cfg["client_config"]["data_config"]["train"],
                    cfg["server_config"]["data_config"]["val"],
                    cfg["server_config"]["data_config"]["test"]):
            sec["total_num_clients"] = 8
        sc["num_clients_per_iteration"] = 2
    p = tmp_path / f"{task}.yaml"
    with open(p, "w") as f:
        yaml.safe_dump(cfg, f)
    return str(p)


@pytest.mark.parametrize("task", sorted(TASK_MATRIX))
def test_task_trains(task, tmp_path):
    data_dir = str(tmp_path / "data")
    _make_data(task, data_dir)
    cfg = _shrink_config(task, tmp_path)
    out = str(tmp_path / "out")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "e2e_trainer.py", "-dataPath", data_dir,
         "-outputPath", out, "-config", cfg, "-task", task,
         "-backend", "gloo"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-3000:])
    # a round ran and produced the latest checkpoint
    assert os.path.exists(os.path.join(
        out, "msrflute_amd", "models", "latest_model.tar"))
