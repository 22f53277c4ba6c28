#!/usr/bin/env python3
"""Head-to-head behavior parity: the REFERENCE (msrflute/FLUTE at
/root/reference, pure Python, CPU/gloo) vs THIS engine, on identical
synthetic LR-MNIST federated data from identical initial weights.

Design:

* Synthetic MNIST-shaped shards (U users x S samples of 784 features,
  separable class means so learning is visible) are written BOTH in the
  FedML zip layout the reference's `experiments/cv_lr_mnist/dataloaders/
  preprocessing.py:40-68` reads AND as this engine's JSON blobs.
* One initial checkpoint (the shared `.tar` dict format,
  reference trainer.py:753-770) seeds both runs via
  `resume_from_checkpoint: true`, so both start from the SAME weights.
* Every round samples ALL clients with full-shard batches, making the
  round a deterministic full-batch FedAvg step on both sides: the
  per-round "Training loss" series and final "Test acc" must then agree
  to floating-point tolerance, and so must the final checkpoints.
* The reference runs unmodified; only its absent third-party deps
  (cerberus, easydict, azureml, wget) are satisfied by the minimal shims
  in tools/parity/shims/ (see each shim's docstring).

Usage: python tools/parity/run_parity.py [--workdir DIR] [--rounds N]
Writes <workdir>/PARITY_REPORT.md and exits nonzero on mismatch.
"""

import argparse
import json
import os
import shutil
import socket
import subprocess
import sys
import zipfile

import numpy as np
import torch
import yaml

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), "..", ".."))
REFERENCE = "/root/reference"
SHIMS = os.path.join(REPO, "tools", "parity", "shims")

N_USERS = 20
N_PER_USER = 12
N_TEST = 200
N_CLASSES = 10
CLIENT_LR = 0.05
ROUNDS = 10


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def gen_data(rng):
    """Separable 784-dim class clusters (so loss visibly decreases)."""
    means = rng.standard_normal((N_CLASSES, 784)) * 0.8
    users, num_samples, xs, ys = [], [], {}, {}
    for u in range(N_USERS):
        name = f"{u:04d}"
        y = rng.integers(0, N_CLASSES, size=N_PER_USER)
        x = means[y] + rng.standard_normal((N_PER_USER, 784)) * 0.5
        users.append(name)
        num_samples.append(N_PER_USER)
        xs[name] = np.round(x, 4).tolist()
        ys[name] = y.tolist()
    yt = rng.integers(0, N_CLASSES, size=N_TEST)
    xt = np.round(means[yt] + rng.standard_normal((N_TEST, 784)) * 0.5,
                  4).tolist()
    return users, num_samples, xs, ys, xt, yt.tolist()


def write_reference_zip(data_dir, users, num_samples, xs, ys, xt, yt):
    """FedML MNIST layout: pre-extracted MNIST/{train,test}/*.json plus an
    EMPTY MNIST.zip — the reference re-extracts the zip on every run and
    every rank (preprocessing.py:30-40), so a populated zip makes the two
    gloo ranks race on extraction; an empty zip extracts nothing and
    read_data() finds the pre-seeded files."""
    train = {"users": users, "num_samples": num_samples,
             "user_data": {u: {"x": xs[u], "y": ys[u]} for u in users}}
    test = {"users": ["test0"], "num_samples": [len(yt)],
            "user_data": {"test0": {"x": xt, "y": yt}}}
    for split, blob in [("train", train), ("test", test)]:
        d = os.path.join(data_dir, "MNIST", split)
        os.makedirs(d, exist_ok=True)
        with open(os.path.join(d, "all_data.json"), "w") as f:
            json.dump(blob, f)
    with zipfile.ZipFile(os.path.join(data_dir, "MNIST.zip"), "w"):
        pass


def write_our_blobs(data_dir, users, num_samples, xs, ys, xt, yt):
    os.makedirs(data_dir, exist_ok=True)
    train = {"users": users, "num_samples": num_samples,
             "user_data": xs, "user_data_label": ys}
    test = {"users": ["test0"], "num_samples": [len(yt)],
            "user_data": {"test0": xt}, "user_data_label": {"test0": yt}}
    with open(os.path.join(data_dir, "mnist_train.json"), "w") as f:
        json.dump(train, f)
    with open(os.path.join(data_dir, "mnist_test.json"), "w") as f:
        json.dump(test, f)


def write_initial_ckpt(paths):
    """One .tar checkpoint (shared format) seeding both runs."""
    torch.manual_seed(42)
    lin = torch.nn.Linear(784, N_CLASSES)
    opt = torch.optim.SGD(lin.parameters(), lr=1.0)
    sched = torch.optim.lr_scheduler.StepLR(opt, step_size=100, gamma=1.0)
    ckpt = {
        "model_state_dict": {"net.linear.weight": lin.weight.detach().clone(),
                             "net.linear.bias": lin.bias.detach().clone()},
        "optimizer_state_dict": opt.state_dict(),
        "lr_scheduler_state_dict": sched.state_dict(),
    }
    for p in paths:
        os.makedirs(os.path.dirname(p), exist_ok=True)
        torch.save(ckpt, p)
    return ckpt


def base_config(rounds):
    return {
        "model_config": {"model_type": "LR",
                         "model_folder": "experiments/cv_lr_mnist/model.py",
                         "input_dim": 784, "output_dim": N_CLASSES},
        "dp_config": {"enable_local_dp": False},
        "privacy_metrics_config": {"apply_metrics": False},
        "strategy": "FedAvg",
        "server_config": {
            "wantRL": False, "resume_from_checkpoint": True,
            "do_profiling": False,
            "optimizer_config": {"type": "sgd", "lr": 1.0},
            "annealing_config": {"type": "step_lr", "step_interval": "epoch",
                                 "gamma": 1.0, "step_size": 100},
            "val_freq": 10000, "rec_freq": rounds,
            "initial_val": False, "initial_rec": False,
            "max_iteration": rounds,
            "num_clients_per_iteration": N_USERS,  # ALL clients -> no sampling
            "data_config": {"val": {"batch_size": 256, "val_data": None},
                            "test": {"batch_size": 256, "test_data": None}},
            "type": "model_optimization", "aggregate_median": "mean",
            "weight_train_loss": "train_loss", "softmax_beta": 1.0,
            "initial_lr_client": CLIENT_LR, "lr_decay_factor": 1.0,
            "best_model_criterion": "loss", "fall_back_to_best_model": False,
        },
        "client_config": {
            "do_profiling": False, "ignore_subtask": False,
            "data_config": {"train": {"batch_size": 64,  # >= shard: full batch
                                      "list_of_train_data": None,
                                      "desired_max_samples": 5000}},
            "type": "optimization",
            "optimizer_config": {"type": "sgd", "lr": CLIENT_LR},
        },
    }


def run(cmd, cwd, env, log_path):
    with open(log_path, "w") as lf:
        r = subprocess.run(cmd, cwd=cwd, env=env, stdout=lf,
                           stderr=subprocess.STDOUT, timeout=900)
    if r.returncode != 0:
        tail = open(log_path).read()[-4000:]
        raise RuntimeError(f"{' '.join(cmd[:6])}... failed "
                           f"rc={r.returncode}\n{tail}")


def parse_jsonl(path):
    recs = []
    with open(path) as f:
        for line in f:
            line = line.strip()
            if line:
                recs.append(json.loads(line))
    return recs


def series(recs, key):
    return [r["value"] for r in recs if r["key"] == key]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workdir", default=os.path.join(REPO, "gpurun_out",
                                                      "parity"))
    ap.add_argument("--rounds", type=int, default=ROUNDS)
    args = ap.parse_args()
    wd = os.path.abspath(args.workdir)
    shutil.rmtree(wd, ignore_errors=True)
    os.makedirs(wd, exist_ok=True)

    rng = np.random.default_rng(7)
    users, num_samples, xs, ys, xt, yt = gen_data(rng)

    # ---- reference sandbox (symlinks; reference reads ./core/schema.py
    # and ./data relative to cwd, and we must not write /root/reference) --
    sandbox = os.path.join(wd, "ref_sandbox")
    os.makedirs(sandbox, exist_ok=True)
    for name in ["core", "utils", "experiments", "extensions",
                 "e2e_trainer.py"]:
        dst = os.path.join(sandbox, name)
        if not os.path.exists(dst):
            os.symlink(os.path.join(REFERENCE, name), dst)
    write_reference_zip(os.path.join(sandbox, "data"), users, num_samples,
                        xs, ys, xt, yt)
    our_data = os.path.join(wd, "our_data")
    write_our_blobs(our_data, users, num_samples, xs, ys, xt, yt)

    # experiment name comes from the azureml shim's run id on the
    # reference side; ours via FLUTE_EXPERIMENT_NAME
    ref_out = os.path.join(wd, "out_ref")
    our_out = os.path.join(wd, "out_ours")
    ref_models = os.path.join(ref_out, "parity-run", "models")
    our_models = os.path.join(our_out, "parity-run", "models")
    write_initial_ckpt([os.path.join(ref_models, "latest_model.tar"),
                        os.path.join(our_models, "latest_model.tar")])

    cfg = base_config(args.rounds)
    ref_cfg_path = os.path.join(wd, "ref_config.yaml")
    with open(ref_cfg_path, "w") as f:
        yaml.safe_dump(cfg, f)
    our_cfg = base_config(args.rounds)
    our_cfg["server_config"]["seed"] = 42
    dc = our_cfg["client_config"]["data_config"]["train"]
    dc["list_of_train_data"] = "mnist_train.json"
    our_cfg["server_config"]["data_config"]["val"]["val_data"] = \
        "mnist_test.json"
    our_cfg["server_config"]["data_config"]["test"]["test_data"] = \
        "mnist_test.json"
    our_cfg_path = os.path.join(wd, "our_config.yaml")
    with open(our_cfg_path, "w") as f:
        yaml.safe_dump(our_cfg, f)

    ref_metrics = os.path.join(wd, "ref_metrics.jsonl")
    env_ref = dict(os.environ)
    env_ref.update({
        "PYTHONPATH": SHIMS, "AZUREML_STUB_LOGFILE": ref_metrics,
        "MASTER_ADDR": "127.0.0.1", "HSA_ENABLE_IPC_MODE_LEGACY": "0",
    })
    env_ref.pop("RANK", None), env_ref.pop("WORLD_SIZE", None)
    print("[parity] running REFERENCE (gloo, 2 procs)...", flush=True)
    run([sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc_per_node=2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "e2e_trainer.py",
         "-config", ref_cfg_path, "-outputPath", ref_out,
         "-dataPath", ".", "-task", "cv_lr_mnist", "-backend", "gloo"],
        cwd=sandbox, env=env_ref, log_path=os.path.join(wd, "ref_run.log"))

    env_ours = dict(os.environ)
    env_ours.update({"FLUTE_EXPERIMENT_NAME": "parity-run",
                     "MASTER_ADDR": "127.0.0.1",
                     "HSA_ENABLE_IPC_MODE_LEGACY": "0"})
    env_ours.pop("RANK", None), env_ours.pop("WORLD_SIZE", None)
    print("[parity] running THIS ENGINE (gloo, 2 procs)...", flush=True)
    run([sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc_per_node=2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "e2e_trainer.py",
         "-config", our_cfg_path, "-outputPath", our_out,
         "-dataPath", our_data, "-task", "cv_lr_mnist", "-backend", "gloo"],
        cwd=REPO, env=env_ours, log_path=os.path.join(wd, "our_run.log"))

    # ---- compare ---------------------------------------------------------
    ref_recs = parse_jsonl(ref_metrics)
    our_recs = parse_jsonl(os.path.join(our_out, "parity-run", "log",
                                        "metrics_rank0.jsonl"))
    ref_loss = series(ref_recs, "Training loss")
    our_loss = series(our_recs, "Training loss")
    ref_acc = series(ref_recs, "Test acc")
    our_acc = series(our_recs, "Test acc")

    report = [
        "# Reference head-to-head parity (LR-MNIST, CPU/gloo, 2 procs)", "",
        f"identical synthetic shards ({N_USERS} users x {N_PER_USER}), "
        f"identical initial checkpoint, all clients sampled every round, "
        f"full-batch client SGD lr={CLIENT_LR}, {args.rounds} rounds.", "",
        "| round | reference training loss | this engine | rel diff |",
        "|---|---|---|---|",
    ]
    ok = True
    if len(ref_loss) != len(our_loss) or not ref_loss:
        ok = False
        report.append(f"| LENGTH MISMATCH | {len(ref_loss)} | "
                      f"{len(our_loss)} | — |")
    else:
        for i, (a, b) in enumerate(zip(ref_loss, our_loss)):
            rd = abs(a - b) / max(abs(a), 1e-12)
            mark = "" if rd < 2e-3 else "  **MISMATCH**"
            if rd >= 2e-3:
                ok = False
            report.append(f"| {i} | {a:.6f} | {b:.6f} | {rd:.2e}{mark} |")
        first, last = ref_loss[0], ref_loss[-1]
        if not last < first:
            ok = False
            report.append("")
            report.append(f"loss did not decrease ({first} -> {last})")
    report.append("")
    report.append(f"Test acc: reference={ref_acc} ours={our_acc}")
    if not (ref_acc and our_acc and
            abs(ref_acc[-1] - our_acc[-1]) < 0.02):
        ok = False
        report.append("**final test accuracy diverged (or missing)**")

    # final checkpoint weights
    rck = torch.load(os.path.join(ref_models, "latest_model.tar"),
                     map_location="cpu", weights_only=False)
    ock = torch.load(os.path.join(our_models, "latest_model.tar"),
                     map_location="cpu", weights_only=False)
    wdiff = max((rck["model_state_dict"][k].float()
                 - ock["model_state_dict"][k].float()).abs().max().item()
                for k in rck["model_state_dict"])
    report.append("")
    report.append(f"final checkpoint max|w_ref - w_ours| = {wdiff:.3e} "
                  f"(tolerance 1e-3)")
    if wdiff > 1e-3:
        ok = False
    report.append("")
    report.append("**PARITY: " + ("PASS" if ok else "FAIL") + "**")
    out_path = os.path.join(wd, "PARITY_REPORT.md")
    with open(out_path, "w") as f:
        f.write("\n".join(report) + "\n")
    print("\n".join(report))
    print(f"[parity] report: {out_path}")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
