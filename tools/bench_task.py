#!/usr/bin/env python3
"""Per-task FL-round benchmark (BASELINE configs 2-4 measured on MI355X).

Like bench.py (the driver contract stays CNN-FEMNIST) but parameterized
over any shipped task: synthetic shards, timed rounds, one JSON line.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--task", required=True)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--users", type=int, default=None)
    ap.add_argument("--samples", type=int, default=None)
    args = ap.parse_args()

    from msrflute_amd.comm import runtime as rt_mod
    from msrflute_amd.config import FLUTEConfig
    from msrflute_amd.core.client import Client
    from msrflute_amd.core.server import OptimizationServer
    from msrflute_amd.models import make_model
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.fused_optim import make_arena_optimizer
    from msrflute_amd.utils import make_optimizer
    from tools import create_data as cd

    with open(os.path.join(REPO, "configs", f"{args.task}.yaml")) as f:
        cfg = yaml.safe_load(f)
    sc = cfg["server_config"]
    sc.update(max_iteration=args.warmup + args.steps, val_freq=10 ** 9,
              rec_freq=10 ** 9, initial_val=False, initial_rec=False,
              seed=1234)

    fn, kw = cd.TASKS[args.task]
    kw = dict(kw)
    if args.users:
        kw["n_users"] = args.users
    if args.samples and "samples_per_user" in kw:
        kw["samples_per_user"] = args.samples
    data_dir = "/tmp/bench_task_data"
    cd.save_blob(fn(seed=7, **kw),
                 os.path.join(data_dir, args.task, "train_data.pt"))

    config = FLUTEConfig.from_dict(cfg)
    config["model_path"] = "/tmp/bench_task_models"
    os.makedirs(config["model_path"], exist_ok=True)
    config["client_config"]["task"] = args.task
    config["server_config"]["task"] = args.task

    backend = "nccl" if torch.cuda.is_available() else "gloo"
    rt = rt_mod.init_runtime(backend=backend, seed=1234)
    num_clients = Client.get_train_dataset(data_dir, config, args.task)

    torch.manual_seed(1234)
    model = make_model(config["model_config"])
    arena = ParameterArena(model, bind_grads=True)
    optimizer = make_arena_optimizer(
        dict(config["server_config"]["optimizer_config"]), arena)
    if optimizer is None:
        optimizer = make_optimizer(
            config["server_config"]["optimizer_config"], model)

    server = OptimizationServer(
        num_clients=num_clients, model=model, optimizer=optimizer,
        ss_scheduler=None, data_path=data_dir,
        model_path=config["model_path"], server_train_dataloader=None,
        config=config, idx_val_clients=[], idx_test_clients=[],
        runtime=rt, arena=arena, task=args.task)
    server.run_stats = {k: [] for k in [
        "secsPerClientRound", "secsPerClient", "secsPerClientTraining",
        "secsPerClientSetup", "secsPerClientFull",
        "secsPerRoundHousekeeping", "secsPerRoundTotal",
        "communicationCosts"]}
    server.worker_trainer.model.train()

    for i in range(args.warmup):
        server.run_one_round(i, housekeeping=False)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.time()
    for i in range(args.warmup, args.warmup + args.steps):
        server.run_one_round(i, housekeeping=False)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    elapsed = time.time() - t0
    print(json.dumps({
        "task": args.task, "rounds_per_sec": args.steps / elapsed,
        "ms_per_round": elapsed / args.steps * 1000,
        "clients_per_round": server.num_clients_per_iteration[0],
        "peak_gpu_mem_mb": round(torch.cuda.max_memory_allocated() / 2**20, 1)
        if torch.cuda.is_available() else 0.0}))
    rt.shutdown()


if __name__ == "__main__":
    main()
