#!/usr/bin/env python3
"""FLUTE-compatible entrypoint (reference: e2e_trainer.py:77-253).

Same CLI contract as the reference::

    python -m torch.distributed.run --nproc_per_node=K e2e_trainer.py \
        -dataPath <data> -outputPath <out> -config <yaml> -task <task> \
        -backend nccl|gloo

Differences from the reference: no AzureML coupling (local JSONL metrics
sink), and the distributed topology is the symmetric-replica runtime of
msrflute_amd.comm.runtime — every rank is a full replica that trains its
share of each round's clients; there is no dedicated parameter-server
process.
"""

import argparse
import logging
import os
import shutil

import torch
import yaml

from msrflute_amd.comm import runtime as federated
from msrflute_amd.config import FLUTEConfig
from msrflute_amd.core.client import Client
from msrflute_amd.core.evaluation import make_eval_clients
from msrflute_amd.core.server import select_server
from msrflute_amd.models import make_model
from msrflute_amd.ops.arena import ParameterArena
from msrflute_amd.ops.fused_optim import make_arena_optimizer
from msrflute_amd.utils import (init_logging, init_metrics_sink, log_metric,
                                make_optimizer, print_rank)
from msrflute_amd.utils.dataloaders_utils import (get_dataset,
                                                  make_train_dataloader)


def find_pretrained_model(model_path, model_config):
    """Look for a pretrained model path (reference: utils/utils.py)."""
    p = model_config.get("pretrained_model_path", None)
    if p and os.path.exists(p):
        return p
    return None


def log_run_properties(config):
    """Log run properties to the local metrics sink
    (reference: e2e_trainer.py:40-74 logged to AzureML)."""
    for key in ["strategy"]:
        log_metric(key, config.get(key))
    log_metric("Max iterations", config.lookup("server_config.max_iteration"))
    log_metric("Server optimizer",
               config.lookup("server_config.optimizer_config.type"))


def run_worker(model_path, config, task, data_path, local_rank, backend):
    """Bring-up on every rank (reference: e2e_trainer.py:77-195)."""
    model_config = config["model_config"]
    server_config = config["server_config"]

    print_rank(f"Backend: {backend}")
    seed = int(config["server_config"].get("seed", 0) or 0)
    rt = federated.init_runtime(backend=backend, seed=seed)

    # identical model init on every rank (then one broadcast to be safe
    # against nondeterministic user model constructors)
    torch.manual_seed(seed + 12345)
    model = make_model(model_config)

    val_dataset = get_dataset(data_path, config, task, mode="val", test_only=True)
    test_dataset = get_dataset(data_path, config, task, mode="test", test_only=True)
    val_clients = list(make_eval_clients(val_dataset, config))
    test_clients = list(make_eval_clients(test_dataset, config))

    num_clients = Client.get_train_dataset(data_path, config, task)
    config["server_config"]["data_config"]["num_clients"] = num_clients

    # flat arena for the global model; one startup broadcast syncs replicas
    arena = ParameterArena(model, bind_grads=True)
    rt.broadcast_(arena.data, src=0)

    if "train" in config["server_config"]["data_config"]:
        server_train_dataloader = make_train_dataloader(
            config["server_config"]["data_config"]["train"], data_path,
            task=task, clientx=None)
    else:
        server_train_dataloader = None

    optimizer = make_arena_optimizer(
        dict(server_config["optimizer_config"]), arena)
    if optimizer is None:
        optimizer = make_optimizer(server_config["optimizer_config"], model)

    best_trained_model = find_pretrained_model(model_path, model_config)
    if best_trained_model is not None:
        model_state_dict = torch.load(
            best_trained_model,
            map_location=None if torch.cuda.is_available() else torch.device("cpu"),
            weights_only=False)
        model.load_state_dict(model_state_dict)

    server_setup = select_server(server_config["type"])
    server = server_setup(
        num_clients=num_clients,
        model=model,
        optimizer=optimizer,
        ss_scheduler=None,
        data_path=data_path,
        model_path=model_path,
        server_train_dataloader=server_train_dataloader,
        config=config,
        idx_val_clients=val_clients,
        idx_test_clients=test_clients,
        runtime=rt,
        arena=arena,
        val_dataset=val_dataset,
        test_dataset=test_dataset,
        task=task,
    )
    if rt.rank == 0:
        log_run_properties(config)
    print_rank("Launching server")
    try:
        server.run()
    finally:
        rt.shutdown()


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("-config")
    parser.add_argument("-outputPath")
    parser.add_argument("-dataPath", default=None)
    parser.add_argument("-task", default=None, help="Define the task for the run")
    parser.add_argument("-backend", default=None,
                        help="Define the communication protocol")
    parser.add_argument("-num_skip_decoding", default=-1, type=int,
                        help="Skip decoding in unsupervised learning mode")
    parser.add_argument("--local_rank", default=-1, type=int)
    args = parser.parse_args()

    data_path = args.dataPath
    task = args.task
    assert args.backend in ["nccl", "gloo"], \
        f"Backend {args.backend} not recognized, please select nccl or gloo"

    experiment_name = os.environ.get("FLUTE_EXPERIMENT_NAME", "msrflute_amd")
    experiment_root = os.path.join(args.outputPath, experiment_name)
    model_path = os.path.join(experiment_root, "models")
    log_path = os.path.join(experiment_root, "log")
    os.makedirs(model_path, exist_ok=True)
    os.makedirs(log_path, exist_ok=True)

    if federated.rank() == 0:
        shutil.copyfile(args.config,
                        os.path.join(experiment_root, "FLUTE_config.yaml"))

    init_logging(log_path, loglevel=logging.INFO)
    init_metrics_sink(os.path.join(log_path, f"metrics_rank{federated.rank()}.jsonl"))

    with open(args.config) as f:
        cfg_dict = yaml.safe_load(f)
    config = FLUTEConfig.from_dict(cfg_dict)
    config["data_path"] = data_path
    config["output_path"] = args.outputPath
    config["model_path"] = model_path
    config["experiment_name"] = experiment_name
    config["client_config"]["task"] = task
    config["server_config"]["task"] = task
    config.validate()

    run_worker(model_path, config, task, data_path, args.local_rank, args.backend)


if __name__ == "__main__":
    main()
