"""ClientPool (stream-parallel clients) vs single executor: identical
round results on a dropout-free model (GPU only).

FedAvg aggregation is a weighted sum, so client concurrency must not
change the aggregate beyond fp reduction order; with LR-MNIST (no
dropout) and the per-client seed discipline the aggregate matches
tightly.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_rounds(parallel_clients, n_rounds=3):
    import bench
    from msrflute_amd.comm import runtime as rt_mod
    from msrflute_amd.core import client as client_mod
    from msrflute_amd.core.server import OptimizationServer
    from msrflute_amd.models import make_model
    from msrflute_amd.models.generic_data import ArrayDataset
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.fused_optim import make_arena_optimizer
    from tools.create_data import make_mnist_blob

    class A:
        warmup = 0
        steps = n_rounds
        clients_per_round = 8

    config = bench.build_config(A())
    config["model_config"] = {
        "model_type": "LR",
        "model_folder": "experiments/cv_lr_mnist/model.py",
        "input_dim": 784, "output_dim": 10}
    config["client_config"]["parallel_clients"] = parallel_clients
    config["model_path"] = "/tmp/pool_models"
    os.makedirs(config["model_path"], exist_ok=True)

    blob = make_mnist_blob(n_users=40, samples_per_user=64, seed=3)
    ds = ArrayDataset(blob, test_only=False, user_idx=-1, args={})
    ds.user_data = blob["user_data"]
    ds.user_data_label = blob["user_data_label"]
    client_mod.train_dataset = ds

    rt = rt_mod.init_runtime(backend="gloo", seed=777)
    torch.manual_seed(777 + 12345)
    model = make_model(config["model_config"])
    arena = ParameterArena(model, bind_grads=True)
    optimizer = make_arena_optimizer(
        dict(config["server_config"]["optimizer_config"]), arena)
    server = OptimizationServer(
        num_clients=40, model=model, optimizer=optimizer, ss_scheduler=None,
        data_path=None, model_path=config["model_path"],
        server_train_dataloader=None, config=config, idx_val_clients=[],
        idx_test_clients=[], runtime=rt, arena=arena, task="cv_lr_mnist")
    server.run_stats = {k: [] for k in [
        "secsPerClientRound", "secsPerClient", "secsPerClientTraining",
        "secsPerClientSetup", "secsPerClientFull",
        "secsPerRoundHousekeeping", "secsPerRoundTotal",
        "communicationCosts"]}
    for i in range(n_rounds):
        server.run_one_round(i, housekeeping=False)
    torch.cuda.synchronize()
    out = arena.data.clone().cpu()
    rt.shutdown()
    rt_mod.set_runtime(None)
    return out


def test_pool_matches_single_executor():
    w1 = _run_rounds(parallel_clients=1)
    w4 = _run_rounds(parallel_clients=4)
    assert torch.allclose(w1, w4, rtol=1e-5, atol=1e-6), \
        (w1 - w4).abs().max().item()
