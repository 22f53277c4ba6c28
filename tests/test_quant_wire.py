"""Quantized transport: 8-bit codes + per-segment
scales on the wire for the round reduce, dequant-accumulated in fixed
rank order.  2-proc gloo: error bound vs the exact fp32 reduce, bitwise
identical result on both ranks, exact reconstruction for
scale-multiple inputs, and ~4x wire-byte reduction."""

import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import os, sys, torch
sys.path.insert(0, os.environ["REPO"])
import torch.distributed as dist
from msrflute_amd.comm.runtime import FedRuntime

rt = FedRuntime(backend="gloo", seed=0)
r = rt.rank
torch.manual_seed(100 + r)

# two segments of different scales (arena-style layout)
seg_expand = torch.cat([torch.zeros(1000, dtype=torch.int64),
                        torch.ones(500, dtype=torch.int64)])
g = torch.cat([torch.randn(1000) * 3.0, torch.randn(500) * 0.01])
g_ref = g.clone()
# exact reduce
exact = g_ref.clone(); rt.all_reduce_(exact)

h = rt.begin_grad_reduce_quant(g, float(r + 1), seg_expand)
wsum = rt.finish_grad_reduce(h)
assert wsum == 3.0, wsum

# per-element error bound: each rank contributes <= scale_r/2 rounding
scales0 = g_ref[:1000].abs().max() / 127.0
scales1 = g_ref[500:].abs().max() / 127.0
# gather both ranks' scales for the bound
smax = torch.tensor([max(float(scales0), float(scales1))])
rt.all_reduce_(smax)  # sum of per-rank max scales >= sum of per-rank scale/2*2
err = (g - exact).abs()
assert float(err.max()) <= float(smax) , (float(err.max()), float(smax))

# both ranks hold the bitwise-identical dequantized sum
g_other = g.clone()
dist.broadcast(g_other, src=0)
assert torch.equal(g, g_other), "replicas diverged"

# exact reconstruction: codes that are exact multiples of the scale
q = torch.cat([torch.arange(-127, 127, dtype=torch.float32) / 127.0 * 5.0,
               torch.zeros(1246)])
seg1 = torch.zeros(1500, dtype=torch.int64)
q_in = q.clone()
h2 = rt.begin_grad_reduce_quant(q_in, 1.0, seg1)
rt.finish_grad_reduce(h2)
assert torch.allclose(q_in, q * 2, atol=1e-6), "scale-multiple values must survive the wire"

# wire bytes: int8 codes = numel bytes vs fp32's 4x
codes = torch.clamp(torch.round(torch.randn(1500)), -127, 127).to(torch.int8)
assert codes.element_size() * codes.numel() == 1500
print("RANK_OK", r)
rt.shutdown()
"""


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def test_quant_wire_two_proc(tmp_path):
    script = tmp_path / "w.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env.update(REPO=REPO, PYTHONPATH=REPO, MASTER_ADDR="127.0.0.1")
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), str(script)],
        env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert r.stdout.count("RANK_OK") == 2


def test_quant_wire_e2e_close_to_fp32(tmp_path):
    """A 2-proc gloo FL run with quant_wire tracks the fp32-wire run."""
    script = tmp_path / "w.py"
    script.write_text(r"""
import os, sys, json, torch
sys.path.insert(0, os.environ["REPO"])
import numpy as np
from msrflute_amd.comm import runtime as rt_mod
from msrflute_amd.config import FLUTEConfig
from msrflute_amd.core import client as client_mod
from msrflute_amd.core.client import Client
from msrflute_amd.core.server import OptimizationServer
from msrflute_amd.models import make_model
from msrflute_amd.ops.arena import ParameterArena
from msrflute_amd.utils import make_optimizer
from tools.create_data import make_femnist_blob
from msrflute_amd.models.generic_data import ArrayDataset

rt = rt_mod.init_runtime(backend="gloo", seed=7)
results = {}
for quant_wire in [False, True]:
    cfg = {
        "model_config": {"model_type": "LR",
                         "model_folder": "experiments/cv_lr_mnist/model.py",
                         "input_dim": 784, "output_dim": 62},
        "dp_config": {"enable_local_dp": False},
        "privacy_metrics_config": {"apply_metrics": False},
        "strategy": "FedAvg",
        "server_config": {
            "wantRL": False, "resume_from_checkpoint": False,
            "do_profiling": False,
            "optimizer_config": {"type": "sgd", "lr": 1.0},
            "annealing_config": {"type": "step_lr", "step_interval": "epoch",
                                 "gamma": 1.0, "step_size": 100},
            "val_freq": 10**9, "rec_freq": 10**9,
            "initial_val": False, "initial_rec": False,
            "max_iteration": 4, "num_clients_per_iteration": 8,
            "data_config": {"val": {"batch_size": 64, "val_data": None},
                            "test": {"batch_size": 64, "test_data": None}},
            "type": "model_optimization", "aggregate_median": "mean",
            "weight_train_loss": "train_loss", "softmax_beta": 1.0,
            "initial_lr_client": 0.05, "lr_decay_factor": 1.0,
            "best_model_criterion": "loss", "fall_back_to_best_model": False,
            "seed": 7},
        "client_config": {
            "quant_wire": quant_wire,
            "do_profiling": False, "ignore_subtask": False,
            "data_config": {"train": {"batch_size": 64,
                                      "list_of_train_data": None,
                                      "desired_max_samples": 10000}},
            "type": "optimization",
            "optimizer_config": {"type": "sgd", "lr": 0.05}},
    }
    config = FLUTEConfig.from_dict(cfg)
    config["model_path"] = os.environ["OUT"] + f"/m{int(quant_wire)}"
    os.makedirs(config["model_path"], exist_ok=True)
    blob = make_femnist_blob(n_users=16, samples_per_user=12, seed=3)
    ds = ArrayDataset(blob, test_only=False, user_idx=-1, args={},
                      x_shape=(784,))
    ds.user_data = blob["user_data"]; ds.user_data_label = blob["user_data_label"]
    # LR model expects 784-feature rows; blob x is (n,784) already
    client_mod.train_dataset = ds
    torch.manual_seed(5)
    model = make_model(cfg["model_config"])
    arena = ParameterArena(model, bind_grads=True)
    rt.broadcast_(arena.data, src=0)
    opt = make_optimizer(dict(cfg["server_config"]["optimizer_config"]), model)
    server = OptimizationServer(
        num_clients=16, model=model, optimizer=opt, ss_scheduler=None,
        data_path=None, model_path=config["model_path"],
        server_train_dataloader=None, config=config, idx_val_clients=[],
        idx_test_clients=[], runtime=rt, arena=arena, task="cv_lr_mnist")
    server.run_stats = {k: [] for k in [
        "secsPerClientRound", "secsPerClient", "secsPerClientTraining",
        "secsPerClientSetup", "secsPerClientFull",
        "secsPerRoundHousekeeping", "secsPerRoundTotal",
        "communicationCosts"]}
    for i in range(4):
        server.run_one_round(i, housekeeping=False)
    results[quant_wire] = (arena.data.clone(), sum(server.train_loss))

w_fp, l_fp = results[False]
w_q, l_q = results[True]
rel = float((w_fp - w_q).norm() / w_fp.norm())
assert rel < 2e-2, rel
assert abs(l_fp - l_q) / abs(l_fp) < 5e-2, (l_fp, l_q)
print("RANK_OK")
""")
    env = dict(os.environ)
    env.update(REPO=REPO, PYTHONPATH=REPO, OUT=str(tmp_path),
               MASTER_ADDR="127.0.0.1")
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), str(script)],
        env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-2000:])
    assert r.stdout.count("RANK_OK") == 2
