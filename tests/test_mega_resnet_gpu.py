"""Cross-client ResNet MEGA round (ops/mega_resnet.py; OPT-IN via
use_mega_round_resnet — measured slower than the per-client epoch-graph
path on ROCm 7.2, kept as an exact alternative formulation).

Equivalence methodology: the model math is proven EXACT on CPU in f64
(tests/test_mega_cpu.py).  On GPU in fp32 this task is chaotically
sensitive — the per-batch clip-to-norm-10 rescales every gradient by
10/||g||, so conv-algorithm fp noise amplifies multiplicatively; even
the production graphed and eager per-client paths (identical semantics)
diverge ~1.3e-3 after ONE step and ~1.3e-2 after two 3-batch rounds
(tools/diag_resnet_mega.py, measured).  The assertable contract is
therefore ONE round / one batch per client: per-round losses match
tightly and weights within the same envelope the production paths
occupy relative to each other."""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))

WORKER = r"""
import os, sys, torch, yaml
sys.path.insert(0, os.environ["REPO"])
from msrflute_amd.comm import runtime as rt_mod
from msrflute_amd.config import FLUTEConfig
from msrflute_amd.core import client as client_mod
from msrflute_amd.core.server import OptimizationServer
from msrflute_amd.models import make_model
from msrflute_amd.models.generic_data import ArrayDataset
from msrflute_amd.ops.arena import ParameterArena
from msrflute_amd.ops.fused_optim import make_arena_optimizer
from tools.create_data import make_fedcifar100_blob

def run(mega, samples=20):
    rt_mod.set_runtime(None)
    rt = rt_mod.init_runtime(backend="nccl", seed=99)
    with open(os.path.join(os.environ["REPO"],
                           "configs/cv_resnet_fedcifar100.yaml")) as f:
        cfg = yaml.safe_load(f)
    cfg["server_config"].update(
        max_iteration=1, num_clients_per_iteration=5, seed=99,
        val_freq=10**9, rec_freq=10**9, initial_val=False,
        initial_rec=False)
    cfg["server_config"]["data_config"]["val"]["val_data"] = None
    cfg["server_config"]["data_config"]["test"]["test_data"] = None
    cfg["client_config"]["use_mega_round"] = mega
    cfg["client_config"]["use_mega_round_resnet"] = mega
    cfg["client_config"]["parallel_clients"] = 4 if mega else 1
    cfg["client_config"]["data_config"]["train"]["list_of_train_data"] = None
    cfg["client_config"]["data_config"]["train"]["batch_size"] = 20
    config = FLUTEConfig.from_dict(cfg)
    config["model_path"] = os.environ["OUT"] + f"/m_{int(mega)}"
    os.makedirs(config["model_path"], exist_ok=True)
    blob = make_fedcifar100_blob(n_users=10, samples_per_user=samples,
                                 seed=3)
    ds = ArrayDataset(blob, test_only=False, user_idx=-1, args={},
                      x_shape=(3, 24, 24))
    ds.user_data = blob["user_data"]
    ds.user_data_label = blob["user_data_label"]
    client_mod.train_dataset = ds
    torch.manual_seed(5)
    model = make_model(cfg["model_config"])
    arena = ParameterArena(model, bind_grads=True)
    opt = make_arena_optimizer(dict(cfg["server_config"]["optimizer_config"]),
                               arena)
    server = OptimizationServer(
        num_clients=10, model=model, optimizer=opt, ss_scheduler=None,
        data_path=None, model_path=config["model_path"],
        server_train_dataloader=None, config=config, idx_val_clients=[],
        idx_test_clients=[], runtime=rt, arena=arena,
        task="cv_resnet_fedcifar100")
    server.run_stats = {k: [] for k in [
        "secsPerClientRound", "secsPerClient", "secsPerClientTraining",
        "secsPerClientSetup", "secsPerClientFull",
        "secsPerRoundHousekeeping", "secsPerRoundTotal",
        "communicationCosts"]}
    server.run_one_round(0, housekeeping=False)
    torch.cuda.synchronize()
    if mega:
        mr = getattr(server.executor, "_mega_resnet", None)
        assert mr not in (None, False), "resnet mega did not engage"
        assert mr._graphs, "resnet mega never ran"
    return arena.data.clone(), sum(server.train_loss)

w_ref, l_ref = run(mega=False)
w_mega, l_mega = run(mega=True)
rel = float((w_ref - w_mega).norm() / w_ref.norm())
print("rel weight diff:", rel, "losses:", l_ref, l_mega)
# measured envelope: graphed-vs-eager production paths differ ~1.3e-3
# on this exact config; mega-vs-exact-eager is ~3e-4
assert rel < 5e-3, rel
assert abs(l_ref - l_mega) / abs(l_ref) < 1e-4, (l_ref, l_mega)
print("MEGA_RESNET_OK")
"""


def test_mega_resnet_matches_per_client(tmp_path):
    env = dict(os.environ)
    env.update(REPO=REPO, PYTHONPATH=REPO, OUT=str(tmp_path))
    r = subprocess.run([sys.executable, "-c", WORKER], env=env,
                       capture_output=True, text=True, timeout=900,
                       cwd=REPO)
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-3000:])
    assert "MEGA_RESNET_OK" in r.stdout
