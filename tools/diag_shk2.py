"""Where does the Shakespeare round go? Real server path, pool vs single."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, yaml
from msrflute_amd.comm import runtime as rt_mod
from msrflute_amd.config import FLUTEConfig
from msrflute_amd.core import client as client_mod
from msrflute_amd.core.client import Client
from msrflute_amd.core.server import OptimizationServer
from msrflute_amd.models import make_model
from msrflute_amd.ops.arena import ParameterArena
from msrflute_amd.utils import make_optimizer
from tools import create_data as cd

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
with open(os.path.join(REPO, "configs", "nlp_rnn_fedshakespeare.yaml")) as f:
    cfg = yaml.safe_load(f)
sc = cfg["server_config"]
sc.update(max_iteration=100, val_freq=10**9, rec_freq=10**9,
          initial_val=False, initial_rec=False, seed=1234)
par = int(os.environ.get("PAR", "8"))
cfg["client_config"]["parallel_clients"] = par

blob = cd.make_shakespeare_blob(n_users=100, samples_per_user=50, seed=7)
data_dir = "/tmp/shk_diag"
cd.save_blob(blob, os.path.join(data_dir, "nlp_rnn_fedshakespeare",
                                "train_data.pt"))
config = FLUTEConfig.from_dict(cfg)
config["model_path"] = "/tmp/shk_models"
os.makedirs(config["model_path"], exist_ok=True)
rt = rt_mod.init_runtime(backend="nccl", seed=1234)
n_users = Client.get_train_dataset(
    data_dir, config, "nlp_rnn_fedshakespeare")
torch.manual_seed(99)
model = make_model(config["model_config"])
arena = ParameterArena(model, bind_grads=True)
opt = make_optimizer(dict(sc["optimizer_config"]), model)
server = OptimizationServer(
    num_clients=n_users, model=model, optimizer=opt, ss_scheduler=None,
    data_path=data_dir, model_path=config["model_path"],
    server_train_dataloader=None, config=config, idx_val_clients=[],
    idx_test_clients=[], runtime=rt, arena=arena,
    task="nlp_rnn_fedshakespeare")
server.run_stats = {k: [] for k in [
    "secsPerClientRound", "secsPerClient", "secsPerClientTraining",
    "secsPerClientSetup", "secsPerClientFull",
    "secsPerRoundHousekeeping", "secsPerRoundTotal", "communicationCosts"]}
for i in range(3):
    server.run_one_round(i, housekeeping=False)
torch.cuda.synchronize()
import cProfile, pstats, io
pr = cProfile.Profile()
t0 = time.time(); N = 10
pr.enable()
for i in range(3, 3 + N):
    server.run_one_round(i, housekeeping=False)
pr.disable()
torch.cuda.synchronize()
buf = io.StringIO()
pstats.Stats(pr, stream=buf).sort_stats("cumulative").print_stats(22)
print(buf.getvalue()[:3200])
ms = (time.time() - t0) / N * 1000
print(f"PAR={par} ms/round={ms:.1f}")
acc = dict(server.executor.perf_acc)
n = max(acc.pop("clients", 1), 1)
print("clients:", n, {k: round(v / n * 1000, 2) for k, v in acc.items()})
ex = (server.executor.executors[0] if hasattr(server.executor, "executors")
      else server.executor)
gc = ex.graph_cache
print("graph cache:", None if gc is None else list(gc._graphs.keys())[:4])
