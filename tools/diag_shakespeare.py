"""Diagnose the Shakespeare round: where do the 300 ms go?"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from msrflute_amd.models import make_model
from msrflute_amd.ops.arena import ParameterArena
from tools import create_data as cd

blob = cd.make_shakespeare_blob(n_users=20, samples_per_user=50, seed=7)
mc = {"model_type": "RNN",
      "model_folder": "experiments/nlp_rnn_fedshakespeare/model.py",
      "vocab_size": 90, "embed_dim": 8, "hidden_dim": 256}
torch.manual_seed(0)
m = make_model(mc)
arena = ParameterArena(m, bind_grads=True)
u = blob["users"][0]
xs = blob["user_data"][u]["x"]
ys = blob["user_data_label"][u]
x = torch.tensor(xs[:4], dtype=torch.long, device="cuda")
y = torch.tensor(ys[:4], dtype=torch.long, device="cuda")
print("batch x", x.shape)

def step():
    arena.grad.zero_()
    loss = m.loss({"x": x, "y": y})
    loss.backward()
    return loss

for _ in range(3):
    step()
torch.cuda.synchronize()
t0 = time.time(); N = 20
for _ in range(N):
    step()
torch.cuda.synchronize()
print(f"eager fwd+bwd per batch: {(time.time()-t0)/N*1000:.2f} ms")

# kernel-level: time the lstm fwd/bwd alone
from msrflute_amd import _C
B, T, H = 4, x.shape[1], 256
xp = torch.randn(B, T, 4*H, device="cuda")
whh = torch.randn(4*H, H, device="cuda")
torch.cuda.synchronize(); t0=time.time()
for _ in range(N):
    from msrflute_amd.ops.lstm import _pack_fwd, _pack_bwd
    h_seq, gates, c_seq = _C.lstm_seq_fwd(xp, _pack_fwd(whh))
torch.cuda.synchronize()
print(f"lstm_seq_fwd per call: {(time.time()-t0)/N*1000:.3f} ms (T={T})")
dh = torch.randn_like(h_seq)
torch.cuda.synchronize(); t0=time.time()
for _ in range(N):
    dg = _C.lstm_seq_bwd(gates, c_seq, _pack_bwd(whh), dh)
torch.cuda.synchronize()
print(f"lstm_seq_bwd per call: {(time.time()-t0)/N*1000:.3f} ms")

# graph path check
from msrflute_amd.ops.graphs import GraphCache
gc = GraphCache(m, arena, {"type": "sgd", "lr": 0.8}, None)
print("graph supports:", gc.supports())
g = gc.get(x, y)
print("per-batch graph captured:", g is not None)
if g is not None:
    gc.set_lr(0.8)
    torch.cuda.synchronize(); t0=time.time()
    for _ in range(N):
        g.run(x, y)
    torch.cuda.synchronize()
    print(f"graph replay per batch: {(time.time()-t0)/N*1000:.3f} ms")
