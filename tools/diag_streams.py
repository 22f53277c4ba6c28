"""Do concurrent stream workloads overlap? raw LSTM kernels vs graphs."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from msrflute_amd import _C

B, T, H = 4, 80, 256
NS = int(os.environ.get("NS", "8"))
xps = [torch.randn(B, T, 4*H, device="cuda") for _ in range(NS)]
whts = [torch.randn(H, 4*H, device="cuda") for _ in range(NS)]
streams = [torch.cuda.Stream() for _ in range(NS)]

def run_serial(n_iter=10):
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(n_iter):
        for k in range(NS):
            _C.lstm_seq_fwd(xps[k], whts[k])
    torch.cuda.synchronize()
    return (time.time()-t0)/n_iter*1000

def run_streams(n_iter=10):
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(n_iter):
        for k in range(NS):
            with torch.cuda.stream(streams[k]):
                _C.lstm_seq_fwd(xps[k], whts[k])
        for st in streams:
            torch.cuda.current_stream().wait_stream(st)
    torch.cuda.synchronize()
    return (time.time()-t0)/n_iter*1000

run_serial(3); run_streams(3)
s = run_serial(); p = run_streams()
print(f"NS={NS} serial {s:.2f} ms, streams {p:.2f} ms, overlap factor {s/p:.2f}")
