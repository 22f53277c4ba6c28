"""2-process gloo tests for the runtime collectives (all_gather_rows,
weighted all-reduce equivalence of the round math)."""

import os
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

# variable-row gather: rank0 sends 2 rows, rank1 sends 3
rows = torch.arange(6, dtype=torch.float64).reshape(3, 2) + 10 * r
mine = rows[: 2 + r]
got = rt.all_gather_rows(mine, [2, 3])
assert [g.shape[0] for g in got] == [2, 3], got
assert torch.allclose(got[0], torch.arange(4, dtype=torch.float64).reshape(2, 2))
assert torch.allclose(got[1], torch.arange(6, dtype=torch.float64).reshape(3, 2) + 10)

# round math: weighted sum over ranks == serial weighted sum
g = torch.full((5,), float(r + 1))
w = float(r + 1)
t = g * w
rt.all_reduce_(t)
wsum = torch.tensor([w]); rt.all_reduce_(wsum)
agg = t / wsum
assert torch.allclose(agg, torch.tensor([(1.0 + 4.0) / 3.0] * 5)), agg
print("RANK_OK", r)
rt.shutdown()
"""


def test_runtime_collectives_two_proc(tmp_path):
    script = tmp_path / "w.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env.update(REPO=REPO, PYTHONPATH=REPO)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", "29811", str(script)],
        env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert r.stdout.count("RANK_OK") == 2
