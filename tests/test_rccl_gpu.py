"""RCCL hardening on single-GPU hardware: a 1-rank nccl(=RCCL) process
group with ``FedRuntime._force_collectives`` runs every device-side
collective branch that an 8-GPU round uses — real RCCL init, device
all_reduce (fp32 arena + fp64 scalar), the overlapped comm-stream round
reduce, device all_gather_rows, broadcast and barrier(device_ids) — so
the nccl code paths are executed on hardware, not just on gloo/CPU
(VERDICT round-1 item 1)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def rt():
    import torch.distributed as dist

    from msrflute_amd.comm.runtime import FedRuntime
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
    r = FedRuntime(backend="nccl", seed=0)
    r._force_collectives = True
    yield r
    r._force_collectives = False


def test_device_all_reduce_fp32_and_fp64(rt):
    g = torch.randn(1_000_000, device="cuda")
    ref = g.clone()
    rt.all_reduce_(g)  # 1-rank sum == identity, but runs RCCL
    assert torch.equal(g, ref)
    t = torch.tensor([3.5], dtype=torch.float64, device="cuda")
    rt.all_reduce_(t)
    assert float(t.item()) == 3.5


def test_overlapped_round_reduce(rt):
    g = torch.randn(2_000_000, device="cuda")
    ref = g.clone()
    h = rt.begin_grad_reduce(g, 7.0)
    # host-side work the reduce overlaps (the round's real overlap is the
    # lazy-stats finalize + metadata gather)
    busy = torch.randn(4096, 4096, device="cuda")
    busy = busy @ busy
    wsum = rt.finish_grad_reduce(h)
    torch.cuda.synchronize()
    assert wsum == 7.0
    assert torch.equal(g, ref)
    assert rt._comm_stream is not None  # reduce really used the comm stream


def test_device_all_gather_rows(rt):
    t = torch.arange(12, dtype=torch.float64).reshape(3, 4)
    out = rt.all_gather_rows(t, [3])
    assert len(out) == 1
    assert torch.allclose(out[0], t)


def test_broadcast_and_barrier(rt):
    x = torch.randn(1000, device="cuda")
    ref = x.clone()
    rt.broadcast_(x, src=0)
    assert torch.equal(x, ref)
    rt.barrier()  # barrier(device_ids=[...]) branch


def test_full_round_through_server_with_forced_collectives(rt):
    """One FL round end-to-end with every collective live on RCCL."""
    import subprocess
    import sys
    # run in a subprocess so the module-level dataset cache and runtime
    # singleton of other tests don't interfere
    code = """
import os, socket, torch, torch.distributed as dist
os.environ["MASTER_ADDR"] = "127.0.0.1"
s = socket.socket(); s.bind(("127.0.0.1", 0))
os.environ["MASTER_PORT"] = str(s.getsockname()[1])  # parent holds 29571
s.close()
dist.init_process_group("nccl", rank=0, world_size=1)
import bench
import sys
sys.argv = ["bench.py", "--steps", "2", "--warmup", "1", "--clients", "40",
            "--samples-per-client", "40"]
from msrflute_amd.comm import runtime as rt_mod
orig_init = rt_mod.init_runtime
def patched(backend="nccl", seed=0):
    r = orig_init(backend=backend, seed=seed)
    r._force_collectives = True
    return r
rt_mod.init_runtime = patched
bench.main()
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=420,
                       cwd=os.path.dirname(os.path.dirname(__file__)))
    assert r.returncode == 0, r.stderr[-3000:]
    assert '"metric"' in r.stdout


def test_quant_wire_device_path(rt):
    """Device-side quantized transport (int8 codes + segment scales) on a
    1-rank RCCL group: dequant(quant(g)) within per-segment bound."""
    torch.manual_seed(3)
    seg_expand = torch.cat([torch.zeros(100_000, dtype=torch.int64),
                            torch.ones(50_000, dtype=torch.int64)]).cuda()
    g = torch.cat([torch.randn(100_000, device="cuda") * 2.0,
                   torch.randn(50_000, device="cuda") * 0.05])
    ref = g.clone()
    h = rt.begin_grad_reduce_quant(g, 4.0, seg_expand)
    wsum = rt.finish_grad_reduce(h)
    torch.cuda.synchronize()
    assert wsum == 4.0
    s0 = ref[:100_000].abs().max() / 127.0
    s1 = ref[100_000:].abs().max() / 127.0
    err0 = (g[:100_000] - ref[:100_000]).abs().max()
    err1 = (g[100_000:] - ref[100_000:]).abs().max()
    assert float(err0) <= float(s0) / 2 + 1e-7
    assert float(err1) <= float(s1) / 2 + 1e-7
