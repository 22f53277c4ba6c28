"""Three-way divergence triangulation for the ResNet mega round:
production per-client GRAPHED path vs production per-client EAGER path
(use_hip_graphs=false) vs the mega round, via the full server harness.
Whichever pair disagrees localizes the defect."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import tests.test_mega_resnet_gpu as t

worker = t.WORKER
worker = worker.replace(
    "def run(mega, samples=45):",
    "def run(mega, samples=45, graphs=True):")
worker = worker.replace(
    'cfg["client_config"]["use_mega_round"] = mega',
    'cfg["client_config"]["use_mega_round"] = mega\n'
    '    cfg["client_config"]["use_hip_graphs"] = graphs')
worker = worker.replace(
    "for i in range(2):",
    "for i in range(int(os.environ.get('DIAG_ROUNDS', '2'))):")
worker = worker.replace(
    'config["model_path"] = os.environ["OUT"] + f"/m_{int(mega)}"',
    'config["model_path"] = os.environ["OUT"] + '
    'f"/m_{int(mega)}_{int(graphs)}"')
tail = worker.index("w_ref, l_ref = run(mega=False)")
worker = worker[:tail] + """
S = int(os.environ.get("DIAG_SAMPLES", "45"))
w_g, l_g = run(mega=False, graphs=True, samples=S)
w_e, l_e = run(mega=False, graphs=False, samples=S)
w_m, l_m = run(mega=True, samples=S)
import torch
def rel(a, b):
    return float((a - b).norm() / a.norm())
print("graphed vs eager :", rel(w_g, w_e), "losses", l_g, l_e)
print("eager   vs mega  :", rel(w_e, w_m), "losses", l_e, l_m)
print("graphed vs mega  :", rel(w_g, w_m), "losses", l_g, l_m)
print("TRIAGE_DONE")
"""

env = dict(os.environ)
env.update(REPO=REPO, PYTHONPATH=REPO, OUT="/tmp/diag_rn")
os.makedirs("/tmp/diag_rn", exist_ok=True)
r = subprocess.run([sys.executable, "-c", worker], env=env,
                   capture_output=True, text=True, timeout=800, cwd=REPO)
print("rc", r.returncode)
print(r.stdout[-2000:])
if r.returncode != 0:
    print(r.stderr[-2000:])
