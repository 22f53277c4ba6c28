"""Head-to-head behavior parity against the UNMODIFIED reference
(msrflute/FLUTE at /root/reference) on CPU/gloo.

Runs both frameworks' e2e_trainer via torch.distributed.run (2 procs,
gloo) on identical synthetic LR-MNIST shards from one shared initial
checkpoint and asserts the per-round training-loss series, the final
test accuracy and the final checkpoint weights agree to fp tolerance.
Harness: tools/parity/run_parity.py (reference deps cerberus/easydict/
azureml/wget satisfied by the minimal shims in tools/parity/shims/).
"""

import os
import subprocess
import sys

import pytest

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))
REFERENCE = "/root/reference"


@pytest.mark.skipif(not os.path.isdir(REFERENCE),
                    reason="reference checkout not present")
def test_reference_headtohead_parity(tmp_path):
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "parity",
                                      "run_parity.py"),
         "--workdir", str(tmp_path / "parity"), "--rounds", "6"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, f"parity FAIL:\n{r.stdout[-4000:]}\n{r.stderr[-2000:]}"
    assert "**PARITY: PASS**" in r.stdout
