"""azureml.core shim: an offline Run whose ``log`` appends JSONL to
``$AZUREML_STUB_LOGFILE`` — this is how the parity harness captures the
reference's per-round metrics ("Training loss", "Test acc", ...).

Reference call sites: ``Run.get_context()`` + ``run.log(k, v)``
(e2e_trainer.py:21,72-74,218-222; server.py:524-525; evaluation.py:87-90).
"""

import json
import os


class _OfflineRun:
    # id shaped so that e2e_trainer.py's
    # `"-".join(id.split("-")[-4:-2])` yields the experiment name "parity-run"
    id = "offline-stub-parity-run-0-0"
    input_datasets = {}

    def log(self, key, value, **kwargs):
        path = os.environ.get("AZUREML_STUB_LOGFILE")
        if path:
            with open(path, "a", encoding="utf-8") as f:
                f.write(json.dumps({"key": key, "value": value}) + "\n")

    def log_row(self, key, **kwargs):
        self.log(key, kwargs)

    def flush(self):
        pass


_RUN = _OfflineRun()


class Run:
    @staticmethod
    def get_context():
        return _RUN
