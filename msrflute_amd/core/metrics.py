"""Generic inference/metrics loop (reference: core/metrics.py:19-73).

Differences from the reference: no ``torch.cuda.empty_cache()`` per batch
(a reference perf crutch), and metric accumulation happens
in device tensors moved to host once at the end.
"""

from __future__ import annotations

import numpy as np
import torch

from ..utils import print_rank


class Metrics:
    """Computes batch-size-weighted metric averages over a dataloader."""

    def compute_metrics(self, dataloader, model):
        return self.call_inference(dataloader, model)

    def call_inference(self, dataloader, model):
        metrics, sum_metrics = {}, {}
        output_tot = {"probabilities": [], "predictions": [], "labels": []}
        counter = 0

        model.set_eval()
        inf_results = {}
        with torch.no_grad():
            for batch in dataloader:
                val_loss = model.loss(batch).item()
                inf_results = model.inference(batch)
                inf_results["loss"] = {"value": val_loss, "higher_is_better": False}
                output = inf_results.pop("output")
                batch_size = inf_results.pop("batch_size")

                for key, v in list(inf_results.items()):
                    if not isinstance(v, dict):
                        inf_results[key] = {"value": v, "higher_is_better": True}
                    sum_metrics.setdefault(key, [])

                if isinstance(output, dict):
                    output_tot["probabilities"].append(output["probabilities"])
                    output_tot["predictions"].append(output["predictions"])
                    output_tot["labels"].append(output["labels"])

                for q in inf_results:
                    sum_metrics[q].append(float(inf_results[q]["value"]) * batch_size)
                counter += batch_size

        for k in output_tot:
            output_tot[k] = np.concatenate(output_tot[k]) if output_tot[k] else []

        model.set_train()
        for k in inf_results:
            metrics[k] = dict(inf_results[k])
            metrics[k]["value"] = sum(sum_metrics[k]) / counter if counter else 0.0

        print_rank(f"validation examples {counter}")
        return output_tot, metrics
