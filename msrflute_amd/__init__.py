"""msrflute_amd — an MI355X-native federated-learning simulation engine.

A from-scratch rebuild of the capabilities of FLUTE (Federated Learning
Utilities for Testing and Experimentation, microsoft/msrflute): same
``e2e_trainer.py`` entrypoint, YAML-config schema, task-plugin layout and
checkpoint format — but designed for one AMD Instinct MI355X node:

* one process per GPU over ``torch.distributed`` (RCCL over xGMI),
* a flat contiguous parameter arena per model replica so every kernel and
  collective is one contiguous op (no per-tensor shape handshakes),
* round-level collectives (all-reduce of the weighted pseudo-gradient sum)
  instead of per-client point-to-point transfers,
* hand-written CDNA4 (gfx950) HIP kernels for the hot flat-arena ops:
  pseudo-gradient, weighted accumulate, fused gradient statistics, norm
  clipping, DP noise, update quantization, and fused optimizers.

Reference layer map: see SURVEY.md §1 (citations into /root/reference).
"""

__version__ = "0.1.0"
