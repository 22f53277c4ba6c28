"""Strategy base contract (reference: core/strategies/base.py:8-57).

Server-mode strategies in this framework run SYMMETRICALLY on every rank
(SURVEY.md §2.5 redesign): each rank processes the payloads of its local
clients, and ``combine_payloads`` performs the single round-level
all-reduce of the flat gradient arena + scalar weight sum before applying
the (replicated, deterministic) server update.  Method names and call
order match the reference so the orchestration loop reads the same.
"""

from __future__ import annotations

from abc import ABC, abstractmethod


class BaseStrategy(ABC):
    def __init__(self, mode, config, model_path=None, runtime=None):
        if mode not in ("client", "server"):
            raise ValueError("mode in strategy must be either `client` or `server`")
        self.mode = mode
        self.config = config
        self.model_path = model_path
        self.runtime = runtime

    @abstractmethod
    def generate_client_payload(self, trainer):
        """Client side: turn the trainer's pseudo-gradient into a payload."""

    @abstractmethod
    def process_individual_payload(self, worker_trainer, payload):
        """Server side (per local client): accumulate or stack a payload.
        Returns False if the payload is rejected (zero weight)."""

    @abstractmethod
    def combine_payloads(self, worker_trainer, curr_iter,
                         num_clients_curr_iter, total_clients, client_stats,
                         logger=None):
        """Server side (once per round, every rank): all-reduce + normalize
        + server optimizer step.  Returns losses for the LR scheduler."""
