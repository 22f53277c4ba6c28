"""LR / sampling schedulers (reference: utils/utils.py:151-294)."""

from __future__ import annotations

import copy
import math

import torch
from torch.optim.lr_scheduler import MultiStepLR, ReduceLROnPlateau, StepLR


def make_lr_scheduler(annealing_config, optimizer, num_batches=1):
    """Build an LR scheduler; epoch-interval steps are converted to
    iteration steps via ``num_batches`` (reference: utils/utils.py:151-186).
    """
    cfg = copy.deepcopy(dict(annealing_config))
    annealing_type = cfg.pop("type")
    step_interval = cfg.pop("step_interval", "epoch")

    if annealing_type == "step_lr":
        if step_interval == "epoch":
            cfg["step_size"] = max(1, int(num_batches * cfg["step_size"]))
        return StepLR(optimizer=optimizer, **cfg)
    if annealing_type == "multi_step_lr":
        if step_interval == "epoch":
            cfg["milestones"] = [int(i * num_batches) for i in cfg["milestones"]]
        return MultiStepLR(optimizer=optimizer, **cfg)
    if annealing_type == "rampup-keep-expdecay-keep":
        return RampupKeepExpdecayKeepLRScheduler(optimizer=optimizer, **cfg)
    if annealing_type == "val_loss":
        cfg.pop("gamma", None)
        cfg.pop("step_size", None)
        return ReduceLROnPlateau(optimizer, **cfg)
    raise ValueError(f"{annealing_type} LR scheduler not supported")


class RampupKeepExpdecayKeepLRScheduler(torch.optim.lr_scheduler._LRScheduler):
    """SpecAugment-style LR schedule: linear ramp to ``peak_lr`` over ``sr``
    steps, hold to ``si``, exponential decay to ``floor_lr`` at ``sf``, hold.
    Reference: utils/utils.py:189-224.
    """

    def __init__(self, optimizer, peak_lr=0.001, floor_lr=0.00001,
                 sr=1000, si=40000, sf=160000, last_epoch=-1):
        assert peak_lr >= floor_lr and sr <= si <= sf
        self.peak_lr = peak_lr
        self.floor_lr = floor_lr
        self.sr, self.si, self.sf = sr, si, sf
        self.gamma = math.log(floor_lr / peak_lr) / float(sf - si)
        self.step_count = 0
        super().__init__(optimizer, last_epoch=last_epoch)

    def step(self, epoch=None):
        for p, lr in zip(self.optimizer.param_groups, self.get_lr()):
            p["lr"] = lr
        self.step_count += 1

    def get_lr(self):
        if self.step_count < self.sr:
            lr = self.peak_lr * float(self.step_count) / float(self.sr)
        elif self.step_count < self.si:
            lr = self.peak_lr
        elif self.step_count < self.sf:
            lr = self.peak_lr * math.exp(self.gamma * float(self.step_count - self.si))
        else:
            lr = self.floor_lr
        return [lr for _ in self.base_lrs]


class ScheduledSamplingScheduler:
    """Anneals ``model.scheduled_sampling_rate`` from ``initial_rate`` to
    ``final_rate`` between rounds ``ramp_start`` and ``ramp_stop``.
    Reference: utils/utils.py:228-260.
    """

    def __init__(self, model, ramp_start, ramp_stop, initial_rate, final_rate):
        self.model = model
        self.ramp_start = ramp_start
        self.ramp_stop = ramp_stop
        self.initial_rate = initial_rate
        self.final_rate = final_rate
        self.iter = 0

    def step(self):
        if self.iter < self.ramp_start:
            rate = self.initial_rate
        elif self.iter <= self.ramp_stop:
            frac = (self.iter - self.ramp_start) / (self.ramp_stop - self.ramp_start)
            rate = self.initial_rate + (self.final_rate - self.initial_rate) * frac
        else:
            rate = self.final_rate
        self.model.scheduled_sampling_rate = rate
        self.model.scheduled_sampling = rate != 0
        self.iter += 1

    def state_dict(self):
        return {k: v for k, v in self.__dict__.items() if k != "model"}

    def load_state_dict(self, state_dict):
        self.__dict__.update(state_dict)


class NBestTaskScheduler:
    """Multi-task stage scheduler (reference: utils/utils.py:263-294)."""

    def __init__(self, num_tasks, iteration_per_task):
        assert len(num_tasks) == len(iteration_per_task), \
            f"Mismatched length {len(num_tasks)}!={len(iteration_per_task)}"
        self.iter = 0
        self.stagex = 0
        self.num_tasks = num_tasks
        self.iteration_per_task = iteration_per_task

    def current_num_tasks(self):
        return self.num_tasks[self.stagex]

    def no_label_updates(self):
        return (self.iter // self.iteration_per_task[-1]) + 1

    def set_iteration_no(self, iter_no):
        self.iter = iter_no

    def step(self):
        local_iter = self.iter % self.iteration_per_task[-1]
        if local_iter == 0:
            self.stagex = 0
        elif local_iter >= self.iteration_per_task[self.stagex]:
            self.stagex += 1
        self.iter += 1

    def state_dict(self):
        return dict(self.__dict__)

    def load_state_dict(self, state_dict):
        self.__dict__.update(state_dict)
