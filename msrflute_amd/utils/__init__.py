from .logging import (init_logging, init_metrics_sink, log_metric, print_cuda_stats,
                      print_rank)
from .misc import AverageMeter, alpha_update, softmax_weights, to_device, update_json_log
from .optimizers import LAMB, LarsSGD, make_optimizer
from .schedulers import (NBestTaskScheduler, RampupKeepExpdecayKeepLRScheduler,
                         ScheduledSamplingScheduler, make_lr_scheduler)

__all__ = [
    "init_logging", "init_metrics_sink", "log_metric", "print_cuda_stats", "print_rank",
    "AverageMeter", "alpha_update", "softmax_weights", "to_device", "update_json_log",
    "LAMB", "LarsSGD", "make_optimizer",
    "NBestTaskScheduler", "RampupKeepExpdecayKeepLRScheduler",
    "ScheduledSamplingScheduler", "make_lr_scheduler",
]
