"""Declarative schema for FLUTE-style YAML configs.

Mirrors the key space of the reference schema (reference:
core/schema.py:9-300) but is consumed by the dependency-free validator in
``msrflute_amd.config.validator`` instead of cerberus (which is not part of
this image).  Semantics kept: required/optional keys, defaults filled into
the normalized dict, ``allowed`` value lists, nullable strings, unknown keys
permitted everywhere (FLUTE sets ``allow_unknown`` on every dict rule).
"""

OPTIMIZER_TYPES = ["sgd", "adam", "adamax", "lars", "LarsSGD", "lamb", "adamW"]

_DATA_PART_COMMON = {
    "batch_size": {"type": "integer", "default": 40},
    "tokenizer_type": {"type": "string"},
    "prepend_datapath": {"type": "boolean", "default": False},
    "vocab_dict": {"type": "string"},
    "pin_memory": {"type": "boolean", "default": True},
    "num_workers": {"type": "integer", "default": 1},
    "num_frames": {"type": "integer", "default": 0},
    "max_batch_size": {"type": "integer", "default": 0},
    "max_num_words": {"type": "integer"},
    "max_grad_norm": {"type": "float", "default": 5.0},
    "unsorted_batch": {"type": "boolean", "default": False},
    "cache_dir": {"type": "string"},
}


def _data_part(extra):
    d = dict(_DATA_PART_COMMON)
    d.update(extra)
    return d


SCHEMA = {
    "model_config": {
        "required": True,
        "type": "dict",
        "schema": {
            "model_type": {"required": True, "type": "string"},
            "model_folder": {"required": True, "type": "string"},
            "BERT": {
                "type": "dict",
                "schema": {
                    "loader_type": {"type": "string"},
                    "model": {
                        "required": True,
                        "type": "dict",
                        "schema": {
                            "model_name_or_path": {"type": "string"},
                            "model_name": {"required": True, "type": "string"},
                            "process_line_by_line": {"type": "boolean",
                                                     "default": False},
                        },
                    },
                },
            },
        },
    },
    "dp_config": {
        "required": True,
        "type": "dict",
        "schema": {
            "enable_local_dp": {"required": True, "type": "boolean"},
            "enable_global_dp": {"type": "boolean"},
            "eps": {"type": "float"},
            "delta": {"type": "float"},
            "global_sigma": {"type": "float"},
            "max_grad": {"type": "float"},
            "max_weight": {"type": "float"},
            "weight_scaler": {"type": "float"},
            "min_weight": {"type": "float"},
        },
    },
    "privacy_metrics_config": {
        "required": True,
        "type": "dict",
        "schema": {
            "apply_metrics": {"required": True, "type": "boolean"},
            "apply_indices_extraction": {"type": "boolean"},
            "allowed_word_rank": {"type": "integer"},
            "apply_leakage_metric": {"type": "boolean"},
            "max_leakage": {"type": "float"},
            "adaptive_leakage_threshold": {"type": "float"},
            "is_leakage_weighted": {"type": "boolean"},
            "attacker_optimizer_config": {"type": "dict"},
        },
    },
    "strategy": {"required": True, "type": "string"},
    "server_config": {
        "required": True,
        "type": "dict",
        "schema": {
            "wantRL": {"required": True, "type": "boolean"},
            "RL": {"type": "dict"},
            "resume_from_checkpoint": {"required": True, "type": "boolean"},
            "do_profiling": {"required": True, "type": "boolean"},
            "optimizer_config": {
                "required": True,
                "type": "dict",
                "schema": {
                    "type": {"required": True, "type": "string", "allowed": OPTIMIZER_TYPES},
                    "lr": {"required": True, "type": "float"},
                    "weight_decay": {"type": "float"},
                },
            },
            "annealing_config": {
                "required": True,
                "type": "dict",
                "schema": {
                    "type": {"required": True, "type": "string"},
                    "step_interval": {"required": True, "type": "string"},
                    "gamma": {"required": True, "type": "float"},
                    "step_size": {"required": True, "type": "integer"},
                },
            },
            "val_freq": {"type": "integer", "default": 1},
            "rec_freq": {"type": "integer", "default": 8},
            "initial_val": {"type": "boolean", "default": True},
            "initial_rec": {"type": "boolean", "default": False},
            "max_iteration": {"type": "integer", "default": 10000},
            # int, or "min,max" string for a per-round random range
            # (reference: core/server.py:84-86, 284-291).
            "num_clients_per_iteration": {"type": ["integer", "string"], "default": 1},
            "data_config": {
                "required": True,
                "type": "dict",
                "forbidden_keys": ["num_clients"],
                "schema": {
                    "val": {
                        "required": True,
                        "type": "dict",
                        "schema": _data_part({"val_data": {"required": True, "type": "string", "nullable": True}}),
                    },
                    "test": {
                        "required": True,
                        "type": "dict",
                        "schema": _data_part({"test_data": {"required": True, "type": "string", "nullable": True}}),
                    },
                    "train": {
                        "type": "dict",
                        "schema": _data_part({
                            "train_data_server": {"type": "string", "nullable": True},
                            "desired_max_samples": {"type": "integer"},
                        }),
                    },
                },
            },
            "type": {
                "type": "string",
                "allowed": ["model_optimization", "personalization"],
                "default": "model_optimization",
            },
            "aggregate_median": {"type": "string"},
            "initial_lr_client": {"required": True, "type": "float"},
            "lr_decay_factor": {"required": True, "type": "float"},
            "weight_train_loss": {"required": True, "type": "string"},
            "best_model_criterion": {"type": "string", "default": "loss"},
            "fall_back_to_best_model": {"type": "boolean", "default": False},
            "softmax_beta": {"required": True, "type": "float"},
            "server_replay_config": {
                "type": "dict",
                "schema": {
                    "server_iterations": {"required": True, "type": "integer"},
                    "optimizer_config": {
                        "required": True,
                        "type": "dict",
                        "schema": {
                            "type": {"required": True, "type": "string", "allowed": OPTIMIZER_TYPES},
                            "lr": {"required": True, "type": "float"},
                            "weight_decay": {"type": "float"},
                            "amsgrad": {"type": "boolean"},
                        },
                    },
                },
            },
            "nbest_task_scheduler": {
                "type": "dict",
                "schema": {
                    "num_tasks": {"required": True, "type": "integer"},
                    "iteration_per_task": {"required": True, "type": "integer"},
                },
            },
        },
    },
    "client_config": {
        "required": True,
        "type": "dict",
        "schema": {
            "meta_learning": {"type": "string"},
            "stats_on_smooth_grad": {"type": "boolean"},
            "ignore_subtask": {"required": True, "type": "boolean"},
            "num_skips_threshold": {"type": "integer"},
            "copying_train_data": {"type": "boolean"},
            "do_profiling": {"required": True, "type": "boolean"},
            "data_config": {
                "required": True,
                "type": "dict",
                "forbidden_keys": ["num_clients"],
                "schema": {
                    "train": {
                        "required": True,
                        "type": "dict",
                        "schema": _data_part({
                            "list_of_train_data": {"required": True, "type": "string", "nullable": True},
                            "desired_max_samples": {"type": "integer"},
                        }),
                    },
                },
            },
            "type": {
                "type": "string",
                "allowed": ["optimization", "gradient_computation"],
                "default": "gradient_computation",
            },
            "meta_optimizer_config": {
                "type": "dict",
                "schema": {
                    "type": {"required": True, "type": "string", "allowed": OPTIMIZER_TYPES},
                    "lr": {"required": True, "type": "float"},
                },
            },
            "optimizer_config": {
                "required": True,
                "type": "dict",
                "schema": {
                    "type": {"required": True, "type": "string", "allowed": OPTIMIZER_TYPES},
                    "lr": {"type": "float"},
                    "weight_decay": {"type": "float"},
                },
            },
            "annealing_config": {
                "type": "dict",
                "schema": {
                    "type": {"required": True, "type": "string"},
                    "step_interval": {"required": True, "type": "string"},
                    "gamma": {"type": "float"},
                    "step_size": {"type": "integer"},
                },
            },
            "ss_config": {"type": "dict"},
        },
    },
}
