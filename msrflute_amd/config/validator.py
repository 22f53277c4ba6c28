"""Dependency-free schema validator with default normalization.

Replaces the cerberus dependency of the reference (core/config.py:762-779):
validates a config dict against ``msrflute_amd.config.schema.SCHEMA`` and
returns a normalized copy with schema defaults filled in.  Unknown keys are
always allowed (the reference sets ``allow_unknown`` on every rule).
"""

from __future__ import annotations

import copy
from typing import Any, Dict, List


class ConfigValidationError(ValueError):
    def __init__(self, errors: List[str]):
        self.errors = errors
        super().__init__("config validation failed:\n  " + "\n  ".join(errors))


_TYPE_CHECKS = {
    "string": lambda v: isinstance(v, str),
    "boolean": lambda v: isinstance(v, bool),
    # bool is an int subclass; exclude it explicitly
    "integer": lambda v: isinstance(v, int) and not isinstance(v, bool),
    "float": lambda v: isinstance(v, (int, float)) and not isinstance(v, bool),
    "dict": lambda v: isinstance(v, dict),
    "list": lambda v: isinstance(v, list),
}


def _check_type(value: Any, types) -> bool:
    if isinstance(types, str):
        types = [types]
    return any(_TYPE_CHECKS[t](value) for t in types)


def _validate_dict(data: Dict, schema: Dict, path: str, errors: List[str]) -> Dict:
    out = dict(data)
    for key, rule in schema.items():
        here = f"{path}.{key}" if path else key
        if key not in data:
            if rule.get("required", False):
                errors.append(f"missing required key: {here}")
            elif "default" in rule:
                out[key] = copy.deepcopy(rule["default"])
            continue
        value = data[key]
        if value is None:
            if rule.get("nullable", False) or "dict" not in _as_list(rule.get("type", [])):
                # FLUTE configs routinely carry explicit nulls for data paths.
                continue
            continue
        rtype = rule.get("type")
        if rtype is not None and not _check_type(value, rtype):
            errors.append(f"{here}: expected {rtype}, got {type(value).__name__}")
            continue
        if "allowed" in rule and value not in rule["allowed"]:
            errors.append(f"{here}: value {value!r} not in allowed set {rule['allowed']}")
        if isinstance(value, dict):
            for fk in rule.get("forbidden_keys", []):
                if fk in value:
                    errors.append(f"{here}: forbidden key present: {fk}")
            if "schema" in rule:
                out[key] = _validate_dict(value, rule["schema"], here, errors)
    return out


def _as_list(x):
    return [x] if isinstance(x, str) else list(x)


def validate_config(data: Dict, schema: Dict) -> Dict:
    """Validate ``data`` against ``schema``; return a normalized copy.

    Raises ConfigValidationError listing every violation found.
    """
    errors: List[str] = []
    out = _validate_dict(data, schema, "", errors)
    if errors:
        raise ConfigValidationError(errors)
    return out
