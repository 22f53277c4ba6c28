"""Minimal cerberus shim for running the REFERENCE (/root/reference) in
this offline container (real cerberus is not installed, no network).

Implements exactly the subset FLUTE's `core/config.py:762-779` uses on its
`core/schema.py` dialect: ``Validator(schema)``, ``validate(doc, schema)``
(recursive *required*-field checking) and ``normalized(doc)`` (recursive
default filling, including dict-typed sub-schemas and list item schemas).
Type coercion/validation is intentionally lenient — the parity harness
feeds well-formed configs; we only need FLUTE to run, not to be guarded.
"""

from copy import deepcopy


class Validator:
    def __init__(self, schema=None, allow_unknown=True):
        self.schema = schema
        self.allow_unknown = allow_unknown
        self.errors = {}

    # -- validation (required fields only) --------------------------------
    def validate(self, document, schema=None):
        schema = schema if schema is not None else self.schema
        self.errors = {}
        self._validate_level(document, schema, self.errors)
        return not self.errors

    def _validate_level(self, doc, schema, errors):
        if not isinstance(schema, dict):
            return
        for key, rule in schema.items():
            if not isinstance(rule, dict):
                continue
            if rule.get("required") and (not isinstance(doc, dict)
                                         or key not in doc):
                errors[key] = ["required field"]
                continue
            if not isinstance(doc, dict) or key not in doc:
                continue
            sub = rule.get("schema")
            val = doc[key]
            if isinstance(sub, dict):
                if rule.get("type") == "dict" and isinstance(val, dict):
                    sub_err = {}
                    self._validate_level(val, sub, sub_err)
                    if sub_err:
                        errors[key] = [sub_err]
                elif rule.get("type") == "list" and isinstance(val, list):
                    item_schema = sub.get("schema")
                    if isinstance(item_schema, dict) and \
                            sub.get("type") == "dict":
                        for item in val:
                            sub_err = {}
                            self._validate_level(item, item_schema, sub_err)
                            if sub_err:
                                errors.setdefault(key, []).append(sub_err)

    # -- normalization (default filling) ----------------------------------
    def normalized(self, document, schema=None):
        schema = schema if schema is not None else self.schema
        return self._normalize_level(deepcopy(document), schema)

    def _normalize_level(self, doc, schema):
        if not isinstance(doc, dict) or not isinstance(schema, dict):
            return doc
        for key, rule in schema.items():
            if not isinstance(rule, dict):
                continue
            if key not in doc and "default" in rule:
                doc[key] = deepcopy(rule["default"])
            if key in doc:
                sub = rule.get("schema")
                if isinstance(sub, dict):
                    if rule.get("type") == "dict" and isinstance(doc[key], dict):
                        doc[key] = self._normalize_level(doc[key], sub)
                    elif rule.get("type") == "list" and \
                            isinstance(doc[key], list):
                        item_schema = sub.get("schema")
                        if isinstance(item_schema, dict) and \
                                sub.get("type") == "dict":
                            doc[key] = [self._normalize_level(it, item_schema)
                                        for it in doc[key]]
        return doc
