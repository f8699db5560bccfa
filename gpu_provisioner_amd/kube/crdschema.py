"""CRD openAPIV3Schema validation — server-side enforcement for the fake.

A real apiserver validates every custom-resource write against the CRD's
structural schema (enums, patterns, lengths, required fields, quantity
shapes) and its CEL rules. The in-memory apiserver enforces the same
contract through this module so a controller write that a real cluster
would 422 fails in the unit/e2e suites too (apiserver-conformance tier).

The validator covers the structural subset the NodeClaim CRD uses —
type/properties/required/additionalProperties/items, enum, pattern,
maxLength/minLength, maxItems, minimum/maximum, anyOf (int-or-string
quantities), format passthrough — plus two CEL rules special-cased
because evaluating CEL in Python is out of scope:

  * ``self == oldSelf`` on .spec (spec immutability);
  * ``self != ''`` non-empty checks on nodeClassRef fields.

Defaults (``default:`` markers, e.g. spec.expireAfter 720h) are applied on
create, as a real apiserver does.
"""
from __future__ import annotations

import re
from typing import Optional


def validate(obj, schema: dict, path: str = "$") -> list:
    """Structural openAPI v3 check; returns a list of violation strings."""
    errs: list = []
    t = schema.get("type")
    if t == "object" or ("properties" in schema and t is None):
        if not isinstance(obj, dict):
            return [f"{path}: expected object, got {type(obj).__name__}"]
        props = schema.get("properties", {})
        extra_ok = (
            schema.get("x-kubernetes-preserve-unknown-fields")
            or "additionalProperties" in schema
            or not props
        )
        for k, v in obj.items():
            if k in props:
                errs += validate(v, props[k], f"{path}.{k}")
            elif isinstance(schema.get("additionalProperties"), dict):
                errs += validate(v, schema["additionalProperties"], f"{path}.{k}")
            elif not extra_ok:
                errs.append(f"{path}.{k}: unknown field")
        for req in schema.get("required", []):
            if req not in obj:
                errs.append(f"{path}.{req}: required field missing")
    elif t == "array":
        if not isinstance(obj, list):
            return [f"{path}: expected array, got {type(obj).__name__}"]
        if "maxItems" in schema and len(obj) > schema["maxItems"]:
            errs.append(f"{path}: {len(obj)} items exceeds maxItems {schema['maxItems']}")
        for i, item in enumerate(obj):
            errs += validate(item, schema.get("items", {}), f"{path}[{i}]")
    elif t == "string":
        if not isinstance(obj, str):
            errs.append(f"{path}: expected string, got {type(obj).__name__}")
        else:
            if "enum" in schema and obj not in schema["enum"]:
                errs.append(f"{path}: {obj!r} not in enum {schema['enum']}")
            if "pattern" in schema and re.search(schema["pattern"], obj) is None:
                errs.append(f"{path}: {obj!r} does not match pattern {schema['pattern']!r}")
            if "maxLength" in schema and len(obj) > schema["maxLength"]:
                errs.append(f"{path}: length {len(obj)} exceeds maxLength {schema['maxLength']}")
            if "minLength" in schema and len(obj) < schema["minLength"]:
                errs.append(f"{path}: length {len(obj)} below minLength {schema['minLength']}")
    elif t == "integer":
        if not isinstance(obj, int) or isinstance(obj, bool):
            errs.append(f"{path}: expected integer, got {type(obj).__name__}")
        else:
            if "minimum" in schema and obj < schema["minimum"]:
                errs.append(f"{path}: {obj} below minimum {schema['minimum']}")
            if "maximum" in schema and obj > schema["maximum"]:
                errs.append(f"{path}: {obj} above maximum {schema['maximum']}")
    elif t == "boolean":
        if not isinstance(obj, bool):
            errs.append(f"{path}: expected boolean, got {type(obj).__name__}")
    # anyOf (e.g. int-or-string quantities): pass if any branch passes.
    # String validators alongside x-kubernetes-int-or-string apply only to
    # the string branch (pattern on quantities).
    if "anyOf" in schema:
        branches = []
        for b in schema["anyOf"]:
            merged = {**b}
            if b.get("type") == "string":
                for f in ("pattern", "maxLength", "minLength", "enum"):
                    if f in schema:
                        merged[f] = schema[f]
            branches.append(validate(obj, merged, path))
        if all(b for b in branches):
            errs.append(
                f"{path}: no anyOf branch matched ({'; '.join(branches[0][:1])})"
            )
    return errs


def _apply_defaults(obj, schema: dict) -> None:
    """Real-apiserver create behavior: fill `default:` markers in place."""
    if not isinstance(obj, dict):
        return
    for name, sub in (schema.get("properties") or {}).items():
        if name not in obj and "default" in sub:
            obj[name] = sub["default"]
        if name in obj:
            if isinstance(obj[name], dict):
                _apply_defaults(obj[name], sub)
            elif isinstance(obj[name], list) and isinstance(sub.get("items"), dict):
                for item in obj[name]:
                    _apply_defaults(item, sub["items"])


class CRDValidator:
    """Server-side validator for one CRD version: structural schema,
    defaults on create, and the special-cased CEL rules."""

    def __init__(self, crd: dict):
        versions = crd["spec"]["versions"]
        assert len(versions) == 1
        self.schema = versions[0]["schema"]["openAPIV3Schema"]
        spec_schema = self.schema.get("properties", {}).get("spec", {})
        self.spec_immutable = any(
            r.get("rule") == "self == oldSelf"
            for r in spec_schema.get("x-kubernetes-validations", [])
        )

    @classmethod
    def from_file(cls, path: str) -> "CRDValidator":
        import yaml

        return cls(next(yaml.safe_load_all(open(path))))

    def __call__(self, new: dict, old: Optional[dict]) -> list:
        errs = []
        if old is None:
            _apply_defaults(new, self.schema)
        errs += validate(new, self.schema)
        if old is not None and self.spec_immutable:
            if new.get("spec") != old.get("spec"):
                errs.append("$.spec: spec is immutable (CEL: self == oldSelf)")
        # nodeClassRef non-empty CEL rules
        ref = (new.get("spec") or {}).get("nodeClassRef")
        if isinstance(ref, dict):
            for f in ("group", "kind", "name"):
                if f in ref and ref[f] == "":
                    errs.append(f"$.spec.nodeClassRef.{f}: may not be empty (CEL)")
        return errs
