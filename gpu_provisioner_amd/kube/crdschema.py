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


def _compile_schema(schema: dict, cache: Optional[dict] = None) -> dict:
    """One-time pass replacing `pattern` strings with compiled regexes
    (stored under `_pattern_re`) and precomputing enum sets — re.search on
    a string pattern pays the regex-cache lookup on EVERY leaf check,
    which dominated per-write validation cost."""
    if cache is None:
        cache = {}
    sid = id(schema)
    if sid in cache:
        return cache[sid]
    out = dict(schema)
    cache[sid] = out
    if "pattern" in out:
        out["_pattern_re"] = re.compile(out["pattern"])
    if "enum" in out:
        out["_enum_set"] = frozenset(out["enum"])
    if "properties" in out:
        out["properties"] = {
            k: _compile_schema(v, cache) for k, v in out["properties"].items()
        }
    if isinstance(out.get("items"), dict):
        out["items"] = _compile_schema(out["items"], cache)
    if isinstance(out.get("additionalProperties"), dict):
        out["additionalProperties"] = _compile_schema(out["additionalProperties"], cache)
    for branch in ("anyOf", "oneOf", "allOf"):
        if branch in out:
            out[branch] = [_compile_schema(b, cache) for b in out[branch]]
    return out


def validate(obj, schema: dict, path: str = "$") -> list:
    """Structural openAPI v3 check; returns a list of violation strings.
    Accepts raw or _compile_schema'd schemas (compiled is much faster)."""
    errs: list = []
    _validate(obj, schema, (path,), errs)
    return errs


def _render(parts: tuple) -> str:
    return "".join(parts)


def _validate(obj, schema: dict, path: tuple, errs: list) -> None:
    t = schema.get("type")
    if t == "object" or ("properties" in schema and t is None):
        if not isinstance(obj, dict):
            errs.append(f"{_render(path)}: expected object, got {type(obj).__name__}")
            return
        props = schema.get("properties")
        ap = schema.get("additionalProperties")
        if props:
            extra_ok = schema.get("x-kubernetes-preserve-unknown-fields") or ap is not None
            for k, v in obj.items():
                sub = props.get(k)
                if sub is not None:
                    _validate(v, sub, path + (".", k), errs)
                elif isinstance(ap, dict):
                    _validate(v, ap, path + (".", k), errs)
                elif not extra_ok:
                    errs.append(f"{_render(path)}.{k}: unknown field")
        elif isinstance(ap, dict):
            for k, v in obj.items():
                _validate(v, ap, path + (".", k), errs)
        for req in schema.get("required", ()):
            if req not in obj:
                errs.append(f"{_render(path)}.{req}: required field missing")
    elif t == "array":
        if not isinstance(obj, list):
            errs.append(f"{_render(path)}: expected array, got {type(obj).__name__}")
            return
        if "maxItems" in schema and len(obj) > schema["maxItems"]:
            errs.append(
                f"{_render(path)}: {len(obj)} items exceeds maxItems {schema['maxItems']}"
            )
        items = schema.get("items")
        if items:
            for i, item in enumerate(obj):
                _validate(item, items, path + (f"[{i}]",), errs)
    elif t == "string":
        if not isinstance(obj, str):
            errs.append(f"{_render(path)}: expected string, got {type(obj).__name__}")
        else:
            enum_set = schema.get("_enum_set")
            if enum_set is not None:
                if obj not in enum_set:
                    errs.append(f"{_render(path)}: {obj!r} not in enum {schema['enum']}")
            elif "enum" in schema and obj not in schema["enum"]:
                errs.append(f"{_render(path)}: {obj!r} not in enum {schema['enum']}")
            pat = schema.get("_pattern_re")
            if pat is not None:
                if pat.search(obj) is None:
                    errs.append(
                        f"{_render(path)}: {obj!r} does not match pattern {schema['pattern']!r}"
                    )
            elif "pattern" in schema and re.search(schema["pattern"], obj) is None:
                errs.append(
                    f"{_render(path)}: {obj!r} does not match pattern {schema['pattern']!r}"
                )
            if "maxLength" in schema and len(obj) > schema["maxLength"]:
                errs.append(
                    f"{_render(path)}: length {len(obj)} exceeds maxLength {schema['maxLength']}"
                )
            if "minLength" in schema and len(obj) < schema["minLength"]:
                errs.append(
                    f"{_render(path)}: length {len(obj)} below minLength {schema['minLength']}"
                )
    elif t == "integer":
        if not isinstance(obj, int) or isinstance(obj, bool):
            errs.append(f"{_render(path)}: expected integer, got {type(obj).__name__}")
        else:
            if "minimum" in schema and obj < schema["minimum"]:
                errs.append(f"{_render(path)}: {obj} below minimum {schema['minimum']}")
            if "maximum" in schema and obj > schema["maximum"]:
                errs.append(f"{_render(path)}: {obj} above maximum {schema['maximum']}")
    elif t == "boolean":
        if not isinstance(obj, bool):
            errs.append(f"{_render(path)}: expected boolean, got {type(obj).__name__}")
    # anyOf (e.g. int-or-string quantities): pass if any branch passes.
    # String validators alongside x-kubernetes-int-or-string apply only to
    # the string branch (pattern on quantities).
    if "anyOf" in schema:
        branch_errs: list = []
        ok = False
        for b in schema["anyOf"]:
            merged = b
            if b.get("type") == "string" and any(
                f in schema for f in ("pattern", "_pattern_re", "maxLength", "minLength", "enum")
            ):
                merged = {**b}
                for f in ("pattern", "_pattern_re", "maxLength", "minLength", "enum", "_enum_set"):
                    if f in schema:
                        merged[f] = schema[f]
            be: list = []
            _validate(obj, merged, path, be)
            if not be:
                ok = True
                break
            if not branch_errs:
                branch_errs = be
        if not ok:
            errs.append(
                f"{_render(path)}: no anyOf branch matched ({'; '.join(branch_errs[:1])})"
            )


_MISSING = object()


def _validate_changed(new, old, schema: dict, path: str, errs: list) -> None:
    """Validate only the parts of `new` that differ from `old` (validity is
    compositional over the tree; an unchanged subtree cannot become
    invalid). Falls back to full validation at the first non-dict or
    schema-opaque level."""
    if new is old:
        return
    props = schema.get("properties")
    if isinstance(new, dict) and isinstance(old, dict) and props is not None:
        ap = schema.get("additionalProperties")
        extra_ok = schema.get("x-kubernetes-preserve-unknown-fields") or ap is not None
        for k, v in new.items():
            sub = props.get(k)
            if sub is None and isinstance(ap, dict):
                sub = ap
            if sub is None:
                if props and not extra_ok:
                    errs.append(f"{path}.{k}: unknown field")
                continue
            ov = old.get(k, _MISSING)
            if ov is _MISSING:
                _validate(v, sub, (path, ".", k), errs)
            else:
                _validate_changed(v, ov, sub, f"{path}.{k}", errs)
        for req in schema.get("required", ()):
            if req not in new:
                errs.append(f"{path}.{req}: required field missing")
        return
    items = schema.get("items")
    if isinstance(new, list) and isinstance(old, list) and isinstance(items, dict):
        # element-wise: an element equal to ANY previously-stored element
        # was already validated (conditions writes touch one entry and
        # carry the rest unchanged)
        if "maxItems" in schema and len(new) > schema["maxItems"]:
            errs.append(f"{path}: {len(new)} items exceeds maxItems {schema['maxItems']}")
        for i, item in enumerate(new):
            if any(item is o or item == o for o in old):
                continue
            _validate(item, items, (path, f"[{i}]"), errs)
        return
    if new == old:
        return
    _validate(new, schema, (path,), errs)


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
        self.schema = _compile_schema(versions[0]["schema"]["openAPIV3Schema"])
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
        errs: list = []
        if old is None:
            _apply_defaults(new, self.schema)
            errs += validate(new, self.schema)
        else:
            # an unchanged subtree cannot become invalid — validate only
            # what this write changed, recursively (the fake re-roots
            # revisions and merge patches share unchanged values, so the
            # `is`/`==` short-circuits prune most of the tree; full-object
            # validation per write cost ~20% of bench throughput)
            props = self.schema.get("properties", {})
            for key in ("spec", "status"):
                ns, olds = new.get(key), old.get(key)
                sub = props.get(key)
                if sub is not None and ns is not None:
                    _validate_changed(ns, olds, sub, f"$.{key}", errs)
            if self.spec_immutable:
                ns, olds = new.get("spec"), old.get("spec")
                if ns is not olds and ns != olds:
                    errs.append("$.spec: spec is immutable (CEL: self == oldSelf)")
        # nodeClassRef non-empty CEL rules
        ref = (new.get("spec") or {}).get("nodeClassRef")
        if isinstance(ref, dict):
            for f in ("group", "kind", "name"):
                if f in ref and ref[f] == "":
                    errs.append(f"$.spec.nodeClassRef.{f}: may not be empty (CEL)")
        return errs
