#!/usr/bin/env python3
"""Field-by-field structural parity diff of the NodeClaim CRD against the
reference chart's CRD.

Walks both openAPIV3Schema trees and compares every VALIDATION feature —
enum, pattern, maxLength/minLength, maxItems/minItems, maximum/minimum,
required (order-insensitive), format, default, and x-kubernetes-validations
CEL rules (by rule text, order-insensitive) — plus printer columns and
subresources. A reference feature missing or different in the repo CRD is a
parity failure unless listed in DELIBERATE_DELTAS with a rationale; extra
repo-side validation is reported but allowed (stricter is safe).

Usage:
    python hack/crd_parity.py [--reference DIR] [--write-parity-md]
Exit 0 = parity (modulo documented deltas), 1 = undocumented divergence.
Also invoked by tests/test_crd_schema.py when the reference tree exists.
"""
from __future__ import annotations

import argparse
import os
import sys

import yaml

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
REPO_CRD = os.path.join(
    REPO_ROOT, "charts", "gpu-provisioner-amd", "crds", "karpenter.sh_nodeclaims.yaml"
)
REF_CRD = "charts/gpu-provisioner/crds/karpenter.sh_nodeclaims.yaml"

FEATURES = (
    "enum",
    "pattern",
    "maxLength",
    "minLength",
    "maxItems",
    "minItems",
    "maximum",
    "minimum",
    "required",
    "format",
    "default",
    "x-kubernetes-validations",
    "x-kubernetes-int-or-string",
    "type",
)

# (path, feature) -> rationale. Every entry here is a DOCUMENTED deliberate
# divergence; anything else missing/different fails the check.
DELIBERATE_DELTAS: dict = {
    # The reference (controller-gen output) stamps these markers on the
    # `values` ARRAY itself, where OpenAPI ignores string validators; the
    # repo applies the identical pattern/maxLength on items[] (see the
    # "extra repo-side validation" section), which actually enforces them
    # per value. Strictly stronger, same intent.
    (".spec.requirements[].values", "maxLength"): (
        "enforced per item at .spec.requirements[].values[] (array-level "
        "string validators are inert in OpenAPI)"
    ),
    (".spec.requirements[].values", "pattern"): (
        "enforced per item at .spec.requirements[].values[] (array-level "
        "string validators are inert in OpenAPI)"
    ),
}


def _norm(feature: str, value):
    if feature == "required" and isinstance(value, list):
        return tuple(sorted(value))
    if feature == "x-kubernetes-validations" and isinstance(value, list):
        return tuple(sorted(r.get("rule", "") for r in value))
    if feature == "enum" and isinstance(value, list):
        return tuple(sorted(map(str, value)))
    return value


def walk(schema: dict, path: str = "") -> dict:
    out: dict = {}
    if not isinstance(schema, dict):
        return out
    for feat in FEATURES:
        if feat in schema:
            out[(path, feat)] = _norm(feat, schema[feat])
    for name, sub in (schema.get("properties") or {}).items():
        out.update(walk(sub, f"{path}.{name}"))
    items = schema.get("items")
    if isinstance(items, dict):
        out.update(walk(items, f"{path}[]"))
    ap = schema.get("additionalProperties")
    if isinstance(ap, dict):
        out.update(walk(ap, f"{path}.*"))
    for branch in ("anyOf", "oneOf", "allOf"):
        for i, sub in enumerate(schema.get(branch) or []):
            out.update(walk(sub, f"{path}<{branch}{i}>"))
    return out


def load_version(path: str) -> dict:
    crd = yaml.safe_load(open(path))
    versions = crd["spec"]["versions"]
    assert len(versions) == 1, f"{path}: expected one version"
    return versions[0]


def compare(ref_path: str, repo_path: str) -> dict:
    ref_v = load_version(ref_path)
    repo_v = load_version(repo_path)
    ref = walk(ref_v["schema"]["openAPIV3Schema"])
    repo = walk(repo_v["schema"]["openAPIV3Schema"])

    missing = {k: ref[k] for k in ref if k not in repo and k not in DELIBERATE_DELTAS}
    different = {
        k: (ref[k], repo[k])
        for k in ref
        if k in repo and ref[k] != repo[k] and k not in DELIBERATE_DELTAS
    }
    extra = {k: repo[k] for k in repo if k not in ref}
    documented = {k: DELIBERATE_DELTAS[k] for k in DELIBERATE_DELTAS}

    ref_cols = [
        (c["name"], c["jsonPath"], c.get("priority", 0))
        for c in ref_v.get("additionalPrinterColumns", [])
    ]
    repo_cols = [
        (c["name"], c["jsonPath"], c.get("priority", 0))
        for c in repo_v.get("additionalPrinterColumns", [])
    ]
    col_missing = [c for c in ref_cols if c not in repo_cols]

    sub_ok = ("status" in ref_v.get("subresources", {})) == (
        "status" in repo_v.get("subresources", {})
    )
    return {
        "missing": missing,
        "different": different,
        "extra": extra,
        "documented": documented,
        "columns_missing": col_missing,
        "subresources_ok": sub_ok,
        "ref_feature_count": len(ref),
        "repo_feature_count": len(repo),
    }


def render_markdown(result: dict, ref_path: str) -> str:
    lines = [
        "# CRD PARITY — NodeClaim CRD vs reference",
        "",
        f"Structural validation-feature diff of `{os.path.relpath(REPO_CRD, REPO_ROOT)}`",
        f"against `{ref_path}`, produced by `hack/crd_parity.py` (run in",
        "`tests/test_crd_schema.py` whenever the reference tree is present).",
        "",
        f"- reference validation features: **{result['ref_feature_count']}**",
        f"- repo validation features: **{result['repo_feature_count']}**",
        f"- reference features missing in repo: **{len(result['missing'])}**",
        f"- features with differing values: **{len(result['different'])}**",
        f"- documented deliberate deltas: **{len(result['documented'])}**",
        f"- printer columns missing: **{len(result['columns_missing'])}**",
        "",
    ]
    if result["missing"]:
        lines.append("## Missing (parity FAILURES)")
        for (p, f), v in sorted(result["missing"].items()):
            lines.append(f"- `{p}` :: `{f}` = `{v}`")
        lines.append("")
    if result["different"]:
        lines.append("## Different (parity FAILURES)")
        for (p, f), (a, b) in sorted(result["different"].items()):
            lines.append(f"- `{p}` :: `{f}`: ref=`{a}` repo=`{b}`")
        lines.append("")
    if result["documented"]:
        lines.append("## Documented deliberate deltas")
        for (p, f), why in sorted(result["documented"].items()):
            lines.append(f"- `{p}` :: `{f}` — {why}")
        lines.append("")
    if result["extra"]:
        lines.append("## Extra repo-side validation (allowed, stricter)")
        for (p, f), v in sorted(result["extra"].items()):
            lines.append(f"- `{p}` :: `{f}` = `{v}`")
        lines.append("")
    lines.append(
        "Parity verdict: **{}**".format(
            "PASS"
            if not result["missing"]
            and not result["different"]
            and not result["columns_missing"]
            and result["subresources_ok"]
            else "FAIL"
        )
    )
    lines.append("")
    return "\n".join(lines)


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--reference", default="/root/reference")
    ap.add_argument("--write-parity-md", action="store_true")
    args = ap.parse_args()
    ref_path = os.path.join(args.reference, REF_CRD)
    if not os.path.exists(ref_path):
        print(f"reference CRD not found at {ref_path}; nothing to compare")
        return 0
    result = compare(ref_path, REPO_CRD)
    md = render_markdown(result, ref_path)
    print(md)
    if args.write_parity_md:
        with open(os.path.join(REPO_ROOT, "CRD_PARITY.md"), "w") as f:
            f.write(md)
    ok = (
        not result["missing"]
        and not result["different"]
        and not result["columns_missing"]
        and result["subresources_ok"]
    )
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
