"""Property-based tests (hypothesis) for the algebraic cores the controllers
lean on: requirement intersection must behave as set intersection, quantity
parsing must round-trip and order consistently, JSON merge patch must obey
RFC 7386, taint merge must be idempotent, and timestamps must round-trip.
These are the pieces where a subtle algebra bug silently mis-schedules or
mis-syncs — the reference trusts upstream karpenter's battle-testing here;
we prove ours instead."""
from datetime import datetime, timezone

from hypothesis import given, settings, strategies as st

from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.kube.client import json_merge_patch
from gpu_provisioner_amd.scheduling.requirements import (
    DOES_NOT_EXIST,
    EXISTS,
    GT,
    IN,
    LT,
    NOT_IN,
    Requirement,
)

# -- requirement algebra ------------------------------------------------------

values = st.sampled_from(["1", "2", "3", "10", "25", "a", "b", "c", "mi355x"])
value_sets = st.lists(values, min_size=0, max_size=4)


@st.composite
def requirements(draw):
    op = draw(st.sampled_from([IN, NOT_IN, EXISTS, DOES_NOT_EXIST, GT, LT]))
    if op in (GT, LT):
        return Requirement("k", op, [str(draw(st.integers(0, 30)))])
    if op in (EXISTS, DOES_NOT_EXIST):
        return Requirement("k", op)
    vals = draw(value_sets)
    return Requirement("k", op, vals)


@given(a=requirements(), b=requirements(), v=values)
@settings(max_examples=300, deadline=None)
def test_intersection_is_set_intersection(a, b, v):
    """(a ∩ b).has(v) == a.has(v) and b.has(v) — the defining property."""
    assert a.intersect(b).has(v) == (a.has(v) and b.has(v))


@given(a=requirements(), b=requirements(), v=values)
@settings(max_examples=200, deadline=None)
def test_intersection_commutes(a, b, v):
    assert a.intersect(b).has(v) == b.intersect(a).has(v)


@given(a=requirements(), b=requirements(), c=requirements(), v=values)
@settings(max_examples=200, deadline=None)
def test_intersection_associates(a, b, c, v):
    assert a.intersect(b).intersect(c).has(v) == a.intersect(b.intersect(c)).has(v)


@given(a=requirements(), v=values)
@settings(max_examples=200, deadline=None)
def test_intersection_idempotent(a, v):
    assert a.intersect(a).has(v) == a.has(v)


@given(a=requirements())
@settings(max_examples=200, deadline=None)
def test_requirement_dict_round_trip(a):
    """to_dict → from_dict preserves membership semantics."""
    b = Requirement.from_dict(a.to_dict())
    for v in ("1", "2", "10", "25", "a", "mi355x", "zz"):
        assert a.has(v) == b.has(v), (a.to_dict(), v)


# -- quantities ---------------------------------------------------------------

suffixes = st.sampled_from(["", "m", "k", "Ki", "M", "Mi", "G", "Gi", "T", "Ti"])


@given(n=st.integers(0, 10**6), suf=suffixes)
@settings(max_examples=300, deadline=None)
def test_quantity_parse_round_trip(n, suf):
    q = ko.qty(f"{n}{suf}")
    assert ko.qty(str(q)) == q


@given(
    a=st.integers(0, 10**6), sa=suffixes,
    b=st.integers(0, 10**6), sb=suffixes,
)
@settings(max_examples=300, deadline=None)
def test_quantity_ordering_matches_values(a, sa, b, sb):
    qa, qb = ko.qty(f"{a}{sa}"), ko.qty(f"{b}{sb}")
    assert (qa < qb) == (qa.value < qb.value)
    assert (qa == qb) == (qa.value == qb.value)


@given(a=st.integers(0, 10**6), b=st.integers(0, 10**6), suf=suffixes)
@settings(max_examples=200, deadline=None)
def test_quantity_subtraction_consistent(a, b, suf):
    hi, lo = max(a, b), min(a, b)
    q = ko.qty(f"{hi}{suf}") - ko.qty(f"{lo}{suf}")
    assert q.value == ko.qty(f"{hi - lo}{suf}").value


# -- JSON merge patch (RFC 7386) ---------------------------------------------

json_scalars = st.one_of(st.integers(-5, 5), st.text(max_size=3), st.booleans())
json_vals = st.recursive(
    json_scalars,
    lambda inner: st.dictionaries(st.text(max_size=2), inner, max_size=3),
    max_leaves=8,
)
json_docs = st.dictionaries(st.text(max_size=2), json_vals, max_size=4)
# patches may include None (= delete key)
patch_vals = st.recursive(
    st.one_of(json_scalars, st.none()),
    lambda inner: st.dictionaries(st.text(max_size=2), inner, max_size=3),
    max_leaves=8,
)
patch_docs = st.dictionaries(st.text(max_size=2), patch_vals, max_size=4)


def rfc7386(target, patch):
    """Reference implementation, straight from the RFC pseudocode."""
    if not isinstance(patch, dict):
        return patch
    if not isinstance(target, dict):
        target = {}
    out = dict(target)
    for k, v in patch.items():
        if v is None:
            out.pop(k, None)
        else:
            out[k] = rfc7386(out.get(k), v)
    return out


@given(target=json_docs, patch=patch_docs)
@settings(max_examples=300, deadline=None)
def test_json_merge_patch_matches_rfc(target, patch):
    assert json_merge_patch(target, patch) == rfc7386(target, patch)


@given(target=json_docs, patch=patch_docs)
@settings(max_examples=200, deadline=None)
def test_json_merge_patch_does_not_mutate_inputs(target, patch):
    import copy

    t0, p0 = copy.deepcopy(target), copy.deepcopy(patch)
    json_merge_patch(target, patch)
    assert target == t0 and patch == p0


# -- taints -------------------------------------------------------------------

taints = st.lists(
    st.builds(
        lambda k, e: {"key": k, "effect": e},
        st.sampled_from(["a", "b", "c"]),
        st.sampled_from(["NoSchedule", "NoExecute"]),
    ),
    max_size=4,
)


@given(existing=taints, desired=taints)
@settings(max_examples=200, deadline=None)
def test_merge_taints_idempotent_and_complete(existing, desired):
    once = ko.merge_taints(existing, desired)
    twice = ko.merge_taints(once, desired)
    assert once == twice  # idempotent
    have = {(t["key"], t["effect"]) for t in once}
    for t in existing + desired:
        assert (t["key"], t["effect"]) in have
    # existing entries keep their position and identity
    assert once[: len(existing)] == existing


# -- timestamps ---------------------------------------------------------------


@given(ts=st.integers(0, 4_102_444_800))  # 1970..2100
@settings(max_examples=300, deadline=None)
def test_fmt_parse_time_round_trip(ts):
    t = datetime.fromtimestamp(ts, tz=timezone.utc)
    assert ko.parse_time(ko.fmt_time(t)) == t


# -- providerID parsing -------------------------------------------------------

pool_names = st.from_regex(r"[a-z][a-z0-9]{0,11}", fullmatch=True)


@given(
    pool=pool_names,
    sub=st.sampled_from(["sub1", "0000-1111"]),
    rg=st.sampled_from(["rg", "MC_rg_cluster_loc"]),
    h=st.from_regex(r"[0-9a-f]{8}", fullmatch=True),
)
@settings(max_examples=200, deadline=None)
def test_provider_id_build_parse_round_trip(pool, sub, rg, h):
    from gpu_provisioner_amd.utils.utils import (
        build_provider_id,
        parse_agent_pool_name_from_id,
    )

    assert parse_agent_pool_name_from_id(build_provider_id(sub, rg, pool, h)) == pool


@given(junk=st.text(max_size=60))
@settings(max_examples=200, deadline=None)
def test_provider_id_parse_never_raises(junk):
    from gpu_provisioner_amd.utils.utils import parse_agent_pool_name_from_id

    parse_agent_pool_name_from_id(junk)  # None or a name, never an exception


# -- label selectors ----------------------------------------------------------

label_keys = st.sampled_from(["app", "env", "tier", "kaito.sh/workspace"])
label_vals = st.sampled_from(["a", "b", "prod", "dev", ""])
label_maps = st.dictionaries(label_keys, label_vals, max_size=3)


@given(labels=label_maps, key=label_keys, val=label_vals)
@settings(max_examples=300, deadline=None)
def test_label_selector_semantics(labels, key, val):
    from gpu_provisioner_amd.kube.client import LabelSelector

    assert LabelSelector.parse(f"{key}={val}").matches(labels) == (
        labels.get(key) == val
    )
    assert LabelSelector.parse(f"{key}!={val}").matches(labels) == (
        labels.get(key) != val
    )
    assert LabelSelector.parse(key).matches(labels) == (key in labels)
    assert LabelSelector.parse(f"!{key}").matches(labels) == (key not in labels)
    assert LabelSelector.parse(f"{key} in (a,prod)").matches(labels) == (
        labels.get(key) in ("a", "prod")
    )


@given(labels=label_maps, k1=label_keys, k2=label_keys, v=label_vals)
@settings(max_examples=200, deadline=None)
def test_label_selector_conjunction(labels, k1, k2, v):
    from gpu_provisioner_amd.kube.client import LabelSelector

    sel = LabelSelector.parse(f"{k1}={v},{k2}")
    assert sel.matches(labels) == (labels.get(k1) == v and k2 in labels)


# -- duration parsing ---------------------------------------------------------


@given(
    h=st.integers(0, 10), m=st.integers(0, 120), s=st.integers(0, 3600),
)
@settings(max_examples=200, deadline=None)
def test_parse_duration_compositional(h, m, s):
    from gpu_provisioner_amd.controllers.termination.controller import parse_duration

    total = h * 3600 + m * 60 + s
    got = parse_duration(f"{h}h{m}m{s}s")
    if total == 0:
        assert got is None  # k8s semantics: zero duration = unset here
    else:
        assert got is not None and got.total_seconds() == total


# ---------------------------------------------------------------------------
# CRD validator: changed-only pruning must be equivalent to full validation
# ---------------------------------------------------------------------------

import os

import yaml as _yaml

from gpu_provisioner_amd.kube.crdschema import CRDValidator, validate as crd_validate

_CRD_PATH = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "charts", "gpu-provisioner-amd", "crds", "karpenter.sh_nodeclaims.yaml",
)
_VALIDATOR = CRDValidator.from_file(_CRD_PATH)


@st.composite
def valid_nodeclaim(draw):
    name = draw(st.text("abcdefgh", min_size=1, max_size=8))
    n_reqs = draw(st.integers(1, 3))
    reqs = []
    for i in range(n_reqs):
        op = draw(st.sampled_from(["In", "NotIn", "Exists"]))
        r = {"key": "node.kubernetes.io/instance-type" if i == 0 else f"amd.com/k{i}",
             "operator": op}
        if op != "Exists":
            r["values"] = draw(
                st.lists(st.text("abcXYZ019", min_size=1, max_size=8), min_size=1, max_size=3)
            )
        reqs.append(r)
    nc = {
        "apiVersion": "karpenter.sh/v1",
        "kind": "NodeClaim",
        "metadata": {"name": name},
        "spec": {
            "requirements": reqs,
            "nodeClassRef": {"group": "kaito.sh", "kind": "KaitoNodeClass", "name": "d"},
        },
        "status": {
            "providerID": draw(st.sampled_from(["", "azure:///x"])),
            "capacity": {"amd.com/gpu": str(draw(st.integers(1, 8)))},
            "conditions": [
                {"type": "Launched", "status": draw(st.sampled_from(["True", "False"])),
                 "reason": "Launched", "lastTransitionTime": "2026-01-01T00:00:00Z"}
            ],
        },
    }
    return nc


@st.composite
def mutation(draw):
    """A mutation applied to a valid claim — possibly invalidating."""
    kind = draw(st.sampled_from([
        "cond_bad_status", "cond_add", "capacity_bad_qty", "capacity_good",
        "req_bad_operator", "req_noop", "nodename", "too_many_reqs",
    ]))
    return kind


def _apply_mutation(nc, kind):
    import copy

    out = copy.deepcopy(nc)
    if kind == "cond_bad_status":
        out["status"]["conditions"][0]["status"] = "Perhaps"  # not in enum
    elif kind == "cond_add":
        out["status"]["conditions"].append(
            {"type": "Registered", "status": "True", "reason": "R",
             "lastTransitionTime": "2026-01-01T00:00:01Z"}
        )
    elif kind == "capacity_bad_qty":
        out["status"]["capacity"]["amd.com/gpu"] = "eight!"
    elif kind == "capacity_good":
        out["status"]["capacity"]["memory"] = "2048Gi"
    elif kind == "req_bad_operator":
        out["spec"]["requirements"][0]["operator"] = "Matches"
    elif kind == "req_noop":
        pass
    elif kind == "nodename":
        out["status"]["nodeName"] = "aks-x-123-vmss000000"
    elif kind == "too_many_reqs":
        out["spec"]["requirements"] = out["spec"]["requirements"] * 51  # > maxItems 100
    return out


@given(valid_nodeclaim(), mutation())
@settings(max_examples=120, deadline=None)
def test_changed_only_validation_equivalent_to_full(nc, kind):
    """The validator's changed-subtree pruning (an optimization) must agree
    with FULL schema validation on whether the new object is valid —
    otherwise the fake admits writes a real apiserver would reject."""
    new = _apply_mutation(nc, kind)
    full_errs = crd_validate(new, _VALIDATOR.schema)
    pruned_errs = _VALIDATOR(new, nc)  # update path: diffs against old
    # immutability: spec mutations are CEL-rejected on update regardless of
    # structural validity — exclude that extra error from the comparison
    pruned_structural = [e for e in pruned_errs if "immutable" not in e]
    assert bool(full_errs) == bool(pruned_structural), (
        f"divergence for {kind}: full={full_errs} pruned={pruned_structural}"
    )
    # and the create path must agree with full validation too
    create_errs = _VALIDATOR(copy_for_create(new), None)
    assert bool(full_errs) == bool(create_errs), (
        f"create divergence for {kind}: full={full_errs} create={create_errs}"
    )


def copy_for_create(obj):
    import copy

    return copy.deepcopy(obj)
