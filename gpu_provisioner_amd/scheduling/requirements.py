"""Node-selector requirement algebra.

The subset of karpenter's scheduling library the provisioner exercises
(reference vendor/sigs.k8s.io/karpenter/pkg/scheduling/requirement.go and
requirements.go): typed requirements with In/NotIn/Exists/DoesNotExist/Gt/Lt
operators and minValues, intersection, compatibility, and label rendering.
Used by the instance provider to extract the VM size
(reference pkg/providers/instance/instance.go:90-95) and by registration to
sync labels.
"""
from __future__ import annotations

from typing import Iterable, Optional

IN = "In"
NOT_IN = "NotIn"
EXISTS = "Exists"
DOES_NOT_EXIST = "DoesNotExist"
GT = "Gt"
LT = "Lt"


class Requirement:
    """A single key requirement. Modeled as (complement, values, gt, lt):
    complement=False → allowed set is `values`; complement=True → allowed set
    is everything except `values` (Exists = complement of empty set)."""

    __slots__ = ("key", "complement", "values", "greater_than", "less_than", "min_values")

    def __init__(
        self,
        key: str,
        operator: str,
        values: Iterable[str] = (),
        min_values: Optional[int] = None,
    ):
        self.key = key
        self.greater_than: Optional[int] = None
        self.less_than: Optional[int] = None
        self.min_values = min_values
        vals = list(values)
        if operator == IN:
            self.complement = False
            self.values = set(vals)
        elif operator == NOT_IN:
            self.complement = True
            self.values = set(vals)
        elif operator == EXISTS:
            self.complement = True
            self.values = set()
        elif operator == DOES_NOT_EXIST:
            self.complement = False
            self.values = set()
        elif operator == GT:
            self.complement = True
            self.values = set()
            self.greater_than = int(vals[0])
        elif operator == LT:
            self.complement = True
            self.values = set()
            self.less_than = int(vals[0])
        else:
            raise ValueError(f"unsupported operator {operator!r}")

    @classmethod
    def from_dict(cls, d: dict) -> "Requirement":
        return cls(
            key=d.get("key", ""),
            operator=d.get("operator", IN),
            values=d.get("values") or [],
            min_values=d.get("minValues"),
        )

    def to_dict(self) -> dict:
        d: dict = {"key": self.key, "operator": self.operator()}
        if self.operator() in (IN, NOT_IN):
            d["values"] = sorted(self.values)
        elif self.operator() in (GT, LT):
            d["values"] = [str(self.greater_than if self.operator() == GT else self.less_than)]
        if self.min_values is not None:
            d["minValues"] = self.min_values
        return d

    def operator(self) -> str:
        if self.greater_than is not None:
            return GT
        if self.less_than is not None:
            return LT
        if self.complement:
            return EXISTS if not self.values else NOT_IN
        return DOES_NOT_EXIST if not self.values else IN

    def has(self, value: str) -> bool:
        if self.greater_than is not None:
            try:
                if int(value) <= self.greater_than:
                    return False
            except ValueError:
                return False
        if self.less_than is not None:
            try:
                if int(value) >= self.less_than:
                    return False
            except ValueError:
                return False
        if self.complement:
            return value not in self.values
        return value in self.values

    def intersect(self, other: "Requirement") -> "Requirement":
        out = Requirement(self.key, EXISTS)
        # numeric bounds: tightest
        for r in (self, other):
            if r.greater_than is not None and (
                out.greater_than is None or r.greater_than > out.greater_than
            ):
                out.greater_than = r.greater_than
            if r.less_than is not None and (out.less_than is None or r.less_than < out.less_than):
                out.less_than = r.less_than
        if self.complement and other.complement:
            out.complement = True
            out.values = self.values | other.values
        elif self.complement:
            out.complement = False
            out.values = {v for v in other.values if v not in self.values}
        elif other.complement:
            out.complement = False
            out.values = {v for v in self.values if v not in other.values}
        else:
            out.complement = False
            out.values = self.values & other.values
        if not out.complement:
            out.values = {v for v in out.values if out._in_bounds(v)}
        out.min_values = max(
            (m for m in (self.min_values, other.min_values) if m is not None), default=None
        )
        return out

    def _in_bounds(self, value: str) -> bool:
        if self.greater_than is None and self.less_than is None:
            return True
        try:
            n = int(value)
        except ValueError:
            return False
        if self.greater_than is not None and n <= self.greater_than:
            return False
        if self.less_than is not None and n >= self.less_than:
            return False
        return True

    def any(self) -> Optional[str]:
        if not self.complement and self.values:
            return sorted(self.values)[0]
        return None

    def is_empty(self) -> bool:
        """Unsatisfiable: a non-complement requirement with no allowed values."""
        return not self.complement and not self.values

    def __repr__(self) -> str:
        return f"Requirement({self.key} {self.operator()} {sorted(self.values)})"


class Requirements:
    """A conjunction of requirements keyed by label key."""

    def __init__(self, reqs: Iterable[Requirement] = ()):
        self._reqs: dict = {}
        for r in reqs:
            self.add(r)

    @classmethod
    def from_nodeclaim(cls, nodeclaim: dict) -> "Requirements":
        """Requirements from spec.requirements + single-value label requirements
        (reference scheduling.NewLabelRequirements + NewNodeSelectorRequirements
        combined, as instance.Provider.Create consumes them)."""
        out = cls()
        for key, val in (nodeclaim.get("metadata", {}).get("labels") or {}).items():
            out.add(Requirement(key, IN, [val]))
        for d in nodeclaim.get("spec", {}).get("requirements") or []:
            out.add(Requirement.from_dict(d))
        return out

    def add(self, req: Requirement) -> None:
        cur = self._reqs.get(req.key)
        self._reqs[req.key] = cur.intersect(req) if cur else req

    def get(self, key: str) -> Optional[Requirement]:
        return self._reqs.get(key)

    def has(self, key: str) -> bool:
        return key in self._reqs

    def keys(self):
        return self._reqs.keys()

    def values_of(self, key: str) -> list:
        r = self._reqs.get(key)
        if r is None or r.complement:
            return []
        return sorted(r.values)

    def compatible(self, labels: dict) -> bool:
        """True if a node with `labels` satisfies these requirements."""
        for key, req in self._reqs.items():
            if key not in labels:
                if req.operator() in (IN, GT, LT):
                    return False
                if req.operator() == EXISTS:
                    return False
                continue
            if req.operator() == DOES_NOT_EXIST:
                return False
            if not req.has(labels[key]):
                return False
        return True

    def labels(self) -> dict:
        """Single-valued In requirements rendered as labels (what registration
        syncs onto the Node)."""
        out = {}
        for key, req in self._reqs.items():
            if not req.complement and len(req.values) == 1:
                out[key] = next(iter(req.values))
        return out

    def __len__(self) -> int:
        return len(self._reqs)

    def __iter__(self):
        return iter(self._reqs.values())
