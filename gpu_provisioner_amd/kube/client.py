"""Typed Kubernetes client interface + error taxonomy + label selectors.

This is the seam every controller talks through. Two implementations exist:
  * ``fake.apiserver.InMemoryClient`` — a full-semantics in-process apiserver
    (optimistic concurrency, finalizers, watches) used by tests and bench.
  * ``kube.http.HTTPClient`` — httpx against a real kube-apiserver.

Replaces client-go/controller-runtime in the reference's stack (reference:
vendor/sigs.k8s.io/controller-runtime, wired in vendor/sigs.k8s.io/karpenter/
pkg/operator/operator.go) with a first-class implementation sized to this
controller's needs.
"""
from __future__ import annotations

import abc
import re
from typing import Any, AsyncIterator, Optional, Sequence


# ---------------------------------------------------------------------------
# Error taxonomy (mirrors apimachinery k8serrors the reference branches on)
# ---------------------------------------------------------------------------


class APIError(Exception):
    code = 500
    reason = "InternalError"

    def __init__(self, message: str = ""):
        super().__init__(message or self.reason)
        self.message = message or self.reason


class NotFoundError(APIError):
    code = 404
    reason = "NotFound"


class AlreadyExistsError(APIError):
    code = 409
    reason = "AlreadyExists"


class ConflictError(APIError):
    code = 409
    reason = "Conflict"


class GoneError(APIError):
    """Watch resourceVersion too old (HTTP 410)."""

    code = 410
    reason = "Gone"


class TooManyRequestsError(APIError):
    code = 429
    reason = "TooManyRequests"

    def __init__(self, message: str = "", retry_after_seconds: float = 1.0):
        super().__init__(message)
        self.retry_after_seconds = retry_after_seconds


class ForbiddenError(APIError):
    code = 403
    reason = "Forbidden"


class InvalidError(APIError):
    code = 422
    reason = "Invalid"


def is_not_found(err: BaseException) -> bool:
    return isinstance(err, NotFoundError)


def is_conflict(err: BaseException) -> bool:
    return isinstance(err, ConflictError)


# ---------------------------------------------------------------------------
# Label selectors: equality + set-based + existence, as kubectl supports.
# ---------------------------------------------------------------------------

_SEL_IN_RE = re.compile(r"^\s*([\w./-]+)\s+(in|notin)\s+\(([^)]*)\)\s*$")


class LabelSelector:
    """Parsed label selector. Supports: ``k=v``, ``k==v``, ``k!=v``, ``k``,
    ``!k``, ``k in (a,b)``, ``k notin (a,b)`` joined by commas (commas inside
    parentheses belong to the value set)."""

    def __init__(self, requirements: Sequence[tuple]):
        self._reqs = list(requirements)

    @classmethod
    def parse(cls, s: str) -> "LabelSelector":
        reqs = []
        for part in _split_selector(s):
            part = part.strip()
            if not part:
                continue
            m = _SEL_IN_RE.match(part)
            if m:
                key, op, vals = m.groups()
                values = {v.strip() for v in vals.split(",") if v.strip()}
                reqs.append((key, op, values))
            elif "!=" in part:
                k, v = part.split("!=", 1)
                reqs.append((k.strip(), "!=", {v.strip()}))
            elif "==" in part:
                k, v = part.split("==", 1)
                reqs.append((k.strip(), "=", {v.strip()}))
            elif "=" in part:
                k, v = part.split("=", 1)
                reqs.append((k.strip(), "=", {v.strip()}))
            elif part.startswith("!"):
                reqs.append((part[1:].strip(), "!exists", set()))
            else:
                reqs.append((part, "exists", set()))
        return cls(reqs)

    def matches(self, labels: dict) -> bool:
        for key, op, values in self._reqs:
            present = key in labels
            val = labels.get(key)
            if op == "=" and (not present or val not in values):
                return False
            if op == "!=" and present and val in values:
                return False
            if op == "in" and (not present or val not in values):
                return False
            if op == "notin" and present and val in values:
                return False
            if op == "exists" and not present:
                return False
            if op == "!exists" and present:
                return False
        return True

    def __str__(self) -> str:
        parts = []
        for key, op, values in self._reqs:
            if op == "=":
                parts.append(f"{key}={next(iter(values))}")
            elif op == "!=":
                parts.append(f"{key}!={next(iter(values))}")
            elif op == "in":
                parts.append(f"{key} in ({','.join(sorted(values))})")
            elif op == "notin":
                parts.append(f"{key} notin ({','.join(sorted(values))})")
            elif op == "exists":
                parts.append(key)
            elif op == "!exists":
                parts.append(f"!{key}")
        return ",".join(parts)


def _split_selector(s: str) -> list:
    """Split on commas not inside parentheses."""
    parts, depth, cur = [], 0, []
    for ch in s:
        if ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
        if ch == "," and depth == 0:
            parts.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
    if cur:
        parts.append("".join(cur))
    return parts


def match_field_selector(obj: dict, selector: Optional[str]) -> bool:
    """Dotted-path equality field selectors (``spec.nodeName=x,metadata.name=y``)."""
    if not selector:
        return True
    for part in selector.split(","):
        part = part.strip()
        if not part:
            continue
        neg = "!=" in part
        k, v = part.split("!=" if neg else "=", 1)
        cur: Any = obj
        for seg in k.strip().split("."):
            if not isinstance(cur, dict):
                cur = None
                break
            cur = cur.get(seg)
        eq = str(cur) == v.strip() if cur is not None else v.strip() == ""
        if neg == eq:
            return False
    return True


# ---------------------------------------------------------------------------
# Watch events
# ---------------------------------------------------------------------------

ADDED = "ADDED"
MODIFIED = "MODIFIED"
DELETED = "DELETED"
BOOKMARK = "BOOKMARK"


# ---------------------------------------------------------------------------
# Client interface
# ---------------------------------------------------------------------------


class KubeClient(abc.ABC):
    """Async typed kube client. Objects are wire-format JSON dicts."""

    @abc.abstractmethod
    async def get(self, api_version: str, kind: str, name: str, namespace: str = "") -> dict:
        ...

    @abc.abstractmethod
    async def list(
        self,
        api_version: str,
        kind: str,
        namespace: str = "",
        label_selector: str = "",
        field_selector: str = "",
    ) -> list:
        ...

    @abc.abstractmethod
    async def create(self, obj: dict) -> dict:
        ...

    @abc.abstractmethod
    async def update(self, obj: dict) -> dict:
        """Replace; enforces optimistic concurrency via metadata.resourceVersion."""

    @abc.abstractmethod
    async def update_status(self, obj: dict) -> dict:
        """Replace only the status subresource."""

    @abc.abstractmethod
    async def patch(
        self,
        api_version: str,
        kind: str,
        name: str,
        patch: dict,
        namespace: str = "",
        subresource: str = "",
    ) -> dict:
        """JSON merge patch (RFC 7386)."""

    @abc.abstractmethod
    async def delete(
        self,
        api_version: str,
        kind: str,
        name: str,
        namespace: str = "",
        uid_precondition: str = "",
        grace_period_seconds: Optional[int] = None,
    ) -> None:
        ...

    @abc.abstractmethod
    async def watch(
        self,
        api_version: str,
        kind: str,
        namespace: str = "",
        resource_version: str = "",
        label_selector: str = "",
    ) -> AsyncIterator[tuple]:
        """Yields (event_type, obj) tuples. Raises GoneError when
        resource_version is too old (caller re-lists)."""

    @abc.abstractmethod
    async def evict(self, pod: dict, grace_period_seconds: Optional[int] = None) -> None:
        """Create an Eviction for the pod (policy/v1 eviction subresource).
        Raises TooManyRequestsError on PDB violation (HTTP 429)."""


# ---------------------------------------------------------------------------
# JSON merge patch (RFC 7386) — used by patch() implementations
# ---------------------------------------------------------------------------


def json_merge_patch(target: dict, patch: dict) -> dict:
    """Apply RFC 7386 merge patch, returning a new dict."""
    if not isinstance(patch, dict):
        return patch
    out = dict(target) if isinstance(target, dict) else {}
    for k, v in patch.items():
        if v is None:
            out.pop(k, None)
        elif isinstance(v, dict):
            out[k] = json_merge_patch(out.get(k, {}), v)
        else:
            out[k] = v
    return out
