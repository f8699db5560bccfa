"""kaito.sh/v1alpha1 KaitoNodeClass — the (empty-spec) NodeClass CRD.

Exists to satisfy GetSupportedNodeClasses / NodeClassRef matching, exactly as
the reference's cluster-scoped KaitoNodeClass
(reference pkg/apis/v1alpha1/kaitonodeclass.go:28-50, register.go:25-39,
kaitonodeclass_status.go:23-33 — conditions are a no-op).
"""
from __future__ import annotations

GROUP = "kaito.sh"
VERSION = "v1alpha1"
API_VERSION = f"{GROUP}/{VERSION}"
KIND_KAITONODECLASS = "KaitoNodeClass"


def new_kaitonodeclass(name: str) -> dict:
    return {
        "apiVersion": API_VERSION,
        "kind": KIND_KAITONODECLASS,
        "metadata": {"name": name},
        "spec": {},
        "status": {"conditions": []},
    }


def node_class_ref(name: str) -> dict:
    """A NodeClassRef pointing at a KaitoNodeClass, for NodeClaim specs."""
    return {"group": GROUP, "kind": KIND_KAITONODECLASS, "name": name}
