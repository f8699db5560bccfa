"""Small shared utilities (reference pkg/utils/utils.go)."""
from __future__ import annotations

import os
import re
from typing import Optional

# providerID format for AKS VMSS-backed nodes:
#   azure:///subscriptions/<sub>/resourceGroups/<rg>/providers/
#     Microsoft.Compute/virtualMachineScaleSets/aks-<pool>-<hash>-vmss/virtualMachines/<idx>
# The agent-pool name is the second dash-token of the VMSS name
# (reference pkg/utils/utils.go:27-46).
_PROVIDER_ID_RE = re.compile(
    r"azure:///subscriptions/[^/]+/resourceGroups/[^/]+/providers/"
    r"Microsoft\.Compute/virtualMachineScaleSets/(?P<vmss>[^/]+)/virtualMachines/\d+"
)


def parse_agent_pool_name_from_id(provider_id: str) -> Optional[str]:
    m = _PROVIDER_ID_RE.match(provider_id or "")
    if not m:
        return None
    vmss = m.group("vmss")
    parts = vmss.split("-")
    # aks-<pool...>-<hash>-vmss : pool name is everything between the leading
    # "aks" and the trailing <hash>, "vmss" tokens. Pool names matching
    # ^[a-z][a-z0-9]{0,11}$ contain no dashes, so this is parts[1].
    if len(parts) < 4 or parts[0] != "aks" or parts[-1] != "vmss":
        return None
    return parts[1]


def build_provider_id(subscription: str, resource_group: str, pool: str, vmss_hash: str) -> str:
    return (
        f"azure:///subscriptions/{subscription}/resourceGroups/{resource_group}/providers/"
        f"Microsoft.Compute/virtualMachineScaleSets/aks-{pool}-{vmss_hash}-vmss/virtualMachines/0"
    )


def env_bool(key: str, default: bool) -> bool:
    """WithDefaultBool (reference pkg/utils/utils.go:50-60)."""
    v = os.environ.get(key)
    if v is None:
        return default
    return v.strip().lower() in ("1", "true", "yes", "on")


def env_str(key: str, default: str = "") -> str:
    return os.environ.get(key, default)


def env_int(key: str, default: int) -> int:
    v = os.environ.get(key)
    if v is None:
        return default
    try:
        return int(v)
    except ValueError:
        return default


def env_float(key: str, default: float) -> float:
    v = os.environ.get(key)
    if v is None:
        return default
    try:
        return float(v)
    except ValueError:
        return default
