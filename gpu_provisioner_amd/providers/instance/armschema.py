"""Pinned Azure agentPools request schema — the ARM payload contract.

Every agent-pool PUT body the provider builds is validated against the
property set of the api-version the client pins, so a field the pinned
schema does not define can never reach ARM silently (real
ARM rejects or silently drops unknown properties, and a dropped gpuProfile
means a node without its driver stack).

Provenance
----------
* ``2024-09-01`` (stable): transcribed from azure-rest-api-specs
  ``specification/containerservice/resource-manager/Microsoft.ContainerService/
  aks/stable/2024-09-01/managedClusters.json`` —
  ``ManagedClusterAgentPoolProfileProperties`` (request fields only;
  read-only fields like provisioningState/nodeImageVersion are accepted on
  GET responses but never emitted). This build is offline, so the
  transcription is from the published swagger as of the knowledge cutoff;
  the per-field comments mark anything extrapolated.
* ``gpu-preview``: the AgentPoolGPUProfile surface from the AKS preview
  api-versions (driver install opt-out). The ROCm-specific fields
  (driverType/driverVersion/rocmVersion) are an EXTRAPOLATION for the
  MI355X generation — they exist in no published stable swagger — and are
  therefore only emitted under this explicitly-selected profile, never
  under the stable default. On the stable profile the ROCm stack is
  installed by the chart's amdgpu-driver + device-plugin DaemonSets
  instead (charts/gpu-provisioner-amd/templates/).

The reference performed no request validation at all (its typed Go SDK
structs pinned the shape implicitly — pkg/providers/instance/armutils.go:28-76);
a dict-based client needs the explicit contract.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field

# -- ManagedClusterAgentPoolProfileProperties, 2024-09-01 stable -------------
# (request-writable properties)

AGENT_POOL_PROPERTIES_2024_09_01 = frozenset(
    {
        "availabilityZones",
        "capacityReservationGroupID",
        "count",
        "creationData",
        "enableAutoScaling",
        "enableEncryptionAtHost",
        "enableFIPS",
        "enableNodePublicIP",
        "enableUltraSSD",
        "gpuInstanceProfile",
        "hostGroupID",
        "kubeletConfig",
        "kubeletDiskType",
        "linuxOSConfig",
        "maxCount",
        "maxPods",
        "minCount",
        "mode",
        "networkProfile",
        "nodeLabels",
        "nodePublicIPPrefixID",
        "nodeTaints",
        "orchestratorVersion",
        "osDiskSizeGB",
        "osDiskType",
        "osSKU",
        "osType",
        "podSubnetID",
        "powerState",
        "proximityPlacementGroupID",
        "scaleDownMode",
        "scaleSetEvictionPolicy",
        "scaleSetPriority",
        "securityProfile",
        "spotMaxPrice",
        "tags",
        "type",
        "upgradeSettings",
        "vmSize",
        "vnetSubnetID",
        "windowsProfile",
        "workloadRuntime",
    }
)

# KubeletConfig (same swagger, definition KubeletConfig)
KUBELET_CONFIG_FIELDS = frozenset(
    {
        "allowedUnsafeSysctls",
        "containerLogMaxFiles",
        "containerLogMaxSizeMB",
        "cpuCfsQuota",
        "cpuCfsQuotaPeriod",
        "cpuManagerPolicy",
        "failSwapOn",
        "imageGcHighThreshold",
        "imageGcLowThreshold",
        "podMaxPids",
        "topologyManagerPolicy",
    }
)

# LinuxOSConfig (same swagger, definition LinuxOSConfig)
LINUX_OS_CONFIG_FIELDS = frozenset(
    {"swapFileSizeMB", "sysctls", "transparentHugePageDefrag", "transparentHugePageEnabled"}
)

# SysctlConfig (same swagger, definition SysctlConfig) — camelCase keys
SYSCTL_CONFIG_FIELDS = frozenset(
    {
        "fsAioMaxNr",
        "fsFileMax",
        "fsInotifyMaxUserWatches",
        "fsNrOpen",
        "kernelThreadsMax",
        "netCoreNetdevMaxBacklog",
        "netCoreOptmemMax",
        "netCoreRmemDefault",
        "netCoreRmemMax",
        "netCoreSomaxconn",
        "netCoreWmemDefault",
        "netCoreWmemMax",
        "netIpv4IpLocalPortRange",
        "netIpv4NeighDefaultGcThresh1",
        "netIpv4NeighDefaultGcThresh2",
        "netIpv4NeighDefaultGcThresh3",
        "netIpv4TcpFinTimeout",
        "netIpv4TcpKeepaliveProbes",
        "netIpv4TcpKeepaliveTime",
        "netIpv4TcpMaxSynBacklog",
        "netIpv4TcpMaxTwBuckets",
        "netIpv4TcpTwReuse",
        "netIpv4TcpkeepaliveIntvl",
        "netNetfilterNfConntrackBuckets",
        "netNetfilterNfConntrackMax",
        "vmMaxMapCount",
        "vmSwappiness",
        "vmVfsCachePressure",
    }
)

# Enum values (same swagger; rejected value → ARM 400 InvalidParameter)
ENUMS = {
    "osType": frozenset({"Linux", "Windows"}),
    "osSKU": frozenset(
        {"AzureLinux", "CBLMariner", "Ubuntu", "Windows2019", "Windows2022"}
    ),
    "mode": frozenset({"System", "User"}),
    "type": frozenset({"AvailabilitySet", "VirtualMachineScaleSets"}),
    "scaleSetPriority": frozenset({"Regular", "Spot"}),
    "scaleSetEvictionPolicy": frozenset({"Deallocate", "Delete"}),
    "scaleDownMode": frozenset({"Deallocate", "Delete"}),
    "kubeletDiskType": frozenset({"OS", "Temporary"}),
    "workloadRuntime": frozenset({"OCIContainer", "WasmWasi"}),
    "gpuInstanceProfile": frozenset({"MIG1g", "MIG2g", "MIG3g", "MIG4g", "MIG7g"}),
}


@dataclass(frozen=True)
class ArmApiProfile:
    """One pinned api-version: its property set and GPU-profile surface."""

    api_version: str
    properties: frozenset
    # gpuProfile sub-fields allowed under this api-version; empty set means
    # the gpuProfile property itself must not be emitted
    gpu_profile_fields: frozenset = field(default_factory=frozenset)


# Default: the GA surface only. No gpuProfile — driver install comes from
# the chart's DaemonSets (see module docstring).
PROFILE_STABLE = ArmApiProfile(
    api_version="2024-09-01",
    properties=AGENT_POOL_PROPERTIES_2024_09_01,
)

# Opt-in: preview GPU-driver surface + the MI355X/ROCm extrapolation.
PROFILE_GPU_PREVIEW = ArmApiProfile(
    api_version="2025-05-02-preview",  # EXTRAPOLATED preview version
    properties=AGENT_POOL_PROPERTIES_2024_09_01 | {"gpuProfile"},
    gpu_profile_fields=frozenset(
        {"driver", "driverType", "driverVersion", "rocmVersion"}
    ),
)

_PROFILES = {
    "stable": PROFILE_STABLE,
    PROFILE_STABLE.api_version: PROFILE_STABLE,
    "gpu-preview": PROFILE_GPU_PREVIEW,
    PROFILE_GPU_PREVIEW.api_version: PROFILE_GPU_PREVIEW,
}


def profile_from_env(environ=None) -> ArmApiProfile:
    """ARM_API_PROFILE selects the pinned profile (default: stable)."""
    env = environ if environ is not None else os.environ
    name = env.get("ARM_API_PROFILE", "stable")
    try:
        return _PROFILES[name]
    except KeyError:
        raise ValueError(
            f"unknown ARM_API_PROFILE {name!r}; valid: {sorted(set(_PROFILES))}"
        ) from None


class SchemaViolation(ValueError):
    """The pool builder emitted a field the pinned api-version doesn't define."""


def validate_agent_pool(pool: dict, profile: ArmApiProfile = PROFILE_STABLE) -> None:
    """Validate an agent-pool PUT body against the pinned schema; raises
    SchemaViolation on any property/enum outside it."""
    errs: list = []
    unknown_top = set(pool) - {"name", "properties"}
    if unknown_top:
        errs.append(f"unknown top-level fields: {sorted(unknown_top)}")
    props = pool.get("properties", {})
    unknown = set(props) - profile.properties
    if unknown:
        errs.append(
            f"properties not in api-version {profile.api_version}: {sorted(unknown)}"
        )
    for key, allowed in ENUMS.items():
        if key in props and props[key] not in allowed:
            errs.append(f"{key}={props[key]!r} not in {sorted(allowed)}")
    kc = props.get("kubeletConfig")
    if kc is not None:
        bad = set(kc) - KUBELET_CONFIG_FIELDS
        if bad:
            errs.append(f"kubeletConfig fields not in schema: {sorted(bad)}")
    loc = props.get("linuxOSConfig")
    if loc is not None:
        bad = set(loc) - LINUX_OS_CONFIG_FIELDS
        if bad:
            errs.append(f"linuxOSConfig fields not in schema: {sorted(bad)}")
        sysctls = (loc or {}).get("sysctls")
        if sysctls is not None:
            bad = set(sysctls) - SYSCTL_CONFIG_FIELDS
            if bad:
                errs.append(f"linuxOSConfig.sysctls keys not in schema: {sorted(bad)}")
    gp = props.get("gpuProfile")
    if gp is not None:
        if not profile.gpu_profile_fields:
            errs.append(
                f"gpuProfile is not defined by api-version {profile.api_version}"
            )
        else:
            bad = set(gp) - profile.gpu_profile_fields
            if bad:
                errs.append(f"gpuProfile fields not in profile: {sorted(bad)}")
    if errs:
        raise SchemaViolation(
            f"agent pool {pool.get('name', '?')!r}: " + "; ".join(errs)
        )
