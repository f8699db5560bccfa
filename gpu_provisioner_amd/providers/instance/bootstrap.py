"""ROCm node bootstrap for MI355X agent pools — net-new vs the reference.

The reference's node-image surface was a single OSSKU switch on the
`kaito.sh/node-image-family` annotation (reference pkg/providers/instance/
instance.go:364,415-441) and relied on AKS's preinstalled NVIDIA driver
images. MI355X nodes need the amdgpu/ROCm stack and the AMD k8s device plugin
before `amd.com/gpu` appears in allocatable (the initialization gate), so the
bootstrap here is explicit:

  * OSSKU selection (Ubuntu default / AzureLinux via annotation), both mapped
    to the AKS GPU image flavor that carries the amdgpu DKMS driver;
  * a gpuProfile asking AKS to install the ROCm driver stack;
  * kubelet/sysctl settings sized for 8× MI355X hosts (huge BAR, many hugepages,
    device-plugin socket dir);
  * node labels advertising the GPU product/vram/xGMI topology so whole-host
    8-GPU workloads can bind to one xGMI island;
  * a taint that keeps ordinary pods off the node until the device plugin
    registers (`amd.com/gpu=present:NoSchedule` is removed by initialization-
    aware workloads tolerating it).
"""
from __future__ import annotations

from typing import Optional

from ...apis import v1 as karpv1
from ..instancetype.catalog import (
    GFX_ARCH,
    GPU_PRODUCT,
    HBM_PER_GPU_GB,
    InstanceTypeProvider,
    XGMI_GBPS_PER_LINK,
    XGMI_LINKS_PER_GPU,
)
from .armschema import ArmApiProfile, PROFILE_STABLE

OSSKU_UBUNTU = "Ubuntu"
OSSKU_AZURELINUX = "AzureLinux"

# ROCm stack versions the preview gpuProfile requests. EXTRAPOLATED for the
# MI355X generation (no published swagger pins these — see armschema.py);
# only emitted under the explicitly-selected gpu-preview ARM profile. On the
# stable profile the same stack is installed by the chart's amdgpu-driver +
# device-plugin DaemonSets.
ROCM_VERSION = "6.4"
AMDGPU_DRIVER_VERSION = "6.10.5"
DEVICE_PLUGIN_IMAGE = "rocm/k8s-device-plugin:latest"

_IMAGE_FAMILY_TO_OSSKU = {
    "": OSSKU_UBUNTU,
    "ubuntu": OSSKU_UBUNTU,
    "ubuntu2204": OSSKU_UBUNTU,
    "azurelinux": OSSKU_AZURELINUX,
    "mariner": OSSKU_AZURELINUX,
}


def determine_os_sku(image_family_annotation: str) -> str:
    """OSSKU from the kaito.sh/node-image-family annotation; unknown values
    fall back to Ubuntu (reference instance.go:415-441 semantics)."""
    return _IMAGE_FAMILY_TO_OSSKU.get((image_family_annotation or "").strip().lower(), OSSKU_UBUNTU)


def rocm_gpu_profile(profile: ArmApiProfile = PROFILE_STABLE) -> Optional[dict]:
    """AgentPool gpuProfile requesting AMD driver installation (the AKS-side
    switch that provisions amdgpu instead of the NVIDIA stack).

    Returns None when the pinned api-version defines no gpuProfile (the
    stable default) — the builder must then omit the property entirely and
    driver installation falls to the chart's DaemonSets. Under gpu-preview
    only the profile's declared sub-fields are emitted."""
    if not profile.gpu_profile_fields:
        return None
    full = {
        "driver": "Install",
        "driverType": "ROCm",
        "driverVersion": AMDGPU_DRIVER_VERSION,
        "rocmVersion": ROCM_VERSION,
    }
    return {k: v for k, v in full.items() if k in profile.gpu_profile_fields}


def rocm_kubelet_config() -> dict:
    return {
        # room for per-GPU device-plugin endpoints + RCCL shared-memory fds
        "podMaxPids": -1,
        "failSwapOn": True,
        "topologyManagerPolicy": "single-numa-node",
    }


def rocm_linux_os_config(gpus: int) -> dict:
    return {
        "sysctls": {
            # xGMI/RDMA pinned allocations for RCCL over large HBM shards
            "vmMaxMapCount": 1048576,
        },
        # 2Mi hugepages for ROCm userptr staging: 1 GiB per GPU
        "transparentHugePageEnabled": "always",
    }


def gpu_node_labels(vm_size: str, catalog: InstanceTypeProvider) -> dict:
    """Node labels the agent pool stamps on MI355X nodes — the scheduling
    topology surface (net-new; the reference stamped none)."""
    it = catalog.get(vm_size)
    if it is None or not catalog.is_gpu_sku(vm_size):
        return {}
    gpus = catalog.gpu_count(vm_size)
    return {
        karpv1.AMD_GPU_PRODUCT_LABEL_KEY: it.requirements.get(
            karpv1.AMD_GPU_PRODUCT_LABEL_KEY, GPU_PRODUCT
        ),
        karpv1.AMD_GPU_COUNT_LABEL_KEY: str(gpus),
        karpv1.AMD_GPU_VRAM_LABEL_KEY: it.requirements.get(
            karpv1.AMD_GPU_VRAM_LABEL_KEY, f"{HBM_PER_GPU_GB}G"
        ),
        karpv1.XGMI_TOPOLOGY_LABEL_KEY: it.requirements.get(
            karpv1.XGMI_TOPOLOGY_LABEL_KEY,
            f"{gpus}x-{XGMI_LINKS_PER_GPU}l-{XGMI_GBPS_PER_LINK}g",
        ),
        "amd.com/compute-arch": it.requirements.get("amd.com/compute-arch", GFX_ARCH),
    }
