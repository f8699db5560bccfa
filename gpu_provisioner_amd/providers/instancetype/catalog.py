"""MI355X VM SKU catalog — the net-new, AMD-native instance-type provider.

The reference never modeled GPU capacity at all: GetInstanceTypes returned an
empty list (reference pkg/cloudprovider/cloudprovider.go:99-101) and GPU
detection was a `Standard_N` prefix match
(pkg/providers/instance/instance.go:335-339) against NVIDIA families. Here
the catalog is first-class: each SKU carries the full MI355X topology —
8× MI355X (gfx950, CDNA4) per host, 288 GB HBM3E per GPU (~8 TB/s), xGMI
point-to-point mesh (7 links × ~153 GB/s per GPU) — surfaced as capacity
(`amd.com/gpu`) and as node labels so schedulers can place whole-host 8-GPU
jobs on one xGMI island.
"""
from __future__ import annotations

from typing import Optional

from ...apis import v1 as karpv1
from ...cloudprovider.types import InstanceType, Offering

GFX_ARCH = "gfx950"
GPU_PRODUCT = "AMD-Instinct-MI355X"
HBM_PER_GPU_GB = 288
XGMI_LINKS_PER_GPU = 7
XGMI_GBPS_PER_LINK = 153

# SKU table — PROVENANCE: Azure's AMD Instinct family uses the NDis-MI
# naming scheme (Standard_ND96isr_MI300X_v5 is the published MI300X SKU);
# the MI355X-generation names/sizes below are an ACKNOWLEDGED EXTRAPOLATION
# of that scheme (no MI355X SKU list is publishable at build time, and this
# environment is offline). Deployments override or extend the table without
# a code change via GPU_PROV_SKU_FILE (YAML/JSON list of SKU dicts, same
# fields) — see InstanceTypeProvider.__init__.
# Fields: vCPU, memory GiB, GPU count, ephemeral OS disk ceiling GiB,
# price-per-hour (on-demand, list), zones.
_SKUS = [
    {
        "name": "Standard_ND128isr_MI355X_v6",
        "vcpu": 128,
        "memory_gib": 2048,
        "gpus": 8,
        "max_os_disk_gib": 4096,
        "price": 72.0,
        "infiniband": True,
    },
    {
        "name": "Standard_ND64is_MI355X_v6",
        "vcpu": 64,
        "memory_gib": 1024,
        "gpus": 4,
        "max_os_disk_gib": 2048,
        "price": 36.0,
        "infiniband": False,
    },
    {
        "name": "Standard_ND32is_MI355X_v6",
        "vcpu": 32,
        "memory_gib": 512,
        "gpus": 2,
        "max_os_disk_gib": 1024,
        "price": 18.0,
        "infiniband": False,
    },
    # Previous generation kept orderable for mixed fleets.
    {
        "name": "Standard_ND96isr_MI300X_v5",
        "vcpu": 96,
        "memory_gib": 1850,
        "gpus": 8,
        "max_os_disk_gib": 4096,
        "price": 48.0,
        "infiniband": True,
        "product": "AMD-Instinct-MI300X",
        "vram_gb": 192,
        "arch": "gfx942",
    },
]

_DEFAULT_ZONES = ("1", "2", "3")

# kubelet overhead reservations (kube-reserved + system-reserved + eviction),
# the allocatable precompute karpenter does in types.go:102-219.
_OVERHEAD_CPU_MILLI = 240
_OVERHEAD_MEM_GIB = 12


def _build(sku: dict, region: str, zones=_DEFAULT_ZONES) -> InstanceType:
    product = sku.get("product", GPU_PRODUCT)
    vram = sku.get("vram_gb", HBM_PER_GPU_GB)
    arch = sku.get("arch", GFX_ARCH)
    gpus = sku["gpus"]
    capacity = {
        "cpu": str(sku["vcpu"]),
        "memory": f"{sku['memory_gib']}Gi",
        "ephemeral-storage": f"{sku['max_os_disk_gib']}Gi",
        "pods": "250",
        karpv1.AMD_GPU_RESOURCE: str(gpus),
    }
    overhead = {
        "cpu": f"{_OVERHEAD_CPU_MILLI}m",
        "memory": f"{_OVERHEAD_MEM_GIB}Gi",
        "pods": "0",
    }
    requirements = {
        karpv1.INSTANCE_TYPE_LABEL_KEY: sku["name"],
        karpv1.ARCH_LABEL_KEY: "amd64",
        karpv1.OS_LABEL_KEY: "linux",
        karpv1.AMD_GPU_PRODUCT_LABEL_KEY: product,
        karpv1.AMD_GPU_VRAM_LABEL_KEY: f"{vram}G",
        karpv1.AMD_GPU_COUNT_LABEL_KEY: str(gpus),
        # xGMI island: all GPUs on one host are fully connected point-to-point;
        # whole-host jobs should require amd.com/gpu: <gpus> on ONE node.
        karpv1.XGMI_TOPOLOGY_LABEL_KEY: f"{gpus}x-{XGMI_LINKS_PER_GPU}l-{XGMI_GBPS_PER_LINK}g",
        "amd.com/compute-arch": arch,
    }
    offerings = []
    for z in zones:
        offerings.append(Offering("on-demand", f"{region}-{z}", sku["price"]))
        offerings.append(Offering("spot", f"{region}-{z}", sku["price"] * 0.35))
    return InstanceType(
        name=sku["name"],
        capacity=capacity,
        overhead=overhead,
        requirements=requirements,
        offerings=offerings,
    )


class InstanceTypeProvider:
    """Catalog provider: list SKUs, resolve one by name, answer GPU questions."""

    def __init__(self, region: str = "eastus2", sku_file: Optional[str] = None):
        import os

        self.region = region
        skus = list(_SKUS)
        path = sku_file if sku_file is not None else os.environ.get("GPU_PROV_SKU_FILE", "")
        if path:
            skus = self._merge_sku_file(skus, path)
        self._types = {s["name"]: _build(s, region) for s in skus}

    @staticmethod
    def _merge_sku_file(skus: list, path: str) -> list:
        """Load a YAML/JSON SKU list; entries override built-ins by name.
        An entry of just {name: ..., remove: true} drops a built-in SKU."""
        import yaml

        with open(path) as f:
            loaded = yaml.safe_load(f) or []
        if not isinstance(loaded, list):
            raise ValueError(f"{path}: expected a list of SKU entries")
        by_name = {s["name"]: s for s in skus}
        for entry in loaded:
            name = entry.get("name")
            if not name:
                raise ValueError(f"{path}: SKU entry missing 'name': {entry}")
            if entry.get("remove"):
                by_name.pop(name, None)
                continue
            required = {"vcpu", "memory_gib", "gpus", "max_os_disk_gib", "price"}
            missing = required - set(entry)
            if missing:
                raise ValueError(f"{path}: SKU {name} missing fields {sorted(missing)}")
            by_name[name] = entry
        return list(by_name.values())

    def list(self) -> list:
        return list(self._types.values())

    def get(self, name: str) -> Optional[InstanceType]:
        return self._types.get(name)

    def is_gpu_sku(self, vm_size: str) -> bool:
        """AMD-native GPU detection: catalog membership with amd.com/gpu
        capacity — replaces the reference's `Standard_N` substring test
        (instance.go:335-339) which both under- and over-matched."""
        it = self._types.get(vm_size)
        if it is not None:
            return karpv1.AMD_GPU_RESOURCE in it.capacity
        # Conservative fallback for SKUs outside the catalog: only Azure AMD
        # Instinct families, which carry the accelerator model in the name
        # (…_MI300X_…, …_MI355X_…). A bare Standard_ND*/Standard_NC* prefix
        # must NOT match — Azure's ND/NC families are mostly NVIDIA (e.g.
        # ND H100 v5, ND A100 v4) and stamping a ROCm gpuProfile on those
        # pools would break them.
        import re

        return re.search(r"_MI\d{3}", vm_size) is not None

    def gpu_count(self, vm_size: str) -> int:
        it = self._types.get(vm_size)
        if it is None:
            return 0
        return int(it.capacity.get(karpv1.AMD_GPU_RESOURCE, "0"))
