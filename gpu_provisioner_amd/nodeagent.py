"""Python binding + CLI for the MI355X node agent (nodeagent/agent.hip).

Runs on provisioned nodes (DaemonSet) to validate the GPU stack before the
node is trusted, and feeds the node.health repair policy with on-node
evidence. Fails LOUDLY if the native library is missing on a GPU machine —
there is no eager/python fallback for the hardware checks.

CLI: ``python -m gpu_provisioner_amd.nodeagent [--json] [--expect-gpus N]``
Exit 0 = healthy, 1 = unhealthy/degraded, 2 = agent error.
"""
from __future__ import annotations

import argparse
import ctypes
import json
import os
import sys
from dataclasses import asdict, dataclass, field

_LIB_NAME = "libmi355x_nodeagent.so"
_SEARCH_PATHS = (
    os.path.join(os.path.dirname(__file__), "_native", _LIB_NAME),
    os.path.join(os.path.dirname(__file__), "..", "nodeagent", _LIB_NAME),
)

EXPECTED_ARCH = "gfx950"
EXPECTED_HBM_GB = 288
# healthy-node floors (MI355X: 8 TB/s HBM peak; the contiguous-chunk nt
# copy probe measures ~5.7 TB/s on healthy silicon — floor leaves margin
# for box-to-box variance; xGMI 7 links x ~153 GB/s guide values)
MIN_HBM_BW_GBS = 4800.0
MIN_XGMI_BW_GBS = 30.0
# D2D copy path (SDMA/blit — the hardware RCCL's transports lean on);
# measured ~4.9 TB/s on healthy silicon (profiles/), floor leaves margin
MIN_SDMA_BW_GBS = 3000.0


class NodeAgentError(RuntimeError):
    pass


def _load_lib() -> ctypes.CDLL:
    for p in _SEARCH_PATHS:
        if os.path.exists(p):
            return ctypes.CDLL(os.path.abspath(p))
    raise NodeAgentError(
        f"{_LIB_NAME} not found (searched {', '.join(_SEARCH_PATHS)}); "
        "build it with __graft_entry__.build() / make -C nodeagent"
    )


@dataclass
class GPUReport:
    index: int
    name: str = ""
    arch: str = ""
    hbm_gb: float = 0.0
    cus: int = 0
    hbm_bw_gbs: float = 0.0
    fma_ok: bool = False
    mfma_ok: bool = False
    mfma_bf16_ok: bool = False
    mfma_fp8_ok: bool = False
    lds_ok: bool = False
    lds_bytes_tested: int = 0
    sdma_bw_gbs: float = 0.0
    healthy: bool = False
    problems: list = field(default_factory=list)


@dataclass
class NodeReport:
    healthy: bool = False
    gpu_count: int = 0
    gpus: list = field(default_factory=list)
    xgmi_p2p: list = field(default_factory=list)
    xgmi_bw_gbs: list = field(default_factory=list)
    problems: list = field(default_factory=list)


class NodeAgent:
    def __init__(self):
        self.lib = _load_lib()
        self.lib.na_last_error.restype = ctypes.c_char_p

    def _err(self) -> str:
        return (self.lib.na_last_error() or b"").decode()

    def device_count(self) -> int:
        n = ctypes.c_int(0)
        if self.lib.na_device_count(ctypes.byref(n)) != 0:
            raise NodeAgentError(f"device_count: {self._err()}")
        return n.value

    def device_info(self, dev: int) -> tuple:
        name = ctypes.create_string_buffer(256)
        arch = ctypes.create_string_buffer(256)
        hbm = ctypes.c_longlong(0)
        cus = ctypes.c_int(0)
        clk = ctypes.c_int(0)
        rc = self.lib.na_device_info(
            dev, name, 256, arch, 256, ctypes.byref(hbm), ctypes.byref(cus), ctypes.byref(clk)
        )
        if rc != 0:
            raise NodeAgentError(f"device_info({dev}): {self._err()}")
        return name.value.decode(), arch.value.decode(), hbm.value, cus.value

    def hbm_bandwidth(self, dev: int, bytes_: int = 1 << 30, iters: int = 10) -> float:
        out = ctypes.c_double(0)
        rc = self.lib.na_hbm_bandwidth(
            dev, ctypes.c_longlong(bytes_), iters, ctypes.byref(out)
        )
        if rc != 0:
            raise NodeAgentError(f"hbm_bandwidth({dev}): {self._err()}")
        return out.value

    def fma_selftest(self, dev: int) -> bool:
        return self.lib.na_fma_selftest(dev) == 0

    def mfma_selftest(self, dev: int) -> bool:
        return self.lib.na_mfma_selftest(dev) == 0

    def mfma_bf16_selftest(self, dev: int) -> bool:
        """v_mfma_f32_16x16x32_bf16 — the CDNA4 training datapath."""
        return self.lib.na_mfma_bf16_selftest(dev) == 0

    def mfma_fp8_selftest(self, dev: int) -> bool:
        """v_mfma_f32_16x16x32_fp8_fp8 (E4M3) — the serving datapath."""
        return self.lib.na_mfma_fp8_selftest(dev) == 0

    def mfma_bf16_tile_check(self, dev: int) -> bool:
        """Layout-correct 16x16x32 bf16 GEMM tile with asymmetric data vs
        an exact host reference — catches fragment-mapping errors the
        uniform-operand self-tests cannot."""
        return self.lib.na_mfma_bf16_tile_check(dev) == 0

    def mfma_fp8_tile_check(self, dev: int) -> bool:
        """E4M3 variant of the layout-correct tile check."""
        return self.lib.na_mfma_fp8_tile_check(dev) == 0

    def mfma_i8_tile_check(self, dev: int) -> bool:
        """int8 K=64 variant (v_mfma_i32_16x16x64_i8) of the tile check."""
        return self.lib.na_mfma_i8_tile_check(dev) == 0

    def mfma_f16_tile_check(self, dev: int) -> bool:
        """f16 variant (v_mfma_f32_16x16x32_f16) of the tile check."""
        return self.lib.na_mfma_f16_tile_check(dev) == 0

    def mfma_bf16_tile32_check(self, dev: int) -> bool:
        """32x32x16 bf16 tile — the other CDNA4 tile geometry, with its
        distinct C/D fragment map."""
        return self.lib.na_mfma_bf16_tile32_check(dev) == 0

    def mfma_mx_tile_check(self, dev: int) -> bool:
        """Block-scaled MX path (v_mfma_scale_f32_16x16x128_f8f6f4, the
        fp4/fp6/MX serving pipe): layout-correct fp8 data + E8M0 scale
        semantics vs exact host references."""
        return self.lib.na_mfma_mx_tile_check(dev) == 0

    def lds_selftest(self, dev: int) -> tuple:
        """(ok, bytes_tested): whole-LDS pattern write/swizzled-read check."""
        tested = ctypes.c_longlong(0)
        rc = self.lib.na_lds_selftest(dev, ctypes.byref(tested))
        return rc == 0, tested.value

    def sdma_bandwidth(self, dev: int, bytes_: int = 512 << 20, iters: int = 10) -> float:
        """D2D copy through the SDMA engines — RCCL's xGMI transport hardware."""
        out = ctypes.c_double(0)
        rc = self.lib.na_sdma_bandwidth(dev, ctypes.c_longlong(bytes_), iters, ctypes.byref(out))
        if rc != 0:
            raise NodeAgentError(f"sdma_bandwidth({dev}): {self._err()}")
        return out.value

    def p2p_matrix(self, n: int) -> list:
        buf = (ctypes.c_int * (n * n))()
        if self.lib.na_p2p_matrix(n, buf) != 0:
            raise NodeAgentError(f"p2p_matrix: {self._err()}")
        return [[buf[i * n + j] for j in range(n)] for i in range(n)]

    def p2p_bandwidth(self, src: int, dst: int, bytes_: int = 256 << 20, iters: int = 5) -> float:
        out = ctypes.c_double(0)
        rc = self.lib.na_p2p_bandwidth(
            src, dst, ctypes.c_longlong(bytes_), iters, ctypes.byref(out)
        )
        if rc != 0:
            raise NodeAgentError(f"p2p_bandwidth({src},{dst}): {self._err()}")
        return out.value

    # -- full health check ---------------------------------------------------

    def check(self, expect_gpus: int = 0, bw_bytes: int = 1 << 30) -> NodeReport:
        report = NodeReport()
        report.gpu_count = self.device_count()
        if expect_gpus and report.gpu_count != expect_gpus:
            report.problems.append(
                f"expected {expect_gpus} GPUs, found {report.gpu_count}"
            )
        for d in range(report.gpu_count):
            g = GPUReport(index=d)
            try:
                g.name, g.arch, hbm, g.cus = self.device_info(d)
                g.hbm_gb = round(hbm / 1e9, 1)
                if EXPECTED_ARCH not in g.arch:
                    g.problems.append(f"arch {g.arch} != {EXPECTED_ARCH}")
                g.hbm_bw_gbs = round(self.hbm_bandwidth(d, bw_bytes), 1)
                if g.hbm_bw_gbs < MIN_HBM_BW_GBS:
                    g.problems.append(
                        f"HBM bandwidth {g.hbm_bw_gbs} GB/s below floor {MIN_HBM_BW_GBS}"
                    )
                g.fma_ok = self.fma_selftest(d)
                if not g.fma_ok:
                    g.problems.append(f"VALU FMA selftest failed: {self._err()}")
                g.mfma_ok = self.mfma_selftest(d)
                if not g.mfma_ok:
                    g.problems.append(f"MFMA matrix-pipe selftest failed: {self._err()}")
                g.mfma_bf16_ok = self.mfma_bf16_selftest(d)
                if not g.mfma_bf16_ok:
                    g.problems.append(f"MFMA bf16 selftest failed: {self._err()}")
                g.mfma_fp8_ok = self.mfma_fp8_selftest(d)
                if not g.mfma_fp8_ok:
                    g.problems.append(f"MFMA fp8 selftest failed: {self._err()}")
                if not self.mfma_bf16_tile_check(d):
                    g.problems.append(f"MFMA bf16 tile check failed: {self._err()}")
                if not self.mfma_fp8_tile_check(d):
                    g.problems.append(f"MFMA fp8 tile check failed: {self._err()}")
                if not self.mfma_i8_tile_check(d):
                    g.problems.append(f"MFMA i8 tile check failed: {self._err()}")
                if not self.mfma_f16_tile_check(d):
                    g.problems.append(f"MFMA f16 tile check failed: {self._err()}")
                if not self.mfma_mx_tile_check(d):
                    g.problems.append(f"MFMA MX-scaled tile check failed: {self._err()}")
                if not self.mfma_bf16_tile32_check(d):
                    g.problems.append(f"MFMA 32x32 tile check failed: {self._err()}")
                g.lds_ok, g.lds_bytes_tested = self.lds_selftest(d)
                if not g.lds_ok:
                    g.problems.append(f"LDS selftest failed: {self._err()}")
                g.sdma_bw_gbs = round(self.sdma_bandwidth(d), 1)
                if g.sdma_bw_gbs < MIN_SDMA_BW_GBS:
                    g.problems.append(
                        f"SDMA D2D bandwidth {g.sdma_bw_gbs} GB/s below floor {MIN_SDMA_BW_GBS}"
                    )
            except NodeAgentError as e:
                g.problems.append(str(e))
            g.healthy = not g.problems
            report.gpus.append(g)
        if report.gpu_count > 1:
            report.xgmi_p2p = self.p2p_matrix(report.gpu_count)
            for i in range(report.gpu_count):
                for j in range(report.gpu_count):
                    if i != j and not report.xgmi_p2p[i][j]:
                        report.problems.append(f"no xGMI peer access {i}->{j}")
        report.problems.extend(
            p for g in report.gpus for p in (f"gpu{g.index}: {q}" for q in g.problems)
        )
        report.healthy = not report.problems and report.gpu_count > 0
        return report


# -- node condition publishing (the repair feedback loop) --------------------
#
# The DaemonSet runs the check in a loop and publishes the result as the
# AMDGPUHealthy condition on its Node. The cloud provider's RepairPolicies
# include (AMDGPUHealthy, False, 5 min), so the node.health controller
# replaces nodes with failed HBM/MFMA/LDS/xGMI self-tests — on-node GPU
# evidence the reference's NodeReady-only repair can't see.


def node_condition_from_report(report: NodeReport) -> dict:
    from .apis import v1 as karpv1

    if report.healthy:
        return {
            "type": karpv1.AMD_GPU_HEALTHY_CONDITION_TYPE,
            "status": "True",
            "reason": "AllChecksPassed",
            "message": f"{report.gpu_count} GPU(s) passed HBM/FMA/MFMA/LDS/xGMI self-tests",
        }
    return {
        "type": karpv1.AMD_GPU_HEALTHY_CONDITION_TYPE,
        "status": "False",
        "reason": "GPUUnhealthy",
        "message": "; ".join(report.problems)[:1024] or "node agent found no GPUs",
    }


async def patch_node_condition(kube, node_name: str, report: NodeReport) -> None:
    """Merge the AMDGPUHealthy condition into the Node's status conditions.

    The status write carries the read's resourceVersion as an optimistic
    lock and retries on conflict: a merge patch of the full conditions list
    from an unconditioned GET races the kubelet's concurrent status writes
    and can clobber or resurrect kubelet-owned conditions (the same
    lost-update class TerminationController._set_nodeclaim_condition guards
    against)."""
    from .kube import objects as ko
    from .kube.client import ConflictError

    cond = node_condition_from_report(report)
    for _ in range(5):
        node = await kube.get("v1", "Node", node_name)
        ko.set_condition(node, cond["type"], cond["status"], cond["reason"], cond["message"])
        try:
            await kube.patch(
                "v1",
                "Node",
                node_name,
                {
                    "metadata": {
                        "resourceVersion": node.get("metadata", {}).get("resourceVersion")
                    },
                    "status": {"conditions": node["status"]["conditions"]},
                },
                subresource="status",
            )
            return
        except ConflictError:
            continue
    raise NodeAgentError(
        f"node {node_name}: AMDGPUHealthy condition patch abandoned after repeated conflicts"
    )


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(description="MI355X node health agent")
    ap.add_argument("--json", action="store_true", help="emit JSON report")
    ap.add_argument("--expect-gpus", type=int, default=0)
    ap.add_argument("--bw-bytes", type=int, default=1 << 30)
    ap.add_argument(
        "--patch-node",
        default="",
        metavar="NODE",
        help="publish the AMDGPUHealthy condition on this Node (in-cluster)",
    )
    args = ap.parse_args(argv)
    try:
        agent = NodeAgent()
        report = agent.check(expect_gpus=args.expect_gpus, bw_bytes=args.bw_bytes)
    except NodeAgentError as e:
        print(json.dumps({"healthy": False, "agent_error": str(e)}))
        return 2
    if args.patch_node:
        import asyncio

        from .kube.http import HTTPClient

        async def publish():
            kube = HTTPClient.from_service_account()
            try:
                await patch_node_condition(kube, args.patch_node, report)
            finally:
                await kube.close()

        try:
            asyncio.run(publish())
        except Exception as e:
            print(json.dumps({"healthy": report.healthy, "patch_error": str(e)}))
            return 2
    out = asdict(report)
    if args.json:
        print(json.dumps(out, indent=2))
    else:
        print(json.dumps(out))
    return 0 if report.healthy else 1


if __name__ == "__main__":
    sys.exit(main())
