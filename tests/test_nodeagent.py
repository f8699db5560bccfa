"""Node-agent tests. CPU: library builds/loads and the CLI handles the
no-GPU case loudly. GPU (@pytest.mark.gpu, run on MI355X via gpurun): device
discovery, HBM bandwidth floor, VALU + MFMA self-tests, full health report."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _ensure_lib():
    import __graft_entry__ as g

    if not os.path.exists(g.LIB_PATH):
        g.build()
    return g.LIB_PATH


def test_hip_library_builds_and_loads():
    """hipcc cross-compiles for gfx950 without a GPU; the .so must load and
    export the C ABI."""
    import ctypes

    lib_path = _ensure_lib()
    lib = ctypes.CDLL(lib_path)
    for sym in (
        "na_device_count",
        "na_device_info",
        "na_hbm_bandwidth",
        "na_fma_selftest",
        "na_mfma_selftest",
        "na_mfma_bf16_selftest",
        "na_mfma_fp8_selftest",
        "na_lds_selftest",
        "na_sdma_bandwidth",
        "na_mfma_bf16_tile_check",
        "na_mfma_fp8_tile_check",
        "na_mfma_i8_tile_check",
        "na_mfma_f16_tile_check",
        "na_mfma_mx_tile_check",
        "na_mfma_bf16_tile32_check",
        "na_p2p_matrix",
        "na_p2p_bandwidth",
        "na_last_error",
    ):
        assert hasattr(lib, sym), f"missing symbol {sym}"


def test_code_object_is_gfx950_only():
    """No multi-arch fatbin, no CUDA: the embedded code object targets gfx950."""
    lib_path = _ensure_lib()
    out = subprocess.run(
        [os.path.join("/opt/rocm/lib/llvm/bin", "llvm-objdump"), "--offloading", lib_path],
        capture_output=True,
        text=True,
    )
    listing = out.stdout + out.stderr
    assert "gfx950" in listing, listing


@pytest.mark.gpu
def test_gpu_device_discovery():
    from gpu_provisioner_amd.nodeagent import NodeAgent

    _ensure_lib()
    agent = NodeAgent()
    n = agent.device_count()
    assert n >= 1
    name, arch, hbm, cus = agent.device_info(0)
    assert "gfx950" in arch, f"unexpected arch {arch}"
    assert hbm > 200e9, f"HBM {hbm} bytes — expected ~288 GB"
    assert cus >= 250, f"CU count {cus} — expected 256"


@pytest.mark.gpu
def test_gpu_hbm_bandwidth_floor():
    from gpu_provisioner_amd.nodeagent import NodeAgent

    _ensure_lib()
    agent = NodeAgent()
    bw = agent.hbm_bandwidth(0, bytes_=1 << 30, iters=10)
    # contiguous-chunk nt copy measures ~5.7 TB/s on healthy MI355X
    assert bw > 4800.0, f"HBM bandwidth {bw:.0f} GB/s below healthy floor"


@pytest.mark.gpu
def test_gpu_fma_and_mfma_selftests():
    from gpu_provisioner_amd.nodeagent import NodeAgent

    _ensure_lib()
    agent = NodeAgent()
    assert agent.fma_selftest(0), "VALU FMA selftest failed"
    assert agent.mfma_selftest(0), "MFMA (v_mfma_f32_16x16x4_f32) selftest failed"


@pytest.mark.gpu
def test_gpu_mfma_datatype_paths():
    """The CDNA4 low-precision pipes ML workloads run on: bf16 (training)
    and fp8 E4M3 (serving) 16x16x32 MFMA forms."""
    from gpu_provisioner_amd.nodeagent import NodeAgent

    _ensure_lib()
    agent = NodeAgent()
    assert agent.mfma_bf16_selftest(0), "MFMA bf16 (v_mfma_f32_16x16x32_bf16) failed"
    assert agent.mfma_fp8_selftest(0), "MFMA fp8 (v_mfma_f32_16x16x32_fp8_fp8) failed"
    # asymmetric-data GEMM tile vs exact host reference: catches fragment
    # layout errors the uniform-operand checks cannot
    assert agent.mfma_bf16_tile_check(0), "MFMA bf16 tile (layout) check failed"
    assert agent.mfma_fp8_tile_check(0), "MFMA fp8 tile (layout) check failed"
    assert agent.mfma_i8_tile_check(0), "MFMA i8 tile (layout) check failed"
    assert agent.mfma_f16_tile_check(0), "MFMA f16 tile (layout) check failed"
    assert agent.mfma_mx_tile_check(0), "MFMA MX-scaled tile (layout+scale) check failed"
    assert agent.mfma_bf16_tile32_check(0), "MFMA 32x32 tile (layout) check failed"


@pytest.mark.gpu
def test_gpu_lds_selftest():
    from gpu_provisioner_amd.nodeagent import NodeAgent

    _ensure_lib()
    agent = NodeAgent()
    ok, tested = agent.lds_selftest(0)
    assert ok, "LDS selftest failed"
    assert tested >= 64 * 1024, f"only {tested} LDS bytes tested"


@pytest.mark.gpu
def test_gpu_sdma_bandwidth():
    """D2D copy through the SDMA engines (RCCL's xGMI transport hardware)."""
    from gpu_provisioner_amd.nodeagent import NodeAgent

    _ensure_lib()
    agent = NodeAgent()
    bw = agent.sdma_bandwidth(0)
    # measured ~4.9 TB/s healthy
    assert bw > 3000.0, f"SDMA D2D bandwidth {bw:.1f} GB/s below floor"


@pytest.mark.gpu
def test_gpu_full_health_report_cli():
    _ensure_lib()
    proc = subprocess.run(
        [sys.executable, "-m", "gpu_provisioner_amd.nodeagent", "--json",
         "--bw-bytes", str(1 << 28)],
        capture_output=True,
        text=True,
        cwd=ROOT,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    report = json.loads(proc.stdout)
    assert report["healthy"] is True
    assert report["gpu_count"] >= 1
    g0 = report["gpus"][0]
    assert g0["mfma_ok"] and g0["fma_ok"]
