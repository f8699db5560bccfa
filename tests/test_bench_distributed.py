"""The driver runs bench.py under torch.distributed.run with one rank per
GPU and aggregates across ranks over gloo. Exercise that exact multi-process
path on CPU (world_size=2, gloo, 127.0.0.1 rendezvous) so the distributed
aggregation (MAX of elapsed over ranks, latency all_gather, whole-job value)
is correct by construction before it ever reaches an 8-GPU node."""
import json
import os
import socket
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_bench_world_size_2_gloo():
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
            "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--concurrent", "2",
        ],
        capture_output=True,
        text=True,
        cwd=ROOT,
        timeout=300,
        env={**os.environ, "MASTER_ADDR": "127.0.0.1"},
    )
    assert proc.returncode == 0, proc.stdout[-2000:] + proc.stderr[-2000:]
    # rank 0 prints exactly one JSON line
    lines = [ln for ln in proc.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, proc.stdout
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["metric"] == "nodeclaims_per_min"
    assert d["value"] > 0
    assert d["scaling"] == "weak"
    # whole-job aggregate: 2 ranks x 2 steps x 2 concurrent claims
    assert d["config"]["parallelism"] == "dp2"
