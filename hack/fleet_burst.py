#!/usr/bin/env python3
"""Fleet-scale burst measurement: N concurrent NodeClaims through one full
provision+teardown cycle, reporting wall time, throughput and peak RSS.
Evidence artifact for profiles/ (the e2e suite carries the 2048 spec; this
script probes bigger fleets)."""
import argparse
import asyncio
import json
import sys
import time

sys.path.insert(0, ".")

from gpu_provisioner_amd.apis import v1 as karpv1  # noqa: E402
from gpu_provisioner_amd.fake.harness import Harness  # noqa: E402
from tests.e2e_env import spec_nodeclaim  # noqa: E402


async def burst(n: int) -> dict:
    import psutil

    h = Harness(node_wait_interval=0.01).add_all_controllers(
        lifecycle_workers=1024, termination_workers=512,
        termination_requeue=0.05, drain_requeue=0.05, instance_poll=0.05,
        gc_interval=300.0, with_health=False,
    )
    await h.start()
    try:
        names = [f"fleet{i:05d}" for i in range(n)]
        t0 = time.monotonic()
        await asyncio.gather(
            *(h.kube.create(spec_nodeclaim(x, {karpv1.KAITO_WORKSPACE_LABEL_KEY: "w"})) for x in names)
        )
        await asyncio.gather(*(h.wait_initialized(x, timeout=900) for x in names))
        t_up = time.monotonic() - t0
        rss = psutil.Process().memory_info().rss
        await asyncio.gather(
            *(h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, x) for x in names)
        )
        await asyncio.gather(
            *(h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, x, timeout=900) for x in names)
        )
        cycle = time.monotonic() - t0
        assert not h.agent_pools.pools
        return {
            "claims": n,
            "gpus_represented": n * 8,
            "provision_s": round(t_up, 2),
            "full_cycle_s": round(cycle, 2),
            "claims_per_min_cycle": round(n / cycle * 60.0, 1),
            "peak_rss_mb": round(rss / 1e6, 1),
        }
    finally:
        await h.stop()


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes", default="1024,2048,4096,8192")
    args = ap.parse_args()
    out = []
    for n in (int(x) for x in args.sizes.split(",")):
        out.append(asyncio.run(burst(n)))
        print(json.dumps(out[-1]), flush=True)


if __name__ == "__main__":
    main()
