"""Live-mode rehearsal: LiveEnv's full machinery exercised WITHOUT a real
cluster by pointing its kubeconfig at the envtest-lite REST server.

The dual-backend e2e suite's live path (tests/e2e_env.py LiveEnv:
kubeconfig client, sanity check, cluster monitor, Eventually with
controller-log dumping, crash detection via container restart counts,
labeled 50-way cleanup) otherwise only runs when a real cluster exists —
an untested claim. Here every line of it runs against the in-memory
apiserver over real HTTP, with the controller manager in-process and the
AKS simulator as the cloud/kubelet actor."""
import asyncio

import pytest
import yaml

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.cloudprovider.azure import AzureCloudProvider
from gpu_provisioner_amd.fake.agentpools import AKSSimulator, FakeAgentPools
from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.fake.harness import install_chart_crd_validators
from gpu_provisioner_amd.fake.restserver import RESTServerHandle
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.kube.http import HTTPClient
from gpu_provisioner_amd.main import build_manager
from gpu_provisioner_amd.operator.options import Options
from gpu_provisioner_amd.providers.instance.provider import InstanceProvider
from gpu_provisioner_amd.providers.instancetype.catalog import InstanceTypeProvider
from tests.conftest import run


def _write_kubeconfig(tmp_path, port: int) -> str:
    cfg = {
        "apiVersion": "v1",
        "kind": "Config",
        "current-context": "rehearsal",
        "contexts": [{"name": "rehearsal", "context": {"cluster": "c", "user": "u"}}],
        "clusters": [{"name": "c", "cluster": {"server": f"http://127.0.0.1:{port}"}}],
        "users": [{"name": "u", "user": {"token": "rehearsal-token"}}],
    }
    p = tmp_path / "kubeconfig"
    p.write_text(yaml.safe_dump(cfg))
    return str(p)


def test_live_env_full_rehearsal(tmp_path, monkeypatch):
    async def main():
        server = InMemoryAPIServer()
        install_chart_crd_validators(server)
        rest = RESTServerHandle(server)
        port = await rest.start()

        # in-process controller stack over its own HTTP client
        ctrl_kube = HTTPClient(f"http://127.0.0.1:{port}")
        actor = InMemoryClient(server)
        catalog = InstanceTypeProvider()
        pools = FakeAgentPools()
        aks = AKSSimulator(
            actor, pools, ready_latency=0.05, plugin_latency=0.05,
            gpu_count_for=catalog.gpu_count,
        )
        instances = InstanceProvider(
            pools, ctrl_kube, catalog, "rg", "cluster", node_wait_interval=0.02
        )
        manager = build_manager(ctrl_kube, Options(), AzureCloudProvider(instances, catalog))
        await manager.start(serve_http=False)

        # a "controller pod" in the controller namespace so diagnostics and
        # crash detection have something to inspect
        await actor.create(
            {
                "apiVersion": "v1",
                "kind": "Pod",
                "metadata": {"name": "gpu-provisioner-amd-0", "namespace": "gpu-provisioner"},
                "spec": {"containers": [{"name": "controller"}]},
                "status": {
                    "phase": "Running",
                    "containerStatuses": [{"name": "controller", "restartCount": 0}],
                },
            }
        )

        monkeypatch.setenv("E2E_LIVE", "1")
        monkeypatch.setenv("KUBECONFIG", _write_kubeconfig(tmp_path, port))
        monkeypatch.setenv("E2E_TIMEOUT_SECONDS", "30")
        monkeypatch.setenv("E2E_CONTROLLER_NAMESPACE", "gpu-provisioner")
        from tests.e2e_env import EventuallyTimeout, LiveEnv, make_env

        env = make_env()
        assert isinstance(env, LiveEnv) and env.is_live
        await env.start()
        try:
            # spec1 body: provision via workspace label through the live surface
            await env.create(env.nodeclaim("live1", {karpv1.KAITO_WORKSPACE_LABEL_KEY: "w"}))
            nc = await env.wait_initialized("live1")
            node = await env.kube.get("v1", "Node", nc["status"]["nodeName"])
            assert ko.node_is_ready(node)
            # monitor sampled a live cluster state
            await asyncio.sleep(0.1)
            assert env.monitor.baseline is not None
            assert "nodeclaims" in env.monitor.report()
            # pool assertions are declared unobservable on live
            assert env.pool_exists("live1") is None
            await env.expect_pool_gone("live1")  # no-op on live
            # diagnostics path: controller pod discovered, log fetched
            restarts = await env._controller_restarts()
            assert restarts == {"gpu-provisioner-amd-0/controller": 0}
            await env.dump_diagnostics("rehearsal")  # prints pod log via /log
            # Eventually timeout path raises after dumping diagnostics
            with pytest.raises(EventuallyTimeout):
                await env.eventually(lambda: _never(), timeout=0.3, desc="never")
        finally:
            # stop() performs labeled cleanup: live1 must be deleted and
            # fully torn down (finalizer → drain → pool delete) by the
            # in-process controllers
            await env.stop()
        claims = await ctrl_kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
        assert claims == []
        assert "live1" not in pools.pools

        await manager.stop()
        await ctrl_kube.close()
        await rest.stop()

    async def _never():
        return None

    run(main(), timeout=120)


def test_live_env_crash_detection_fails_suite(tmp_path, monkeypatch):
    """A controller container restart during the run must fail env.stop()
    (reference expectation.go:364 crash detection)."""

    async def main():
        server = InMemoryAPIServer()
        rest = RESTServerHandle(server)
        port = await rest.start()
        actor = InMemoryClient(server)
        # required CRD presence for LiveEnv.start sanity list
        await actor.create(karpv1.new_nodeclaim("seed", labels={}))
        await actor.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "seed")
        pod = {
            "apiVersion": "v1",
            "kind": "Pod",
            "metadata": {"name": "ctrl-0", "namespace": "gpu-provisioner"},
            "spec": {"containers": [{"name": "controller"}]},
            "status": {"containerStatuses": [{"name": "controller", "restartCount": 0}]},
        }
        await actor.create(pod)

        monkeypatch.setenv("E2E_LIVE", "1")
        monkeypatch.setenv("KUBECONFIG", _write_kubeconfig(tmp_path, port))
        monkeypatch.setenv("E2E_TIMEOUT_SECONDS", "5")
        monkeypatch.setenv("E2E_CONTROLLER_NAMESPACE", "gpu-provisioner")
        from tests.e2e_env import make_env

        env = make_env()
        await env.start()
        # simulate a controller crash-restart mid-suite
        fresh = await actor.get("v1", "Pod", "ctrl-0", "gpu-provisioner")
        fresh["status"]["containerStatuses"][0]["restartCount"] = 2
        await actor.update_status(fresh)
        with pytest.raises(AssertionError, match="restarted"):
            await env.stop()
        await rest.stop()

    run(main(), timeout=60)
