"""HTTP KubeClient tests against a mocked apiserver: REST paths, error
mapping, watch stream decoding (incl. 410 Gone → informer relist), eviction
subresource, and the manager's probe endpoints via ASGI."""
import json

import httpx
import pytest

from gpu_provisioner_amd.kube.client import (
    ConflictError,
    GoneError,
    NotFoundError,
    TooManyRequestsError,
)
from gpu_provisioner_amd.kube.http import HTTPClient, plural_of
from tests.conftest import run


def make_client(handler) -> HTTPClient:
    c = HTTPClient("https://kube.example", token="tok")
    c.http = httpx.AsyncClient(
        base_url="https://kube.example",
        transport=httpx.MockTransport(handler),
        headers={"Authorization": "Bearer tok"},
    )
    return c


def test_plurals():
    assert plural_of("NodeClaim") == "nodeclaims"
    assert plural_of("KaitoNodeClass") == "kaitonodeclasses"
    assert plural_of("Node") == "nodes"
    assert plural_of("SomethingElse") == "somethingelses"


def test_rest_paths_and_verbs():
    seen = []

    def handler(request: httpx.Request) -> httpx.Response:
        seen.append((request.method, request.url.path, request.url.params.get("labelSelector")))
        if request.method == "DELETE":
            return httpx.Response(200, json={"status": "Success"})
        return httpx.Response(
            200, json={"items": [], "metadata": {"resourceVersion": "5"}}
            if "watch" not in str(request.url) and request.method == "GET" and not request.url.path.endswith("n1")
            else {"metadata": {"name": "n1", "resourceVersion": "5"}},
        )

    async def main():
        c = make_client(handler)
        await c.get("karpenter.sh/v1", "NodeClaim", "n1")
        await c.list("v1", "Node", label_selector="agentpool=x")
        await c.list("v1", "Pod", namespace="default")
        await c.patch("v1", "Node", "n1", {"metadata": {}})
        await c.patch("karpenter.sh/v1", "NodeClaim", "n1", {"status": {}}, subresource="status")
        await c.delete("karpenter.sh/v1", "NodeClaim", "n1")
        assert seen[0] == ("GET", "/apis/karpenter.sh/v1/nodeclaims/n1", None)
        assert seen[1] == ("GET", "/api/v1/nodes", "agentpool=x")
        assert seen[2] == ("GET", "/api/v1/namespaces/default/pods", None)
        assert seen[3][:2] == ("PATCH", "/api/v1/nodes/n1")
        assert seen[4][:2] == ("PATCH", "/apis/karpenter.sh/v1/nodeclaims/n1/status")
        assert seen[5][:2] == ("DELETE", "/apis/karpenter.sh/v1/nodeclaims/n1")

    run(main())


def test_error_mapping():
    codes = iter(
        [
            (404, "NotFound"),
            (409, "Conflict"),
            (410, "Expired"),
            (429, "TooManyRequests"),
        ]
    )

    def handler(request: httpx.Request) -> httpx.Response:
        code, reason = next(codes)
        return httpx.Response(code, json={"reason": reason, "message": reason})

    async def main():
        c = make_client(handler)
        with pytest.raises(NotFoundError):
            await c.get("v1", "Node", "x")
        with pytest.raises(ConflictError):
            await c.get("v1", "Node", "x")
        with pytest.raises(GoneError):
            await c.get("v1", "Node", "x")
        with pytest.raises(TooManyRequestsError):
            await c.get("v1", "Node", "x")

    run(main())


def test_watch_stream_decoding_and_gone():
    lines = [
        {"type": "ADDED", "object": {"metadata": {"name": "a", "resourceVersion": "1"}}},
        {"type": "BOOKMARK", "object": {"metadata": {"resourceVersion": "2"}}},
        {"type": "MODIFIED", "object": {"metadata": {"name": "a", "resourceVersion": "3"}}},
        {"type": "ERROR", "object": {"code": 410, "message": "too old"}},
    ]

    def handler(request: httpx.Request) -> httpx.Response:
        assert request.url.params["watch"] == "true"
        assert request.url.params["resourceVersion"] == "7"
        body = "\n".join(json.dumps(l) for l in lines) + "\n"
        return httpx.Response(200, content=body.encode())

    async def main():
        c = make_client(handler)
        got = []
        with pytest.raises(GoneError):
            async for etype, obj in c.watch("v1", "Node", resource_version="7"):
                got.append((etype, obj["metadata"].get("resourceVersion")))
        # bookmarks are SURFACED (the informer advances its resume rv from
        # them) — they are not object events, just rv carriers
        assert got == [("ADDED", "1"), ("BOOKMARK", "2"), ("MODIFIED", "3")]

    run(main())


def test_eviction_subresource():
    def handler(request: httpx.Request) -> httpx.Response:
        assert request.url.path == "/api/v1/namespaces/ns/pods/p1/eviction"
        body = json.loads(request.content)
        assert body["kind"] == "Eviction"
        assert body["deleteOptions"]["gracePeriodSeconds"] == 30
        return httpx.Response(201, json={})

    async def main():
        c = make_client(handler)
        pod = {"metadata": {"name": "p1", "namespace": "ns"}}
        await c.evict(pod, 30)

    run(main())


def test_list_continues_pagination():
    def handler(request: httpx.Request) -> httpx.Response:
        cont = request.url.params.get("continue")
        if cont == "c1":
            return httpx.Response(
                200,
                json={"items": [{"metadata": {"name": "b"}}], "metadata": {"resourceVersion": "9"}},
            )
        return httpx.Response(
            200,
            json={
                "items": [{"metadata": {"name": "a"}}],
                "metadata": {"resourceVersion": "8", "continue": "c1"},
            },
        )

    async def main():
        c = make_client(handler)
        items, rv = await c.list_with_rv("v1", "Node")
        assert [i["metadata"]["name"] for i in items] == ["a", "b"]
        assert rv == "9"
        assert all(i["kind"] == "Node" for i in items)

    run(main())


def test_manager_probes_and_metrics_endpoints():
    """Exercise the manager's starlette apps in-process via ASGI transport."""
    from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
    from gpu_provisioner_amd.kube.informer import InformerFactory
    from gpu_provisioner_amd.operator.manager import Manager
    from gpu_provisioner_amd.operator.options import Options

    async def main():
        server = InMemoryAPIServer()
        kube = InMemoryClient(server)
        informers = InformerFactory(kube)
        inf = informers.informer("v1", "Node")
        opts = Options()
        opts.enable_profiling = True
        mgr = Manager(kube, opts, informers, required_crds=(("v1", "Node"),))
        # before sync: readyz must 503
        probes = httpx.AsyncClient(
            transport=httpx.ASGITransport(app=mgr._probes_app()), base_url="http://t"
        )
        resp = await probes.get("/readyz")
        assert resp.status_code == 503
        resp = await probes.get("/healthz")
        assert resp.status_code == 200
        # after sync: ready
        informers.start_all()
        await informers.wait_for_sync()
        resp = await probes.get("/readyz")
        assert resp.status_code == 200, resp.text
        # metrics endpoint serves prometheus text + debug endpoints exist
        metrics = httpx.AsyncClient(
            transport=httpx.ASGITransport(app=mgr._metrics_app()), base_url="http://t"
        )
        resp = await metrics.get("/metrics")
        assert resp.status_code == 200
        assert b"controller_runtime_reconcile" in resp.content
        resp = await metrics.get("/debug/tasks")
        assert resp.status_code == 200
        await informers.stop_all()

    run(main())


def test_profiling_debug_endpoints():
    """ENABLE_PROFILING exposes pprof-equivalent debug routes on the
    metrics app (reference operator.go:181-197)."""
    from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
    from gpu_provisioner_amd.operator.manager import Manager
    from gpu_provisioner_amd.operator.options import Options

    async def main():
        kube = InMemoryClient(InMemoryAPIServer())
        opts = Options.from_env_and_args(["--enable-profiling"], {})
        mgr = Manager(kube, opts)
        app = httpx.AsyncClient(
            transport=httpx.ASGITransport(app=mgr._metrics_app()), base_url="http://t"
        )
        resp = await app.get("/metrics")
        assert resp.status_code == 200 and b"karpenter" in resp.content
        resp = await app.get("/debug/pprof/goroutine")
        assert resp.status_code == 200 and resp.json()  # per-thread stacks
        resp = await app.get("/debug/tasks")
        assert resp.status_code == 200
        # heap/allocs (tracemalloc) — 503 until tracing starts, live after
        import tracemalloc

        was_tracing = tracemalloc.is_tracing()
        if not was_tracing:
            resp = await app.get("/debug/pprof/heap")
            assert resp.status_code == 503
            tracemalloc.start(5)
        try:
            resp = await app.get("/debug/pprof/heap?n=5")
            body = resp.json()
            assert resp.status_code == 200
            assert body["traced_current_bytes"] > 0
            assert len(body["top"]) <= 5 and body["top"][0]["size_bytes"] > 0
            resp = await app.get("/debug/pprof/allocs")
            assert resp.status_code == 200
        finally:
            if not was_tracing:
                tracemalloc.stop()
        # cpu profile + loop-lag (block) over a short window
        resp = await app.get("/debug/pprof/profile?seconds=0.2")
        assert resp.status_code == 200 and b"cumulative" in resp.content
        resp = await app.get("/debug/pprof/block?seconds=0.2")
        lag = resp.json()["loop_lag_ms"]
        assert resp.status_code == 200 and lag["max"] >= 0
        # without the flag the debug routes are absent
        mgr2 = Manager(kube, Options.from_env_and_args([], {}))
        app2 = httpx.AsyncClient(
            transport=httpx.ASGITransport(app=mgr2._metrics_app()), base_url="http://t"
        )
        resp = await app2.get("/debug/tasks")
        assert resp.status_code == 404
        await app.aclose()
        await app2.aclose()

    run(main())
