"""Manager: the operator runtime — servers, informers, controllers, leases.

Replaces controller-runtime's Manager as the reference wires it
(vendor/sigs.k8s.io/karpenter/pkg/operator/operator.go:155-248): metrics
server on :8080 (prometheus + optional profiling endpoints), health probes on
:8081 (healthz; readyz = informer caches synced + required CRDs present),
informer start/sync, controller start, optional Lease leader election.
"""
from __future__ import annotations

import asyncio
import logging
import sys
import traceback
from typing import Optional

from prometheus_client import generate_latest, CONTENT_TYPE_LATEST
from starlette.applications import Starlette
from starlette.responses import JSONResponse, PlainTextResponse, Response
from starlette.routing import Route
import uvicorn

from ..kube.client import KubeClient, NotFoundError
from ..kube.informer import InformerFactory
from ..metrics.registry import BUILD_INFO
from .gcpacer import GCPacer
from .leaderelection import LeaderElector
from .options import Options

log = logging.getLogger(__name__)


class Manager:
    def __init__(
        self,
        client: KubeClient,
        options: Options,
        informers: Optional[InformerFactory] = None,
        required_crds: tuple = (),
        version: str = "0.1.0",
        lease_duration: Optional[float] = None,
        renew_interval: Optional[float] = None,
    ):
        self.client = client
        self.options = options
        self.informers = informers or InformerFactory(client)
        self.required_crds = required_crds  # [(api_version, kind)] probed via list
        self.version = version
        self.lease_duration = lease_duration
        self.renew_interval = renew_interval
        self.controllers: list = []
        # bounded-tail cyclic GC (see operator/gcpacer.py); opt out via options
        self.gc_pacer: Optional[GCPacer] = (
            GCPacer() if getattr(options, "gc_pacer", True) else None
        )
        self._elector: Optional[LeaderElector] = None
        self._server_tasks: list = []
        self._elector_task: Optional[asyncio.Task] = None
        self._started = asyncio.Event()

    def register(self, *controllers) -> "Manager":
        self.controllers.extend(controllers)
        return self

    # ----------------------------------------------------------- http apps

    def _metrics_app(self) -> Starlette:
        async def metrics(request):
            return Response(generate_latest(), media_type=CONTENT_TYPE_LATEST)

        routes = [Route("/metrics", metrics)]
        if self.options.enable_profiling:
            # pprof-equivalent debug endpoints. The reference exposes
            # heap/block/goroutine/profile/trace via net/http/pprof
            # (vendor/.../operator/operator.go:181-197); the Python
            # analogues: tracemalloc (heap/allocs), per-thread stacks
            # (goroutine), cProfile over a window (profile), task dump
            # (trace-ish) and an event-loop lag sampler (block).
            async def stacks(request):
                frames = sys._current_frames()
                out = {}
                for tid, frame in frames.items():
                    out[str(tid)] = "".join(traceback.format_stack(frame))
                return JSONResponse(out)

            async def tasks(request):
                return JSONResponse(
                    [repr(t) for t in asyncio.all_tasks()], status_code=200
                )

            async def heap(request):
                import tracemalloc

                if not tracemalloc.is_tracing():
                    return PlainTextResponse(
                        "tracemalloc not tracing (profiling just enabled?)",
                        status_code=503,
                    )
                limit = int(request.query_params.get("n", "50"))
                snap = tracemalloc.take_snapshot()
                stats = snap.statistics(
                    "traceback" if request.query_params.get("traceback") else "lineno"
                )
                current, peak = tracemalloc.get_traced_memory()
                return JSONResponse(
                    {
                        "traced_current_bytes": current,
                        "traced_peak_bytes": peak,
                        "top": [
                            {
                                "site": str(s.traceback),
                                "size_bytes": s.size,
                                "count": s.count,
                            }
                            for s in stats[:limit]
                        ],
                    }
                )

            async def profile(request):
                """cProfile the process for ?seconds=N (default 5) and
                return the top cumulative entries as text."""
                import cProfile
                import io
                import pstats

                seconds = min(float(request.query_params.get("seconds", "5")), 60.0)
                prof = cProfile.Profile()
                prof.enable()
                await asyncio.sleep(seconds)
                prof.disable()
                buf = io.StringIO()
                pstats.Stats(prof, stream=buf).sort_stats("cumulative").print_stats(60)
                return PlainTextResponse(buf.getvalue())

            async def block(request):
                """Event-loop lag sampled over ?seconds=N — the asyncio
                analogue of the block profile: how long ready callbacks
                wait for the loop."""
                seconds = min(float(request.query_params.get("seconds", "5")), 60.0)
                import time as _time

                samples: list = []
                deadline = _time.monotonic() + seconds
                while _time.monotonic() < deadline:
                    t0 = _time.monotonic()
                    await asyncio.sleep(0.01)
                    samples.append(max(0.0, _time.monotonic() - t0 - 0.01))
                samples.sort()

                def pct(p):
                    return round(samples[min(len(samples) - 1, int(len(samples) * p))] * 1000, 3)

                return JSONResponse(
                    {
                        "samples": len(samples),
                        "loop_lag_ms": {
                            "p50": pct(0.50),
                            "p90": pct(0.90),
                            "p99": pct(0.99),
                            "max": round(samples[-1] * 1000, 3) if samples else 0,
                        },
                    }
                )

            routes += [
                Route("/debug/pprof/goroutine", stacks),
                Route("/debug/pprof/heap", heap),
                Route("/debug/pprof/allocs", heap),
                Route("/debug/pprof/profile", profile),
                Route("/debug/pprof/block", block),
                Route("/debug/tasks", tasks),
            ]
        return Starlette(routes=routes)

    def _probes_app(self) -> Starlette:
        async def healthz(request):
            return PlainTextResponse("ok")

        async def readyz(request):
            # cache sync check
            for key, inf in self.informers._informers.items():
                if not inf.has_synced:
                    return PlainTextResponse(f"informer {key} not synced", status_code=503)
            # CRD presence check (reference operator.go:203-221, NodeClaim only)
            for api_version, kind in self.required_crds:
                try:
                    await self.client.list(api_version, kind)
                except NotFoundError:
                    return PlainTextResponse(f"CRD {kind} absent", status_code=503)
                except Exception as e:
                    return PlainTextResponse(f"CRD check {kind}: {e}", status_code=503)
            return PlainTextResponse("ok")

        return Starlette(routes=[Route("/healthz", healthz), Route("/readyz", readyz)])

    async def _serve(self, app, port: int) -> None:
        config = uvicorn.Config(app, host="0.0.0.0", port=port, log_level="warning", lifespan="off")
        server = uvicorn.Server(config)
        await server.serve()

    # ------------------------------------------------------------- lifecycle

    async def start(self, serve_http: bool = True) -> None:
        BUILD_INFO.labels(version=self.version).set(1)
        if self.options.enable_profiling:
            # heap endpoint needs allocation tracing from process start-ish;
            # 10 frames keeps per-alloc overhead modest while still
            # attributing sites usefully
            import tracemalloc

            if not tracemalloc.is_tracing():
                tracemalloc.start(10)
        if serve_http:
            self._server_tasks = [
                asyncio.create_task(
                    self._serve(self._metrics_app(), self.options.metrics_port),
                    name="metrics-server",
                ),
                asyncio.create_task(
                    self._serve(self._probes_app(), self.options.health_probe_port),
                    name="probes-server",
                ),
            ]
        self.informers.start_all()
        await self.informers.wait_for_sync()
        if self.options.leader_elect:
            kwargs = {}
            if self.lease_duration is not None:
                kwargs["lease_duration"] = self.lease_duration
            if self.renew_interval is not None:
                kwargs["renew_interval"] = self.renew_interval
            self._elector = LeaderElector(
                self.client,
                self.options.leader_election_name,
                self.options.leader_election_namespace,
                **kwargs,
            )
            self._elector_task = asyncio.create_task(
                self._elector.run(self._start_controllers, self._stop_controllers),
                name="leader-elector",
            )
        else:
            await self._start_controllers()
        if self.gc_pacer is not None:
            # engage at steady state: caches synced, controllers built —
            # that heap is what gets frozen out of future full collections
            self.gc_pacer.engage()
        self._started.set()

    async def _start_controllers(self) -> None:
        for c in self.controllers:
            c.controller.start()
        log.info("started %d controllers", len(self.controllers))

    async def _stop_controllers(self) -> None:
        for c in self.controllers:
            await c.controller.stop()

    async def run_forever(self) -> None:
        """Run until SIGTERM/SIGINT (what kubelet sends on pod stop), then
        shut down gracefully: controllers stop, informers stop, and the
        leader lease is RELEASED so a replica takes over immediately
        instead of waiting out the lease duration."""
        import signal

        stop_event = asyncio.Event()
        loop = asyncio.get_running_loop()
        for sig in (signal.SIGTERM, signal.SIGINT):
            try:
                loop.add_signal_handler(sig, stop_event.set)
            except (NotImplementedError, RuntimeError):
                pass  # non-unix / nested-loop environments
        await self.start()
        await stop_event.wait()
        log.info("shutdown signal received; stopping")
        await self.stop()

    async def stop(self) -> None:
        if self.gc_pacer is not None:
            await self.gc_pacer.disengage()
        if self._elector is not None:
            await self._elector.release()
        if self._elector_task is not None:
            self._elector_task.cancel()
            try:
                await self._elector_task
            except (asyncio.CancelledError, Exception):
                pass
            self._elector_task = None
        await self._stop_controllers()
        await self.informers.stop_all()
        for t in self._server_tasks:
            t.cancel()
