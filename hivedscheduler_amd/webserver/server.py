"""Scheduler-extender HTTP server (L5).

Parity with reference pkg/webserver/webserver.go: routes (l.77-84)
/, /v1/extender/{filter,bind,preempt}, /v1/inspect/{affinitygroups,
clusterstatus[,physicalcluster,virtualclusters]}, JSON decode/validate,
error->HTTP-status translation (l.142-155). Adds /metrics (Prometheus) and
/healthz, which the reference lacks.

The K8s default scheduler POSTs the extender verbs per the policy config
(filterVerb/bindVerb/preemptVerb; examples/deploy/).
"""
from __future__ import annotations

import logging
from typing import Any, Dict, Optional

from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse, PlainTextResponse
from starlette.concurrency import run_in_threadpool

from ..api import constants
from ..api.types import WebServerError
from ..scheduler import HivedScheduler

log = logging.getLogger("hivedscheduler.webserver")

# Scheduler calls run in the threadpool, never on the event loop: they take
# the scheduler lock, and a waiting pod's FIFO block (or a slow filter) would
# otherwise stall every concurrent request — /healthz, health intake, probe
# polling. The reference's Go server serves each request on its own goroutine
# (webserver.go:93-155); run_in_threadpool is the asyncio equivalent.


def create_app(scheduler: HivedScheduler) -> FastAPI:
    app = FastAPI(title="hivedscheduler-amd", docs_url=None, redoc_url=None)

    @app.exception_handler(WebServerError)
    async def _hived_error(request: Request, exc: WebServerError):
        return JSONResponse(status_code=exc.code, content={"error": exc.message})

    @app.get(constants.RootPath)
    async def root():
        return {"component": constants.ComponentName, "paths": [
            constants.FilterPath, constants.BindPath, constants.PreemptPath,
            constants.AffinityGroupsPath, constants.ClusterStatusPath,
            constants.PhysicalClusterPath, constants.VirtualClustersPath,
            constants.MetricsPath,
        ]}

    @app.get("/healthz")
    async def healthz():
        return PlainTextResponse("ok")

    # ---- extender verbs ----
    async def _json_body(request: Request) -> Dict[str, Any]:
        # raw json parse: pydantic body validation costs ~1 ms per call on
        # the filter hot path and the extender args are plain dicts anyway
        try:
            args = await request.json()
        except Exception:
            raise WebServerError.bad_request("request body is not valid JSON")
        if not isinstance(args, dict):
            raise WebServerError.bad_request("request body must be a JSON object")
        return args

    @app.post(constants.FilterPath)
    async def filter_verb(request: Request):
        args = await _json_body(request)
        if not isinstance(args.get("Pod"), dict):
            raise WebServerError.bad_request("ExtenderArgs.Pod is missing")
        return JSONResponse(await run_in_threadpool(scheduler.filter, args))

    @app.post(constants.BindPath)
    async def bind_verb(request: Request):
        args = await _json_body(request)
        for field in ("PodName", "PodNamespace", "PodUID", "Node"):
            if not args.get(field):
                raise WebServerError.bad_request(f"ExtenderBindingArgs.{field} is missing")
        try:
            return await run_in_threadpool(scheduler.bind, args)
        except WebServerError as e:
            # binding errors are returned in-band so the default scheduler
            # surfaces them on the pod (reference webserver.go:194-215)
            return {"Error": e.message}

    @app.post(constants.PreemptPath)
    async def preempt_verb(request: Request):
        args = await _json_body(request)
        if not isinstance(args.get("Pod"), dict):
            raise WebServerError.bad_request("ExtenderPreemptionArgs.Pod is missing")
        return JSONResponse(await run_in_threadpool(scheduler.preempt, args))

    # ---- inspect API ----
    @app.get(constants.AffinityGroupsPath)
    async def affinity_groups():
        return await run_in_threadpool(scheduler.get_all_affinity_groups)

    @app.get(constants.AffinityGroupsPath + "{name:path}")
    async def affinity_group(name: str):
        return await run_in_threadpool(scheduler.get_affinity_group, name)

    @app.get(constants.ClusterStatusPath)
    async def cluster_status():
        return await run_in_threadpool(scheduler.get_cluster_status)

    @app.get(constants.PhysicalClusterPath)
    async def physical_cluster():
        return await run_in_threadpool(scheduler.get_physical_cluster_status)

    @app.get(constants.VirtualClustersPath)
    async def virtual_clusters():
        return await run_in_threadpool(scheduler.get_all_virtual_clusters_status)

    @app.get(constants.VirtualClustersPath + "{vc}")
    async def virtual_cluster(vc: str):
        return await run_in_threadpool(scheduler.get_virtual_cluster_status, vc)

    # ---- GPU-level health intake (node agents) ----
    @app.post("/v1/health/nodes/{node}")
    async def health_report(node: str, report: Dict[str, Any]):
        return await run_in_threadpool(scheduler.on_health_report, node, report)

    @app.get("/v1/inspect/health")
    async def health_reports():
        return await run_in_threadpool(scheduler.get_health_reports)

    @app.get("/v1/inspect/links/{node}")
    async def xgmi_links(node: str):
        # first-class xGMI link state: [{a, b, gbps, healthy}, ...]
        return await run_in_threadpool(scheduler.get_xgmi_links, node)

    # ---- post-bind placement probes (agents poll tasks, post results) ----
    @app.get("/v1/health/probes/{node}")
    async def probe_tasks(node: str):
        return await run_in_threadpool(scheduler.probe_manager.poll, node)

    @app.post("/v1/health/probes")
    async def probe_result(result: Dict[str, Any]):
        return await run_in_threadpool(scheduler.on_probe_result, result)

    @app.get("/v1/inspect/probes/{group:path}")
    async def probe_results(group: str):
        return await run_in_threadpool(scheduler.probe_manager.group_results, group)

    # ---- metrics ----
    @app.get(constants.MetricsPath)
    async def metrics():
        try:
            from prometheus_client import CONTENT_TYPE_LATEST, generate_latest

            return Response(content=generate_latest(), media_type=CONTENT_TYPE_LATEST)
        except ImportError:  # pragma: no cover
            return PlainTextResponse("prometheus_client not available", status_code=501)

    return app


def run_server(scheduler: HivedScheduler, address: Optional[str] = None) -> None:
    """Blocking uvicorn server on config.webServerAddress (":9096" style)."""
    import uvicorn

    addr = address or scheduler.config.webServerAddress or constants.DefaultWebServerAddress
    host, _, port = addr.rpartition(":")
    host = host or "0.0.0.0"
    uvicorn.run(create_app(scheduler), host=host, port=int(port), log_level="info")
