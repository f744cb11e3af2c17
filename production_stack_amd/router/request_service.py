"""Request proxying: streaming, stats hooks, failover, PD orchestration.

Behavioural parity with reference services/request_service/request.py:
  route_general_request (:384-691) incl. model filtering, sleep filtering,
  404-vs-503 (:547-566), failover loop (:611-674); process_request streaming
  generator with TTFT capture (:225-381); orchestrated disaggregated prefill
  (:733-935); sleep/wake passthrough (:1041-1128).
"""

from __future__ import annotations

import json
import logging
import time
import uuid
from typing import Any, Dict, List, Optional

import aiohttp
from fastapi import Request
from fastapi.responses import JSONResponse, Response, StreamingResponse

from production_stack_amd.router.service_discovery import (
    EndpointInfo,
    get_service_discovery,
)
from production_stack_amd.router.stats import (
    get_engine_stats_scraper,
    get_request_stats_monitor,
)

logger = logging.getLogger("router.request")

HOP_BY_HOP = {
    "host",
    "content-length",
    "connection",
    "keep-alive",
    "transfer-encoding",
    "upgrade",
    "proxy-authenticate",
    "proxy-authorization",
    "te",
    "trailers",
}


def _forward_headers(request: Request) -> Dict[str, str]:
    from production_stack_amd.router import tracing

    headers = {
        k: v for k, v in request.headers.items() if k.lower() not in HOP_BY_HOP
    }
    if tracing.is_enabled():
        parent = tracing.extract_context(request.headers)
        headers = tracing.inject_context(headers, parent)
    return headers


def _session() -> aiohttp.ClientSession:
    from production_stack_amd.router.app import get_http_session

    return get_http_session()


async def process_request(
    request: Request,
    body: bytes,
    backend_url: str,
    endpoint: str,
    request_id: str,
    model: Optional[str] = None,
):
    """Stream a request to a backend, yielding (first) status+headers then
    content chunks. Drives the request-stats lifecycle hooks."""
    from production_stack_amd.router import metrics as rm

    monitor = get_request_stats_monitor()
    start = time.time()
    monitor.on_new_request(backend_url, request_id, start)
    session = _session()
    first = True
    async with session.request(
        request.method,
        backend_url + endpoint,
        data=body,
        headers=_forward_headers(request),
        timeout=aiohttp.ClientTimeout(total=None),
    ) as backend_response:
        yield backend_response.status, backend_response.headers
        async for chunk in backend_response.content.iter_any():
            if first:
                monitor.on_request_response(
                    backend_url, request_id, time.time()
                )
                first = False
            yield chunk
    end = time.time()
    monitor.on_request_complete(backend_url, request_id, end)
    rm.request_latency_hist.labels(model=model or "unknown").observe(
        end - start
    )


def filter_endpoints(
    endpoints: List[EndpointInfo],
    model: Optional[str],
    aliases: Optional[Dict[str, str]] = None,
) -> List[EndpointInfo]:
    if aliases and model in aliases:
        model = aliases[model]
    out = []
    for ep in endpoints:
        if ep.sleep:
            continue
        if model is None or not ep.model_names or model in ep.model_names:
            out.append(ep)
    return out


async def route_general_request(
    request: Request, endpoint: str, background_tasks=None
) -> Response:
    """The hot path: parse, route, stream, failover."""
    in_router_time = time.time()
    request_id = request.headers.get("x-request-id") or str(uuid.uuid4())
    body = await request.body()
    content_type = request.headers.get("content-type", "")
    if content_type.startswith("multipart/form-data"):
        # audio/file endpoints: body passes through verbatim (the
        # boundary lives in the forwarded content-type header); routing
        # fields come from the form (reference
        # proxy_multipart_request, request.py:1230-1420)
        from production_stack_amd.router.app import parse_multipart

        parts = parse_multipart(body, content_type)
        is_multipart = True
        request_json = {
            name: payload[1].decode("utf-8", "replace")
            for name, payload in parts.items()
            if payload[0] is None  # plain form fields only
        }
    else:
        is_multipart = False
        try:
            request_json = json.loads(body) if body else {}
        except json.JSONDecodeError:
            return JSONResponse(
                status_code=400,
                content={"error": "invalid JSON body"},
            )
    requested_model = request_json.get("model")
    try:
        request.state.model_name = requested_model
    except AttributeError:
        pass

    app = request.app
    router = app.state.router

    # PII scan (feature-gated)
    pii = getattr(app.state, "pii_analyzer", None)
    if pii is not None and request_json and not is_multipart:
        from production_stack_amd.router.pii import scan_request_body

        allowed, request_json, matches = scan_request_body(
            request_json, pii, getattr(app.state, "pii_action", "block")
        )
        if not allowed:
            return JSONResponse(
                status_code=400,
                content={
                    "error": "request blocked: PII detected",
                    "entities": sorted({m.entity_type for m in matches}),
                },
            )
        if matches:
            body = json.dumps(request_json).encode()

    # semantic cache lookup (feature-gated, chat only, non-streaming)
    sem = getattr(app.state, "semantic_cache", None)
    if (
        sem is not None
        and endpoint == "/v1/chat/completions"
        and not request_json.get("stream")
    ):
        cached = sem.search(request_json)
        if cached is not None:
            return JSONResponse(
                status_code=200,
                content=cached,
                headers={"x-semantic-cache": "hit"},
            )

    # external provider branch
    ext = getattr(app.state, "external_providers", None)
    if ext is not None and ext.has_model(requested_model):
        return await _proxy_external(
            app, ext, endpoint, request_json, request_id
        )
    aliases = getattr(app.state, "model_aliases", None)

    # PD orchestrated short-circuit
    from production_stack_amd.router.routing_logic import (
        DisaggregatedPrefillOrchestratedRouter,
    )

    if isinstance(router, DisaggregatedPrefillOrchestratedRouter):
        return await route_disaggregated_request(
            request, endpoint, body, request_json, request_id
        )

    # callbacks / rewrites
    callbacks = getattr(app.state, "callbacks", None)
    if callbacks and hasattr(callbacks, "pre_request") and not is_multipart:
        maybe = callbacks.pre_request(request, request_json, requested_model)
        if maybe is not None:
            request_json = maybe
            body = json.dumps(request_json).encode()
    rewriter = getattr(app.state, "request_rewriter", None)
    if rewriter is not None and not is_multipart:
        new_body = rewriter.rewrite(endpoint, request_json)
        if new_body is not None:
            request_json = new_body
            body = json.dumps(request_json).encode()

    # alias resolution rewrites the body's model so engines that
    # validate model names accept it (reference
    # utils.replace_model_in_request_body + content-length update)
    if (aliases and requested_model in aliases and not is_multipart
            and request_json):
        request_json["model"] = aliases[requested_model]
        body = json.dumps(request_json).encode()

    endpoints = get_service_discovery().get_endpoint_info()
    candidates = filter_endpoints(endpoints, requested_model, aliases)
    if not candidates:
        seen_any = any(
            requested_model in ep.model_names
            for ep in endpoints
        )
        if endpoints and seen_any:
            return JSONResponse(
                status_code=503,
                content={
                    "error": f"model {requested_model} temporarily unavailable"
                },
            )
        return JSONResponse(
            status_code=404,
            content={"error": f"model {requested_model} not found"},
        )

    scraper = get_engine_stats_scraper()
    engine_stats = scraper.get_engine_stats() if scraper else {}
    request_stats = get_request_stats_monitor().get_request_stats()

    server_url = await router.route_request(
        candidates, engine_stats, request_stats, request, request_json
    )
    logger.info(
        "Routing request %s with session id %s to %s at %f, process time = %f",
        request_id,
        request.headers.get("x-user-id"),
        server_url,
        time.time(),
        time.time() - in_router_time,
    )

    max_attempts = getattr(app.state, "max_failover_attempts", 0)
    tried: set = set()
    attempt = 0
    last_error: Optional[str] = None
    sem_store = (
        sem is not None
        and endpoint == "/v1/chat/completions"
        and not request_json.get("stream")
    )
    while True:
        try:
            if sem_store:
                return await _proxy_buffered_with_cache(
                    request, body, server_url, endpoint, request_id,
                    sem, request_json,
                )
            return await _proxy_streaming(
                request, body, server_url, endpoint, request_id
            )
        except (aiohttp.ClientError, OSError, TimeoutError) as e:
            from production_stack_amd.router import metrics as rm

            get_request_stats_monitor().on_request_failed(
                server_url, request_id
            )
            rm.request_errors.labels(
                model=requested_model or "unknown",
                reason=type(e).__name__,
            ).inc()
            last_error = f"{type(e).__name__}: {e}"
            tried.add(server_url)
            attempt += 1
            if attempt > max_attempts:
                break
            remaining = [ep for ep in candidates if ep.url not in tried]
            if not remaining:
                break
            server_url = await router.route_request(
                remaining, engine_stats, request_stats, request, request_json
            )
            logger.warning(
                "failover: retrying request %s on %s (attempt %d)",
                request_id,
                server_url,
                attempt,
            )
    return JSONResponse(
        status_code=503,
        content={
            "error": f"all backends failed for request {request_id}",
            "detail": last_error,
        },
    )


def request_json_model_for_metrics(request) -> Optional[str]:
    try:
        return getattr(request.state, "model_name", None)
    except AttributeError:
        return None


def _record_usage(model, payload) -> None:
    from production_stack_amd.router import metrics as rm

    usage = payload.get("usage") or {}
    m = model or "unknown"
    if usage.get("prompt_tokens"):
        rm.model_input_tokens.labels(model=m).inc(usage["prompt_tokens"])
    if usage.get("completion_tokens"):
        rm.model_output_tokens.labels(model=m).inc(
            usage["completion_tokens"]
        )


async def _proxy_buffered_with_cache(
    request: Request,
    body: bytes,
    server_url: str,
    endpoint: str,
    request_id: str,
    sem,
    request_json: Dict[str, Any],
) -> Response:
    """Non-streaming chat proxy that stores the answer in the semantic
    cache (reference request.py:360-364 behaviour)."""
    monitor = get_request_stats_monitor()
    monitor.on_new_request(server_url, request_id, time.time())
    session = _session()
    async with session.post(
        server_url + endpoint,
        data=body,
        headers=_forward_headers(request),
        timeout=aiohttp.ClientTimeout(total=None),
    ) as resp:
        monitor.on_request_response(server_url, request_id, time.time())
        data = await resp.read()
        status = resp.status
        ctype = resp.headers.get("content-type", "application/json")
    monitor.on_request_complete(server_url, request_id, time.time())
    if status == 200:
        try:
            payload = json.loads(data)
            sem.store(request_json, payload)
            _record_usage(request_json.get("model"), payload)
        except (ValueError, TypeError):
            pass
    return Response(
        content=data,
        status_code=status,
        media_type=ctype,
        headers={"x-request-id": request_id},
    )


async def _proxy_external(
    app, ext, endpoint: str, request_json: Dict[str, Any], request_id: str
) -> Response:
    provider = ext.provider_for(request_json.get("model"))
    session = _session()
    resp = await provider.forward(session, endpoint, request_json)
    if request_json.get("stream"):

        async def stream():
            try:
                async for chunk in resp.content.iter_any():
                    yield chunk
            finally:
                resp.release()

        return StreamingResponse(
            stream(),
            status_code=resp.status,
            media_type=resp.headers.get("content-type", "application/json"),
            headers={"x-request-id": request_id},
        )
    data = await resp.read()
    out = Response(
        content=data,
        status_code=resp.status,
        media_type=resp.headers.get("content-type", "application/json"),
        headers={"x-request-id": request_id},
    )
    resp.release()
    return out


async def _proxy_streaming(
    request: Request,
    body: bytes,
    server_url: str,
    endpoint: str,
    request_id: str,
) -> Response:
    gen = process_request(
        request, body, server_url, endpoint, request_id,
        model=request_json_model_for_metrics(request),
    )
    status, headers = await gen.__anext__()
    media_type = headers.get("content-type", "application/json")
    out_headers = {
        k: v
        for k, v in headers.items()
        if k.lower() not in HOP_BY_HOP and k.lower() != "content-type"
    }
    out_headers["x-request-id"] = request_id

    async def stream():
        async for chunk in gen:
            yield chunk

    return StreamingResponse(
        stream(),
        status_code=status,
        media_type=media_type,
        headers=out_headers,
    )


# ---------------------------------------------------------------------------
# Orchestrated disaggregated prefill: one client request -> P then D.
# ---------------------------------------------------------------------------
async def route_disaggregated_request(
    request: Request,
    endpoint: str,
    body: bytes,
    request_json: Dict[str, Any],
    request_id: str,
) -> Response:
    app = request.app
    router = app.state.router
    endpoints = filter_endpoints(
        get_service_discovery().get_endpoint_info(),
        request_json.get("model"),
        getattr(app.state, "model_aliases", None),
    )
    prefill_url = router.select_prefill_endpoint(endpoints)
    decode_url = router.select_decode_endpoint(endpoints)
    if prefill_url is None or decode_url is None:
        return JSONResponse(
            status_code=503,
            content={"error": "no prefill/decode endpoints available"},
        )

    session = _session()
    monitor = get_request_stats_monitor()

    # 1. prefill request: 1 token, not streamed, ask engine to keep KV for a
    #    remote decode.
    prefill_json = dict(request_json)
    prefill_json["max_tokens"] = 1
    if "max_completion_tokens" in prefill_json:
        prefill_json["max_completion_tokens"] = 1
    prefill_json["stream"] = False
    prefill_json["kv_transfer_params"] = {
        "do_remote_decode": True,
        "do_remote_prefill": False,
        "remote_engine_id": None,
        "remote_block_ids": None,
        "remote_host": None,
        "remote_port": None,
    }
    headers = _forward_headers(request)
    headers["content-type"] = "application/json"
    monitor.on_new_request(prefill_url, f"{request_id}-prefill", time.time())
    async with session.post(
        prefill_url + endpoint,
        json=prefill_json,
        headers=headers,
        timeout=aiohttp.ClientTimeout(total=None),
    ) as pr:
        if pr.status != 200:
            detail = await pr.text()
            monitor.on_request_failed(prefill_url, f"{request_id}-prefill")
            return JSONResponse(
                status_code=pr.status,
                content={"error": "prefill failed", "detail": detail},
            )
        prefill_out = await pr.json()
    monitor.on_request_complete(
        prefill_url, f"{request_id}-prefill", time.time()
    )

    kv_params = prefill_out.get("kv_transfer_params") or {}
    kv_params["do_remote_decode"] = False
    kv_params["do_remote_prefill"] = True
    if not kv_params.get("remote_host"):
        kv_params["remote_host"] = prefill_url.split("//")[-1].split(":")[0]

    # 2. decode request with the prefill's KV handle; stream to client.
    decode_json = dict(request_json)
    decode_json["kv_transfer_params"] = kv_params

    monitor.on_new_request(decode_url, request_id, time.time())
    resp = await session.post(
        decode_url + endpoint,
        json=decode_json,
        headers=headers,
        timeout=aiohttp.ClientTimeout(total=None),
    )
    if resp.status != 200:
        detail = await resp.text()
        resp.release()
        monitor.on_request_failed(decode_url, request_id)
        return JSONResponse(
            status_code=resp.status,
            content={"error": "decode failed", "detail": detail},
        )

    async def stream():
        first = True
        try:
            async for chunk in resp.content.iter_any():
                if first:
                    monitor.on_request_response(
                        decode_url, request_id, time.time()
                    )
                    first = False
                yield chunk
        finally:
            monitor.on_request_complete(decode_url, request_id, time.time())
            resp.release()

    return StreamingResponse(
        stream(),
        status_code=resp.status,
        media_type=resp.headers.get("content-type", "application/json"),
        headers={"x-request-id": request_id},
    )


# ---------------------------------------------------------------------------
# Sleep / wake passthrough (reference request.py:1041-1128)
# ---------------------------------------------------------------------------
async def route_sleep_wakeup_request(
    request: Request, endpoint: str
) -> Response:
    target = request.query_params.get("url") or request.headers.get(
        "x-engine-url"
    )
    endpoints = get_service_discovery().get_endpoint_info()
    if target:
        matches = [ep for ep in endpoints if ep.url == target.rstrip("/")]
    else:
        matches = endpoints
    if not matches:
        return JSONResponse(
            status_code=404, content={"error": "engine not found"}
        )
    session = _session()
    results = {}
    for ep in matches:
        try:
            if endpoint == "/is_sleeping":
                async with session.get(
                    ep.url + endpoint, params=dict(request.query_params)
                ) as r:
                    results[ep.url] = await r.json()
            else:
                async with session.post(
                    ep.url + endpoint, params=dict(request.query_params)
                ) as r:
                    results[ep.url] = {"status": r.status}
                sd = get_service_discovery()
                if hasattr(sd, "set_sleep"):
                    sd.set_sleep(ep.url, endpoint == "/sleep")
        except (aiohttp.ClientError, OSError) as e:
            results[ep.url] = {"error": str(e)}
    return JSONResponse(status_code=200, content=results)
