"""Router FastAPI app: endpoint table, singleton wiring, uvicorn entry.

Parity map:
  * endpoint table -> reference routers/main_router.py:51-300 +
    files_router.py + batches_router.py + metrics_router.py
  * initialize_all -> reference app.py:161-365
  * lifespan (shared aiohttp session) -> reference app.py:106-158
"""

from __future__ import annotations

import contextlib
import logging
import threading
import time
from typing import Optional

import aiohttp
from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse


def parse_multipart(body: bytes, content_type: str):
    """Minimal multipart/form-data parser (python-multipart is not in the
    offline env). Returns {field_name: (filename | None, content_bytes)}."""
    import re as _re

    m = _re.search(r'boundary="?([^";]+)"?', content_type)
    if not m:
        return {}
    boundary = b"--" + m.group(1).encode()
    parts = {}
    for chunk in body.split(boundary):
        chunk = chunk.strip(b"\r\n")
        if not chunk or chunk == b"--":
            continue
        if b"\r\n\r\n" not in chunk:
            continue
        header_blob, content = chunk.split(b"\r\n\r\n", 1)
        headers = header_blob.decode(errors="replace")
        name_m = _re.search(r'name="([^"]+)"', headers)
        if not name_m:
            continue
        fn_m = _re.search(r'filename="([^"]*)"', headers)
        parts[name_m.group(1)] = (
            fn_m.group(1) if fn_m else None,
            content,
        )
    return parts

from production_stack_amd import __version__
from production_stack_amd.router import request_service
from production_stack_amd.router.metrics import fill_and_render
from production_stack_amd.router.protocols import ModelCard, ModelList
from production_stack_amd.router.service_discovery import (
    get_service_discovery,
)

logger = logging.getLogger("router.app")

_http_session: Optional[aiohttp.ClientSession] = None


def get_http_session() -> aiohttp.ClientSession:
    global _http_session
    if _http_session is None or _http_session.closed:
        _http_session = aiohttp.ClientSession(
            connector=aiohttp.TCPConnector(limit=0)
        )
    return _http_session


@contextlib.asynccontextmanager
async def lifespan(app: FastAPI):
    get_http_session()
    yield
    if _http_session is not None and not _http_session.closed:
        await _http_session.close()


_last_app = None


def build_app() -> FastAPI:
    global _last_app
    app = FastAPI(title="production-stack-amd router", lifespan=lifespan)
    _last_app = app

    # ---- proxied OpenAI API -------------------------------------------
    PROXIED = [
        "/v1/chat/completions",
        "/v1/completions",
        "/v1/embeddings",
        "/v1/rerank",
        "/rerank",
        "/v1/score",
        "/score",
        "/v1/responses",
        "/v1/messages",
        "/v1/images/generations",
        "/v1/images/edits",
        "/v1/images/variations",
    ]

    def make_handler(path: str):
        async def handler(request: Request):
            return await request_service.route_general_request(request, path)

        return handler

    for path in PROXIED:
        app.post(path)(make_handler(path))

    @app.post("/v1/audio/speech")
    async def speech(request: Request):
        return await request_service.route_general_request(
            request, "/v1/audio/speech"
        )

    @app.post("/v1/audio/transcriptions")
    async def transcriptions(request: Request):
        return await request_service.route_general_request(
            request, "/v1/audio/transcriptions"
        )

    @app.post("/v1/audio/translations")
    async def translations(request: Request):
        return await request_service.route_general_request(
            request, "/v1/audio/translations"
        )

    @app.get("/v1/audio/voices")
    async def voices(request: Request):
        return await request_service.route_general_request(
            request, "/v1/audio/voices"
        )

    @app.post("/tokenize")
    async def tokenize(request: Request):
        return await request_service.route_general_request(
            request, "/tokenize"
        )

    @app.post("/detokenize")
    async def detokenize(request: Request):
        return await request_service.route_general_request(
            request, "/detokenize"
        )

    # ---- model aggregation --------------------------------------------
    @app.get("/v1/models")
    async def list_models():
        # full backend card payloads pass through (extra fields like
        # max_model_len preserved — reference test_main_router_models)
        cards = {}
        for ep in get_service_discovery().get_endpoint_info():
            for name in ep.model_names:
                if name in cards:
                    continue
                mi = ep.model_info.get(name)
                cards[name] = (
                    mi.to_dict() if mi is not None
                    else ModelCard(id=name).model_dump()
                )
        aliases = getattr(app.state, "model_aliases", None) or {}
        for alias, target in aliases.items():
            if alias not in cards:
                cards[alias] = ModelCard(
                    id=alias, parent=target).model_dump()
        ext = getattr(app.state, "external_providers", None)
        if ext is not None:
            for name in ext.model_names():
                if name not in cards:
                    cards[name] = ModelCard(
                        id=name, owned_by="external").model_dump()
        return {"object": "list", "data": list(cards.values())}

    # ---- ops endpoints -------------------------------------------------
    @app.get("/health")
    async def health():
        problems = []
        try:
            sd = get_service_discovery()
            if not sd.get_health():
                problems.append("service discovery unhealthy")
        except RuntimeError:
            problems.append("service discovery not initialized")
        from production_stack_amd.router.dynamic_config import (
            get_dynamic_config_watcher,
        )
        from production_stack_amd.router.stats import (
            get_engine_stats_scraper,
        )

        scraper = get_engine_stats_scraper()
        if scraper and not scraper.get_health():
            problems.append("engine stats scraper dead")
        watcher = get_dynamic_config_watcher()
        if watcher and not watcher.get_health():
            problems.append("dynamic config watcher dead")
        if problems:
            return JSONResponse(
                status_code=503, content={"status": ", ".join(problems)}
            )
        return {"status": "healthy"}

    @app.get("/version")
    async def version():
        return {"version": __version__}

    @app.get("/engines")
    async def engines():
        from production_stack_amd.router.stats import (
            get_engine_stats_scraper,
            get_request_stats_monitor,
        )

        scraper = get_engine_stats_scraper()
        engine_stats = scraper.get_engine_stats() if scraper else {}
        request_stats = get_request_stats_monitor().get_request_stats()
        out = {}
        for ep in get_service_discovery().get_endpoint_info():
            es = engine_stats.get(ep.url)
            rs = request_stats.get(ep.url)
            out[ep.url] = {
                "model_names": ep.model_names,
                "model_label": ep.model_label,
                "sleep": ep.sleep,
                "engine_stats": es.__dict__ if es else None,
                "request_stats": rs.__dict__ if rs else None,
            }
        return out

    @app.get("/metrics")
    async def metrics():
        return Response(
            content=fill_and_render(), media_type="text/plain; version=0.0.4"
        )

    @app.post("/sleep")
    async def sleep(request: Request):
        return await request_service.route_sleep_wakeup_request(
            request, "/sleep"
        )

    @app.post("/wake_up")
    async def wake_up(request: Request):
        return await request_service.route_sleep_wakeup_request(
            request, "/wake_up"
        )

    @app.get("/is_sleeping")
    async def is_sleeping(request: Request):
        return await request_service.route_sleep_wakeup_request(
            request, "/is_sleeping"
        )

    # ---- files + batches ----------------------------------------------
    @app.post("/v1/files")
    async def upload_file(request: Request):
        storage = getattr(app.state, "file_storage", None)
        if storage is None:
            return JSONResponse(
                status_code=501, content={"error": "file API disabled"}
            )
        body = await request.body()
        ctype = request.headers.get("content-type", "")
        parts = parse_multipart(body, ctype)
        file_part = parts.get("file")
        if file_part is None:
            return JSONResponse(
                status_code=400, content={"error": "missing file field"}
            )
        purpose = parts.get("purpose")
        meta = storage.save_file(
            file_part[1],
            file_part[0] or "upload",
            purpose=(purpose[1].decode() if purpose else "batch"),
        )
        return meta.metadata()

    @app.get("/v1/files/{file_id}")
    async def get_file(file_id: str):
        storage = getattr(app.state, "file_storage", None)
        meta = storage.get_file_metadata(file_id) if storage else None
        if meta is None:
            return JSONResponse(
                status_code=404, content={"error": "file not found"}
            )
        return meta.metadata()

    @app.get("/v1/files/{file_id}/content")
    async def get_file_content(file_id: str):
        storage = getattr(app.state, "file_storage", None)
        content = storage.get_file_content(file_id) if storage else None
        if content is None:
            return JSONResponse(
                status_code=404, content={"error": "file not found"}
            )
        return Response(content=content, media_type="application/octet-stream")

    @app.post("/v1/batches")
    async def create_batch(request: Request):
        proc = getattr(app.state, "batch_processor", None)
        if proc is None:
            return JSONResponse(
                status_code=501, content={"error": "batch API disabled"}
            )
        body = await request.json()
        b = proc.create_batch(
            input_file_id=body["input_file_id"],
            endpoint=body.get("endpoint", "/v1/chat/completions"),
            completion_window=body.get("completion_window", "24h"),
            metadata=body.get("metadata"),
        )
        import asyncio

        asyncio.get_running_loop().create_task(proc.run_batch(b.id))
        return b.to_dict()

    @app.get("/v1/batches/{batch_id}")
    async def get_batch(batch_id: str):
        proc = getattr(app.state, "batch_processor", None)
        b = proc.get_batch(batch_id) if proc else None
        if b is None:
            return JSONResponse(
                status_code=404, content={"error": "batch not found"}
            )
        return b.to_dict()

    @app.get("/v1/batches")
    async def list_batches():
        proc = getattr(app.state, "batch_processor", None)
        return {"object": "list", "data": proc.list_batches() if proc else []}

    @app.post("/v1/batches/{batch_id}/cancel")
    async def cancel_batch(batch_id: str):
        proc = getattr(app.state, "batch_processor", None)
        b = proc.cancel_batch(batch_id) if proc else None
        if b is None:
            return JSONResponse(
                status_code=404, content={"error": "batch not found"}
            )
        return b.to_dict()

    return app


def initialize_all(app: FastAPI, args) -> None:
    """Wire every singleton from parsed args (reference app.py:161-365)."""
    from production_stack_amd.router import routing_logic as rl
    from production_stack_amd.router import service_discovery as sd
    from production_stack_amd.router import stats
    from production_stack_amd.router.utils import (
        parse_comma_separated,
        parse_static_aliases,
        parse_static_model_names,
        parse_static_urls,
    )

    # reference-compatible alias for the discovery mode
    if getattr(args, "k8s_service_discovery_type", None):
        args.service_discovery = (
            "k8s_service_name"
            if args.k8s_service_discovery_type == "service-name"
            else "k8s"
        )
    if args.service_discovery == "static":
        sd.initialize_service_discovery(
            "static",
            urls=parse_static_urls(args.static_backends),
            models=parse_static_model_names(args.static_models),
            aliases=parse_static_aliases(args.static_aliases),
            model_labels=(
                parse_comma_separated(args.static_model_labels) or None
            ),
            model_types=(
                parse_comma_separated(args.static_model_types) or None
            ),
            health_check=args.static_backend_health_checks,
            health_check_interval=args.health_check_interval,
            api_key=args.api_key,
        )
        app.state.model_aliases = parse_static_aliases(args.static_aliases)
    elif args.service_discovery in ("k8s", "k8s_pod_ip"):
        sd.initialize_service_discovery(
            "k8s",
            namespace=args.k8s_namespace,
            port=args.k8s_port,
            label_selector=args.k8s_label_selector,
            api_key=args.api_key,
            insecure_skip_tls_verify=getattr(
                args, "k8s_insecure_skip_tls_verify", False),
            watcher_timeout_seconds=getattr(
                args, "k8s_watcher_timeout_seconds", 30),
        )
        app.state.model_aliases = {}
    elif args.service_discovery == "k8s_service_name":
        sd.initialize_service_discovery(
            "k8s_service_name",
            namespace=args.k8s_namespace,
            port=args.k8s_port,
            label_selector=args.k8s_label_selector,
            api_key=args.api_key,
            insecure_skip_tls_verify=getattr(
                args, "k8s_insecure_skip_tls_verify", False),
        )
        app.state.model_aliases = {}
    else:
        sd.initialize_service_discovery("external")
        app.state.model_aliases = {}

    stats.initialize_engine_stats_scraper(args.engine_stats_interval)
    stats.initialize_request_stats_monitor(args.request_stats_window)

    kwargs = dict(
        session_key=args.session_key,
        prefix_min_match_length=args.prefix_min_match_length,
        kv_controller_host=args.kv_controller_host,
        kv_controller_port=args.lmcache_controller_port,
        kv_match_threshold=args.kv_aware_threshold,
        prefill_model_labels=parse_comma_separated(args.prefill_model_labels)
        or None,
        decode_model_labels=parse_comma_separated(args.decode_model_labels)
        or None,
    )
    app.state.router = rl.initialize_routing_logic(
        args.routing_logic, **kwargs
    )
    app.state.max_failover_attempts = (
        args.max_instance_failover_reroute_attempts
    )

    from production_stack_amd.router.experimental import (
        configure_custom_callbacks,
        get_request_rewriter,
        initialize_feature_gates,
    )

    gates = initialize_feature_gates(args.feature_gates)

    if getattr(args, "sentry_dsn", None):
        try:  # sentry-sdk is optional (absent in the offline image)
            import sentry_sdk

            sentry_sdk.init(
                dsn=args.sentry_dsn,
                traces_sample_rate=args.sentry_traces_sample_rate,
                profile_session_sample_rate=getattr(
                    args, "sentry_profile_session_sample_rate", 0.0),
            )
        except ImportError:
            logger.warning("--sentry-dsn set but sentry_sdk not installed")

    from production_stack_amd.router import tracing

    tracing.initialize_tracing(
        getattr(args, "otel_endpoint", None),
        getattr(args, "otel_service_name", "vllm-router"),
        getattr(args, "otel_secure", False),
    )

    if getattr(args, "external_providers_config", None):
        from production_stack_amd.router.external_providers import (
            ExternalProviderManager,
        )

        app.state.external_providers = ExternalProviderManager.from_yaml(
            args.external_providers_config
        )
    else:
        app.state.external_providers = None

    if gates.is_enabled("PIIDetection"):
        from production_stack_amd.router.pii import create_analyzer

        app.state.pii_analyzer = create_analyzer(
            getattr(args, "pii_analyzer", "regex")
        )
        app.state.pii_action = getattr(args, "pii_action", "block")
    else:
        app.state.pii_analyzer = None

    if gates.is_enabled("SemanticCache"):
        from production_stack_amd.router.semantic_cache import SemanticCache

        app.state.semantic_cache = SemanticCache(
            threshold=getattr(args, "semantic_cache_threshold", 0.95)
        )
    else:
        app.state.semantic_cache = None
    app.state.request_rewriter = (
        get_request_rewriter(args.request_rewriter)
        if args.request_rewriter != "noop"
        else None
    )
    app.state.callbacks = (
        configure_custom_callbacks(args.callbacks) if args.callbacks else None
    )

    if args.enable_batch_api:
        from production_stack_amd.router.batches import BatchProcessor
        from production_stack_amd.router.files import FileStorage

        app.state.file_storage = FileStorage(args.file_storage_path)
        app.state.batch_processor = BatchProcessor(
            app.state.file_storage, args.batch_processor_db
        )
    else:
        from production_stack_amd.router.files import FileStorage

        app.state.file_storage = FileStorage(args.file_storage_path)
        app.state.batch_processor = None

    if getattr(args, "dynamic_config_yaml", None) and not \
            args.dynamic_config_json:
        args.dynamic_config_json = args.dynamic_config_yaml
    if args.dynamic_config_json:
        from production_stack_amd.router.dynamic_config import (
            initialize_dynamic_config_watcher,
        )

        initialize_dynamic_config_watcher(
            args.dynamic_config_json, 10.0, app
        )

    if args.log_stats:
        start_log_stats_thread(args.log_stats_interval)


def start_log_stats_thread(interval: float) -> threading.Thread:
    from production_stack_amd.router.stats import (
        get_engine_stats_scraper,
        get_request_stats_monitor,
    )

    def worker():
        while True:
            time.sleep(interval)
            try:
                scraper = get_engine_stats_scraper()
                es = scraper.get_engine_stats() if scraper else {}
                rs = get_request_stats_monitor().get_request_stats()
                lines = ["--- router stats ---"]
                for url in sorted(set(es) | set(rs)):
                    e, r = es.get(url), rs.get(url)
                    lines.append(
                        f"{url}: running={getattr(e, 'num_running_requests', '?')} "
                        f"waiting={getattr(e, 'num_queuing_requests', '?')} "
                        f"qps={getattr(r, 'qps', 0):.2f} "
                        f"ttft={getattr(r, 'ttft', -1):.3f}"
                    )
                logger.info("\n".join(lines))
            except Exception as e:
                logger.debug("log stats failed: %s", e)

    t = threading.Thread(target=worker, daemon=True)
    t.start()
    return t


def main() -> None:
    import uvicorn

    from production_stack_amd.router.log import init_logger
    from production_stack_amd.router.parser import parse_args
    from production_stack_amd.router.utils import set_ulimit

    args = parse_args()
    init_logger(args.log_level, args.log_format)
    app = build_app()
    initialize_all(app, args)
    set_ulimit()
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning",
                root_path=args.root_path or "")


if __name__ == "__main__":
    main()
