"""Router CLI (flag surface mirrors reference parsers/parser.py:125-495 so
helm's values->args mapping carries over unchanged)."""

from __future__ import annotations

import argparse
from typing import Optional


def parse_args(argv: Optional[list] = None) -> argparse.Namespace:
    p = argparse.ArgumentParser(description="MI355X-native vLLM-router-compatible request router")
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8001)

    # service discovery
    p.add_argument(
        "--service-discovery",
        choices=["static", "k8s", "k8s_pod_ip", "k8s_service_name",
                 "external", "external-only"],
        default="static",
    )
    p.add_argument("--static-backends", type=str, default=None,
                   help="comma-separated backend URLs")
    p.add_argument("--static-models", type=str, default=None,
                   help="comma-separated model names aligned with backends")
    p.add_argument("--static-aliases", type=str, default=None,
                   help="alias1:model1,alias2:model2")
    p.add_argument("--static-model-labels", type=str, default=None)
    p.add_argument("--static-model-types", type=str, default=None)
    p.add_argument("--static-backend-health-checks", action="store_true")
    p.add_argument("--health-check-interval", type=float, default=60.0)
    p.add_argument("--static-backend-health-check-interval", type=float,
                   default=None, dest="health_check_interval_alias",
                   help="alias for --health-check-interval")
    p.add_argument("--static-backend-health-check-timeout-seconds",
                   type=float, default=10.0)
    p.add_argument("--backend-health-check-timeout-seconds", type=float,
                   default=None,
                   help="alias for --static-backend-health-check-timeout-seconds")
    p.add_argument("--k8s-namespace", default="default")
    p.add_argument("--k8s-port", type=int, default=8000)
    p.add_argument("--k8s-label-selector", default=None)
    p.add_argument("--k8s-service-discovery-type", default=None,
                   choices=[None, "pod-ip", "service-name"],
                   help="reference-compatible alias: pod-ip == "
                        "--service-discovery k8s, service-name == "
                        "k8s_service_name")
    p.add_argument("--k8s-watcher-timeout-seconds", type=int, default=30,
                   help="server-side timeoutSeconds for the pod watch "
                        "stream (reconnects after)")
    p.add_argument("--k8s-insecure-skip-tls-verify", action="store_true",
                   help="explicitly disable K8s API TLS verification "
                        "(test clusters only; never silently implied)")

    # routing
    p.add_argument(
        "--routing-logic",
        choices=[
            "roundrobin",
            "session",
            "kvaware",
            "prefixaware",
            "disaggregated_prefill",
            "disaggregated_prefill_orchestrated",
        ],
        default="roundrobin",
    )
    p.add_argument("--session-key", default=None)
    p.add_argument("--prefix-min-match-length", type=int, default=128)
    p.add_argument("--kv-aware-threshold", type=int, default=2000)
    p.add_argument("--lmcache-controller-port", type=int, default=9000,
                   help="KV-pool controller port (kvaware routing)")
    p.add_argument("--kv-controller-host", default="127.0.0.1")
    # accepted for reference CLI compatibility; the in-house msgpack-TCP
    # controller uses a single port (no separate ZMQ reply/heartbeat
    # sockets), and health/timeout knobs map onto its client timeouts
    p.add_argument("--lmcache-controller-reply-port", type=int, default=None)
    p.add_argument("--lmcache-controller-heartbeat-port", type=int,
                   default=None)
    p.add_argument("--lmcache-health-check-interval", type=float,
                   default=30.0)
    p.add_argument("--lmcache-worker-timeout", type=float, default=10.0)
    p.add_argument("--prefill-model-labels", type=str, default=None)
    p.add_argument("--decode-model-labels", type=str, default=None)
    p.add_argument("--max-instance-failover-reroute-attempts", type=int,
                   default=0)

    # stats / metrics
    p.add_argument("--engine-stats-interval", type=float, default=10.0)
    p.add_argument("--request-stats-window", type=float, default=60.0)
    p.add_argument("--log-stats", action="store_true")
    p.add_argument("--log-stats-interval", type=float, default=10.0)

    # dynamic config
    p.add_argument("--dynamic-config-json", default=None)
    p.add_argument("--root-path", default=None,
                   help="ASGI root_path when served behind a path prefix")
    p.add_argument("--version", action="version",
                   version="production-stack-amd router 0.1")
    p.add_argument("--dynamic-config-yaml", default=None,
                   help="YAML form of --dynamic-config-json; supports the "
                        "structured models:/aliases: schema")

    # features
    p.add_argument("--feature-gates", default=None,
                   help="e.g. SemanticCache=true")
    p.add_argument("--callbacks", default=None,
                   help="module path exposing pre/post request hooks")
    p.add_argument("--request-rewriter", default="noop")
    p.add_argument("--file-storage-path", default="/tmp/vllm_files")
    p.add_argument("--file-storage-class", default="local_file",
                   choices=["local_file"])
    p.add_argument("--batch-processor", default="local",
                   choices=["local"])
    p.add_argument("--enable-batch-api", action="store_true")
    p.add_argument("--batch-processor-db", default="/tmp/vllm_batches.sqlite")

    # auth / api
    p.add_argument("--api-key", default=None,
                   help="bearer token forwarded to engines")

    # observability / experimental
    p.add_argument("--sentry-dsn", default=None)
    p.add_argument("--sentry-traces-sample-rate", type=float, default=0.0)
    p.add_argument("--sentry-profile-session-sample-rate", type=float,
                   default=0.0)
    p.add_argument("--otel-endpoint", default=None)
    p.add_argument("--otel-service-name", default="vllm-router")
    p.add_argument("--otel-secure", action="store_true")
    p.add_argument("--external-providers-config", default=None,
                   help="YAML file of external OpenAI-compatible providers")
    p.add_argument("--pii-analyzer", default="regex",
                   choices=["regex", "presidio"])
    p.add_argument("--pii-action", default="block",
                   choices=["block", "redact"])
    p.add_argument("--semantic-cache-threshold", type=float, default=0.95)

    # logging
    p.add_argument("--log-level", default="info",
                   choices=["trace", "debug", "info", "warning", "error"])
    p.add_argument("--log-format", default="text", choices=["text", "json"])

    args = p.parse_args(argv)
    # flag aliases (reference parser compatibility)
    if getattr(args, "health_check_interval_alias", None):
        args.health_check_interval = args.health_check_interval_alias
    if getattr(args, "backend_health_check_timeout_seconds", None):
        args.static_backend_health_check_timeout_seconds = \
            args.backend_health_check_timeout_seconds
    validate_args(args)
    return args


def validate_args(args: argparse.Namespace) -> None:
    if args.service_discovery == "static":
        if not args.static_backends:
            raise ValueError(
                "--static-backends is required with static discovery"
            )
        if not args.static_models:
            raise ValueError(
                "--static-models is required with static discovery"
            )
        n_b = len([u for u in args.static_backends.split(",") if u.strip()])
        n_m = len([m for m in args.static_models.split(",") if m.strip()])
        if n_m not in (1, n_b):
            raise ValueError(
                "number of --static-models must be 1 or match "
                "--static-backends"
            )
    if args.routing_logic in (
        "disaggregated_prefill",
        "disaggregated_prefill_orchestrated",
    ):
        if not (args.prefill_model_labels and args.decode_model_labels):
            raise ValueError(
                "disaggregated prefill routing requires "
                "--prefill-model-labels and --decode-model-labels"
            )
