"""Router utilities: singletons, model-type probes, parsing helpers.

Behavioural parity: reference src/vllm_router/utils.py (SingletonMeta with
_create-gated lookup :36-65, ModelType probe payloads :68-158, URL parsing
:201-230, set_ulimit :183-198).
"""

from __future__ import annotations

import enum
import resource
from typing import Dict, List, Optional


class SingletonMeta(type):
    _instances: Dict[type, object] = {}

    def __call__(cls, *args, **kwargs):
        if cls not in cls._instances:
            cls._instances[cls] = super().__call__(*args, **kwargs)
        return cls._instances[cls]


class SingletonABCMeta(SingletonMeta):
    pass


def get_singleton(cls):
    return SingletonMeta._instances.get(cls)


def clear_singletons() -> None:
    SingletonMeta._instances.clear()


class ModelType(enum.Enum):
    chat = "/v1/chat/completions"
    completion = "/v1/completions"
    embeddings = "/v1/embeddings"
    rerank = "/v1/rerank"
    transcription = "/v1/audio/transcriptions"

    @staticmethod
    def get_test_payload(model_type: str, model: str) -> dict:
        payloads = {
            "chat": {
                "model": model,
                "messages": [{"role": "user", "content": "ping"}],
                "max_tokens": 2,
            },
            "completion": {"model": model, "prompt": "ping", "max_tokens": 2},
            "embeddings": {"model": model, "input": "ping"},
            "rerank": {
                "model": model,
                "query": "ping",
                "documents": ["pong"],
            },
        }
        return payloads.get(model_type, payloads["completion"])

    @staticmethod
    def get_all_fields() -> List[str]:
        return [m.name for m in ModelType]


def parse_static_urls(value: str) -> List[str]:
    return [u.strip().rstrip("/") for u in value.split(",") if u.strip()]


def parse_comma_separated(value: Optional[str]) -> List[str]:
    if not value:
        return []
    return [v.strip() for v in value.split(",") if v.strip()]


def parse_static_model_names(value: str) -> List[str]:
    return parse_comma_separated(value)


def parse_static_aliases(value: Optional[str]) -> Dict[str, str]:
    """"alias1:model1,alias2:model2" -> {alias: model}"""
    out: Dict[str, str] = {}
    for pair in parse_comma_separated(value):
        if ":" in pair:
            alias, model = pair.split(":", 1)
            out[alias.strip()] = model.strip()
    return out


def set_ulimit(target: int = 65535) -> None:
    soft, hard = resource.getrlimit(resource.RLIMIT_NOFILE)
    if soft < target:
        try:
            resource.setrlimit(
                resource.RLIMIT_NOFILE, (min(target, hard), hard)
            )
        except ValueError:
            pass


def update_content_length(request, body: bytes) -> dict:
    headers = dict(request.headers)
    headers["content-length"] = str(len(body))
    return headers
