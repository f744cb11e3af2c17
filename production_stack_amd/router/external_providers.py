"""External OpenAI-compatible providers.

Parity: reference src/vllm_router/external_providers/ — route selected model
names to SaaS endpoints instead of local engines; model index, alias
resolution, live model validation, auth header injection.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import yaml

logger = logging.getLogger("router.external")


@dataclass
class ExternalProviderConfig:
    name: str
    base_url: str
    api_key: Optional[str] = None
    api_key_env: Optional[str] = None
    models: List[str] = field(default_factory=list)
    model_aliases: Dict[str, str] = field(default_factory=dict)
    provider_type: str = "openai"

    def resolve_api_key(self) -> Optional[str]:
        if self.api_key:
            return self.api_key
        if self.api_key_env:
            import os

            return os.environ.get(self.api_key_env)
        return None


class OpenAIProvider:
    def __init__(self, config: ExternalProviderConfig) -> None:
        self.config = config

    @property
    def base_url(self) -> str:
        return self.config.base_url.rstrip("/")

    def headers(self) -> Dict[str, str]:
        h = {"content-type": "application/json"}
        key = self.config.resolve_api_key()
        if key:
            h["Authorization"] = f"Bearer {key}"
        return h

    def resolve_model(self, model: str) -> str:
        return self.config.model_aliases.get(model, model)

    async def validate_models(self, session) -> List[str]:
        """Live-check which configured models the provider actually serves."""
        import aiohttp

        try:
            async with session.get(
                self.base_url + "/v1/models",
                headers=self.headers(),
                timeout=aiohttp.ClientTimeout(total=10),
            ) as r:
                if r.status != 200:
                    return list(self.config.models)
                data = await r.json()
                live = {m.get("id") for m in data.get("data", [])}
                return [
                    m
                    for m in self.config.models
                    if self.resolve_model(m) in live or not live
                ]
        except Exception:
            return list(self.config.models)

    async def forward(
        self, session, endpoint: str, body: Dict[str, Any]
    ):
        import aiohttp

        body = dict(body)
        if "model" in body:
            body["model"] = self.resolve_model(body["model"])
        return await session.post(
            self.base_url + endpoint,
            json=body,
            headers=self.headers(),
            timeout=aiohttp.ClientTimeout(total=None),
        )


class ExternalProviderManager:
    def __init__(self, configs: List[ExternalProviderConfig]) -> None:
        self.providers = [OpenAIProvider(c) for c in configs]
        self._index: Dict[str, OpenAIProvider] = {}
        for p in self.providers:
            for m in p.config.models:
                self._index.setdefault(m, p)
            for alias in p.config.model_aliases:
                self._index.setdefault(alias, p)

    @staticmethod
    def from_yaml(path: str) -> "ExternalProviderManager":
        with open(path) as f:
            raw = yaml.safe_load(f) or {}
        configs = [
            ExternalProviderConfig(**p) for p in raw.get("providers", [])
        ]
        return ExternalProviderManager(configs)

    def has_model(self, model: Optional[str]) -> bool:
        return model in self._index

    def provider_for(self, model: str) -> Optional[OpenAIProvider]:
        return self._index.get(model)

    def model_names(self) -> List[str]:
        return sorted(self._index)
