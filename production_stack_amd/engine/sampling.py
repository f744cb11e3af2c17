"""Sampling parameters (OpenAI-compatible subset the router proxies)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class SamplingParams:
    max_tokens: int = 16
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1
    stop_token_ids: List[int] = field(default_factory=list)
    stop: List[str] = field(default_factory=list)
    ignore_eos: bool = False
    seed: Optional[int] = None
    logprobs: Optional[int] = None
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    logit_bias: Optional[dict] = None  # token_id -> bias
    min_tokens: int = 0
    # vLLM-style min_p: drop tokens with prob < min_p * max_prob
    min_p: float = 0.0
    # vLLM-style prompt_logprobs: log-probs of each prompt token under
    # the model (position 0 has no prediction -> None), computed during
    # the prefill chunks; any non-None value enables it
    prompt_logprobs: Optional[int] = None
    # OpenAI structured outputs: {"type": "json_object"} or
    # {"type": "json_schema", "json_schema": {...}} (engine/guided.py)
    response_format: Optional[dict] = None

    @property
    def needs_penalties(self) -> bool:
        return (
            self.presence_penalty != 0.0
            or self.frequency_penalty != 0.0
            or self.repetition_penalty != 1.0
            or bool(self.logit_bias)
            or self.min_tokens > 0
        )

    @property
    def greedy(self) -> bool:
        return self.temperature == 0.0

    def validate(self, max_model_len: int) -> None:
        if self.max_tokens < 1:
            raise ValueError("max_tokens must be >= 1")
        if self.temperature < 0:
            raise ValueError("temperature must be >= 0")
        if not (0 < self.top_p <= 1.0):
            raise ValueError("top_p must be in (0, 1]")
        if not (0.0 <= self.min_p <= 1.0):
            raise ValueError("min_p must be in [0, 1]")
        if self.top_k == 0 or self.top_k < -1:
            raise ValueError("top_k must be -1 (off) or >= 1")
        if not (-2.0 <= self.presence_penalty <= 2.0):
            raise ValueError("presence_penalty must be in [-2, 2]")
        if not (-2.0 <= self.frequency_penalty <= 2.0):
            raise ValueError("frequency_penalty must be in [-2, 2]")
        if self.repetition_penalty <= 0:
            raise ValueError("repetition_penalty must be > 0")
