"""OpenAI-compatible pydantic models (reference protocols.py parity)."""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from pydantic import BaseModel, ConfigDict, Field


class ModelCard(BaseModel):
    id: str
    object: str = "model"
    created: int = Field(default_factory=lambda: int(time.time()))
    owned_by: str = "production-stack-amd"
    parent: Optional[str] = None
    root: Optional[str] = None


class ModelList(BaseModel):
    object: str = "list"
    data: List[ModelCard] = Field(default_factory=list)


class ErrorResponse(BaseModel):
    object: str = "error"
    message: str
    type: str = "invalid_request_error"
    code: int = 400
    param: Optional[str] = None


class ChatMessage(BaseModel):
    role: str
    content: Any = None


class ChatCompletionRequest(BaseModel):
    model: str
    messages: List[ChatMessage]
    max_tokens: Optional[int] = None
    max_completion_tokens: Optional[int] = None
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    stream: bool = False
    stream_options: Optional[Dict[str, Any]] = None
    stop: Optional[Any] = None
    user: Optional[str] = None
    session_id: Optional[str] = None

    model_config = ConfigDict(extra="allow")


class CompletionRequest(BaseModel):
    model: str
    prompt: Any
    max_tokens: Optional[int] = None
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    stream: bool = False
    stream_options: Optional[Dict[str, Any]] = None
    user: Optional[str] = None

    model_config = ConfigDict(extra="allow")
