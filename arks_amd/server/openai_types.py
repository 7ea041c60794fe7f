"""OpenAI API request/response schemas (the subset the reference gateway
parses — reference handle_request.go:87-104 reads {model, stream,
stream_options.include_usage}; handle_response.go:113-182 reads usage)."""

from __future__ import annotations

import time
import uuid
from typing import Any, Literal

from pydantic import BaseModel, Field


class StreamOptions(BaseModel):
    include_usage: bool = False


class ChatMessage(BaseModel):
    role: str
    content: str | None = ""


class ChatCompletionRequest(BaseModel):
    model: str
    messages: list[ChatMessage]
    max_tokens: int | None = None
    max_completion_tokens: int | None = None
    temperature: float = 1.0
    top_p: float = 1.0
    n: int = 1
    stream: bool = False
    stream_options: StreamOptions | None = None
    stop: list[str] | str | None = None
    logprobs: bool = False
    top_logprobs: int = 0
    top_k: int = 0  # extension (vLLM-compatible)
    min_p: float = 0.0  # extension (vLLM-compatible)
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0  # extension (vLLM-compatible)
    seed: int | None = None
    ignore_eos: bool = False  # extension (load testing)
    logit_bias: dict[str, float] | None = None
    min_tokens: int = 0  # extension (vLLM-compatible)


class CompletionRequest(BaseModel):
    model: str
    prompt: str | list[str] | list[int] | list[list[int]]
    max_tokens: int | None = 16
    temperature: float = 1.0
    top_p: float = 1.0
    n: int = 1
    stream: bool = False
    stream_options: StreamOptions | None = None
    stop: list[str] | str | None = None
    logprobs: int | None = None
    top_k: int = 0
    min_p: float = 0.0
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    seed: int | None = None
    ignore_eos: bool = False
    logit_bias: dict[str, float] | None = None
    min_tokens: int = 0  # extension (vLLM-compatible)
    echo: bool = False  # return the prompt (+ its logprobs) too


class Usage(BaseModel):
    prompt_tokens: int = 0
    completion_tokens: int = 0
    total_tokens: int = 0


class ChatChoice(BaseModel):
    index: int = 0
    message: ChatMessage | None = None
    finish_reason: str | None = None
    logprobs: dict | None = None


class ChatDeltaChoice(BaseModel):
    index: int = 0
    delta: dict[str, Any] = Field(default_factory=dict)
    logprobs: dict[str, Any] | None = None  # {"content": [...]} when requested
    finish_reason: str | None = None


class ChatCompletionResponse(BaseModel):
    id: str = Field(default_factory=lambda: f"chatcmpl-{uuid.uuid4().hex}")
    object: Literal["chat.completion"] = "chat.completion"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str = ""
    choices: list[ChatChoice] = Field(default_factory=list)
    usage: Usage | None = None


class ChatCompletionChunk(BaseModel):
    id: str = ""
    object: Literal["chat.completion.chunk"] = "chat.completion.chunk"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str = ""
    choices: list[ChatDeltaChoice] = Field(default_factory=list)
    usage: Usage | None = None


class CompletionChoice(BaseModel):
    index: int = 0
    text: str = ""
    finish_reason: str | None = None
    logprobs: dict | None = None


class CompletionResponse(BaseModel):
    id: str = Field(default_factory=lambda: f"cmpl-{uuid.uuid4().hex}")
    object: Literal["text_completion"] = "text_completion"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str = ""
    choices: list[CompletionChoice] = Field(default_factory=list)
    usage: Usage | None = None


class ModelCard(BaseModel):
    id: str
    object: Literal["model"] = "model"
    created: int = Field(default_factory=lambda: int(time.time()))
    owned_by: str = "arks_amd"


class ModelList(BaseModel):
    object: Literal["list"] = "list"
    data: list[ModelCard] = Field(default_factory=list)


class ErrorResponse(BaseModel):
    error: dict[str, Any]

    @classmethod
    def make(cls, message: str, code: int, type_: str = "invalid_request_error"):
        return cls(error={"message": message, "type": type_, "code": code})
