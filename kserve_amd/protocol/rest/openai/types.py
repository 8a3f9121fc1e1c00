"""OpenAI API types (pydantic).

Reference parity: python/kserve protocol/rest/openai/types/openapi.py (generated
pydantic) + vLLM protocol re-exports. This is a fresh, compact definition of
the fields the serving paths use: /completions, /chat/completions, /embeddings,
/rerank, /models.
"""

from __future__ import annotations

import time
import uuid
from typing import Any, Dict, List, Literal, Optional, Union

from pydantic import BaseModel, Field


def _id(prefix: str) -> str:
    return f"{prefix}-{uuid.uuid4().hex}"


def _now() -> int:
    return int(time.time())


# ---------------------------------------------------------------------------
# Completions
# ---------------------------------------------------------------------------

class CompletionRequest(BaseModel):
    model: str
    prompt: Union[str, List[str], List[int], List[List[int]]]
    best_of: Optional[int] = None
    echo: Optional[bool] = False
    frequency_penalty: Optional[float] = 0.0
    logit_bias: Optional[Dict[str, float]] = None
    logprobs: Optional[int] = None
    max_tokens: Optional[int] = 16
    n: int = 1
    presence_penalty: Optional[float] = 0.0
    repetition_penalty: Optional[float] = 1.0
    seed: Optional[int] = None
    stop: Optional[Union[str, List[str]]] = None
    stream: Optional[bool] = False
    stream_options: Optional[Dict[str, Any]] = None
    suffix: Optional[str] = None
    temperature: Optional[float] = 1.0
    top_p: Optional[float] = 1.0
    top_k: Optional[int] = -1
    min_p: Optional[float] = 0.0
    # {"type": "json_object"} constrains output to valid JSON
    response_format: Optional[Dict[str, Any]] = None
    min_tokens: Optional[int] = 0
    ignore_eos: Optional[bool] = False
    # scheduling priority (vLLM extension: lower value runs first)
    priority: Optional[int] = 0
    user: Optional[str] = None


class CompletionLogprobs(BaseModel):
    text_offset: List[int] = Field(default_factory=list)
    token_logprobs: List[Optional[float]] = Field(default_factory=list)
    tokens: List[str] = Field(default_factory=list)
    top_logprobs: Optional[List[Optional[Dict[str, float]]]] = None


class CompletionChoice(BaseModel):
    index: int
    text: str
    logprobs: Optional[CompletionLogprobs] = None
    finish_reason: Optional[Literal["stop", "length", "abort"]] = None


class UsageInfo(BaseModel):
    prompt_tokens: int = 0
    completion_tokens: Optional[int] = 0
    total_tokens: int = 0


class Completion(BaseModel):
    id: str = Field(default_factory=lambda: _id("cmpl"))
    object: Literal["text_completion"] = "text_completion"
    created: int = Field(default_factory=_now)
    model: str = ""
    choices: List[CompletionChoice] = Field(default_factory=list)
    usage: Optional[UsageInfo] = None


# ---------------------------------------------------------------------------
# Chat completions
# ---------------------------------------------------------------------------

class ChatMessage(BaseModel):
    role: Literal["system", "user", "assistant", "tool"]
    content: Optional[Union[str, List[Dict[str, Any]]]] = None
    name: Optional[str] = None


class ChatCompletionRequest(BaseModel):
    model: str
    messages: List[ChatMessage]
    frequency_penalty: Optional[float] = 0.0
    logit_bias: Optional[Dict[str, float]] = None
    logprobs: Optional[bool] = False
    top_logprobs: Optional[int] = None
    max_tokens: Optional[int] = None
    max_completion_tokens: Optional[int] = None
    n: int = 1
    presence_penalty: Optional[float] = 0.0
    repetition_penalty: Optional[float] = 1.0
    seed: Optional[int] = None
    stop: Optional[Union[str, List[str]]] = None
    stream: Optional[bool] = False
    stream_options: Optional[Dict[str, Any]] = None
    response_format: Optional[Dict[str, Any]] = None
    temperature: Optional[float] = 1.0
    top_p: Optional[float] = 1.0
    top_k: Optional[int] = -1
    min_p: Optional[float] = 0.0
    min_tokens: Optional[int] = 0
    ignore_eos: Optional[bool] = False
    user: Optional[str] = None


class ChatCompletionChoiceMessage(BaseModel):
    role: Literal["assistant"] = "assistant"
    content: Optional[str] = None


class ChatCompletionChoice(BaseModel):
    index: int
    message: ChatCompletionChoiceMessage
    logprobs: Optional[Dict[str, Any]] = None
    finish_reason: Optional[Literal["stop", "length", "abort"]] = None


class ChatCompletion(BaseModel):
    id: str = Field(default_factory=lambda: _id("chatcmpl"))
    object: Literal["chat.completion"] = "chat.completion"
    created: int = Field(default_factory=_now)
    model: str = ""
    choices: List[ChatCompletionChoice] = Field(default_factory=list)
    usage: Optional[UsageInfo] = None


class ChatCompletionChunkDelta(BaseModel):
    role: Optional[str] = None
    content: Optional[str] = None


class ChatCompletionChunkChoice(BaseModel):
    index: int
    delta: ChatCompletionChunkDelta
    finish_reason: Optional[str] = None


class ChatCompletionChunk(BaseModel):
    id: str = Field(default_factory=lambda: _id("chatcmpl"))
    object: Literal["chat.completion.chunk"] = "chat.completion.chunk"
    created: int = Field(default_factory=_now)
    model: str = ""
    choices: List[ChatCompletionChunkChoice] = Field(default_factory=list)
    usage: Optional[UsageInfo] = None


# ---------------------------------------------------------------------------
# Embeddings / rerank / models
# ---------------------------------------------------------------------------

class EmbeddingRequest(BaseModel):
    model: str
    input: Union[str, List[str], List[int], List[List[int]]]
    encoding_format: Literal["float", "base64"] = "float"
    dimensions: Optional[int] = None
    user: Optional[str] = None


class EmbeddingObject(BaseModel):
    object: Literal["embedding"] = "embedding"
    index: int
    embedding: Union[List[float], str]


class Embedding(BaseModel):
    object: Literal["list"] = "list"
    data: List[EmbeddingObject] = Field(default_factory=list)
    model: str = ""
    usage: Optional[UsageInfo] = None


class RerankRequest(BaseModel):
    model: str
    query: str
    documents: List[str]
    top_n: Optional[int] = None
    return_documents: bool = True


class RerankResult(BaseModel):
    index: int
    relevance_score: float
    document: Optional[Dict[str, str]] = None


class Rerank(BaseModel):
    id: str = Field(default_factory=lambda: _id("rerank"))
    results: List[RerankResult] = Field(default_factory=list)
    usage: Optional[UsageInfo] = None


class ModelCard(BaseModel):
    id: str
    object: Literal["model"] = "model"
    created: int = Field(default_factory=_now)
    owned_by: str = "kserve-amd"


class ModelList(BaseModel):
    object: Literal["list"] = "list"
    data: List[ModelCard] = Field(default_factory=list)


class OpenAIErrorResponse(BaseModel):
    message: str
    type: str = "invalid_request_error"
    param: Optional[str] = None
    code: Optional[int] = None
