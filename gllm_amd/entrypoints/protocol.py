"""OpenAI-compatible pydantic schemas (reference: entrypoints/protocol.py)."""

import time
import uuid
from typing import Any, Dict, List, Literal, Optional, Union

from pydantic import BaseModel, Field


def random_id(prefix: str) -> str:
    return f"{prefix}-{uuid.uuid4().hex[:24]}"


class FunctionCall(BaseModel):
    name: str
    arguments: str


class ToolCall(BaseModel):
    id: str = Field(default_factory=lambda: random_id("call"))
    type: Literal["function"] = "function"
    function: FunctionCall


class ChatMessage(BaseModel):
    role: str
    content: Optional[Union[str, List[Dict[str, Any]]]] = None
    tool_calls: Optional[List[ToolCall]] = None
    tool_call_id: Optional[str] = None
    name: Optional[str] = None


class FunctionDef(BaseModel):
    name: str
    description: Optional[str] = None
    parameters: Optional[Dict[str, Any]] = None


class ToolDef(BaseModel):
    type: Literal["function"] = "function"
    function: FunctionDef


class ChatCompletionRequest(BaseModel):
    model: str = ""
    messages: List[ChatMessage]
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    top_k: Optional[int] = None
    max_tokens: Optional[int] = None
    max_completion_tokens: Optional[int] = None
    min_tokens: Optional[int] = 0
    stream: bool = False
    stream_options: Optional[Dict[str, Any]] = None
    stop: Optional[Union[str, List[str]]] = None
    stop_token_ids: Optional[List[int]] = None
    n: int = 1
    seed: Optional[int] = None
    min_p: Optional[float] = None
    repetition_penalty: Optional[float] = None
    presence_penalty: Optional[float] = 0.0
    frequency_penalty: Optional[float] = 0.0
    ignore_eos: bool = False
    logprobs: Optional[bool] = False
    top_logprobs: Optional[int] = None
    tools: Optional[List[ToolDef]] = None
    tool_choice: Optional[Union[str, Dict[str, Any]]] = None
    chat_template_kwargs: Optional[Dict[str, Any]] = None
    add_generation_prompt: bool = True
    continue_final_message: bool = False
    prompt_logprobs: Optional[int] = None
    include_stop_str_in_output: bool = False
    logit_bias: Optional[Dict[str, float]] = None
    allowed_token_ids: Optional[List[int]] = None
    bad_words: Optional[List[str]] = None
    skip_special_tokens: bool = True
    truncate_prompt_tokens: Optional[int] = Field(default=None, ge=1)


class CompletionRequest(BaseModel):
    model: str = ""
    prompt: Union[str, List[str], List[int], List[List[int]]]
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    top_k: Optional[int] = None
    max_tokens: Optional[int] = 16
    min_tokens: Optional[int] = 0
    stream: bool = False
    stop: Optional[Union[str, List[str]]] = None
    stop_token_ids: Optional[List[int]] = None
    n: int = 1
    seed: Optional[int] = None
    min_p: Optional[float] = None
    repetition_penalty: Optional[float] = None
    presence_penalty: Optional[float] = 0.0
    frequency_penalty: Optional[float] = 0.0
    ignore_eos: bool = False
    echo: bool = False
    stream_options: Optional[Dict[str, Any]] = None
    logprobs: Optional[int] = None
    prompt_logprobs: Optional[int] = None
    include_stop_str_in_output: bool = False
    logit_bias: Optional[Dict[str, float]] = None
    allowed_token_ids: Optional[List[int]] = None
    bad_words: Optional[List[str]] = None
    skip_special_tokens: bool = True
    truncate_prompt_tokens: Optional[int] = Field(default=None, ge=1)


class UsageInfo(BaseModel):
    prompt_tokens: int = 0
    completion_tokens: int = 0
    total_tokens: int = 0


class ChatCompletionResponseChoice(BaseModel):
    index: int
    message: ChatMessage
    finish_reason: Optional[str] = None
    logprobs: Optional[Dict[str, Any]] = None


class ChatCompletionResponse(BaseModel):
    id: str = Field(default_factory=lambda: random_id("chatcmpl"))
    object: Literal["chat.completion"] = "chat.completion"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str = ""
    choices: List[ChatCompletionResponseChoice]
    usage: UsageInfo = Field(default_factory=UsageInfo)


class DeltaMessage(BaseModel):
    role: Optional[str] = None
    content: Optional[str] = None
    tool_calls: Optional[List[Dict[str, Any]]] = None


class ChatCompletionStreamChoice(BaseModel):
    index: int
    delta: DeltaMessage
    finish_reason: Optional[str] = None
    logprobs: Optional[Dict[str, Any]] = None


class ChatCompletionStreamResponse(BaseModel):
    id: str = Field(default_factory=lambda: random_id("chatcmpl"))
    object: Literal["chat.completion.chunk"] = "chat.completion.chunk"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str = ""
    choices: List[ChatCompletionStreamChoice]
    usage: Optional[UsageInfo] = None


class CompletionResponseChoice(BaseModel):
    index: int
    text: str
    finish_reason: Optional[str] = None
    logprobs: Optional[Dict[str, Any]] = None
    prompt_logprobs: Optional[List[Any]] = None


class CompletionResponse(BaseModel):
    id: str = Field(default_factory=lambda: random_id("cmpl"))
    object: Literal["text_completion"] = "text_completion"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str = ""
    choices: List[CompletionResponseChoice]
    usage: UsageInfo = Field(default_factory=UsageInfo)


class ModelCard(BaseModel):
    id: str
    object: Literal["model"] = "model"
    created: int = Field(default_factory=lambda: int(time.time()))
    owned_by: str = "gllm_amd"


class ModelList(BaseModel):
    object: Literal["list"] = "list"
    data: List[ModelCard] = []
