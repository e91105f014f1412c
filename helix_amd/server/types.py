"""Control-plane types — behavioral parity with the reference's
api/pkg/types (types.go): AppHelixConfig / AssistantConfig (helix.yaml),
Session / Interaction, runner state. Pydantic instead of Go structs.
"""
from __future__ import annotations

import time
import uuid
from enum import Enum
from typing import Any, Dict, List, Optional

from pydantic import BaseModel, Field


def new_id(prefix: str) -> str:
    return f"{prefix}_{uuid.uuid4().hex[:24]}"


def now_ms() -> int:
    return int(time.time() * 1000)


# ---------------------------------------------------------------------------
# helix.yaml app schema (reference types.go:1561-1688, 1951-1986)
# ---------------------------------------------------------------------------

class ToolAPIConfig(BaseModel):
    name: str = ""
    description: str = ""
    schema_: str = Field("", alias="schema")   # OpenAPI spec (inline or ref)
    url: str = ""
    headers: Dict[str, str] = {}
    query: Dict[str, str] = {}
    oauth_provider: str = ""

    class Config:
        populate_by_name = True


class KnowledgeSource(BaseModel):
    name: str = ""
    description: str = ""
    rag_settings: Dict[str, Any] = {}
    source: Dict[str, Any] = {}     # {filestore|web|text: {...}}
    refresh_schedule: str = ""


class AgentModelConfig(BaseModel):
    """The 4-slot agent LLM config (reference llm_client.go:14-19)."""
    provider: str = ""
    model: str = ""
    reasoning_effort: str = ""      # low|medium|high ('' = n/a)


class AssistantTest(BaseModel):
    name: str = ""
    steps: List[Dict[str, str]] = []


class AssistantConfig(BaseModel):
    id: str = ""
    name: str = ""
    description: str = ""
    avatar: str = ""
    provider: str = ""
    model: str = ""
    agent_mode: bool = False
    agent_type: str = ""            # '' | 'helix' | 'zed_external'
    # 4 model slots (reference types.go:1583-1607)
    reasoning_model: AgentModelConfig = AgentModelConfig()
    generation_model: AgentModelConfig = AgentModelConfig()
    small_reasoning_model: AgentModelConfig = AgentModelConfig()
    small_generation_model: AgentModelConfig = AgentModelConfig()
    system_prompt: str = ""
    context_limit: int = 0
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    presence_penalty: Optional[float] = None
    frequency_penalty: Optional[float] = None
    max_tokens: Optional[int] = None
    reasoning_effort: str = ""
    knowledge: List[KnowledgeSource] = []
    apis: List[ToolAPIConfig] = []
    zapier: List[Dict[str, Any]] = []
    mcps: List[Dict[str, Any]] = []
    browser: Dict[str, Any] = {}
    web_search: Dict[str, Any] = {}
    calculator: Dict[str, Any] = {}
    email: Dict[str, Any] = {}
    memory: Dict[str, Any] = {}
    repository: Dict[str, Any] = {}
    project: Dict[str, Any] = {}
    sandbox: Dict[str, Any] = {}
    tests: List[AssistantTest] = []
    is_default: bool = False


class AppHelixConfig(BaseModel):
    name: str = ""
    description: str = ""
    avatar: str = ""
    assistants: List[AssistantConfig] = []
    triggers: List[Dict[str, Any]] = []
    secrets: Dict[str, str] = {}
    allowed_domains: List[str] = []


class App(BaseModel):
    id: str = Field(default_factory=lambda: new_id("app"))
    owner: str = ""
    owner_type: str = "user"
    organization_id: str = ""
    created: int = Field(default_factory=now_ms)
    updated: int = Field(default_factory=now_ms)
    config: AppHelixConfig = AppHelixConfig()
    global_: bool = Field(False, alias="global")

    class Config:
        populate_by_name = True


# ---------------------------------------------------------------------------
# Sessions (reference session types + response entries)
# ---------------------------------------------------------------------------

class InteractionState(str, Enum):
    WAITING = "waiting"
    EDITING = "editing"
    COMPLETE = "complete"
    ERROR = "error"


class Interaction(BaseModel):
    id: str = Field(default_factory=lambda: new_id("int"))
    session_id: str = ""
    created: int = Field(default_factory=now_ms)
    updated: int = Field(default_factory=now_ms)
    prompt_message: str = ""
    response_message: str = ""
    state: InteractionState = InteractionState.WAITING
    error: str = ""
    usage: Dict[str, int] = {}
    ttft_ms: int = 0
    duration_ms: int = 0


class Session(BaseModel):
    id: str = Field(default_factory=lambda: new_id("ses"))
    name: str = ""
    owner: str = ""
    parent_app: str = ""
    organization_id: str = ""
    created: int = Field(default_factory=now_ms)
    updated: int = Field(default_factory=now_ms)
    provider: str = ""
    model_name: str = ""
    type: str = "text"
    metadata: Dict[str, Any] = {}


# ---------------------------------------------------------------------------
# Runner / scheduler state (replaces reference types/runner.go; the deleted
# scheduler is reinstated HBM-aware per SURVEY.md §2.8)
# ---------------------------------------------------------------------------

class GPUStatus(BaseModel):
    index: int = 0
    vendor: str = "amd"
    arch: str = "cdna4"             # gfx950
    name: str = "MI355X"
    total_memory: int = 0           # bytes
    free_memory: int = 0
    used_memory: int = 0


class ModelStatus(BaseModel):
    model_id: str = ""
    state: str = "loading"          # loading|ready|evicting|error
    memory_bytes: int = 0
    gpu_indices: List[int] = []
    last_used: float = 0.0
    error: str = ""


class RunnerHeartbeat(BaseModel):
    runner_id: str = ""
    address: str = ""               # http base the control plane dials back
    gpus: List[GPUStatus] = []
    models: List[ModelStatus] = []
    ts: float = Field(default_factory=time.time)


class RunnerState(BaseModel):
    runner_id: str = ""
    address: str = ""
    status: str = "online"
    gpus: List[GPUStatus] = []
    models: List[ModelStatus] = []
    last_seen: float = Field(default_factory=time.time)


# ---------------------------------------------------------------------------
# OpenAI-compatible wire types (subset; the API is the contract)
# ---------------------------------------------------------------------------

class ChatMessage(BaseModel):
    role: str
    content: Any = ""
    name: Optional[str] = None
    tool_calls: Optional[List[Dict[str, Any]]] = None
    tool_call_id: Optional[str] = None


class ChatCompletionRequest(BaseModel):
    model: str = ""
    messages: List[ChatMessage] = []
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    max_tokens: Optional[int] = None
    n: int = 1
    stream: bool = False
    stop: Optional[Any] = None
    presence_penalty: Optional[float] = None
    frequency_penalty: Optional[float] = None
    seed: Optional[int] = None
    tools: Optional[List[Dict[str, Any]]] = None
    tool_choice: Optional[Any] = None
    user: Optional[str] = None


class EmbeddingRequest(BaseModel):
    model: str = ""
    input: Any = ""                 # str | List[str] | List[int]
    encoding_format: str = "float"


class LLMCall(BaseModel):
    id: str = Field(default_factory=lambda: new_id("llmc"))
    created: int = Field(default_factory=now_ms)
    session_id: str = ""
    interaction_id: str = ""
    owner: str = ""
    provider: str = ""
    model: str = ""
    step: str = ""
    request: Dict[str, Any] = {}
    response: Dict[str, Any] = {}
    duration_ms: int = 0
    first_token_ms: int = 0         # TTFT (reference openai_logger.go:249)
    prompt_tokens: int = 0
    completion_tokens: int = 0
    error: str = ""
