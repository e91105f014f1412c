"""Env-first server configuration (parity with the reference's envconfig
tree, api/pkg/config/config.go). Every knob has an env var + default."""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import List, Optional


def _env(name: str, default: str = "") -> str:
    return os.environ.get(name, default)


def _env_int(name: str, default: int) -> int:
    v = os.environ.get(name)
    return int(v) if v else default


def _env_bool(name: str, default: bool = False) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v.lower() in ("1", "true", "yes", "on")


@dataclass
class ProviderConfig:
    openai_api_key: str = field(default_factory=lambda: _env("OPENAI_API_KEY"))
    openai_base_url: str = field(
        default_factory=lambda: _env("OPENAI_BASE_URL",
                                     "https://api.openai.com/v1"))
    together_api_key: str = field(
        default_factory=lambda: _env("TOGETHER_API_KEY"))
    together_base_url: str = field(
        default_factory=lambda: _env("TOGETHER_BASE_URL",
                                     "https://api.together.xyz/v1"))
    anthropic_api_key: str = field(
        default_factory=lambda: _env("ANTHROPIC_API_KEY"))
    anthropic_base_url: str = field(
        default_factory=lambda: _env("ANTHROPIC_BASE_URL",
                                     "https://api.anthropic.com"))


@dataclass
class InferenceConfig:
    default_provider: str = field(
        default_factory=lambda: _env("HELIX_DEFAULT_PROVIDER", "helix"))
    default_model: str = field(
        default_factory=lambda: _env("HELIX_DEFAULT_MODEL", "llama3-8b"))
    # dispatch budget / client wait (reference helix_openai_server.go:260,
    # helix_openai_client.go:31 — 5 min / 180 s)
    dispatch_timeout_s: int = field(
        default_factory=lambda: _env_int("HELIX_DISPATCH_TIMEOUT", 300))
    client_timeout_s: int = field(
        default_factory=lambda: _env_int("HELIX_CLIENT_TIMEOUT", 180))


@dataclass
class RunnerPlaneConfig:
    # heartbeat cadence / staleness (reference config.go:75-92: 30 s beat,
    # 90 s dispatch-stale filter, 5 min offline)
    heartbeat_interval_s: int = field(
        default_factory=lambda: _env_int("HELIX_HEARTBEAT_INTERVAL", 30))
    dispatch_stale_s: int = field(
        default_factory=lambda: _env_int("HELIX_DISPATCH_STALE", 90))
    offline_after_s: int = field(
        default_factory=lambda: _env_int("HELIX_RUNNER_OFFLINE", 300))
    runner_token: str = field(
        default_factory=lambda: _env("HELIX_RUNNER_TOKEN", "runner-token"))
    local_runner: bool = field(
        default_factory=lambda: _env_bool("HELIX_LOCAL_RUNNER", False))
    local_runner_device: str = field(
        default_factory=lambda: _env("HELIX_LOCAL_RUNNER_DEVICE", "cuda:0"))


@dataclass
class RAGConfig:
    embeddings_provider: str = field(
        default_factory=lambda: _env("RAG_EMBEDDINGS_PROVIDER", "helix"))
    embeddings_model: str = field(
        default_factory=lambda: _env("RAG_EMBEDDINGS_MODEL", "bge-base"))
    chunk_size: int = field(
        default_factory=lambda: _env_int("RAG_CHUNK_SIZE", 512))
    chunk_overlap: int = field(
        default_factory=lambda: _env_int("RAG_CHUNK_OVERLAP", 64))
    results_count: int = field(
        default_factory=lambda: _env_int("RAG_RESULTS_COUNT", 4))
    distance_threshold: float = 0.0


@dataclass
class WebServerConfig:
    # a streaming turn with no partial-persist pulse for this long is
    # wedged (reference auto_wake_stuck_interactions)
    wedge_timeout_s: float = float(os.environ.get(
        "HELIX_WEDGE_TIMEOUT", "600"))
    host: str = field(default_factory=lambda: _env("SERVER_HOST", "0.0.0.0"))
    port: int = field(default_factory=lambda: _env_int("SERVER_PORT", 8080))
    admin_api_key: str = field(
        default_factory=lambda: _env("HELIX_ADMIN_API_KEY", "admin-key"))


@dataclass
class OIDCConfig:
    """OIDC login mode (reference api/pkg/auth/oidc.go OIDCConfig:
    KEYCLOAK/OIDC issuer + client credentials + allowed email
    domains)."""
    enabled: bool = field(
        default_factory=lambda: _env("OIDC_ENABLED", "") in
        ("1", "true", "yes"))
    issuer: str = field(default_factory=lambda: _env("OIDC_ISSUER", ""))
    client_id: str = field(
        default_factory=lambda: _env("OIDC_CLIENT_ID", "helix"))
    client_secret: str = field(
        default_factory=lambda: _env("OIDC_CLIENT_SECRET", ""))
    redirect_url: str = field(
        default_factory=lambda: _env("OIDC_REDIRECT_URL", ""))
    allowed_domains: str = field(
        default_factory=lambda: _env("OIDC_ALLOWED_DOMAINS", ""))


@dataclass
class StoreConfig:
    path: str = field(
        default_factory=lambda: _env("HELIX_STORE_PATH", "helix.db"))


@dataclass
class FileStoreConfig:
    path: str = field(
        default_factory=lambda: _env("HELIX_FILESTORE_PATH", "filestore"))


@dataclass
class ServerConfig:
    providers: ProviderConfig = field(default_factory=ProviderConfig)
    inference: InferenceConfig = field(default_factory=InferenceConfig)
    runner_plane: RunnerPlaneConfig = field(default_factory=RunnerPlaneConfig)
    rag: RAGConfig = field(default_factory=RAGConfig)
    web: WebServerConfig = field(default_factory=WebServerConfig)
    store: StoreConfig = field(default_factory=StoreConfig)
    oidc: OIDCConfig = field(default_factory=OIDCConfig)
    filestore: FileStoreConfig = field(default_factory=FileStoreConfig)
    # agent loop cap (reference agent.go:26 maxIterations default 10)
    agent_max_iterations: int = field(
        default_factory=lambda: _env_int("HELIX_AGENT_MAX_ITERATIONS", 10))
    # per-user daily token quota per provider (0 = unlimited; reference
    # api/pkg/quota + controller/balance_check.go)
    daily_token_limit: int = field(
        default_factory=lambda: _env_int("HELIX_DAILY_TOKEN_LIMIT", 0))


def load_config() -> ServerConfig:
    return ServerConfig()
