"""Env-var-driven configuration (the reference's scheme: .env is the schema,
reference: .env.example — no config files).  Engine knobs added per
SURVEY.md §5.6 'Rebuild'."""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Optional


def _b(name: str, default: bool) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v.strip().lower() in ("1", "true", "yes", "on")


def _i(name: str, default: int) -> int:
    try:
        return int(os.environ.get(name, default))
    except ValueError:
        return default


@dataclass
class Settings:
    # storage
    db_path: str = field(default_factory=lambda: os.environ.get("AURORA_DB", "aurora.db"))
    storage_dir: str = field(default_factory=lambda: os.environ.get("AURORA_STORAGE_DIR", "aurora_storage"))
    secrets_path: str = field(default_factory=lambda: os.environ.get("AURORA_SECRETS", "aurora_secrets.json"))
    # engine / models (reference: llm.py:32-67 ModelConfig)
    main_model: str = field(default_factory=lambda: os.environ.get("MAIN_MODEL", "tiny-llama"))
    guardrails_model: str = field(default_factory=lambda: os.environ.get("GUARDRAILS_LLM", ""))
    summarization_model: str = field(default_factory=lambda: os.environ.get("SUMMARIZATION_MODEL", ""))
    engine_device: str = field(default_factory=lambda: os.environ.get("ENGINE_DEVICE", "auto"))
    kv_memory_fraction: float = field(default_factory=lambda: float(os.environ.get("KV_MEMORY_FRACTION", "0.55")))
    max_num_seqs: int = field(default_factory=lambda: _i("ENGINE_MAX_SEQS", 64))
    tp_size: int = field(default_factory=lambda: _i("ENGINE_TP", 1))
    # production weights: HF-layout safetensors dir + tokenizer.json
    # (empty = random init / byte tokenizer, this environment's default)
    checkpoint_path: str = field(default_factory=lambda: os.environ.get("MODEL_CHECKPOINT", ""))
    tokenizer_path: str = field(default_factory=lambda: os.environ.get("MODEL_TOKENIZER", ""))
    # agent loop (reference: .env.example:111,160-161)
    agent_recursion_limit: int = field(default_factory=lambda: _i("AGENT_RECURSION_LIMIT", 240))
    max_input_tokens: int = field(default_factory=lambda: _i("MAX_INPUT_TOKENS", 20000))
    workflow_timeout_s: int = field(default_factory=lambda: _i("WORKFLOW_TIMEOUT", 1800))
    rca_max_turns: int = field(default_factory=lambda: _i("RCA_MAX_TURNS", 0))  # 0 = workflow default
    rca_skills_token_budget: int = field(default_factory=lambda: _i("RCA_SKILLS_TOKEN_BUDGET", 4000))
    # guardrails (reference: .env.example:123-128)
    guardrails_enabled: bool = field(default_factory=lambda: _b("GUARDRAILS_ENABLED", True))
    sigma_enabled: bool = field(default_factory=lambda: _b("GUARDRAILS_SIGMA_ENABLED", True))
    # orchestrator (reference: .env.example:141)
    orchestrator_enabled: bool = field(default_factory=lambda: _b("ORCHESTRATOR_ENABLED", False))
    # service
    api_port: int = field(default_factory=lambda: _i("AURORA_API_PORT", 5080))
    ws_port: int = field(default_factory=lambda: _i("AURORA_WS_PORT", 5006))
    internal_api_secret: str = field(default_factory=lambda: os.environ.get("INTERNAL_API_SECRET", ""))
    worker_concurrency: int = field(default_factory=lambda: _i("CELERY_CONCURRENCY", 4))
    # production deployments set this to reject webhooks with no configured
    # secret instead of merely auditing the unsigned acceptance
    require_webhook_secrets: bool = field(default_factory=lambda: _b("AURORA_REQUIRE_WEBHOOK_SECRETS", False))
    ws_rate_limit: int = field(default_factory=lambda: _i("WS_RATE_LIMIT", 5))  # msgs / 60 s
    rca_rate_window_s: int = field(default_factory=lambda: _i("RCA_RATE_WINDOW", 300))
    max_concurrent_rca: int = field(default_factory=lambda: _i("MAX_CONCURRENT_RCA", 256))
    # feature flags (reference: utils/flags/feature_flags.py:6-36)
    flags: dict = field(default_factory=lambda: {
        name: _b(f"FLAG_{name.upper()}", True)
        for name in ("ovh", "pagerduty_oauth", "confluence", "jira", "sharepoint",
                     "spinnaker", "incident_prevention", "visualization")
    })


_settings: Optional[Settings] = None


def settings() -> Settings:
    global _settings
    if _settings is None:
        _settings = Settings()
    return _settings


def reset_settings() -> None:
    global _settings
    _settings = None
