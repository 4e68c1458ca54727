"""LLMManager: the model registry + invoke surface for the agent stack.

The reference's ModelConfig maps purposes (MAIN/RCA/ORCHESTRATOR/SUBAGENT/
SUMMARIZATION/GUARDRAILS) to rented provider models (reference:
llm.py:32-67); here every purpose resolves to a purpose-tag over the shared
in-process engine (one engine per GPU; different purposes share weights,
which is the right MI355X design: 288 GB holds the model once, concurrency
comes from continuous batching, not from more model copies).  Usage is
tracked per call into llm_usage (reference: llm.py:270-350).
"""
from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass
from typing import Optional

from ..engine.sampling_params import SamplingParams
from ..engine.server import AsyncEngineServer, GenResult


@dataclass
class ModelConfig:
    main_model: str = "tiny-llama"
    rca_model: str = ""
    orchestrator_model: str = ""
    subagent_model: str = ""
    summarization_model: str = ""
    guardrails_model: str = ""
    vision_model: str = ""

    def resolve(self, purpose: str) -> str:
        return getattr(self, f"{purpose}_model", "") or self.main_model


class LLMManager:
    def __init__(self, server: AsyncEngineServer, config: Optional[ModelConfig] = None,
                 store=None, engine_resolver=None):
        self.server = server
        self.config = config or ModelConfig()
        self.store = store  # OrgStore for usage tracking (optional)
        # optional model-name -> engine mapping (per-purpose aux models;
        # reference llm.py:32-67 ModelConfig); default: everything on the
        # shared main engine
        self.engine_resolver = engine_resolver

    def _server_for(self, purpose: str) -> AsyncEngineServer:
        if self.engine_resolver is None:
            return self.server
        return self.engine_resolver(self.config.resolve(purpose))

    @property
    def tokenizer(self):
        return self.server.tokenizer()

    def count_tokens(self, text: str) -> int:
        return self.tokenizer.count(text)

    async def invoke(self, prompt: str, params: Optional[SamplingParams] = None,
                     purpose: str = "main", session_id: Optional[str] = None) -> GenResult:
        params = params or SamplingParams(temperature=0.2, max_tokens=512)
        t0 = time.monotonic()
        srv = self._server_for(purpose)
        # transient-failure retry (reference: agent.py:1031-1047 retries
        # network errors 3x; here the analog is an engine step fault that
        # failed the in-flight request while the stepping thread recovered)
        last: Optional[BaseException] = None
        for attempt in range(3):
            try:
                res = await srv.complete_text(prompt, params,
                                              session_id=session_id)
                break
            except (RuntimeError, TimeoutError) as e:
                last = e
                if attempt == 2:
                    raise
                await asyncio.sleep(0.5 * (attempt + 1))
        else:  # pragma: no cover
            raise last  # type: ignore[misc]
        if self.store is not None:
            self.store.track_usage(
                model=self.config.resolve(purpose),
                prompt_tokens=res.prompt_tokens,
                completion_tokens=len(res.token_ids),
                response_ms=(time.monotonic() - t0) * 1000.0,
                purpose=purpose, session_id=session_id)
        return res

    async def complete_text(self, prompt: str, params: SamplingParams,
                            session_id: Optional[str] = None) -> GenResult:
        """Raw passthrough (guardrail judges call this shape)."""
        return await self.invoke(prompt, params, purpose="guardrails",
                                 session_id=session_id)

    async def summarize(self, text: str, max_tokens: int = 256,
                        instruction: str = "Summarize the following:") -> str:
        """Tool-output summarisation with context capping
        (reference: llm.py:356-445)."""
        budget = 6000
        if self.count_tokens(text) > budget:
            # head+tail cap before summarising (reference: tool_output_cap.py:22)
            text = text[: budget * 2] + "\n...[truncated]...\n" + text[-budget:]
        res = await self.invoke(
            f"{instruction}\n---\n{text}\n---\nSummary:",
            SamplingParams(temperature=0.0, max_tokens=max_tokens, stop=["\n---"]),
            purpose="summarization")
        return res.text.strip()
