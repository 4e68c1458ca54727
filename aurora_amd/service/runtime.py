"""Process runtime: wires store + queue + engine + per-org services.

The composition root the API server, WS chatbot and background workers all
share (the reference splits this across main_compute.py, main_chatbot.py
and celery_config.py worker prewarm — reference: celery_config.py:272-304).
"""
from __future__ import annotations

import asyncio
import collections
import time
from typing import Any, Dict, List, Optional

from ..agent.llm import LLMManager, ModelConfig
from ..agent.tools.registry import GLOBAL_REGISTRY, ToolContext
from ..engine.engine import EngineConfig
from ..engine.server import AsyncEngineServer
from ..graph.infra_graph import InfraGraph
from ..rag.index import RagIndex
from ..utils.config import Settings, settings as get_settings
from .queue import TaskQueue
from .store import AdminStore, Database, OrgStore

# tool modules register themselves on import
from ..agent.tools import core as _core_tools  # noqa: F401


class EventBus:
    """In-process pub/sub feeding the SSE stream and WS status pushes
    (reference: incidents_sse.py:20,34)."""

    def __init__(self):
        self._subs: Dict[str, List[asyncio.Queue]] = collections.defaultdict(list)
        self.history: Dict[str, List[dict]] = collections.defaultdict(list)

    def publish(self, org_id: str, event: dict) -> None:
        event = dict(event, ts=time.time())
        hist = self.history[org_id]
        hist.append(event)
        del hist[:-200]
        for q in list(self._subs.get(org_id, [])):
            try:
                q.put_nowait(event)
            except asyncio.QueueFull:
                pass

    def subscribe(self, org_id: str) -> asyncio.Queue:
        q: asyncio.Queue = asyncio.Queue(maxsize=500)
        self._subs[org_id].append(q)
        return q

    def unsubscribe(self, org_id: str, q: asyncio.Queue) -> None:
        try:
            self._subs[org_id].remove(q)
        except ValueError:
            pass


class Runtime:
    def __init__(self, cfg: Optional[Settings] = None,
                 engine_server: Optional[AsyncEngineServer] = None,
                 engine_config: Optional[EngineConfig] = None):
        self.settings = cfg or get_settings()
        self.db = Database(self.settings.db_path)
        self.admin = AdminStore(self.db)
        self.queue = TaskQueue(self.db, concurrency=self.settings.worker_concurrency,
                               default_time_limit_s=self.settings.workflow_timeout_s)
        self.events = EventBus()
        if engine_server is None:
            ecfg = engine_config or EngineConfig(
                model=self.settings.main_model,
                device=self.settings.engine_device,
                kv_memory_fraction=self.settings.kv_memory_fraction,
                max_num_seqs=self.settings.max_num_seqs,
                tp_size=self.settings.tp_size,
                checkpoint_path=self.settings.checkpoint_path or None,
                tokenizer_path=self.settings.tokenizer_path or None,
            )
            engine_server = AsyncEngineServer(ecfg)
        self.engine = engine_server
        # per-purpose aux engines (reference ModelConfig maps MAIN /
        # SUMMARIZATION / GUARDRAILS to different models, llm.py:32-67;
        # 288 GB HBM co-hosts a small aux model beside the main one)
        self._aux_engines: Dict[str, AsyncEngineServer] = {}
        self._rag: Dict[str, RagIndex] = {}
        self._encoder = None  # None = unprobed, False = unavailable (CPU)
        self._embedding_rail = None
        self._graph: Dict[str, InfraGraph] = {}
        self._correlators: Dict[str, Any] = {}
        self._rca_rate: Dict[str, float] = {}  # user -> last background RCA ts
        self.kubectl_agents: Dict[str, Any] = {}  # cluster_id -> AgentConn
        from ..connectors.oauth import StateCache
        self.oauth_states = StateCache()  # one-shot OAuth CSRF states
        from . import worker as _worker  # registers task handlers
        _worker.register_tasks(self)

    # ---- per-org services ----
    def org(self, org_id: str) -> OrgStore:
        return self.admin.org_store(org_id)

    def engine_for(self, model: str) -> AsyncEngineServer:
        """Engine serving `model`: the main engine when the name matches
        (or is empty), else a lazily-started aux engine with a small KV
        slice of the remaining HBM."""
        if not model or model == self.settings.main_model:
            return self.engine
        srv = self._aux_engines.get(model)
        if srv is None:
            srv = AsyncEngineServer(EngineConfig(
                model=model, device=self.settings.engine_device,
                kv_memory_fraction=0.08, max_num_seqs=64))
            self._aux_engines[model] = srv
        return srv

    def llm(self, org_id: str) -> LLMManager:
        mc = ModelConfig(
            main_model=self.settings.main_model,
            guardrails_model=self.settings.guardrails_model,
            summarization_model=self.settings.summarization_model)
        return LLMManager(self.engine, mc, store=self.org(org_id),
                          engine_resolver=self.engine_for)

    def encoder(self):
        """Shared on-GPU sentence encoder for RAG + the embedding rail
        (the Weaviate text2vec sidecar replacement, SURVEY preamble table).
        Built lazily on first use and only when the engine runs on a GPU —
        CPU deployments fall back to BM25-only search and the hashed-BoW
        rail (a random-init fp32 encoder on CPU would cost seconds per
        query for meaningless vectors)."""
        if self._encoder is False:
            return None
        if self._encoder is None:
            import torch as _t
            dev = str(getattr(self.engine.engine, "device", "cpu"))
            if not dev.startswith("cuda"):
                self._encoder = False
                return None
            from ..models.encoder import SentenceEncoder
            enc = SentenceEncoder("bge-base", dtype=_t.bfloat16, device=dev)
            enc.init_random(3)  # no network: random-init, BASELINE.md
            self._encoder = enc
        return self._encoder

    def rag(self, org_id: str) -> RagIndex:
        if org_id not in self._rag:
            enc = self.encoder()
            self._rag[org_id] = RagIndex.from_store(
                self.org(org_id), encoder=enc,
                tokenizer=self.engine.tokenizer(),
                device=str(getattr(self.engine.engine, "device", "cpu"))
                if enc is not None else "cpu")
        return self._rag[org_id]

    def graph(self, org_id: str) -> InfraGraph:
        if org_id not in self._graph:
            self._graph[org_id] = InfraGraph(self.org(org_id))
        return self._graph[org_id]

    def correlator(self, org_id: str):
        """Per-org alert correlator (holds the per-incident embedding
        cache across webhooks). Embedding similarity rides the shared GPU
        encoder when opted in (AURORA_CORRELATOR_EMBED=1); default stays
        lexical+topology so the webhook hot path never blocks on encoder
        work."""
        if org_id not in self._correlators:
            import os as _os
            from .correlation import AlertCorrelator
            rag = (self.rag(org_id)
                   if _os.environ.get("AURORA_CORRELATOR_EMBED", "0") == "1"
                   else None)
            self._correlators[org_id] = AlertCorrelator(
                self.org(org_id), graph=self.graph(org_id), rag=rag)
        return self._correlators[org_id]

    def embedding_rail(self):
        """Shared prompt-injection embedding classifier (lazy singleton;
        guardrails/embedding_rail.py). AURORA_RAIL_BACKEND=encoder runs it
        on the shared GPU encoder (BASELINE config 3: classifier on every
        agent turn); default is the hashed-BoW backend — a random-init
        encoder makes similarity meaningless, so the encoder backend is
        opt-in for trained weights / perf measurement."""
        if self._embedding_rail is None:
            import os as _os
            from ..guardrails.embedding_rail import EmbeddingRail, hashed_bow_embed
            embed_fn = hashed_bow_embed
            threshold = 0.55
            if _os.environ.get("AURORA_RAIL_BACKEND", "") == "encoder":
                enc = self.encoder()
                if enc is not None:
                    tok = self.engine.tokenizer()

                    def embed_fn(texts):  # noqa: F811 - encoder backend
                        return enc.encode_ids(
                            [tok.encode(t)[:256] for t in texts])
                    threshold = 0.86
            self._embedding_rail = EmbeddingRail(embed_fn, threshold=threshold)
        return self._embedding_rail

    def tool_ctx(self, org_id: str, user_id: Optional[str] = None,
                 session_id: Optional[str] = None,
                 incident_id: Optional[str] = None, mode: str = "agent",
                 notifier=None, depth: int = 0) -> ToolContext:
        store = self.org(org_id)
        return ToolContext(
            org_store=store, user_id=user_id, session_id=session_id,
            incident_id=incident_id, llm=self.llm(org_id), mode=mode,
            rag=self.rag(org_id), graph=self.graph(org_id), queue=self.queue,
            notifier=notifier, settings=self.settings, depth=depth,
            extra={"kubectl_agents": self.kubectl_agents, "runtime": self,
                   "embedding_rail": self.embedding_rail()})

    # ---- RCA trigger rate limit (reference: task.py:336) ----
    def rca_allowed(self, user_key: str) -> bool:
        now = time.time()
        last = self._rca_rate.get(user_key, 0.0)
        if now - last < self.settings.rca_rate_window_s:
            return False
        self._rca_rate[user_key] = now
        return True

    def start(self) -> None:
        """Start workers + beats (call from a running event loop)."""
        # beat schedules (reference: celery_config.py:113-146)
        self.queue.add_beat("cleanup_stale_investigations", 300.0, {})
        self.queue.add_beat("action_scheduler", 60.0, {})
        self.queue.add_beat("run_prediscovery", 3600.0, {})
        self.queue.add_beat("cleanup_sandboxes", 600.0, {})  # reference: 10 min  # hourly (reference: celery_config.py:126-129)
        self.queue.start()
        # external MCP servers (reference: mcp_preloader started at boot,
        # main_chatbot.py:1684): spawn + register their tools when
        # AURORA_MCP_SERVERS is configured
        from ..agent.tools.mcp_client import MCPManager
        from ..utils.secrets import secrets_backend
        self.mcp = MCPManager(secrets=secrets_backend())
        if self.mcp.specs:
            asyncio.get_event_loop().create_task(
                self.mcp.register_tools(GLOBAL_REGISTRY))

    async def stop(self) -> None:
        await self.queue.stop()
        mcp = getattr(self, "mcp", None)
        if mcp is not None:
            await mcp.stop_all()
        for srv in self._aux_engines.values():
            srv.shutdown()
