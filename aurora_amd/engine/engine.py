"""The in-process LLM engine: continuous batching over a paged KV cache.

This replaces the reference's provider-API boundary (reference:
server/chat/backend/agent/llm.py:192 LLMManager.invoke — an HTTP call to a
rented model) with an engine the agent loop steps directly.  Concurrency
that the reference buys with Celery worker processes (celery_config.py:66-148)
is in-engine here: every concurrent investigation is a sequence in one
continuously-batched decode step.

Design (MI355X-first):
* packed varlen everywhere — no [B, S] padding;
* chunked prefill: new sequences stream into the batch without stalling
  running decodes; append-prefill reuses the session's cached KV pages
  (the in-engine analog of the reference's prompt-prefix cache,
  utils/prefix_cache.py:39-158);
* decode steps batch every running sequence into one kernel launch set;
* sampling supports grammar-guided tool-JSON decoding (engine/grammar.py).
"""
from __future__ import annotations

import codecs
import dataclasses
import itertools
import time
from collections import OrderedDict, deque
from dataclasses import dataclass
from typing import Deque, Dict, List, Optional, Tuple

import torch

from ..models.config import ModelSpec, get_spec
from ..models.llama import AttnMetadata, LlamaForCausalLM
from ..parallel import dist as pdist
from .kv_cache import PagedKVCache
from .sampling_params import SamplingParams
from .tokenizer import Tokenizer, get_tokenizer

WAITING, RUNNING, FINISHED = "waiting", "running", "finished"


@dataclass
class EngineConfig:
    model: str = "tiny-llama"
    dtype: str = "bfloat16"
    device: str = "auto"  # "auto" -> cuda:local_rank if available else cpu
    page_size: int = 0  # 0 -> 256 on GPU, 16 on CPU
    kv_cache_pages: int = 0  # 0 -> sized from kv_memory_fraction (GPU) / 512 pages (CPU)
    kv_memory_fraction: float = 0.55
    max_num_seqs: int = 64
    max_batched_tokens: int = 8192
    max_model_len: int = 32768
    tp_size: int = 1
    tokenizer_path: Optional[str] = None
    checkpoint_path: Optional[str] = None  # HF-layout safetensors dir
    seed: int = 0
    grammar_topk: int = 32  # candidates tried per step under a grammar
    use_hip_graphs: bool = True  # capture the decode forward per batch bucket
    # cross-request prefix caching: identical full prompt pages (the shared
    # system+skills prefix of concurrent investigations — measured 92% of a
    # first-turn RCA prompt) are computed once and refcount-shared.
    # AURORA_PREFIX_CACHE=0 is the kill switch.
    prefix_cache: bool = True


@dataclass
class StepOutput:
    request_id: str
    new_token_ids: List[int]
    new_text: str
    finished: bool
    finish_reason: Optional[str] = None


_SEQ_SERIAL = itertools.count()


def _chain_hash(prev: bytes, tokens: List[int]) -> bytes:
    """Rolling hash of one full page of prompt tokens, chained so a page's
    identity includes everything before it (vLLM-style block hashing)."""
    import hashlib
    from array import array
    h = hashlib.blake2b(prev, digest_size=16)
    h.update(array("q", tokens).tobytes())
    return h.digest()


class Sequence:
    def __init__(self, req_id: str, token_ids: List[int], params: SamplingParams,
                 session_id: Optional[str] = None):
        self.req_id = req_id
        # unique per Sequence OBJECT (req_id strings may be reused by callers)
        self.serial = next(_SEQ_SERIAL)
        self.session_id = session_id
        self.token_ids: List[int] = list(token_ids)
        self.num_prompt = len(token_ids)
        self.num_cached = 0  # tokens whose KV is already in cache pages
        self.pages: List[int] = []
        self.params = params
        self.status = WAITING
        self.finish_reason: Optional[str] = None
        self.out_bytes = bytearray()
        # streaming detokenizer: holds partial UTF-8 sequences across tokens
        self.text_dec = codecs.getincrementaldecoder("utf-8")(errors="replace")
        self.grammar = params.grammar.fork() if params.grammar is not None else None
        self.arrival = time.monotonic()
        self.first_token_time: Optional[float] = None
        # prefix-cache registration progress (engine._register_prefix)
        self.reg_pages = 0
        self.prefix_hash = b""
        # bumped whenever this seq's pages are freed (preemption/session evict)
        # so graph block-table staging keyed on (req_id, gen, npages) can never
        # reuse a stale row after the free list hands back different page ids
        self.page_gen = 0

    @property
    def num_tokens(self) -> int:
        return len(self.token_ids)

    @property
    def num_output(self) -> int:
        return len(self.token_ids) - self.num_prompt

    @property
    def prefill_target(self) -> int:
        """Tokens whose KV must be cached before this seq can decode.
        A preempted seq that already generated tokens re-prefills everything
        but its last token (which becomes the next decode input — no
        re-sampling)."""
        return self.num_prompt if self.num_output == 0 else self.num_tokens - 1


class LLMEngine:
    def __init__(self, config: EngineConfig, model: Optional[LlamaForCausalLM] = None,
                 tokenizer: Optional[Tokenizer] = None):
        self.config = config
        self.spec: ModelSpec = get_spec(config.model)
        use_cuda = torch.cuda.is_available() and config.device != "cpu"
        if config.device in ("auto", "cuda") and use_cuda:
            self.device = f"cuda:{torch.cuda.current_device()}"
        else:
            self.device = "cpu"
        self.dtype = getattr(torch, config.dtype) if self.device != "cpu" else torch.float32
        pdist.init_distributed(tp_size=config.tp_size)
        self.tokenizer = tokenizer or get_tokenizer(config.tokenizer_path)

        if model is None:
            model = LlamaForCausalLM(self.spec, dtype=self.dtype, device=self.device)
            if config.checkpoint_path:
                from ..models.loader import load_llama_safetensors
                load_llama_safetensors(model, config.checkpoint_path)
            else:
                model.init_random(config.seed)
        self.model = model.eval()

        page_size = config.page_size or (256 if self.device != "cpu" else 16)
        tp = pdist.tp_world_size()
        kv_heads_local = max(self.spec.n_kv_heads // tp, 1)
        # fp8 (e4m3) KV cache, env-gated (AURORA_KV_FP8=1): attention is at
        # the HBM stream limit in bf16 (6.15 TB/s measured), so fewer bytes
        # is the remaining lever — and the page budget doubles
        # (docs/ROUND2.md §3). Compute stays bf16; KV is converted at the
        # cache boundary by the HIP kernels.
        import os as _os
        kv_fp8 = (_os.environ.get("AURORA_KV_FP8", "0") == "1"
                  and self.device != "cpu")
        kv_dtype = torch.float8_e4m3fn if kv_fp8 else self.dtype
        kv_bytes = 1 if kv_fp8 else (2 if self.dtype != torch.float32 else 4)
        if config.kv_cache_pages:
            pages = config.kv_cache_pages
        elif self.device != "cpu":
            free, total = torch.cuda.mem_get_info()
            budget = int(free * config.kv_memory_fraction)
            pages = PagedKVCache.pages_for_memory(
                budget, self.spec.n_layers, kv_heads_local, self.spec.head_dim,
                page_size, kv_bytes,
            )
        else:
            pages = 512
        self.kv = PagedKVCache(
            self.spec.n_layers, kv_heads_local, self.spec.head_dim,
            page_size=page_size, num_pages=pages, device=self.device,
            dtype=kv_dtype,
        )
        self.max_model_len = min(config.max_model_len, self.spec.max_position)

        self.waiting: Deque[Sequence] = deque()
        self.running: List[Sequence] = []
        self.seqs: Dict[str, Sequence] = {}
        # session KV reuse: session_id -> finished Sequence kept warm in cache
        self.sessions: "OrderedDict[str, Sequence]" = OrderedDict()
        # cross-request prefix cache: chain-hash of full prompt pages ->
        # resident page id (refcount-shared across sequences). Entries die
        # with the page (kv.on_final_free).
        self.prefix_cache = (config.prefix_cache
                             and _os.environ.get("AURORA_PREFIX_CACHE", "1") == "1")
        self._prefix_table: Dict[bytes, int] = {}
        self._page_hash: Dict[int, bytes] = {}
        self.kv.on_final_free = self._retain_page
        self.kv.on_cache_evict = self._drop_page_hash
        self.gen = torch.Generator(device="cpu").manual_seed(config.seed)
        self._graph_cache: Dict[int, tuple] = {}
        self._graph_pool = None
        if self._graphs_enabled():
            # reserve the idle-row scratch page + pinned host staging now,
            # while the cache is guaranteed non-full
            mp = (self.max_model_len + self.kv.page_size - 1) // self.kv.page_size
            self._scratch_page = self.kv.alloc(1)[0]
            self._scratch_slot = self._scratch_page * self.kv.page_size
            maxb = self._BUCKETS[-1]
            self._host_i64 = torch.empty(maxb, dtype=torch.int64, pin_memory=True)
            self._host_i32 = torch.empty(maxb, 3, dtype=torch.int32, pin_memory=True)
            self._host_bt = torch.full((maxb, mp), self._scratch_page,
                                       dtype=torch.int32, pin_memory=True)
            self._bt_staged = [None] * maxb
        self._dec_staged_ids = None
        # counters (engine observability; surfaced by service.metrics)
        self.counters = {
            "prompt_tokens": 0, "generated_tokens": 0, "prefill_steps": 0,
            "decode_steps": 0, "preemptions": 0, "grammar_fallbacks": 0,
            "grammar_bucket_picks": 0, "prefix_cache_pages": 0,
            "steps_since_finish": 0,
        }
        # phase timers (seconds, cumulative) for perf triage
        self.timers = {"schedule": 0.0, "build": 0.0, "forward": 0.0,
                       "sample": 0.0, "prefill_forward": 0.0}

    # ------------------------------------------------------------------ API

    def add_request(self, request_id: str, prompt_token_ids: List[int],
                    params: SamplingParams, session_id: Optional[str] = None) -> None:
        if request_id in self.seqs:
            raise ValueError(f"duplicate request id {request_id}")
        if not prompt_token_ids:
            raise ValueError("empty prompt")
        v = self.spec.vocab_size
        if min(prompt_token_ids) < 0 or max(prompt_token_ids) >= v:
            # fail at the API boundary, not as an embedding IndexError deep
            # inside the stepping thread
            raise ValueError(f"prompt token id outside vocab [0, {v})")
        # hard length ceiling: the model's positions AND the cache itself —
        # a seq whose prompt+max_tokens exceeds total KV capacity can never
        # complete and would preempt-cycle forever (found by the seed sweep)
        reserved = 1 if self._graphs_enabled() else 0  # scratch page
        capacity = max((self.kv.num_pages - reserved) * self.kv.page_size,
                       self.kv.page_size)
        hard = min(self.max_model_len, capacity)
        if params.max_tokens >= hard:
            params = dataclasses.replace(params, max_tokens=max(hard - 1, 1))
        if len(prompt_token_ids) >= hard:
            prompt_token_ids = prompt_token_ids[-max(hard - params.max_tokens - 1, 1):]
        # clamp so prompt + output can never exceed the ceiling (the
        # per-step max_len check is the backstop; this keeps max_tokens honest)
        room = hard - len(prompt_token_ids)
        if params.max_tokens > room:
            params = dataclasses.replace(params, max_tokens=max(room, 1))
        seq = Sequence(request_id, prompt_token_ids, params, session_id)
        self.seqs[request_id] = seq
        self.waiting.append(seq)

    def _adopt_session(self, seq: Sequence) -> None:
        """Adopt the warm session KV at SCHEDULE time, not admission:
        admission-time adoption let every queued turn-N+1 request hoard
        unpreemptable pages and livelock the running seq. Pages stay in
        the evictable session tier until a seq is actually considered;
        at most ONE deferred seq holds adopted pages at a time (the
        headroom guard breaks the admission loop right after it)."""
        sid = seq.session_id
        if sid is None or sid not in self.sessions:
            return
        cached = self.sessions.pop(sid)
        match = 0
        limit = min(cached.num_cached, seq.num_prompt - 1)
        while match < limit and cached.token_ids[match] == seq.token_ids[match]:
            match += 1
        # adopt whole pages covered by the matching prefix
        keep_pages = match // self.kv.page_size
        if keep_pages > 0:
            seq.pages = cached.pages[:keep_pages]
            self.kv.free(cached.pages[keep_pages:])
            seq.num_cached = keep_pages * self.kv.page_size
        else:
            self.kv.free(cached.pages)

    # ------------------------------------------------- prefix cache (global)

    def _attach_prefix(self, seq: Sequence) -> None:
        """Adopt already-resident pages for the longest matching full-page
        prompt prefix. Never the page holding the final prompt token (at
        least one token must run to produce logits, and the seq must only
        ever write K/V into pages it owns exclusively)."""
        P = self.kv.page_size
        usable = (seq.num_prompt - 1) // P
        h = b""
        attached: List[int] = []
        for k in range(usable):
            h2 = _chain_hash(h, seq.token_ids[k * P:(k + 1) * P])
            pid = self._prefix_table.get(h2)
            if pid is None:
                break
            attached.append(pid)
            h = h2  # hash of exactly the attached pages
        if attached:
            self.kv.incref(attached)
            seq.pages = list(attached)
            seq.num_cached = len(attached) * P
            seq.reg_pages = len(attached)
            seq.prefix_hash = h
            self.counters["prefix_cache_pages"] += len(attached)

    def _register_prefix(self, seq: Sequence) -> None:
        """Publish this seq's freshly-prefilled full prompt pages so later
        requests with the same prefix can share them."""
        if not self.prefix_cache:
            return
        P = self.kv.page_size
        complete = min(seq.num_cached, seq.num_prompt) // P
        while seq.reg_pages < complete:
            k = seq.reg_pages
            seq.prefix_hash = _chain_hash(seq.prefix_hash,
                                          seq.token_ids[k * P:(k + 1) * P])
            pid = seq.pages[k]
            if (seq.prefix_hash not in self._prefix_table
                    and pid not in self._page_hash):
                self._prefix_table[seq.prefix_hash] = pid
                self._page_hash[pid] = seq.prefix_hash
            seq.reg_pages += 1

    def _retain_page(self, page: int) -> bool:
        """Last ref dropped: keep registered prefix pages warm (LRU tier,
        still counted free) so later identical prompts attach; others
        recycle immediately."""
        return page in self._page_hash

    def _drop_page_hash(self, page: int) -> None:
        """A retained page is being reclaimed for new content."""
        h = self._page_hash.pop(page, None)
        if h is not None and self._prefix_table.get(h) == page:
            del self._prefix_table[h]

    def abort_request(self, request_id: str) -> None:
        seq = self.seqs.pop(request_id, None)
        if seq is None:
            return
        if seq in self.waiting:
            self.waiting.remove(seq)
        if seq in self.running:
            self.running.remove(seq)
        self.kv.free(seq.pages)
        seq.pages = []
        seq.status = FINISHED
        seq.finish_reason = "aborted"

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def num_running(self) -> int:
        return len(self.running)

    # ------------------------------------------------------------- schedule

    def _evict_sessions(self, need_pages: int) -> None:
        while self.kv.num_free() < need_pages and self.sessions:
            _, old = self.sessions.popitem(last=False)
            self.kv.free(old.pages)
            old.pages = []

    def _release_waiting_pages(self, keep: Sequence) -> None:
        """Strip adopted/attached pages from WAITING seqs (they re-adopt or
        re-attach when scheduled). A deferred holder's pages must never cap
        a running seq below its need — that is a livelock (a waiting seq is
        not preemptable, so nothing else can reclaim them)."""
        for w in self.waiting:
            if w is keep or not w.pages:
                continue
            self.kv.free(w.pages)
            w.pages = []
            w.num_cached = 0
            w.reg_pages = 0
            w.prefix_hash = b""
            w.page_gen += 1

    def _try_reserve(self, seq: Sequence, up_to_tokens: int) -> bool:
        need = self.kv.pages_needed(up_to_tokens) - len(seq.pages)
        if need <= 0:
            return True
        self._evict_sessions(need)
        if self.kv.num_free() < need:
            self._release_waiting_pages(keep=seq)
        if self.kv.num_free() < need:
            return False
        seq.pages.extend(self.kv.alloc(need))
        return True

    def _preempt_lowest(self, exclude=()) -> bool:
        """Evict the most recently admitted running seq (not in ``exclude``)
        back to waiting, freeing its pages for recompute later."""
        for j in range(len(self.running) - 1, -1, -1):
            victim = self.running[j]
            if victim in exclude:
                continue
            self.running.pop(j)
            self.kv.free(victim.pages)
            victim.pages = []
            victim.num_cached = 0
            victim.reg_pages = 0
            victim.prefix_hash = b""
            victim.page_gen += 1
            victim.status = WAITING
            self.waiting.appendleft(victim)
            self.counters["preemptions"] += 1
            return True
        return False

    def _schedule(self) -> Tuple[str, List[Sequence], List[int]]:
        """Returns (mode, seqs, chunk_lens). Prefill takes priority so new
        investigations reach first token fast; decode batches all running."""
        budget = self.config.max_batched_tokens
        prefill: List[Sequence] = []
        chunks: List[int] = []
        while self.waiting and len(self.running) + len(prefill) < self.config.max_num_seqs:
            seq = self.waiting.popleft()
            # session adoption + prefix attach at SCHEDULE time, not
            # admission: waiting seqs must hold no pages, and a burst of
            # identical-prefix requests all arrives before any prefill has
            # registered pages — by the time the 2nd..Nth are scheduled,
            # the 1st has published the shared prefix
            if seq.num_cached == 0 and not seq.pages:
                self._adopt_session(seq)
                if self.prefix_cache and seq.num_cached == 0 and not seq.pages:
                    self._attach_prefix(seq)
            remaining = seq.prefill_target - seq.num_cached
            chunk = min(remaining, budget)
            if chunk <= 0:
                self.waiting.appendleft(seq)
                break
            # Admission never preempts running seqs (that thrashes); decode
            # growth below does the preempting. If nothing is running at all
            # and the seq still does not fit, it never will.
            # Decode headroom: admitting a seq that leaves no room for the
            # running batch to grow causes an immediate preemption storm
            # (prefix sharing makes admission look cheap on tiny caches) —
            # defer instead, unless nothing is running (progress guarantee).
            if self.running or prefill:
                need = self.kv.pages_needed(seq.num_cached + chunk) - len(seq.pages)
                headroom = len(self.running) + len(prefill) + 1
                if self.kv.num_free() < max(need, 0) + headroom:
                    self.waiting.appendleft(seq)
                    break
            reserved = self._try_reserve(seq, seq.num_cached + chunk)
            if not reserved:
                self.waiting.appendleft(seq)
                if not prefill and not self.running:
                    raise MemoryError(
                        f"sequence {seq.req_id} needs more KV pages than the whole cache "
                        f"({self.kv.num_pages} pages x {self.kv.page_size} tokens)"
                    )
                break
            prefill.append(seq)
            chunks.append(chunk)
            budget -= chunk
            if budget <= 0:
                break
        # decode: every running seq (not being prefilled) advances one token.
        # Prefill does NOT stall decodes: both regimes batch into one mixed
        # forward (projections share the packed batch; attention splits).
        in_prefill = set(prefill)
        decode: List[Sequence] = []
        for seq in list(self.running):
            if seq not in self.running or seq in in_prefill:
                continue
            ok = self._try_reserve(seq, seq.num_tokens + 1)
            while not ok:
                # evict others first; as a last resort evict this seq itself
                if self._preempt_lowest(exclude=set(decode) | in_prefill | {seq}):
                    ok = self._try_reserve(seq, seq.num_tokens + 1)
                else:
                    self._preempt_lowest(exclude=set(decode) | in_prefill)
                    break
            if ok:
                decode.append(seq)
        if prefill and decode:
            return "mixed", (prefill, decode), chunks
        if prefill:
            return "prefill", prefill, chunks
        return "decode", decode, []

    # -------------------------------------------------------------- tensors

    def _slot(self, seq: Sequence, pos: int) -> int:
        page = seq.pages[pos // self.kv.page_size]
        return page * self.kv.page_size + pos % self.kv.page_size

    def _block_table_tensor(self, seqs: List[Sequence]) -> torch.Tensor:
        maxp = max(len(s.pages) for s in seqs)
        bt = torch.zeros(len(seqs), maxp, dtype=torch.int32)
        for i, s in enumerate(seqs):
            if s.pages:
                bt[i, : len(s.pages)] = torch.tensor(s.pages, dtype=torch.int32)
        return bt.to(self.device)

    # ------------------------------------------------------------------ step

    @torch.inference_mode()
    def step(self) -> List[StepOutput]:
        _t0 = time.monotonic()
        mode, seqs, chunks = self._schedule()
        self.timers["schedule"] += time.monotonic() - _t0
        if not seqs:
            return []
        # thrash indicator: steps since ANY request finished. Steady growth
        # alongside climbing preemptions is the signature of the livelock
        # class fixed this round — surfaced so ops can see it, never acted on
        self.counters["steps_since_finish"] += 1
        dev = self.device
        if mode == "mixed":
            self._dec_staged_ids = None
            return self._step_mixed(*seqs, chunks)
        if mode == "prefill":
            self._dec_staged_ids = None
            self.counters["prefill_steps"] += 1
            ids: List[int] = []
            positions: List[int] = []
            slots: List[int] = []
            cu = [0]
            ctx_lens: List[int] = []
            for seq, chunk in zip(seqs, chunks):
                a = seq.num_cached
                b = a + chunk
                ids.extend(seq.token_ids[a:b])
                positions.extend(range(a, b))
                slots.extend(self._slot(seq, p) for p in range(a, b))
                cu.append(cu[-1] + chunk)
                ctx_lens.append(b)
            meta = AttnMetadata(
                mode="prefill",
                positions=torch.tensor(positions, dtype=torch.int32, device=dev),
                slot_mapping=torch.tensor(slots, dtype=torch.int32, device=dev),
                block_tables=self._block_table_tensor(seqs),
                context_lens=torch.tensor(ctx_lens, dtype=torch.int32, device=dev),
                cu_seqlens_q=torch.tensor(cu, dtype=torch.int32, device=dev),
            )
            input_ids = torch.tensor(ids, dtype=torch.long, device=dev)
            _tp = time.monotonic()
            hidden = self.model(input_ids, self.kv.caches, meta)
            if dev != "cpu":
                torch.cuda.synchronize()
            self.timers["prefill_forward"] += time.monotonic() - _tp
            outputs: List[StepOutput] = []
            sample_rows = []
            sample_seqs = []
            for i, (seq, chunk) in enumerate(zip(seqs, chunks)):
                had_output = seq.num_output > 0
                seq.num_cached += chunk
                self._register_prefix(seq)
                self.counters["prompt_tokens"] += chunk
                if seq.num_cached >= seq.prefill_target:
                    seq.status = RUNNING
                    if seq not in self.running:
                        self.running.append(seq)
                    if not had_output:  # preempted seqs resume decode, no re-sample
                        sample_rows.append(cu[i + 1] - 1)
                        sample_seqs.append(seq)
                else:
                    self.waiting.appendleft(seq)  # more chunks to go
            if sample_seqs:
                rows = hidden[torch.tensor(sample_rows, dtype=torch.long, device=dev)]
                logits = self.model.compute_logits(rows)
                outputs.extend(self._sample_and_append(sample_seqs, logits))
            return outputs

        # decode
        self.counters["decode_steps"] += 1
        _t1 = time.monotonic()
        if self._graphs_enabled() and len(seqs) <= self._BUCKETS[-1]:
            logits = self._decode_with_graph(seqs)
        else:
            self._dec_staged_ids = None
            ids = [s.token_ids[-1] for s in seqs]
            positions = [s.num_tokens - 1 for s in seqs]
            # note: the last token was appended by sampling but its KV is
            # not yet written; it is this step's input token.
            slots = [self._slot(s, s.num_tokens - 1) for s in seqs]
            ctx_lens = [s.num_tokens for s in seqs]
            meta = AttnMetadata(
                mode="decode",
                positions=torch.tensor(positions, dtype=torch.int32, device=dev),
                slot_mapping=torch.tensor(slots, dtype=torch.int32, device=dev),
                block_tables=self._block_table_tensor(seqs),
                context_lens=torch.tensor(ctx_lens, dtype=torch.int32, device=dev),
            )
            input_ids = torch.tensor(ids, dtype=torch.long, device=dev)
            hidden = self.model(input_ids, self.kv.caches, meta)
            logits = self.model.compute_logits(hidden)
        if dev != "cpu":
            torch.cuda.synchronize()
        _t2 = time.monotonic()
        self.timers["forward"] += _t2 - _t1
        for s in seqs:
            s.num_cached = s.num_tokens
        outs = self._sample_and_append(seqs, logits)
        self.timers["sample"] += time.monotonic() - _t2
        return outs

    @torch.inference_mode()
    def _step_mixed(self, prefill: List[Sequence], decode: List[Sequence],
                    chunks: List[int]) -> List[StepOutput]:
        """One forward for prefill chunks + decode tokens: the projections
        and norms run on the whole packed batch, attention splits per
        regime — decode streams no longer stall while new investigations
        prefill (this was ~30% of the bench step as a serial bubble)."""
        dev = self.device
        self.counters["prefill_steps"] += 1
        self.counters["decode_steps"] += 1
        ids: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        cu = [0]
        pre_ctx: List[int] = []
        for seq, chunk in zip(prefill, chunks):
            a = seq.num_cached
            b = a + chunk
            ids.extend(seq.token_ids[a:b])
            positions.extend(range(a, b))
            slots.extend(self._slot(seq, p) for p in range(a, b))
            cu.append(cu[-1] + chunk)
            pre_ctx.append(b)
        Tp = len(ids)
        ids.extend(s.token_ids[-1] for s in decode)
        positions.extend(s.num_tokens - 1 for s in decode)
        slots.extend(self._slot(s, s.num_tokens - 1) for s in decode)
        dec_ctx = [s.num_tokens for s in decode]
        meta = AttnMetadata(
            mode="mixed",
            positions=torch.tensor(positions, dtype=torch.int32, device=dev),
            slot_mapping=torch.tensor(slots, dtype=torch.int32, device=dev),
            block_tables=self._block_table_tensor(prefill),
            context_lens=torch.tensor(pre_ctx, dtype=torch.int32, device=dev),
            cu_seqlens_q=torch.tensor(cu, dtype=torch.int32, device=dev),
            num_prefill_tokens=Tp,
            dec_block_tables=self._block_table_tensor(decode),
            dec_context_lens=torch.tensor(dec_ctx, dtype=torch.int32, device=dev),
        )
        input_ids = torch.tensor(ids, dtype=torch.long, device=dev)
        _t1 = time.monotonic()
        hidden = self.model(input_ids, self.kv.caches, meta)
        # sample rows: completed prefills + every decode row
        sample_rows: List[int] = []
        sample_seqs: List[Sequence] = []
        for i, (seq, chunk) in enumerate(zip(prefill, chunks)):
            had_output = seq.num_output > 0
            seq.num_cached += chunk
            self._register_prefix(seq)
            self.counters["prompt_tokens"] += chunk
            if seq.num_cached >= seq.prefill_target:
                seq.status = RUNNING
                if seq not in self.running:
                    self.running.append(seq)
                if not had_output:
                    sample_rows.append(cu[i + 1] - 1)
                    sample_seqs.append(seq)
            else:
                self.waiting.appendleft(seq)
        for j, seq in enumerate(decode):
            sample_rows.append(Tp + j)
            sample_seqs.append(seq)
        rows = hidden[torch.tensor(sample_rows, dtype=torch.long, device=dev)]
        logits = self.model.compute_logits(rows)
        if dev != "cpu":
            torch.cuda.synchronize()
        self.timers["forward"] += time.monotonic() - _t1
        for seq in decode:
            seq.num_cached = seq.num_tokens
        _t2 = time.monotonic()
        outs = self._sample_and_append(sample_seqs, logits)
        self.timers["sample"] += time.monotonic() - _t2
        return outs

    # ----------------------------------------------------- hipGraph decode

    _BUCKETS = (1, 2, 4, 8, 16, 32, 64, 96, 128, 192, 256)

    def _graphs_enabled(self) -> bool:
        # MoE decode uses the fixed-capacity batched dispatch (static
        # shapes), so it is capturable too. TP decode is capturable as of
        # round 2: the RCCL communicator is initialized before capture (the
        # two warmup forwards in _capture_decode_graph run a real
        # all-reduce), and ROCm supports collectives inside hipGraph
        # capture when the communicator pre-exists (docs/ROUND2.md §2);
        # every TP rank captures/replays the same bucket in lockstep
        # because scheduling inputs are mirrored. AURORA_GRAPH_TP=0
        # restores the round-1 eager-TP behavior.
        if not (self.config.use_hip_graphs and self.device != "cpu"):
            return False
        if self.config.tp_size != 1:
            import os
            return os.environ.get("AURORA_GRAPH_TP", "1") != "0"
        return True

    def _decode_with_graph(self, seqs):
        """Replay a hipGraph-captured decode forward (per batch-size bucket,
        fixed-shape static buffers; idle rows decode a reserved scratch page
        and are ignored).  Removes the per-layer kernel-launch host time
        that dominates small-batch decode.

        Staging is incremental: when the batch membership is unchanged
        since the previous graph decode, position/slot/context rows each
        advance by exactly one token, so the pinned staging matrix gets a
        single vectorized += 1 instead of three B-element Python list
        builds (only page-boundary rows re-derive their slot); any
        prefill/mixed/eager step invalidates the staged state."""
        B = len(seqs)
        bucket = next(b for b in self._BUCKETS if b >= B)
        g = self._graph_cache.get(bucket)
        if g is None:
            g = self._capture_decode_graph(bucket)
            self._graph_cache[bucket] = g
        (graph, in_ids, in_pos, in_slot, in_bt, in_ctx, out_logits) = g
        mp = in_bt.shape[1]
        P = self.kv.page_size
        ids_t = self._host_i64
        ids_t[:B] = torch.tensor([s.token_ids[-1] for s in seqs],
                                 dtype=torch.int64)
        pos = self._host_i32
        batch_ids = [(s.serial, s.page_gen) for s in seqs]
        if batch_ids == self._dec_staged_ids:
            pos[:B] += 1  # position, slot and context each advance one token
            for i, s in enumerate(seqs):
                if (s.num_tokens - 1) % P == 0:  # slot jumped to a new page
                    pos[i, 1] = self._slot(s, s.num_tokens - 1)
        else:
            pos[:B, 0] = torch.tensor([s.num_tokens - 1 for s in seqs],
                                      dtype=torch.int32)
            pos[B:bucket, 0] = 0
            # note: the last token was appended by sampling but its KV is
            # not yet written; it is this step's input token.
            pos[:B, 1] = torch.tensor(
                [self._slot(s, s.num_tokens - 1) for s in seqs],
                dtype=torch.int32)
            pos[B:bucket, 1] = self._scratch_slot
            pos[:B, 2] = torch.tensor([s.num_tokens for s in seqs],
                                      dtype=torch.int32)
            pos[B:bucket, 2] = 1
            self._dec_staged_ids = batch_ids
        # block-table rows change only on page append / batch-slot change:
        # rewrite a staged row only when its (seq, npages) key moved
        bt_host = self._host_bt
        staged = self._bt_staged
        for i, s in enumerate(seqs):
            key = (s.serial, s.page_gen, len(s.pages))
            if staged[i] != key:
                n = min(len(s.pages), mp)
                bt_host[i, :n] = torch.tensor(s.pages[:n], dtype=torch.int32)
                bt_host[i, n:] = self._scratch_page
                staged[i] = key
        for i in range(len(seqs), bucket):
            if staged[i] is not None:
                bt_host[i].fill_(self._scratch_page)
                staged[i] = None
        in_ids.copy_(ids_t[:bucket], non_blocking=True)
        in_pos.copy_(pos[:bucket, 0], non_blocking=True)
        in_slot.copy_(pos[:bucket, 1], non_blocking=True)
        in_ctx.copy_(pos[:bucket, 2], non_blocking=True)
        in_bt.copy_(bt_host[:bucket], non_blocking=True)
        graph.replay()
        return out_logits[:B]

    def _capture_decode_graph(self, bucket: int):
        dev = self.device
        mp = (self.max_model_len + self.kv.page_size - 1) // self.kv.page_size
        if not hasattr(self, "_scratch_page"):
            self._scratch_page = self.kv.alloc(1)[0]
            self._scratch_slot = self._scratch_page * self.kv.page_size
            maxb = self._BUCKETS[-1]
            self._host_i64 = torch.empty(maxb, dtype=torch.int64, pin_memory=True)
            self._host_i32 = torch.empty(maxb, 3, dtype=torch.int32, pin_memory=True)
            self._host_bt = torch.full((maxb, mp), self._scratch_page,
                                       dtype=torch.int32, pin_memory=True)
            self._bt_staged = [None] * maxb
        self._dec_staged_ids = None
        in_ids = torch.zeros(bucket, dtype=torch.int64, device=dev)
        in_pos = torch.zeros(bucket, dtype=torch.int32, device=dev)
        in_slot = torch.full((bucket,), self._scratch_slot, dtype=torch.int32,
                             device=dev)
        in_bt = torch.full((bucket, mp), self._scratch_page, dtype=torch.int32,
                           device=dev)
        in_ctx = torch.ones(bucket, dtype=torch.int32, device=dev)
        meta = AttnMetadata(mode="decode", positions=in_pos, slot_mapping=in_slot,
                            block_tables=in_bt, context_lens=in_ctx)

        def fwd():
            hidden = self.model(in_ids, self.kv.caches, meta)
            return self.model.compute_logits(hidden)

        for _ in range(2):  # warmup outside capture
            fwd()
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        if self._graph_pool is None:
            with torch.cuda.graph(graph):
                out_logits = fwd()
            self._graph_pool = graph.pool()
        else:
            with torch.cuda.graph(graph, pool=self._graph_pool):
                out_logits = fwd()
        return (graph, in_ids, in_pos, in_slot, in_bt, in_ctx, out_logits)

    # --------------------------------------------------------------- sample

    def _grammar_pick(self, seq: Sequence, cand: List[int], row: torch.Tensor) -> int:
        """Walk candidate tokens in probability order; keep the first whose
        bytes keep the JSON FSM valid."""
        for tok in cand:
            b = self.tokenizer.token_bytes(tok)
            if tok == self.tokenizer.eos_id and seq.grammar.is_complete():
                return tok
            if not b:
                continue
            g = seq.grammar.fork()
            if g.advance(b):
                seq.grammar = g
                return tok
        # byte-constrained pick: when the FSM pins the legal next bytes
        # (literal phases of ToolCallGrammar, JSON punctuation states),
        # gather only tokens starting with those bytes — O(bucket) instead
        # of a full-vocab argsort. With the byte tokenizer a bucket is ONE
        # token, so this is O(|allowed|).
        allowed = None
        fn = getattr(seq.grammar, "allowed_next_bytes", None)
        if fn is not None:
            allowed = fn()
        if fn is not None:
            if allowed == b"":
                return self.tokenizer.eos_id
            self.counters["grammar_bucket_picks"] += 1
            buckets = self.tokenizer.first_byte_buckets()
            ids: List[int] = []
            if allowed is None:
                # unconstrained state (inside a string): every token with a
                # byte expansion is a candidate — for the byte tokenizer
                # that is 256 ids vs a full-vocab argsort
                for lst in buckets.values():
                    ids.extend(lst)
            else:
                for by in allowed:
                    ids.extend(buckets.get(by, ()))
            if ids:
                if len(ids) > 1:
                    t = torch.tensor(ids, device=row.device)
                    vals = row[t]
                    temp = seq.params.temperature
                    if temp > 0:  # gumbel-max keeps sampling stochastic
                        u = torch.rand(vals.shape, device=vals.device,
                                       generator=self._dev_gen(vals.device))
                        vals = (vals / max(temp, 1e-6)
                                - torch.log(-torch.log(u.clamp(min=1e-20))))
                    order = t[vals.argsort(descending=True)].tolist()
                else:
                    order = ids
                for tok in order:
                    g = seq.grammar.fork()
                    if g.advance(self.tokenizer.token_bytes(tok)):
                        seq.grammar = g
                        return tok
        # fallback: full-vocab scan in logit order (rare)
        self.counters["grammar_fallbacks"] += 1
        order = row.argsort(descending=True).tolist()
        for tok in order:
            b = self.tokenizer.token_bytes(tok)
            if not b:
                continue
            g = seq.grammar.fork()
            if g.advance(b):
                seq.grammar = g
                return tok
        return self.tokenizer.eos_id

    def _dev_gen(self, device) -> torch.Generator:
        if device.type == "cpu":
            return self.gen
        if getattr(self, "_cuda_gen", None) is None:
            self._cuda_gen = torch.Generator(device=device).manual_seed(
                self.config.seed)
        return self._cuda_gen

    def _batch_plain_sample(self, logits: torch.Tensor, seqs: List[Sequence]) -> torch.Tensor:
        """Fully-batched temperature/top-k/top-p sampling on the logits'
        device (gumbel-max for stochastic rows) — one sort + one argmax for
        the whole batch, no per-row python work, no host randoms."""
        B, V = logits.shape
        dev = logits.device
        all_greedy = all(s.params.temperature <= 0 for s in seqs)
        if all_greedy:
            return logits.argmax(-1).cpu()
        temps = torch.tensor([s.params.temperature for s in seqs], device=dev)
        greedy = temps <= 0
        r = logits / temps.clamp(min=1e-6).unsqueeze(1)
        top_ks = torch.tensor([s.params.top_k or V for s in seqs], device=dev)
        top_ps = torch.tensor([s.params.top_p for s in seqs], device=dev)
        # restrict to the top-K0 candidates: at temp <= 1 the nucleus mass
        # beyond rank 1024 is negligible, and topk beats a full-vocab sort
        # (the sort was 0.9 ms/step at V=128k, ~15% of the decode step)
        K0 = min(1024, V)
        vals, idx = r.topk(K0, dim=-1)  # sorted descending
        probs = vals.softmax(-1)
        cum = probs.cumsum(-1) - probs
        ranks = torch.arange(K0, device=dev).unsqueeze(0)
        cut = (cum >= top_ps.unsqueeze(1)) | (ranks >= top_ks.clamp(max=K0).unsqueeze(1))
        cut[:, 0] = False  # always keep the top token
        vals = vals.masked_fill(cut, float("-inf"))
        u = torch.rand(B, K0, device=dev, generator=self._dev_gen(dev))
        gumbel = -torch.log(-torch.log(u.clamp(min=1e-20)).clamp(min=1e-20))
        pick = (vals + gumbel).argmax(-1, keepdim=True)
        sampled = idx.gather(1, pick).squeeze(1)
        return torch.where(greedy, logits.argmax(-1), sampled).cpu()

    def _sample_and_append(self, seqs: List[Sequence], logits: torch.Tensor) -> List[StepOutput]:
        outputs = []
        lf = logits.float()
        now = time.monotonic()
        # plain rows: one batched device-side sample; grammar rows: top-k walk
        plain_idx = [i for i, s in enumerate(seqs) if s.grammar is None]
        gram_idx = [i for i, s in enumerate(seqs) if s.grammar is not None]
        toks: Dict[int, int] = {}
        if plain_idx:
            sel = lf[plain_idx] if len(plain_idx) < len(seqs) else lf
            ids = self._batch_plain_sample(sel, [seqs[i] for i in plain_idx])
            for j, i in enumerate(plain_idx):
                toks[i] = int(ids[j])
        if gram_idx:
            k = min(self.config.grammar_topk, lf.shape[1])
            sub = lf[gram_idx]
            seq_list = [seqs[i] for i in gram_idx]
            cand: List[List[int]] = []
            stochastic = any(s.params.temperature > 0 for s in seq_list)
            if stochastic:
                u = torch.rand(sub.shape, device=sub.device,
                               generator=self._dev_gen(sub.device))
                g = -torch.log(-torch.log(u.clamp(min=1e-20)).clamp(min=1e-20))
                temps = torch.tensor(
                    [max(s.params.temperature, 1e-6) for s in seq_list], device=sub.device
                ).unsqueeze(1)
                noisy = sub / temps + g
                cand_t = noisy.topk(k, dim=-1).indices.cpu()
            else:
                cand_t = sub.topk(k, dim=-1).indices.cpu()
            for j, i in enumerate(gram_idx):
                toks[i] = self._grammar_pick(seqs[i], cand_t[j].tolist(), sub[j])
        for i, seq in enumerate(seqs):
            tok = toks[i]
            seq.token_ids.append(tok)
            if seq.first_token_time is None:
                seq.first_token_time = now
            self.counters["generated_tokens"] += 1
            tb = self.tokenizer.token_bytes(tok)
            seq.out_bytes.extend(tb)
            # incremental UTF-8: a codepoint split across tokens (every
            # non-ASCII char under the byte tokenizer) must surface once
            # complete, not be dropped byte-by-byte
            new_text = seq.text_dec.decode(bytes(tb))
            finish = None
            p = seq.params
            if tok == self.tokenizer.eos_id and not p.ignore_eos:
                finish = "eos"
            elif tok in p.stop_token_ids:
                finish = "stop_token"
            elif seq.grammar is not None and seq.grammar.is_complete():
                finish = "grammar"
            elif seq.num_output >= p.max_tokens:
                finish = "max_tokens"
            elif seq.num_tokens >= self.max_model_len:
                # checked BEFORE stop strings so it can never be shadowed:
                # growing past max_position would index RoPE tables OOB
                finish = "max_len"
            elif p.stop:
                tail = seq.out_bytes[-256:].decode("utf-8", errors="ignore")
                for s in p.stop:
                    if s in tail:
                        finish = "stop"
                        break
            if finish:
                new_text += seq.text_dec.decode(b"", True)  # flush partial char
                self._finish(seq, finish)
            outputs.append(StepOutput(seq.req_id, [tok], new_text, bool(finish), finish))
        return outputs

    def _finish(self, seq: Sequence, reason: str) -> None:
        self.counters["steps_since_finish"] = 0
        seq.status = FINISHED
        seq.finish_reason = reason
        if seq in self.running:
            self.running.remove(seq)
        self.seqs.pop(seq.req_id, None)
        if seq.session_id is not None:
            seq.num_cached = seq.num_tokens - 1  # last token's KV not written
            old = self.sessions.pop(seq.session_id, None)
            if old is not None:
                self.kv.free(old.pages)
            self.sessions[seq.session_id] = seq
        else:
            self.kv.free(seq.pages)
            seq.pages = []

    # --------------------------------------------------------- convenience

    def generate(self, prompts: List[List[int]], params: SamplingParams,
                 max_steps: int = 100000) -> List[List[int]]:
        """Blocking batch generate (tests / bench)."""
        for i, p in enumerate(prompts):
            self.add_request(f"gen-{id(prompts)}-{i}", p, params)
        done: Dict[str, List[int]] = {}
        outs: Dict[str, List[int]] = {f"gen-{id(prompts)}-{i}": [] for i in range(len(prompts))}
        steps = 0
        while self.has_work() and steps < max_steps:
            for out in self.step():
                if out.request_id in outs:
                    outs[out.request_id].extend(out.new_token_ids)
            steps += 1
        return [outs[f"gen-{id(prompts)}-{i}"] for i in range(len(prompts))]
