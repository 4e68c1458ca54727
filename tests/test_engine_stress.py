"""Randomized engine stress: arbitrary interleavings of add / abort /
step on a tiny-KV engine must preserve invariants — page accounting
never goes negative or leaks, every request terminates exactly once,
and the engine always drains (engine.py scheduler + kv_cache.py)."""
import random

import pytest

from aurora_amd.engine.engine import EngineConfig, LLMEngine
from aurora_amd.engine.sampling_params import SamplingParams


@pytest.mark.parametrize("seed", [1, 7, 23, 99, 1234])
def test_random_interleavings_preserve_invariants(seed):
    rng = random.Random(seed)
    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=0,
                                 kv_cache_pages=24, max_num_seqs=6))
    total_pages = eng.kv.num_free()
    finished = {}
    live = set()
    next_id = 0
    V = eng.spec.vocab_size

    for step in range(400):
        op = rng.random()
        if op < 0.25 and len(live) < 10:
            rid = f"r{next_id}"
            next_id += 1
            plen = rng.randint(1, 60)
            sess = f"s{rng.randint(0, 3)}" if rng.random() < 0.3 else None
            eng.add_request(rid, [rng.randint(2, V - 2) for _ in range(plen)],
                            SamplingParams(temperature=rng.choice([0.0, 0.8]),
                                           max_tokens=rng.randint(1, 8),
                                           ignore_eos=True),
                            session_id=sess)
            live.add(rid)
        elif op < 0.32 and live:
            victim = rng.choice(sorted(live))
            eng.abort_request(victim)
            finished[victim] = "aborted"
            live.discard(victim)
        else:
            for out in eng.step():
                if out.finished:
                    assert out.request_id not in finished, out.request_id
                    finished[out.request_id] = out.finish_reason
                    live.discard(out.request_id)
        assert eng.kv.num_free() >= 0

    # drain everything still in flight
    for _ in range(2000):
        if not eng.has_work():
            break
        for out in eng.step():
            if out.finished:
                assert out.request_id not in finished
                finished[out.request_id] = out.finish_reason
                live.discard(out.request_id)
    assert not eng.has_work()
    assert not live, live
    # all pages are free/retained or held by warm session caches (prefix
    # sharing: a page held by several sessions counts once)
    held = {p for s in eng.sessions.values() for p in s.pages}
    assert eng.kv.num_free() + len(held) == total_pages, (
        eng.kv.num_free(), len(held), total_pages)


def test_staging_key_uniquely_identifies_page_assignment():
    """The hipGraph block-table staging key (serial, page_gen, len(pages))
    must change whenever a sequence's page list changes identity — under
    preemption/eviction churn on a tiny KV (stale rows
    attended over freed pages)."""
    import random
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    rnd = random.Random(7)
    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=3,
                                 kv_cache_pages=12, max_num_seqs=6))
    key_to_pages = {}
    live = 0
    for step in range(400):
        if live < 5 and rnd.random() < 0.5:
            rid = f"pk-{step}"
            # short prompts admit cheaply; long decodes grow the running
            # batch past the page budget so decode-growth preemption fires
            eng.add_request(rid, [5] * rnd.randint(3, 10),
                            SamplingParams(max_tokens=rnd.randint(30, 60),
                                           ignore_eos=True),
                            session_id=f"s-{step % 3}")
            live += 1
        try:
            outs = eng.step()
        except MemoryError:
            break
        live -= sum(1 for o in outs if o.finished)
        for seq in list(eng.running) + list(eng.waiting):
            key = (seq.serial, seq.page_gen, len(seq.pages))
            pages = tuple(seq.pages)
            if key in key_to_pages:
                assert key_to_pages[key] == pages, (
                    f"staging key {key} reused for different pages "
                    f"{key_to_pages[key]} -> {pages}")
            else:
                key_to_pages[key] = pages
    assert eng.counters["preemptions"] > 0  # churn actually happened


def test_kv_page_accounting_invariant_fuzz():
    """Randomized add/step/abort/session churn on a tight page budget:
    after every operation, every page is owned by exactly one of
    {free list, a live sequence, a cached session, the scratch page} and
    the total equals num_pages (no leaks, no double ownership)."""
    import random

    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=3,
                                 kv_cache_pages=24, max_model_len=256,
                                 max_num_seqs=8))
    rnd = random.Random(42)
    n_added = 0

    def check(tag):
        from collections import Counter
        owners = Counter()
        if hasattr(eng, "_scratch_page"):
            owners[eng._scratch_page] += 1
        for s in eng.seqs.values():
            owners.update(s.pages)
        for s in eng.sessions.values():
            owners.update(s.pages)
        free = set(eng.kv._free)
        cached = set(eng.kv._cached)
        assert len(free) == len(eng.kv._free), f"{tag}: dup page in free list"
        assert not (set(owners) & free), f"{tag}: page both owned and free"
        assert not (set(owners) & cached), f"{tag}: page owned and cached"
        assert not (free & cached), f"{tag}: page free and cached"
        assert set(owners) | free | cached == set(range(eng.kv.num_pages)), (
            f"{tag}: pages leaked")
        # refcounts mirror ownership exactly (prefix sharing: rc = #owners)
        assert dict(owners) == eng.kv._rc, f"{tag}: refcount drift"
        # every retained page still has a live prefix-table identity
        for pg in cached:
            assert pg in eng._page_hash, f"{tag}: cached page without hash"

    # fixed pool so many prompts share full 16-token pages -> the global
    # prefix cache actually fires and the refcount invariant is exercised
    common = [rnd.randrange(1, 500) for _ in range(96)]

    for it in range(400):
        op = rnd.random()
        if op < 0.45 and len(eng.seqs) < 8:
            n_added += 1
            sess = f"s{rnd.randrange(6)}" if rnd.random() < 0.5 else None
            tail = [rnd.randrange(1, 500) for _ in range(rnd.randrange(1, 40))]
            prompt = (common[:16 * rnd.randrange(0, 7)] + tail
                      if rnd.random() < 0.6 else tail)
            try:
                eng.add_request(
                    f"r{n_added}", prompt,
                    SamplingParams(max_tokens=rnd.randrange(1, 20),
                                   ignore_eos=True),
                    session_id=sess)
            except MemoryError:
                pass  # admission may reject under pressure; nothing leaks
        elif op < 0.55 and eng.seqs:
            eng.abort_request(rnd.choice(list(eng.seqs)))
        elif eng.has_work():
            eng.step()
        check(f"iter {it}")
    # drain everything; all pages return except scratch + session cache
    while eng.has_work():
        eng.step()
        check("drain")
    for rid in list(eng.seqs):
        eng.abort_request(rid)
    check("end")


def test_session_reuse_equivalence_fuzz():
    """Randomized multi-turn sessions — extending, partially diverging,
    and fully replacing the cached prefix — must generate exactly what a
    cold engine does for every request (the keep_pages common-prefix
    carve in engine.add_request is the code under test)."""
    import random

    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    rnd = random.Random(7)
    cfg = EngineConfig(model="tiny-llama", device="cpu", seed=5,
                       kv_cache_pages=96, max_model_len=512)
    eng = LLMEngine(cfg)
    p = SamplingParams(temperature=0.0, max_tokens=4, ignore_eos=True)
    history = {}  # session -> last prompt
    n = 0
    for it in range(30):
        sess = f"s{rnd.randrange(4)}"
        prev = history.get(sess)
        if prev is None or rnd.random() < 0.25:
            prompt = [rnd.randrange(2, 500) for _ in range(rnd.randrange(4, 60))]
        elif rnd.random() < 0.6:
            # extend the cached turn (the common fast path)
            prompt = prev + [rnd.randrange(2, 500)
                             for _ in range(rnd.randrange(1, 40))]
        else:
            # diverge mid-prefix: only part of the cache is reusable
            keep = rnd.randrange(0, len(prev) + 1)
            prompt = prev[:keep] + [rnd.randrange(2, 500)
                                    for _ in range(rnd.randrange(1, 30))]
        history[sess] = prompt
        n += 1
        eng.add_request(f"r{n}", prompt, p, session_id=sess)
        got = []
        while eng.has_work():
            for o in eng.step():
                got.extend(o.new_token_ids)
        fresh = LLMEngine(cfg).generate([prompt], p)[0]
        assert got == fresh, (it, sess, len(prompt))


def test_turn_burst_oversubscribed_sessions_no_livelock():
    """Turn-2 burst: 8 sessions each re-submitting an extended prompt on a
    cache that fits only a few at once. Admission-time session adoption
    let the WAITING seqs hoard pages (they are never preemptable), so the
    one running seq could never grow and the engine preempt-thrashed
    forever. Schedule-time adoption keeps the pages in the evictable
    session tier; every request must finish."""
    import time

    import torch

    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu",
                                 max_num_seqs=10, max_batched_tokens=16384))
    rng = torch.Generator().manual_seed(1)
    p = SamplingParams(temperature=0.7, top_p=0.95, max_tokens=128,
                       ignore_eos=True)
    ctx = {}
    outs = {}
    for i in range(8):
        ctx[i] = torch.randint(0, 250, (768,), generator=rng).tolist()
        eng.add_request(f"t1-{i}", ctx[i], p, session_id=f"inc-{i}")
    while eng.has_work():
        for o in eng.step():
            outs.setdefault(o.request_id, []).extend(o.new_token_ids)
    for i in range(8):  # extended turn-2 prompts, all queued at once
        ctx[i].extend(outs[f"t1-{i}"][-128:])
        ctx[i].extend(torch.randint(0, 250, (512,), generator=rng).tolist())
        eng.add_request(f"t2-{i}", ctx[i], p, session_id=f"inc-{i}")
    done = 0
    t0 = time.time()
    while eng.has_work():
        assert time.time() - t0 < 120, (
            f"livelock: {done}/8 done, {eng.counters['preemptions']} preemptions")
        done += sum(1 for o in eng.step() if o.finished)
    assert done == 8


def test_request_exceeding_cache_capacity_is_clamped_and_completes():
    """A request whose prompt+max_tokens exceeds TOTAL KV capacity can
    never finish as asked — it used to preempt-cycle forever (found by a
    seed sweep). Admission must clamp to cache capacity so the request
    completes with what the cache can actually hold."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=2,
                                 kv_cache_pages=10, max_model_len=256))
    cap = 10 * eng.kv.page_size  # 160 token slots total
    # prompt 144 + max_tokens 24 = 168 > 160: the old engine livelocked
    eng.add_request("big", [5] * 144,
                    SamplingParams(max_tokens=24, ignore_eos=True))
    seq = eng.seqs["big"]
    assert seq.num_prompt + seq.params.max_tokens <= cap
    out = []
    for _ in range(500):
        for o in eng.step():
            out.extend(o.new_token_ids)
        if not eng.has_work():
            break
    assert not eng.has_work()
    assert len(out) == seq.params.max_tokens > 0
