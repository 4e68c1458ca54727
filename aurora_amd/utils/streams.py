"""Auxiliary HIP stream for service-side GPU work.

The engine's step loop enqueues decode forwards on the default stream
continuously; any service-side GPU call (guardrail embedding, RAG
encoder/kNN) issued on the default stream queues BEHIND those steps, and
its host sync (`.item()`, `.tolist()`) then waits out every queued engine
step — serializing the whole investigation pipeline on one stream. Aux
work has no data dependency on the engine, so it runs on its own stream
and syncs only itself (HIP streams overlap copies/compute — the platform
guideline's 'overlap collectives and copies with compute on separate HIP
streams').
"""
from __future__ import annotations

import contextlib

import torch

_aux_stream = None


def aux_stream():
    global _aux_stream
    if _aux_stream is None and torch.cuda.is_available():
        _aux_stream = torch.cuda.Stream()
    return _aux_stream


@contextlib.contextmanager
def on_aux_stream():
    """Run GPU work on the aux stream (no-op on CPU). The context exit
    does NOT synchronize — the caller's host read (.item()/.cpu()) syncs
    the aux stream only, never the engine's default stream."""
    s = aux_stream()
    if s is None:
        yield
        return
    with torch.cuda.stream(s):
        yield
