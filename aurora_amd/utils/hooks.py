"""Extensibility hooks loaded from an external module (reference:
utils/hooks.py:55-90 — before_llm_call / after_llm_call / before_add_member /
report_usage / verify_entitlement, gated at main_chatbot.py:1494-1505 and
task.py:677-697).  Set AURORA_HOOKS_MODULE to a dotted module path exposing
any subset of the hook names."""
from __future__ import annotations

import importlib
import os
from typing import Any, Callable, Dict, Optional

HOOK_NAMES = ("before_llm_call", "after_llm_call", "before_add_member",
              "report_usage", "verify_entitlement")

_hooks: Optional[Dict[str, Callable]] = None


def load_hooks(module_path: Optional[str] = None) -> Dict[str, Callable]:
    global _hooks
    if _hooks is not None and module_path is None:
        return _hooks
    path = module_path or os.environ.get("AURORA_HOOKS_MODULE", "")
    hooks: Dict[str, Callable] = {}
    if path:
        try:
            mod = importlib.import_module(path)
            for name in HOOK_NAMES:
                fn = getattr(mod, name, None)
                if callable(fn):
                    hooks[name] = fn
        except ImportError:
            pass
    _hooks = hooks
    return hooks


def reset_hooks() -> None:
    global _hooks
    _hooks = None


async def fire(name: str, *args, default: Any = True, **kwargs) -> Any:
    """Run a hook if installed; async or sync; failures return the default
    (hook errors must never take down an investigation)."""
    fn = load_hooks().get(name)
    if fn is None:
        return default
    try:
        res = fn(*args, **kwargs)
        if hasattr(res, "__await__"):
            res = await res
        return res
    except Exception:  # noqa: BLE001
        return default
