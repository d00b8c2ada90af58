"""Generic utility functions (parity: reference utils.py:13-61)."""
from __future__ import annotations

import asyncio
import logging
from typing import Callable, Iterable, Optional, TypeVar

T = TypeVar("T")
_log = logging.getLogger(__file__)

__all__ = ["argmin_none_or_func", "get_useful_event_loop"]


def argmin_none_or_func(
    items: Iterable[Optional[T]],
    func: Callable[[T], float],
) -> Optional[int]:
    """Argmin of ``func`` over non-``None`` items; ``None`` if all are ``None``.

    Used by the balanced-connect logic to pick the least-loaded live worker
    while ignoring dead ones (reference utils.py:13-34).  Ties resolve to the
    earliest index, matching the shuffled-server-list semantics.
    """
    best_idx: Optional[int] = None
    best_val: Optional[float] = None
    for i, item in enumerate(items):
        if item is None:
            continue
        val = float(func(item))
        if best_val is None or val < best_val:
            best_idx, best_val = i, val
    return best_idx


def get_useful_event_loop() -> asyncio.AbstractEventLoop:
    """Like ``asyncio.get_event_loop()`` but usable from sync code everywhere.

    If called while a loop is already running (Jupyter, PyMC samplers), that
    loop is patched with ``nest_asyncio`` so a synchronous ``evaluate`` can
    still ``run_until_complete`` on it (reference utils.py:37-61).  On the
    GPU worker the same discipline applies in reverse: never block the event
    loop on a HIP synchronization -- the engine uses HIP events + executor
    threads instead.
    """
    loop = asyncio._get_running_loop()
    if loop is not None:
        if not hasattr(loop, "_nest_patched"):
            import nest_asyncio

            _log.debug("A loop is running here; applying nest_asyncio re-entrance patch.")
            nest_asyncio.apply(loop)
        return loop
    try:
        loop = asyncio.get_event_loop_policy().get_event_loop()
    except RuntimeError:
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
    if loop.is_closed():
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
    return loop
