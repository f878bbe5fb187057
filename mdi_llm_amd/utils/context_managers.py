"""Loop guards for the node runtimes.

Capability parity with the reference's ``catch_loop_errors``
(/root/reference/src/sub/utils/context_managers.py:16-56): convert
exceptions / Ctrl-C inside a serving loop into a clean event-based
shutdown instead of a stack trace and a wedged peer.
"""

from __future__ import annotations

import contextlib
import sys
import threading
from typing import Iterable, Optional

__all__ = ["catch_loop_errors"]


@contextlib.contextmanager
def catch_loop_errors(
    running_event: Optional[threading.Event] = None,
    stop_events: Iterable[threading.Event] = (),
    label: str = "loop",
):
    """On exception or KeyboardInterrupt: clear ``running_event``, set every
    event in ``stop_events``, log, and suppress KeyboardInterrupt (other
    exceptions propagate after cleanup)."""
    try:
        yield
    except KeyboardInterrupt:
        print(f"[{label}] interrupted — shutting down", file=sys.stderr)
        if running_event is not None:
            running_event.clear()
        for ev in stop_events:
            ev.set()
    except Exception:
        if running_event is not None:
            running_event.clear()
        for ev in stop_events:
            ev.set()
        raise
