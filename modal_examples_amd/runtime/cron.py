"""Schedule runner: ``modal.Cron`` / ``modal.Period`` on deployed apps.

Reference: 05_scheduling/schedule_simple.py:27,34 (Period + Cron),
13_sandboxes/sandbox_pool.py:74 (cron pool maintenance).  ``app.deploy()``
starts a daemon thread per scheduled function; ``stop_schedules()`` (used at
teardown) signals them to exit at the next wakeup.
"""
from __future__ import annotations

import threading
import time

_started: dict = {}  # (app_name, fn_name) -> stop Event


def start_schedules(app) -> None:
    for name, fn in app.functions.items():
        sched = fn.opts.schedule
        if sched is None or (app.name, name) in _started:
            continue
        stop = threading.Event()
        _started[(app.name, name)] = stop
        t = threading.Thread(
            target=_loop, args=(fn, sched, stop), daemon=True,
            name=f"sched-{app.name}.{name}"
        )
        t.start()


def stop_schedules(app_name: str = None) -> None:
    """Stop schedule threads (all, or one app's) and allow re-deploys."""
    for key, ev in list(_started.items()):
        if app_name is None or key[0] == app_name:
            ev.set()
            _started.pop(key, None)


def _loop(fn, sched, stop: threading.Event):
    from ..app import Cron, Period

    if isinstance(sched, Period):
        while not stop.wait(sched.total_seconds):
            _fire(fn)
    elif isinstance(sched, Cron):
        last_min = None
        while not stop.wait(5):
            now = time.localtime()
            key = (now.tm_year, now.tm_yday, now.tm_hour, now.tm_min)
            if key != last_min and sched.matches(now):
                last_min = key
                _fire(fn)


def _fire(fn):
    try:
        fn.spawn()
    except Exception:
        import traceback

        traceback.print_exc()
