"""Schedule runner: ``modal.Cron`` / ``modal.Period`` on deployed apps.

Reference: 05_scheduling/schedule_simple.py:27,34 (Period + Cron),
13_sandboxes/sandbox_pool.py:74 (cron pool maintenance).  ``app.deploy()``
starts a daemon thread that fires scheduled functions.
"""
from __future__ import annotations

import threading
import time

_started = set()


def start_schedules(app) -> None:
    for name, fn in app.functions.items():
        sched = fn.opts.schedule
        if sched is None or (app.name, name) in _started:
            continue
        _started.add((app.name, name))
        t = threading.Thread(
            target=_loop, args=(fn, sched), daemon=True, name=f"sched-{app.name}.{name}"
        )
        t.start()


def _loop(fn, sched):
    from ..app import Cron, Period

    if isinstance(sched, Period):
        while True:
            time.sleep(sched.total_seconds)
            _fire(fn)
    elif isinstance(sched, Cron):
        last_min = None
        while True:
            now = time.localtime()
            key = (now.tm_year, now.tm_yday, now.tm_hour, now.tm_min)
            if key != last_min and sched.matches(now):
                last_min = key
                _fire(fn)
            time.sleep(5)


def _fire(fn):
    try:
        fn.spawn()
    except Exception:
        import traceback

        traceback.print_exc()
