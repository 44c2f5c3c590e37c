"""Communication wall-clock timer.

Same surface as the reference's CommTimer singleton
(/root/reference/helper/timer/comm_timer.py): named (t0, t1) intervals with a
duplicate-name guard. In pipeline mode it measures only the *wait* time at
the staleness boundary — the point being that it goes to ~0 when overlap is
perfect. The HIP-event-based comm-stream busy time (for the overlap-% metric)
lives in Buffer.pop_comm_stats().
"""
import time
from contextlib import contextmanager


class CommTimer:
    def __init__(self):
        self._time = {}

    @contextmanager
    def timer(self, name):
        if name in self._time:
            raise RuntimeError(f"timer {name!r} already exists")
        t0 = time.time()
        yield
        self._time[name] = (t0, time.time())

    def tot_time(self):
        return sum(t1 - t0 for t0, t1 in self._time.values())

    def print_time(self):
        for k, (t0, t1) in self._time.items():
            print(f"Communication time of {k}: {t1 - t0} seconds.")

    def clear(self):
        self._time = {}


comm_timer = CommTimer()
