"""Query tracer — mirror of lib/querytracer (tracer.go:25 Tracer tree).

Same shape as the reference: a nil-safe tree of spans, activated
per-request; disabled tracers are no-ops with zero overhead.  The engine
adds per-kernel timings (hipEvent wall time from vmgpu_last_kernel_ms)
into the span that wraps each exec — the GPU analog of the reference's
per-stage children (eval.go:1931, netstorage.go:220).

Usage (mirrors querytracer.New / NewChild / Donef / ToJSON):

    qt = Tracer.new(enabled=True, "promql query %s", expr)
    child = qt.new_child("rollup %s() on %d series", func, n)
    out = batch.exec(plan, tracer=child)
    child.donef("kernel %.3f ms", engine.last_kernel_ms())
    qt.done()
    print(qt.to_json())
"""
import json
import time


class Tracer:
    """Nil-safe span tree; a None/disabled tracer swallows every call."""

    __slots__ = ("_msg", "_start", "_elapsed_s", "_children", "_done")

    def __init__(self, msg):
        self._msg = msg
        self._start = time.monotonic()
        self._elapsed_s = None
        self._children = []
        self._done = False

    @classmethod
    def new(cls, enabled, fmt, *args):
        """querytracer.New: returns None when tracing is disabled — every
        method on None is routed through the module-level nil-safe
        helpers below (mirroring Go's nil-receiver methods)."""
        if not enabled:
            return None
        return cls(fmt % args if args else fmt)

    def new_child(self, fmt, *args):
        child = Tracer(fmt % args if args else fmt)
        self._children.append(child)
        return child

    def donef(self, fmt, *args):
        self._msg = f"{self._msg}: {fmt % args if args else fmt}"
        self.done()

    def done(self):
        if not self._done:
            self._elapsed_s = time.monotonic() - self._start
            self._done = True

    def add_json(self, obj, fmt, *args):
        c = self.new_child(fmt, *args)
        c._children.append(obj)
        c.done()

    def _tree(self):
        out = {
            "message": self._msg,
            "duration_msec": round((self._elapsed_s
                                    if self._elapsed_s is not None
                                    else time.monotonic() - self._start)
                                   * 1e3, 3),
        }
        kids = [c._tree() if isinstance(c, Tracer) else c
                for c in self._children]
        if kids:
            out["children"] = kids
        return out

    def to_json(self):
        return json.dumps(self._tree())


# nil-safe helpers (Go nil-receiver method analogs) -------------------------

def new_child(qt, fmt, *args):
    return qt.new_child(fmt, *args) if qt is not None else None


def donef(qt, fmt, *args):
    if qt is not None:
        qt.donef(fmt, *args)


def done(qt):
    if qt is not None:
        qt.done()
