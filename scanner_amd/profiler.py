"""Profile: engine-interval analysis + Chrome trace export (parity:
python/scannerpy/profiler.py write_trace + statistics)."""
import json


class Profile:
    def __init__(self, profilers, node=0):
        # profilers: list (per pipeline instance) of
        # {"intervals": [(label, start_ns, end_ns)], "counters": {...}}
        self._profilers = profilers or []
        self._node = node

    def statistics(self):
        """Total time per interval label, across instances."""
        totals = {}
        counts = {}
        for p in self._profilers:
            for label, s, e in p["intervals"]:
                totals[label] = totals.get(label, 0) + (e - s)
                counts[label] = counts.get(label, 0) + 1
        return {
            label: {"total_ms": totals[label] / 1e6, "count": counts[label]}
            for label in totals
        }

    def counters(self):
        out = {}
        for p in self._profilers:
            for k, v in p.get("counters", {}).items():
                out[k] = out.get(k, 0) + v
        return out

    def write_trace(self, path):
        """Chrome trace (chrome://tracing / perfetto) with one thread per
        pipeline instance (parity: profiler.py:57-198)."""
        events = []
        for tid, p in enumerate(self._profilers):
            events.append({
                "name": "thread_name", "ph": "M", "pid": self._node,
                "tid": tid,
                "args": {"name": f"Pipeline[{tid}]"},
            })
            for label, s, e in p["intervals"]:
                events.append({
                    "name": label, "ph": "X", "pid": self._node, "tid": tid,
                    "ts": s / 1e3, "dur": (e - s) / 1e3,
                })
        with open(path, "w") as f:
            json.dump({"traceEvents": events}, f)
        return path
