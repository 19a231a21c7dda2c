"""Observability: scalar metrics writer + rocprof-friendly trace ranges.

The reference logs scalars via tensorboardX (ref:train.py:552-554,
710-712).  tensorboardX is not in this image; MetricsWriter uses it when
importable and otherwise falls back to JSONL under the log dir (same
add_scalar API, so the engine does not care).

``trace_range`` emits roctx markers (via torch.cuda.nvtx, which maps to
roctx on ROCm) so rocprofv3 traces show named fwd/bwd/allreduce/opt
phases.
"""

import contextlib
import json
import os
import time

import torch

try:
    from tensorboardX import SummaryWriter as _TBWriter  # type: ignore
except ImportError:
    _TBWriter = None


class MetricsWriter:
    def __init__(self, log_dir):
        os.makedirs(log_dir, exist_ok=True)
        self._tb = _TBWriter(log_dir) if _TBWriter is not None else None
        self._f = open(os.path.join(log_dir, "metrics.jsonl"), "a")

    def add_scalar(self, tag, value, step):
        if self._tb is not None:
            self._tb.add_scalar(tag, value, step)
        self._f.write(json.dumps({
            "t": time.time(), "tag": tag,
            "value": float(value), "step": int(step)}) + "\n")
        self._f.flush()

    def close(self):
        if self._tb is not None:
            self._tb.close()
        self._f.close()


_TRACE = os.environ.get("BDBNN_TRACE", "0") == "1"


@contextlib.contextmanager
def trace_range(name):
    """roctx range around a phase; enabled with BDBNN_TRACE=1."""
    if _TRACE and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield
