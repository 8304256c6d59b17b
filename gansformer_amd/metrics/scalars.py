"""Scalar logging: moving averages -> metrics.jsonl (+ TensorBoard when
available). Parity with the reference's autosummary (ref
src/dnnlib/tflib/autosummary.py [R]): collect scalars
anywhere, report smoothed values per tick.
"""

from __future__ import annotations

import json
import os
import time


class ScalarLogger:
    def __init__(self, run_dir=None, use_tensorboard=True):
        self.run_dir = run_dir
        self.accum = {}
        self.file = None
        self.tb = None
        if run_dir is not None:
            self.file = open(os.path.join(run_dir, "metrics.jsonl"), "a",
                             buffering=1)
            if use_tensorboard:
                try:
                    from torch.utils.tensorboard import SummaryWriter
                    self.tb = SummaryWriter(log_dir=run_dir)
                except Exception:
                    self.tb = None

    def report(self, name, value):
        """Accumulate one observation of a scalar.

        Tensors are stored as-is and only converted to floats in
        flush(), so reporting a GPU tensor costs no host-device sync on
        the hot path (one sync per tick instead of one per report).
        """
        self.accum.setdefault(name, []).append(value)

    def means(self):
        out = {}
        for k, vals in self.accum.items():
            total, n = 0.0, 0
            for v in vals:
                try:
                    total += float(v)
                    n += 1
                except (TypeError, ValueError):
                    pass
            out[k] = total / max(n, 1)
        return out

    def flush(self, step, **extra):
        """Write the averaged scalars for this tick and reset."""
        row = {"step": step, "time": time.time()}
        row.update(self.means())
        row.update(extra)
        if self.file is not None:
            self.file.write(json.dumps(row) + "\n")
        if self.tb is not None:
            for k, v in row.items():
                if isinstance(v, (int, float)) and k not in ("step", "time"):
                    self.tb.add_scalar(k, v, step)
        self.accum.clear()
        return row

    def close(self):
        if self.file is not None:
            self.file.close()
        if self.tb is not None:
            self.tb.close()
