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
        """Accumulate one observation of a scalar."""
        try:
            value = float(value)
        except (TypeError, ValueError):
            return
        s = self.accum.setdefault(name, [0.0, 0])
        s[0] += value
        s[1] += 1

    def means(self):
        return {k: v[0] / max(v[1], 1) for k, v in self.accum.items()}

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
