"""Run-dir management: numbered results dirs + stdout tee.

Capability parity with the reference's dnnlib.submit_run (ref
src/dnnlib/submit.py [R]): create `results/NNNNN-<desc>/`,
tee stdout/stderr to `log.txt`, record the submit config. The dotted-name
dispatch and cluster-submission machinery of the reference is dead weight
for a single-node framework and is not rebuilt; training calls the loop
directly.
"""

from __future__ import annotations

import json
import os
import re
import sys
import time


def create_run_dir(result_dir: str, desc: str) -> str:
    """Create the next numbered run dir `result_dir/NNNNN-desc`."""
    os.makedirs(result_dir, exist_ok=True)
    run_id = 0
    for name in os.listdir(result_dir):
        m = re.match(r"^(\d+)-", name)
        if m:
            run_id = max(run_id, int(m.group(1)) + 1)
    run_dir = os.path.join(result_dir, f"{run_id:05d}-{desc}")
    os.makedirs(run_dir)
    return run_dir


class Tee:
    """Mirror a text stream into a file (the reference's Logger [R])."""

    def __init__(self, stream, path, mode="a"):
        self.stream = stream
        self.file = open(path, mode, buffering=1)

    def write(self, data):
        self.stream.write(data)
        self.file.write(data)

    def flush(self):
        self.stream.flush()
        self.file.flush()

    def close(self):
        self.file.close()

    def isatty(self):
        return False


def tee_stdout(run_dir: str, fname: str = "log.txt"):
    """Redirect stdout+stderr through a tee into run_dir/log.txt."""
    path = os.path.join(run_dir, fname)
    sys.stdout = Tee(sys.__stdout__, path)
    sys.stderr = Tee(sys.__stderr__, path)


def save_submit_config(run_dir: str, cfg: dict):
    def default(o):
        try:
            json.dumps(o)
            return o
        except TypeError:
            return repr(o)

    with open(os.path.join(run_dir, "submit_config.json"), "w") as f:
        json.dump(cfg, f, indent=2, default=default)


class RunContext:
    """Progress heartbeat (parity with dnnlib.RunContext [R]).

    Tracks wall time and writes a small JSON heartbeat so an external
    watcher can see progress; `should_stop()` honours an abort file.
    """

    def __init__(self, run_dir: str | None = None, total_kimg: float = 0.0):
        self.run_dir = run_dir
        self.total_kimg = total_kimg
        self.start_time = time.time()
        self.last_update = self.start_time

    def update(self, cur_kimg: float = 0.0, **extra):
        self.last_update = time.time()
        if self.run_dir is not None:
            hb = dict(
                cur_kimg=cur_kimg,
                total_kimg=self.total_kimg,
                elapsed_sec=self.last_update - self.start_time,
                **extra,
            )
            try:
                with open(os.path.join(self.run_dir, "heartbeat.json"), "w") as f:
                    json.dump(hb, f)
            except OSError:
                pass

    def should_stop(self) -> bool:
        return self.run_dir is not None and os.path.exists(
            os.path.join(self.run_dir, "abort.txt")
        )

    def get_time_since_start(self) -> float:
        return time.time() - self.start_time
