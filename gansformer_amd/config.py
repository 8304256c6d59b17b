"""Config system: EasyDict + helpers.

The reference drove everything through nested attribute-access dicts
(`dnnlib.EasyDict`, ref src/dnnlib/__init__.py [R] per SURVEY.md #2).
EasyDict IS the config system: train.py builds a cascade of these and the
training loop consumes them as kwargs.
"""

from __future__ import annotations

import copy


class EasyDict(dict):
    """dict with attribute access. The universal config object."""

    def __getattr__(self, name):
        try:
            return self[name]
        except KeyError as e:
            raise AttributeError(name) from e

    def __setattr__(self, name, value):
        self[name] = value

    def __delattr__(self, name):
        try:
            del self[name]
        except KeyError as e:
            raise AttributeError(name) from e

    def clone(self) -> "EasyDict":
        return copy.deepcopy(self)


def parse_comma_list(s):
    """'fid1k,ppl' -> ['fid1k', 'ppl']; passes lists through; '' / 'none' -> []."""
    if s is None:
        return []
    if isinstance(s, (list, tuple)):
        return list(s)
    s = str(s).strip()
    if s in ("", "none", "None"):
        return []
    return [t.strip() for t in s.split(",") if t.strip()]


def nearest_pow2(x: int) -> int:
    p = 1
    while p * 2 <= x:
        p *= 2
    return p


def res_log2(resolution: int) -> int:
    r = int(resolution)
    lg = r.bit_length() - 1
    if 2 ** lg != r or r < 4:
        raise ValueError(f"resolution must be a power of 2 >= 4, got {resolution}")
    return lg
