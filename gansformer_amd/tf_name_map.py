"""Best-effort mapping of TF-scoped reference checkpoint variable names
onto this framework's module tree.

The reference stored `variables` as TF graph-scoped names like
`G_synthesis/8x8/Conv0_up/weight` (ref src/dnnlib/tflib/network.py [R];
the snapshot itself contains no pickles to verify against, so this map
encodes the upstream StyleGAN2/GANsformer-TF1 naming scheme [R] and is
explicitly best-effort: mapped variables load with the correct
transposition, unmapped ones are reported, never silently dropped when
strict).

Shape conventions translated here:
  TF conv weight  [kh, kw, in, out] -> torch [out, in, kh, kw]
  TF dense weight [in, out]         -> torch [out, in]
  TF mod_weight   [w_dim, C]        -> affine.weight [C, w_dim]
"""

from __future__ import annotations

import re

import numpy as np


def _t_conv(a):
    return np.transpose(a, (3, 2, 0, 1)) if a.ndim == 4 else a


def _t_dense(a):
    return np.transpose(a, (1, 0)) if a.ndim == 2 else a


_G_CONV_LEAF = {
    "weight": ("weight", _t_conv),
    "bias": ("bias", None),
    "noise_strength": ("noise_strength", lambda a: np.reshape(a, (-1,))),
    "mod_weight": ("affine.weight", _t_dense),
    "mod_bias": ("affine.bias", None),
}

_D_CONV_LEAF = {
    "weight": ("weight", _t_conv),
    "bias": ("bias", None),
}


def map_tf_name(tf_name: str, top_res: int = None):
    """-> (our_name, transform_fn | None) or None if unmapped."""
    parts = tf_name.split("/")
    # ---- G mapping net ----
    m = re.fullmatch(r"G_mapping/Dense(\d+)/(weight|bias)", tf_name)
    if m:
        j, leaf = int(m.group(1)), m.group(2)
        fn = _t_dense if leaf == "weight" else None
        return f"mapping.layers.{j}.{leaf}", fn
    if tf_name in ("dlatent_avg", "G/dlatent_avg"):
        # TF: [w_dim]; ours: [num_latents, w_dim] -> broadcast
        return "mapping.w_avg", "broadcast_w_avg"
    # ---- G synthesis ----
    m = re.fullmatch(
        r"G_synthesis/(\d+)x\1/(Const|Conv|Conv0_up|Conv1|ToRGB)/(\w+)",
        tf_name)
    if m:
        res, unit, leaf = int(m.group(1)), m.group(2), m.group(3)
        bi = max(0, res.bit_length() - 3)  # 4->0, 8->1, 16->2, ...
        if unit == "Const":
            return (f"synthesis.blocks.{bi}.const", None) \
                if leaf == "const" else None
        if unit == "ToRGB":
            tgt = _G_CONV_LEAF.get(leaf)
            if tgt is None:
                return None
            return f"synthesis.blocks.{bi}.torgb.{tgt[0]}", tgt[1]
        conv = {"Conv": "conv1", "Conv0_up": "conv0", "Conv1": "conv1"}[unit]
        tgt = _G_CONV_LEAF.get(leaf)
        if tgt is None:
            return None
        return f"synthesis.blocks.{bi}.{conv}.{tgt[0]}", tgt[1]
    # ---- D ----
    m = re.fullmatch(r"D/(\d+)x\1/FromRGB/(weight|bias)", tf_name)
    if m:
        leaf = m.group(2)
        fn = _t_conv if leaf == "weight" else None
        return f"frgb.{leaf}", fn
    m = re.fullmatch(r"D/(\d+)x\1/(Conv0|Conv1_down|Skip)/(\w+)", tf_name)
    if m and top_res:
        res, unit, leaf = int(m.group(1)), m.group(2), m.group(3)
        bi = top_res.bit_length() - res.bit_length()  # top->0, halved -> +1
        name = {"Conv0": "conv0", "Conv1_down": "conv1", "Skip": "skip"}[unit]
        tgt = _D_CONV_LEAF.get(leaf)
        if tgt is None or (unit == "Skip" and leaf != "weight"):
            return (f"blocks.{bi}.skip.weight", _t_conv) \
                if unit == "Skip" and leaf == "weight" else None
        return f"blocks.{bi}.{name}.{tgt[0]}", tgt[1]
    m = re.fullmatch(r"D/4x4/(Conv|Dense0)/(weight|bias)", tf_name)
    if m:
        unit, leaf = m.group(1), m.group(2)
        if unit == "Conv":
            fn = _t_conv if leaf == "weight" else None
            return f"conv_out.{leaf}", fn
        fn = _t_dense if leaf == "weight" else None
        return f"fc.{leaf}", fn
    m = re.fullmatch(r"D/Output/(weight|bias)", tf_name)
    if m:
        leaf = m.group(1)
        fn = _t_dense if leaf == "weight" else None
        return f"out.{leaf}", fn
    return None


def map_tf_variables(variables, state_dict, top_res=None):
    """variables: list of (tf_name, ndarray). Returns
    (mapped: dict our_name -> ndarray, unmapped: list of tf names)."""
    mapped, unmapped = {}, []
    for name, arr in variables:
        hit = map_tf_name(name, top_res=top_res)
        if hit is None or hit[0] not in state_dict:
            unmapped.append(name)
            continue
        our, fn = hit
        a = np.array(arr)
        if fn == "broadcast_w_avg":
            tgt = state_dict[our]
            a = np.broadcast_to(a.reshape(1, -1), tuple(tgt.shape)).copy()
        elif fn is not None:
            a = fn(a)
        mapped[our] = a
    return mapped, unmapped
