"""`.pkl` network checkpoint layout, compatible with the reference's
`misc.save_pkl((G, D, Gs), 'network-snapshot-%06d.pkl')` (SURVEY.md §5
Checkpoint; ref src/dnnlib/tflib/network.py [R]).

Layout: a Python pickle of a 3-tuple (G, D, Gs); each element pickles as
a state dict with keys
    version            int
    name               str
    static_kwargs      dict of build kwargs (k, transformer variant,
                       resolution, ... live here)
    build_module_src   str   (source of the build function; written for
                       layout parity, NEVER executed on load)
    build_func_name    str
    variables          list of (var_name, np.ndarray)

SECURITY: loading uses a restricted Unpickler that maps any
tflib/dnnlib class to an inert state-capturing stub and refuses all
other non-allowlisted globals — unpickled code is never executed.
"""

from __future__ import annotations

import io
import pickle

import numpy as np
import torch

PKL_VERSION = 4

_G_FUNC_NAME = "training.networks.G_GANsformer"
_D_FUNC_NAME = "training.networks.D_GANsformer"


def network_state(module: torch.nn.Module, name: str, func_name: str,
                  static_kwargs: dict) -> dict:
    variables = []
    for k, v in module.state_dict().items():
        variables.append((k, v.detach().cpu().numpy()))
    import inspect
    try:
        src = inspect.getsource(type(module))
    except (OSError, TypeError):
        src = ""
    return {
        "version": PKL_VERSION,
        "name": name,
        "static_kwargs": dict(static_kwargs),
        "build_module_src": src,
        "build_func_name": func_name,
        "variables": variables,
    }


def save_network_pkl(path, G, D, Gs):
    """Write the 3-tuple (G, D, Gs) network pickle."""
    states = (
        network_state(G, "G", _G_FUNC_NAME, getattr(G, "init_kwargs", {})),
        network_state(D, "D", _D_FUNC_NAME, getattr(D, "init_kwargs", {})),
        network_state(Gs, "Gs", _G_FUNC_NAME, getattr(Gs, "init_kwargs", {})),
    )
    with open(path, "wb") as f:
        pickle.dump(states, f, protocol=2)


class NetworkStub:
    """Inert stand-in for tflib.Network-style objects: captures state."""

    def __setstate__(self, state):
        self.state = state

    def __getstate__(self):  # pragma: no cover
        return getattr(self, "state", {})


_ALLOWED = {
    ("collections", "OrderedDict"),
    ("numpy", "ndarray"),
    ("numpy", "dtype"),
    ("numpy.core.multiarray", "_reconstruct"),
    ("numpy.core.multiarray", "scalar"),
    ("numpy._core.multiarray", "_reconstruct"),
    ("numpy._core.multiarray", "scalar"),
    ("builtins", "dict"),
    ("builtins", "list"),
    ("builtins", "tuple"),
    ("builtins", "set"),
    ("builtins", "frozenset"),
    ("builtins", "bytearray"),
    ("builtins", "complex"),
    ("_codecs", "encode"),  # numpy array byte payloads in protocol-2 pickles
}


class _RestrictedUnpickler(pickle.Unpickler):
    def find_class(self, module, name):
        if (module, name) in _ALLOWED:
            return super().find_class(module, name)
        # Any dnnlib/tflib (or our own) network class becomes a stub —
        # never executes pickled code.
        if "dnnlib" in module or "tflib" in module or name in (
                "Network", "EasyDict", "NetworkStub"):
            if name == "EasyDict":
                from .config import EasyDict
                return EasyDict
            return NetworkStub
        raise pickle.UnpicklingError(
            f"refusing to unpickle {module}.{name} (not allowlisted)")


def load_network_states(path):
    """-> list of plain state dicts (see module docstring) for (G, D, Gs)."""
    with open(path, "rb") as f:
        obj = _RestrictedUnpickler(f, encoding="latin1").load()
    if isinstance(obj, (list, tuple)):
        items = list(obj)
    else:
        items = [obj]
    states = []
    for it in items:
        if isinstance(it, NetworkStub):
            states.append(dict(it.state))
        elif isinstance(it, dict):
            states.append(dict(it))
        else:
            raise ValueError(f"unexpected object in network pkl: {type(it)}")
    return states


def build_from_state(state: dict) -> torch.nn.Module:
    """Reconstruct a Generator/Discriminator from a pkl state dict."""
    from .models.networks import Discriminator, Generator

    func = state.get("build_func_name", "")
    kwargs = dict(state.get("static_kwargs", {}))
    if "G_" in func or "Generator" in func or state.get("name", "").startswith("G"):
        net = Generator(**kwargs)
    else:
        net = Discriminator(**kwargs)
    load_variables(net, state.get("variables", []))
    return net


def load_variables(module: torch.nn.Module, variables, strict=True):
    sd = module.state_dict()
    variables = list(variables)
    # Reference pickles carry TF graph-scoped names (G_synthesis/8x8/...)
    # with TF shape conventions; route those through the best-effort name
    # map (tf_name_map.py) instead of failing on every variable.
    n_scoped = sum(1 for n, _ in variables if "/" in n)
    if variables and n_scoped > len(variables) // 2:
        from .tf_name_map import map_tf_variables
        top_res = getattr(module, "img_resolution", None)
        mapped, unmapped = map_tf_variables(variables, sd, top_res=top_res)
        for name, arr in mapped.items():
            t = torch.as_tensor(np.array(arr))
            sd[name].copy_(t.to(sd[name].dtype).reshape(sd[name].shape))
        if unmapped and strict:
            raise KeyError(
                f"{len(unmapped)} TF-scoped pkl variables could not be "
                f"mapped: {unmapped[:5]}... (pass strict=False to load "
                f"the mapped subset)")
        return unmapped
    missing = []
    for name, arr in variables:
        if name in sd:
            t = torch.as_tensor(np.array(arr))
            sd[name].copy_(t.to(sd[name].dtype).reshape(sd[name].shape))
        else:
            missing.append(name)
    if missing and strict:
        raise KeyError(f"{len(missing)} pkl variables not found in module: "
                       f"{missing[:5]}...")
    return missing


def load_network_pkl(path):
    """-> (G, D, Gs) rebuilt modules."""
    states = load_network_states(path)
    assert len(states) == 3, f"expected (G, D, Gs) 3-tuple, got {len(states)}"
    return tuple(build_from_state(s) for s in states)
