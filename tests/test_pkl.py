"""Checkpoint layout: pkl write/read round-trip, layout structure, and
the never-execute-unpickled-code guarantee."""

import io
import pickle

import numpy as np
import pytest
import torch

from gansformer_amd import pkl_compat
from gansformer_amd.models.networks import Discriminator, Generator


def tiny_nets():
    G = Generator(z_dim=32, w_dim=32, img_resolution=16, num_components=2,
                  transformer="simplex", channel_base=512, channel_max=64,
                  bf16_res_count=0, mapping_layers=1,
                  attn_resolutions=[8])
    D = Discriminator(img_resolution=16, channel_base=512, channel_max=64,
                      mbstd_group_size=2, bf16_res_count=0)
    return G, D


def test_pkl_roundtrip(tmp_path):
    torch.manual_seed(0)
    G, D = tiny_nets()
    import copy
    Gs = copy.deepcopy(G)
    path = str(tmp_path / "network-snapshot-000000.pkl")
    pkl_compat.save_network_pkl(path, G, D, Gs)

    G2, D2, Gs2 = pkl_compat.load_network_pkl(path)
    z = G.sample_z(2)
    with torch.no_grad():
        a = G(z, noise_mode="const")
        b = G2(z, noise_mode="const")
    assert torch.allclose(a, b, atol=1e-6)


def test_pkl_layout(tmp_path):
    G, D = tiny_nets()
    import copy
    path = str(tmp_path / "net.pkl")
    pkl_compat.save_network_pkl(path, G, D, copy.deepcopy(G))
    with open(path, "rb") as f:
        obj = pickle.load(f)  # raw stdlib load of OUR file: plain dicts only
    assert isinstance(obj, tuple) and len(obj) == 3
    for state, name in zip(obj, ("G", "D", "Gs")):
        assert state["name"] == name
        for key in ("version", "static_kwargs", "build_module_src",
                    "build_func_name", "variables"):
            assert key in state
        assert isinstance(state["variables"], list)
        vn, arr = state["variables"][0]
        assert isinstance(vn, str) and isinstance(arr, np.ndarray)
    # static kwargs carry the architecture
    assert obj[0]["static_kwargs"]["num_components"] == 2
    assert obj[0]["static_kwargs"]["transformer"] == "simplex"


def test_unpickler_refuses_code_execution():
    class Evil:
        def __reduce__(self):
            return (print, ("pwned",))

    buf = io.BytesIO()
    pickle.dump(Evil(), buf)
    buf.seek(0)
    with pytest.raises(pickle.UnpicklingError):
        pkl_compat._RestrictedUnpickler(buf).load()


def test_unpickler_maps_tflib_network_to_stub():
    # Simulate a reference-style pickle: a class from dnnlib.tflib.network
    payload = (b"\x80\x02cdnnlib.tflib.network\nNetwork\nq\x00)\x81q\x01}"
               b"q\x02X\x04\x00\x00\x00nameq\x03X\x01\x00\x00\x00Gq\x04sb.")
    obj = pkl_compat._RestrictedUnpickler(io.BytesIO(payload)).load()
    assert isinstance(obj, pkl_compat.NetworkStub)
    assert obj.state["name"] == "G"


def test_tf_scoped_variable_loading():
    """A TF-style variables list (reference naming + TF shape
    conventions [R]) loads through the name map with correct
    transposition."""
    import numpy as np
    from gansformer_amd.models.networks import Discriminator, Generator
    from gansformer_amd.pkl_compat import load_variables

    torch.manual_seed(0)
    G = Generator(z_dim=32, w_dim=32, img_resolution=16, num_components=4,
                  transformer="none", channel_base=512, channel_max=64,
                  bf16_res_count=0, mapping_layers=2)
    sd = {k: v.clone() for k, v in G.state_dict().items()}
    tf_vars = []
    # mapping denses (TF [in,out])
    for j in range(2):
        w = sd[f"mapping.layers.{j}.weight"].numpy()
        tf_vars.append((f"G_mapping/Dense{j}/weight", w.T.copy() + 1.0))
        tf_vars.append((f"G_mapping/Dense{j}/bias",
                        sd[f"mapping.layers.{j}.bias"].numpy() + 1.0))
    # one synthesis conv (TF [kh,kw,in,out])
    w = sd["synthesis.blocks.1.conv0.weight"].numpy()
    tf_vars.append(("G_synthesis/8x8/Conv0_up/weight",
                    np.transpose(w, (2, 3, 1, 0)) + 1.0))
    tf_vars.append(("G_synthesis/8x8/Conv0_up/mod_weight",
                    sd["synthesis.blocks.1.conv0.affine.weight"].numpy().T
                    + 1.0))
    tf_vars.append(("G_synthesis/4x4/Const/const",
                    sd["synthesis.blocks.0.const"].numpy() + 1.0))

    unmapped = load_variables(G, tf_vars, strict=True)
    assert unmapped == []
    sd2 = G.state_dict()
    for our in ("mapping.layers.0.weight", "mapping.layers.1.bias",
                "synthesis.blocks.1.conv0.weight",
                "synthesis.blocks.1.conv0.affine.weight",
                "synthesis.blocks.0.const"):
        assert torch.allclose(sd2[our], sd[our] + 1.0), our

    # D-side round trip
    D = Discriminator(img_resolution=16, channel_base=512, channel_max=64,
                      mbstd_group_size=2, bf16_res_count=0)
    dsd = {k: v.clone() for k, v in D.state_dict().items()}
    wd = dsd["blocks.0.conv1.weight"].numpy()
    tf_d = [
        ("D/16x16/FromRGB/weight",
         np.transpose(dsd["frgb.weight"].numpy(), (2, 3, 1, 0)) + 1.0),
        ("D/16x16/Conv1_down/weight", np.transpose(wd, (2, 3, 1, 0)) + 1.0),
        ("D/16x16/Skip/weight",
         np.transpose(dsd["blocks.0.skip.weight"].numpy(),
                      (2, 3, 1, 0)) + 1.0),
        ("D/4x4/Dense0/weight", dsd["fc.weight"].numpy().T + 1.0),
        ("D/Output/bias", dsd["out.bias"].numpy() + 1.0),
    ]
    assert load_variables(D, tf_d, strict=True) == []
    dsd2 = D.state_dict()
    assert torch.allclose(dsd2["blocks.0.conv1.weight"],
                          dsd["blocks.0.conv1.weight"] + 1.0)
    assert torch.allclose(dsd2["fc.weight"], dsd["fc.weight"] + 1.0)


def test_tf_scoped_unmapped_raises():
    from gansformer_amd.models.networks import Generator
    from gansformer_amd.pkl_compat import load_variables
    import numpy as np
    G = Generator(z_dim=32, w_dim=32, img_resolution=16, num_components=4,
                  transformer="none", channel_base=512, channel_max=64,
                  bf16_res_count=0, mapping_layers=2)
    bogus = [(f"G_synthesis/8x8/Mystery/weight", np.zeros((2, 2)))
             for _ in range(3)]
    with pytest.raises(KeyError):
        load_variables(G, bogus, strict=True)
    assert len(load_variables(G, bogus, strict=False)) == 3


def test_conditional_pkl_roundtrip(tmp_path):
    """label_dim travels through static_kwargs; a conditional model
    rebuilt from its pkl produces identical outputs."""
    from gansformer_amd.models.networks import Discriminator, Generator
    from gansformer_amd.pkl_compat import load_network_pkl, save_network_pkl
    torch.manual_seed(0)
    G = Generator(z_dim=32, w_dim=32, img_resolution=16, num_components=4,
                  transformer="simplex", channel_base=512, channel_max=64,
                  bf16_res_count=0, mapping_layers=2, label_dim=5)
    D = Discriminator(img_resolution=16, channel_base=512, channel_max=64,
                      mbstd_group_size=2, bf16_res_count=0, label_dim=5)
    path = str(tmp_path / "cond.pkl")
    save_network_pkl(path, G, D, G)
    G2, D2, _ = load_network_pkl(path)
    assert G2.label_dim == 5 and D2.label_dim == 5
    z = G.sample_z(2)
    lab = torch.zeros(2, 5)
    lab[:, 1] = 1.0
    with torch.no_grad():
        a = G(z, label=lab, noise_mode="const")
        b = G2(z, label=lab, noise_mode="const")
    assert torch.allclose(a, b, atol=1e-6)
    with torch.no_grad():
        la = D(a, label=lab)
        lb = D2(a, label=lab)
    assert torch.allclose(la, lb, atol=1e-6)


def test_tf_name_map_remaining_leaves():
    """ToRGB / Conv(4x4) / noise_strength / dlatent_avg broadcast."""
    import numpy as np
    from gansformer_amd.models.networks import Generator
    from gansformer_amd.pkl_compat import load_variables
    G = Generator(z_dim=32, w_dim=32, img_resolution=16, num_components=4,
                  transformer="none", channel_base=512, channel_max=64,
                  bf16_res_count=0, mapping_layers=2)
    sd = {k: v.clone() for k, v in G.state_dict().items()}
    w = sd["synthesis.blocks.0.torgb.weight"].numpy()  # [3, C, 1, 1]
    tf_vars = [
        ("G_synthesis/4x4/ToRGB/weight",
         np.transpose(w, (2, 3, 1, 0)) + 1.0),
        ("G_synthesis/4x4/ToRGB/mod_bias",
         sd["synthesis.blocks.0.torgb.affine.bias"].numpy() + 1.0),
        ("G_synthesis/4x4/Conv/noise_strength",
         sd["synthesis.blocks.0.conv1.noise_strength"].numpy().reshape(())
         + 1.0),
        ("dlatent_avg", np.full((32,), 0.25, dtype=np.float32)),
    ]
    assert load_variables(G, tf_vars, strict=True) == []
    sd2 = G.state_dict()
    assert torch.allclose(sd2["synthesis.blocks.0.torgb.weight"],
                          sd["synthesis.blocks.0.torgb.weight"] + 1.0)
    assert torch.allclose(sd2["synthesis.blocks.0.torgb.affine.bias"],
                          sd["synthesis.blocks.0.torgb.affine.bias"] + 1.0)
    assert torch.allclose(
        sd2["synthesis.blocks.0.conv1.noise_strength"],
        sd["synthesis.blocks.0.conv1.noise_strength"] + 1.0)
    # dlatent_avg broadcasts over all k+1 latents
    assert torch.allclose(sd2["mapping.w_avg"],
                          torch.full_like(sd["mapping.w_avg"], 0.25))
