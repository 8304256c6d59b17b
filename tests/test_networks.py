"""Network construction + forward/backward on CPU (driver config 1 shape:
CLEVR-64 simplex k=8)."""

import pytest
import torch

from gansformer_amd.models.networks import (BipartiteLayer, Discriminator,
                                            Generator)
from gansformer_amd.training.loss import (PathLengthRegularizer,
                                          d_logistic_loss,
                                          g_nonsaturating_loss, r1_penalty)


def small_G(transformer="simplex", k=8, res=64):
    return Generator(z_dim=64, w_dim=64, img_resolution=res,
                     num_components=k, transformer=transformer,
                     channel_base=2048, channel_max=128,
                     bf16_res_count=0, mapping_layers=2)


def small_D(res=64):
    return Discriminator(img_resolution=res, channel_base=2048,
                         channel_max=128, mbstd_group_size=2,
                         bf16_res_count=0)


@pytest.mark.parametrize("transformer", ["none", "simplex", "duplex"])
def test_generator_forward(transformer):
    G = small_G(transformer)
    z = G.sample_z(2)
    img = G(z)
    assert img.shape == (2, 3, 64, 64)
    assert torch.isfinite(img).all()


def test_generator_truncation():
    G = small_G().eval()
    with torch.no_grad():
        z = G.sample_z(2)
        # prime w_avg
        G.mapping(z, update_w_avg=True)
        img_full = G(z, truncation_psi=1.0, noise_mode="const")
        img_trunc = G(z, truncation_psi=0.0, noise_mode="const")
    # psi=0 collapses both samples onto w_avg -> identical images
    assert torch.allclose(img_trunc[0], img_trunc[1], atol=1e-4)
    assert not torch.allclose(img_full, img_trunc, atol=1e-3)


def test_discriminator_forward():
    D = small_D()
    x = torch.randn(2, 3, 64, 64)
    logits = D(x)
    assert logits.shape == (2, 1)


def test_gan_losses_backward():
    G, D = small_G(), small_D()
    z = G.sample_z(2)
    real = torch.randn(2, 3, 64, 64)
    fake = G(z)
    loss = d_logistic_loss(D(real), D(fake.detach())) \
        + g_nonsaturating_loss(D(fake))
    loss.backward()
    assert any(p.grad is not None and p.grad.abs().sum() > 0
               for p in G.parameters())
    assert any(p.grad is not None and p.grad.abs().sum() > 0
               for p in D.parameters())


def test_r1_second_order():
    D = small_D()
    real = torch.randn(2, 3, 64, 64, requires_grad=True)
    logits = D(real)
    r1 = r1_penalty(logits, real)
    r1.backward()
    grads = [p.grad for p in D.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)


@pytest.mark.parametrize("transformer", ["simplex", "duplex"])
def test_pathreg_second_order(transformer):
    G = small_G(transformer)
    z = G.sample_z(2)
    ws = G.mapping(z)
    fake = G.synthesis(ws)
    pl = PathLengthRegularizer()(fake, ws)
    pl.backward()
    grads = [p.grad for p in G.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)


def test_bipartite_layer_shapes():
    layer = BipartiteLayer(channels=32, latent_dim=16, num_components=4,
                           transformer="duplex")
    x = torch.randn(2, 32, 8, 8)
    y = torch.randn(2, 4, 16)
    x2, y2 = layer(x, y)
    assert x2.shape == x.shape and y2.shape == y.shape
    a = layer.attention_maps(x, y)
    assert a.shape == (2, 64, 4)
    assert torch.allclose(a.sum(-1), torch.ones(2, 64), atol=1e-5)


def test_num_ws_indexing():
    G = small_G(res=64)
    # res 4..64 -> 5 blocks; first block 2 layers (conv1+torgb), rest 3
    assert G.synthesis.num_ws == 2 + 4 * 3
    assert G.mapping.num_ws == G.synthesis.num_ws


@pytest.mark.parametrize("integration", ["mul", "add", "both"])
def test_generator_integration_variants(integration):
    """All three attention-integration modes forward+backward on CPU."""
    torch.manual_seed(0)
    G = Generator(z_dim=32, w_dim=32, img_resolution=32, num_components=4,
                  transformer="duplex", integration=integration,
                  channel_base=2048, channel_max=64, bf16_res_count=0,
                  mapping_layers=2)
    z = G.sample_z(2)
    img = G(z)
    assert img.shape == (2, 3, 32, 32) and torch.isfinite(img).all()
    img.square().mean().backward()
    assert all(p.grad is None or torch.isfinite(p.grad).all()
               for p in G.parameters())


def test_down_conv_exact_ordering_small_res():
    """At feature maps <= 32 the D down-conv uses the exact pad-baked
    blur->VALID-conv ordering: border pixels must match the reference
    composition exactly (the reordered form deviated up to
    0.56 relative on borders where they are 75% of the output)."""
    from gansformer_amd.models.networks import Conv2dLayer as ConvLayer
    from gansformer_amd.ops import conv2d_gradfix, setup_filter, upfirdn2d
    torch.manual_seed(0)
    for H in (8, 16):
        layer = ConvLayer(8, 16, kernel_size=3, act="linear", down=2,
                          bias=False)
        x = torch.randn(2, 8, H, H)
        y = layer(x)
        f = setup_filter([1, 3, 3, 1])
        w = (layer.weight * layer.weight_gain)
        xe = upfirdn2d(x, f, padding=(2, 3, 2, 3))
        ye = conv2d_gradfix(xe, w, stride=2, padding=0)
        assert y.shape == ye.shape
        assert torch.allclose(y, ye, atol=1e-5), f"H={H}"
