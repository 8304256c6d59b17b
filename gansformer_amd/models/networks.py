"""GANsformer generator + discriminator, MI355X-native.

Re-implements the capability surface of the reference's
src/training/networks.py [R] (mapping net M1, StyleGAN2
synthesis blocks M2, bipartite Simplex/Duplex attention M3/M4,
discriminator with minibatch-stddev M5, noise & style mixing M6) on top of
the op layer in gansformer_amd.ops (HIP kernels on gfx950, eager reference
on CPU). Written from the GANsformer paper's math (Hudson & Zitnick,
arXiv:2103.01209) — not a translation of the TF graph code.

Latent layout: z is [B, k+1, z_dim] — k attention components plus one
global component (index k) that drives StyleGAN2-style conv modulation.
The k components are carried through synthesis as a state Y that Duplex
attention updates in place; Simplex attention lets image tokens read from
Y and modulates the feature maps (spatially-varying mean/var modulation).

Precision policy: parameters live in fp32; blocks at the top
`bf16_res_count` resolutions compute in bf16 with fp32
demodulation/softmax and a conv_clamp on activations (the MI355X analog
of the reference's fp16 policy in tflib.Optimizer [R]).
"""

from __future__ import annotations

import math

import numpy as np
import torch
import torch.nn as nn

from ..ops import (
    bias_act,
    mod_bias_act,
    bipartite_attention,
    conv2d_gradfix,
    downsample2d,
    linear_nobias,
    linear_transposed,
    minibatch_stddev,
    modnorm,
    modulated_conv2d,
    setup_filter,
    upsample2d,
    upfirdn2d,
)


def pixel_norm(x, dim=-1, eps=1e-8):
    return x * (x.square().mean(dim=dim, keepdim=True) + eps).rsqrt()


def channels_for(res, channel_base=32768, channel_max=512):
    return int(min(channel_base // res, channel_max))


class FullyConnected(nn.Module):
    """Equalized-LR linear layer with fused bias+act."""

    def __init__(self, in_features, out_features, bias=True, act="linear",
                 lr_mul=1.0, bias_init=0.0):
        super().__init__()
        self.weight = nn.Parameter(torch.randn(out_features, in_features) / lr_mul)
        self.bias = nn.Parameter(torch.full((out_features,), float(bias_init))) if bias else None
        self.act = act
        self.weight_gain = lr_mul / math.sqrt(in_features)
        self.bias_gain = lr_mul

    def forward(self, x):
        w = (self.weight * self.weight_gain).to(x.dtype)
        b = self.bias
        if b is not None:
            b = (b * self.bias_gain).to(x.dtype)
        # split-K weight gradient for the token-side FCs (K = B*HW):
        # hipBLASLt's direct tall-K GEMM runs ~25 TF/s, the split ~250+
        y = linear_nobias(x, w)
        return _fc_bias_act(y, b, self.act)


def _fc_transposed(fc, u):
    """Linear FC applied to u [B,N,d] with TRANSPOSED output [B,F,N]:
    (w*gain) @ u^T + bias — a strided batched GEMM, no permute copy."""
    assert fc.act == "linear"
    w = (fc.weight * fc.weight_gain).to(u.dtype)
    y = linear_transposed(u, w)
    if fc.bias is not None:
        y = y + (fc.bias * fc.bias_gain).to(y.dtype).reshape(1, -1, 1)
    return y


def _fc_bias_act(y, b, act):
    # bias_act broadcasts bias over dim 1; FC outputs are [..., F] — move F
    # to dim 1 by flattening the leading dims.
    shp = y.shape
    y2 = y.reshape(-1, shp[-1])
    y2 = bias_act(y2.unsqueeze(-1), b, act=act).squeeze(-1)
    return y2.reshape(shp)


class Conv2dLayer(nn.Module):
    """Equalized-LR conv layer (shared weights) with resample + fused act."""

    def __init__(self, in_channels, out_channels, kernel_size=3, bias=True,
                 act="linear", up=1, down=1, resample_filter=(1, 3, 3, 1),
                 conv_clamp=None, gain_out=1.0):
        super().__init__()
        self.up, self.down = up, down
        self.act = act
        self.conv_clamp = conv_clamp
        self.gain_out = gain_out
        self.padding = kernel_size // 2
        self.weight_gain = 1.0 / math.sqrt(in_channels * kernel_size ** 2)
        self.weight = nn.Parameter(
            torch.randn(out_channels, in_channels, kernel_size, kernel_size))
        self.bias = nn.Parameter(torch.zeros(out_channels)) if bias else None
        self.register_buffer("resample_filter", setup_filter(resample_filter),
                             persistent=False)

    def forward(self, x):
        w = (self.weight * self.weight_gain).to(x.dtype)
        if self.up > 1:
            x = upsample2d(x, self.resample_filter, up=self.up)
        if self.down > 1:
            fh = self.resample_filter.shape[0]
            if self.padding == 0:
                # 1x1 down-conv: blur-downsample then a plain conv; this
                # is exactly blur + strided conv (1x1 commutes with the
                # subsample) and keeps tensor sizes even
                x = downsample2d(x, self.resample_filter, down=self.down)
                y = conv2d_gradfix(x, w, stride=1, padding=0)
            elif x.shape[-1] <= 16:
                # EXACT pad-baked ordering (reference semantics): bake
                # the conv's padding into the blur so border outputs see
                # blurred contributions, then a VALID strided conv. At
                # these sizes border pixels are 44-75% of the output
                # (measured: ~0.35-0.56 relative border deviation in the
                # reordered form) and the stride-2 slab kernel is not
                # eligible anyway (it needs W >= 32), so exactness is
                # free. At 32 the fast path keeps the s2 slab kernel
                # (a threshold of 32 cost ~6% whole-step throughput).
                # PARITY.md "D resample ordering".
                p0 = (fh - self.down + 1) // 2 + self.padding
                p1 = fh - 1 - (fh - self.down + 1) // 2 + self.padding
                x = upfirdn2d(x, self.resample_filter,
                              padding=(p0, p1, p0, p1))
                y = conv2d_gradfix(x, w, stride=self.down, padding=0)
            else:
                # same-size blur with the conv's own padding kept ON the
                # conv (pads sum to fh-1 so the blur output stays even
                # and the strided conv takes the stride-2 slab kernel);
                # interior-identical to baking the pad into the blur,
                # border deviation <= 12% of pixels at these sizes
                # (PARITY.md quantification)
                p0 = (fh - self.down + 1) // 2
                p1 = fh - 1 - p0
                x = upfirdn2d(x, self.resample_filter,
                              padding=(p0, p1, p0, p1))
                y = conv2d_gradfix(x, w, stride=self.down,
                                   padding=self.padding)
        else:
            y = conv2d_gradfix(x, w, stride=1, padding=self.padding)
        b = self.bias.to(y.dtype) if self.bias is not None else None
        return bias_act(y, b, act=self.act, gain=_act_gain(self.act) * self.gain_out,
                        clamp=self.conv_clamp)


def _act_gain(act):
    from ..ops.fused_act import activation_defs
    return activation_defs[act][2]


def posenc_2d(h, w, dim, device, dtype=torch.float32):
    """Fixed 2D sinusoidal positional encoding, [H*W, dim]."""
    assert dim % 4 == 0
    d4 = dim // 4
    freq = torch.exp(torch.arange(d4, device=device, dtype=torch.float32)
                     * (-math.log(10000.0) / max(d4 - 1, 1)))
    ys = torch.arange(h, device=device, dtype=torch.float32).unsqueeze(1) * freq  # [H,d4]
    xs = torch.arange(w, device=device, dtype=torch.float32).unsqueeze(1) * freq
    pe = torch.zeros(h, w, dim, device=device, dtype=torch.float32)
    pe[:, :, 0*d4:1*d4] = ys.sin().unsqueeze(1).expand(h, w, d4)
    pe[:, :, 1*d4:2*d4] = ys.cos().unsqueeze(1).expand(h, w, d4)
    pe[:, :, 2*d4:3*d4] = xs.sin().unsqueeze(0).expand(h, w, d4)
    pe[:, :, 3*d4:4*d4] = xs.cos().unsqueeze(0).expand(h, w, d4)
    return pe.reshape(h * w, dim).to(dtype)


class BipartiteLayer(nn.Module):
    """One bipartite transformer layer between image tokens and latents.

    transformer='simplex': X <- modulate(X, Attn(X -> Y))
    transformer='duplex' : Y <- Y + gate * Attn(Y -> X), then simplex step.
    integration: 'mul' (spatially-varying mean/var modulation, the
    GANsformer default), 'add' (gated residual), 'both'.
    """

    def __init__(self, channels, latent_dim, num_components, attn_dim=None,
                 transformer="duplex", integration="mul", use_pos=True,
                 ltnt_gate=True):
        super().__init__()
        self.channels = channels
        self.num_components = num_components
        self.transformer = transformer
        self.integration = integration
        self.use_pos = use_pos
        d = attn_dim or min(channels, latent_dim)
        self.attn_dim = d
        # X -> queries (1x1 conv as linear over channels)
        self.to_q = FullyConnected(channels, d)
        self.to_k = FullyConnected(latent_dim, d)
        self.to_v = FullyConnected(latent_dim, d)
        if transformer == "duplex":
            self.y_to_q = FullyConnected(latent_dim, d)
            self.x_to_k = FullyConnected(channels, d)
            self.x_to_v = FullyConnected(channels, d)
            self.y_proj = FullyConnected(d, latent_dim)
            self.y_gate = nn.Parameter(torch.zeros(1))
        if integration in ("mul", "both"):
            self.to_gamma = FullyConnected(d, channels)
            self.to_beta = FullyConnected(d, channels)
        if integration in ("add", "both"):
            self.to_out = FullyConnected(d, channels)
            self.x_gate = nn.Parameter(torch.zeros(1))
        self._pos_cache = {}

    def _pos(self, h, w, device, dtype):
        key = (h, w, device, dtype)
        pe = self._pos_cache.get(key)
        if pe is None:
            pe = posenc_2d(h, w, self.attn_dim, device, dtype)
            self._pos_cache[key] = pe
        return pe

    def forward(self, x, y):
        """x: [B,C,H,W] image features; y: [B,k,latent_dim] components."""
        B, C, H, W = x.shape
        # materialize the token view ONCE: every projection matmul on the
        # strided transpose view would otherwise copy it again (the
        # to_q/x_to_k/x_to_v GEMMs were each re-packing [B,HW,C])
        tokens = x.reshape(B, C, H * W).transpose(1, 2).contiguous()
        pos = self._pos(H, W, x.device, tokens.dtype) if self.use_pos else None

        if self.transformer == "duplex":
            qy = self.y_to_q(y.to(tokens.dtype))
            kx = self.x_to_k(tokens)
            vx = self.x_to_v(tokens)
            if pos is not None:
                kx = kx + pos.unsqueeze(0)
            uy = bipartite_attention(qy, kx, vx)  # [B,k,d] softmax over HW
            y = y + torch.tanh(self.y_gate) * self.y_proj(uy).to(y.dtype)

        qx = self.to_q(tokens)
        if pos is not None:
            qx = qx + pos.unsqueeze(0)
        ky = self.to_k(y.to(tokens.dtype))
        vy = self.to_v(y.to(tokens.dtype))
        if getattr(self, "capture_attn", False):
            u, a = bipartite_attention(qx, ky, vy, need_weights=True)
            self.last_attn = a.detach().reshape(B, H, W, -1)
        else:
            u = bipartite_attention(qx, ky, vy)  # [B,HW,d] softmax over k

        if self.integration in ("mul", "both"):
            # gamma/beta computed TRANSPOSED ([B,C,HW], a strided GEMM —
            # no permute kernels) so the fused instance-norm+modulation
            # runs in NCHW and writes the layer output layout directly
            gamma_t = _fc_transposed(self.to_gamma, u)  # [B,C,HW]
            beta_t = _fc_transposed(self.to_beta, u)
            xr = x.reshape(B, C, H * W)
            out_t = modnorm(xr, gamma_t.to(xr.dtype), beta_t.to(xr.dtype))
            if self.integration == "both":
                out_t = out_t + torch.tanh(self.x_gate) \
                    * _fc_transposed(self.to_out, u).to(out_t.dtype)
            return out_t.reshape(B, C, H, W), y
        out = tokens
        if self.integration in ("add", "both"):
            out = out + torch.tanh(self.x_gate) * self.to_out(u)
        x = out.transpose(1, 2).reshape(B, C, H, W)
        return x, y

    def attention_maps(self, x, y):
        """Simplex attention map [B,HW,k] for visualization."""
        B, C, H, W = x.shape
        tokens = x.reshape(B, C, H * W).transpose(1, 2)
        qx = self.to_q(tokens)
        if self.use_pos:
            qx = qx + self._pos(H, W, x.device, tokens.dtype).unsqueeze(0)
        ky = self.to_k(y.to(tokens.dtype))
        vy = self.to_v(y.to(tokens.dtype))
        _, a = bipartite_attention(qx, ky, vy, need_weights=True)
        return a


class MappingNetwork(nn.Module):
    def __init__(self, z_dim=512, w_dim=512, num_components=16, num_ws=14,
                 num_layers=8, lr_mul=0.01, w_avg_beta=0.995, label_dim=0):
        super().__init__()
        self.z_dim, self.w_dim = z_dim, w_dim
        self.num_components = num_components  # k (attention latents)
        self.num_latents = num_components + 1  # + global
        self.num_ws = num_ws
        self.w_avg_beta = w_avg_beta
        self.label_dim = label_dim
        if label_dim > 0:
            # class-conditional path (SURVEY M1 "+class embed"): the
            # label embedding is normalized and concatenated to every
            # latent component before the shared MLP
            self.label_embed = FullyConnected(label_dim, z_dim)
        layers = []
        dims = [z_dim * (2 if label_dim > 0 else 1)] + [w_dim] * num_layers
        for i in range(num_layers):
            layers.append(FullyConnected(dims[i], dims[i + 1], act="lrelu",
                                         lr_mul=lr_mul))
        self.layers = nn.ModuleList(layers)
        self.register_buffer("w_avg", torch.zeros(self.num_latents, w_dim))

    def forward(self, z, label=None, truncation_psi=1.0,
                truncation_cutoff=None, update_w_avg=False):
        """z: [B, k+1, z_dim] -> ws [B, num_ws, k+1, w_dim]."""
        assert z.ndim == 3 and z.shape[1] == self.num_latents
        x = pixel_norm(z.to(torch.float32), dim=-1)
        if self.label_dim > 0:
            assert label is not None and label.shape[-1] == self.label_dim, \
                "conditional mapping needs a [B, label_dim] label"
            y = pixel_norm(self.label_embed(label.to(torch.float32)), dim=-1)
            x = torch.cat([x, y.unsqueeze(1).expand(-1, self.num_latents,
                                                    -1)], dim=-1)
        for layer in self.layers:
            x = layer(x)
        if update_w_avg:
            with torch.no_grad():
                self.w_avg.copy_(
                    x.detach().mean(dim=0).lerp(self.w_avg, self.w_avg_beta))
        ws = x.unsqueeze(1).repeat(1, self.num_ws, 1, 1)
        if truncation_psi != 1.0:
            cutoff = self.num_ws if truncation_cutoff is None else truncation_cutoff
            ws[:, :cutoff] = self.w_avg.unsqueeze(0).unsqueeze(0).lerp(
                ws[:, :cutoff], truncation_psi)
        return ws


class SynthesisLayer(nn.Module):
    """Modulated 3x3 conv + noise + fused bias/lrelu."""

    def __init__(self, in_channels, out_channels, w_dim, resolution, up=1,
                 kernel_size=3, resample_filter=(1, 3, 3, 1), conv_clamp=256.0):
        super().__init__()
        self.resolution = resolution
        self.up = up
        self.conv_clamp = conv_clamp
        self.affine = FullyConnected(w_dim, in_channels, bias_init=1.0)
        self.weight = nn.Parameter(
            torch.randn(out_channels, in_channels, kernel_size, kernel_size))
        self.weight_gain = 1.0 / math.sqrt(in_channels * kernel_size ** 2)
        self.bias = nn.Parameter(torch.zeros(out_channels))
        self.noise_strength = nn.Parameter(torch.zeros(1))
        self.register_buffer("noise_const",
                             torch.randn(resolution, resolution), persistent=True)
        self.register_buffer("resample_filter", setup_filter(resample_filter),
                             persistent=False)

    def forward(self, x, w, noise_mode="random"):
        styles = self.affine(w.to(torch.float32))  # [B, in]
        # weight_gain is omitted on purpose: demodulation pre-normalizes
        # the weight by rsqrt(mean w^2), which is scale-invariant in both
        # value AND gradient, so multiplying by the equalized-LR gain is
        # an exact no-op here (it still matters for ToRGB / D convs)
        y, dvec = modulated_conv2d(x, self.weight, styles,
                                   demodulate=True, up=self.up,
                                   resample_filter=self.resample_filter,
                                   return_demod=True)
        n = None
        if noise_mode == "random":
            n = torch.randn(y.shape[0], 1, y.shape[2], y.shape[3],
                            device=y.device, dtype=y.dtype)
        elif noise_mode == "const":
            n = self.noise_const.to(y.dtype).reshape(
                1, 1, *self.noise_const.shape).expand(y.shape[0], -1, -1, -1)
        # demod scale + noise + bias + lrelu in ONE fused pass
        return mod_bias_act(y, dvec, noise=n, sigma=self.noise_strength,
                            b=self.bias, act="lrelu", clamp=self.conv_clamp)


class ToRGB(nn.Module):
    def __init__(self, in_channels, img_channels, w_dim, conv_clamp=256.0):
        super().__init__()
        self.affine = FullyConnected(w_dim, in_channels, bias_init=1.0)
        self.weight = nn.Parameter(torch.randn(img_channels, in_channels, 1, 1))
        self.weight_gain = 1.0 / math.sqrt(in_channels)
        self.bias = nn.Parameter(torch.zeros(img_channels))
        self.conv_clamp = conv_clamp

    def forward(self, x, w):
        styles = self.affine(w.to(torch.float32))
        y = modulated_conv2d(x, (self.weight * self.weight_gain), styles,
                             demodulate=False)
        return bias_act(y, self.bias.to(y.dtype), act="linear",
                        clamp=self.conv_clamp)


class SynthesisBlock(nn.Module):
    def __init__(self, in_channels, out_channels, w_dim, resolution,
                 img_channels, num_components, latent_dim,
                 transformer="none", integration="mul", use_pos=True,
                 is_first=False, use_bf16=False, conv_clamp=256.0,
                 resample_filter=(1, 3, 3, 1)):
        super().__init__()
        self.resolution = resolution
        self.is_first = is_first
        self.use_bf16 = use_bf16
        self.num_conv = 0
        self.transformer = transformer
        self.register_buffer("resample_filter", setup_filter(resample_filter),
                             persistent=False)
        if is_first:
            self.const = nn.Parameter(torch.randn(out_channels, resolution, resolution))
        else:
            self.conv0 = SynthesisLayer(in_channels, out_channels, w_dim,
                                        resolution, up=2, conv_clamp=conv_clamp)
            self.num_conv += 1
        self.conv1 = SynthesisLayer(out_channels, out_channels, w_dim,
                                    resolution, conv_clamp=conv_clamp)
        self.num_conv += 1
        if transformer in ("simplex", "duplex"):
            self.attn = BipartiteLayer(out_channels, latent_dim, num_components,
                                       transformer=transformer,
                                       integration=integration, use_pos=use_pos)
        self.torgb = ToRGB(out_channels, img_channels, w_dim, conv_clamp=conv_clamp)

    def forward(self, x, img, ws_block, y, noise_mode="random", force_fp32=False):
        """ws_block: list of per-layer global styles for this block
        ([conv0], conv1, torgb). y: latent components state."""
        dtype = torch.bfloat16 if (self.use_bf16 and not force_fp32
                                   and ws_block[0].is_cuda) else torch.float32
        i = 0
        if self.is_first:
            B = ws_block[0].shape[0]
            x = self.const.to(dtype).unsqueeze(0).repeat(B, 1, 1, 1)
        else:
            x = x.to(dtype)
            x = self.conv0(x, ws_block[i], noise_mode=noise_mode)
            i += 1
        x = self.conv1(x, ws_block[i], noise_mode=noise_mode)
        i += 1
        if self.transformer in ("simplex", "duplex"):
            x, y = self.attn(x, y)
        rgb = self.torgb(x, ws_block[i])
        img = rgb if img is None else upsample2d(
            img, self.resample_filter, up=2).to(rgb.dtype) + rgb
        return x, img, y


class SynthesisNetwork(nn.Module):
    def __init__(self, w_dim=512, img_resolution=256, img_channels=3,
                 num_components=16, transformer="duplex", integration="mul",
                 attn_resolutions=None, use_pos=True, channel_base=32768,
                 channel_max=512, bf16_res_count=16, conv_clamp=256.0):
        super().__init__()
        self.w_dim = w_dim
        self.img_resolution = img_resolution
        self.img_channels = img_channels
        self.num_components = num_components
        self.transformer = transformer
        self.res_log2 = int(math.log2(img_resolution))
        assert 2 ** self.res_log2 == img_resolution and img_resolution >= 4
        self.block_resolutions = [2 ** i for i in range(2, self.res_log2 + 1)]
        if attn_resolutions is None:
            attn_resolutions = [r for r in self.block_resolutions
                                if 8 <= r <= min(128, img_resolution)]
        self.attn_resolutions = list(attn_resolutions) if transformer != "none" else []
        bf16_start = max(self.block_resolutions) / (2 ** (bf16_res_count - 1)) \
            if bf16_res_count > 0 else float("inf")
        self.blocks = nn.ModuleList()
        self.num_ws = 0
        prev_ch = 0
        for res in self.block_resolutions:
            ch = channels_for(res, channel_base, channel_max)
            use_attn = (res in self.attn_resolutions)
            blk = SynthesisBlock(
                prev_ch, ch, w_dim, res, img_channels, num_components,
                latent_dim=w_dim,
                transformer=transformer if use_attn else "none",
                integration=integration, use_pos=use_pos,
                is_first=(res == 4), use_bf16=(res >= bf16_start),
                conv_clamp=conv_clamp)
            self.num_ws += blk.num_conv + 1  # + torgb
            self.blocks.append(blk)
            prev_ch = ch

    def forward(self, ws, noise_mode="random", force_fp32=False):
        """ws: [B, num_ws, k+1, w_dim] -> imgs [B, C, R, R]."""
        B = ws.shape[0]
        assert ws.shape[1] == self.num_ws and ws.shape[2] == self.num_components + 1
        w_global = ws[:, :, self.num_components]     # [B, num_ws, w_dim]
        y = ws[:, 0, : self.num_components].contiguous()  # latent state [B,k,w]
        x = img = None
        idx = 0
        for blk in self.blocks:
            n = blk.num_conv + 1
            ws_block = [w_global[:, idx + j] for j in range(n)]
            x, img, y = blk(x, img, ws_block, y, noise_mode=noise_mode,
                            force_fp32=force_fp32)
            idx += n
        return img.to(torch.float32)


class Generator(nn.Module):
    def __init__(self, z_dim=512, w_dim=512, img_resolution=256, img_channels=3,
                 num_components=16, transformer="duplex", integration="mul",
                 attn_resolutions=None, use_pos=True, channel_base=32768,
                 channel_max=512, bf16_res_count=16, mapping_layers=8,
                 mapping_lr_mul=0.01, w_avg_beta=0.995, conv_clamp=256.0,
                 style_mixing_prob=0.9, label_dim=0):
        super().__init__()
        self.z_dim = z_dim
        self.w_dim = w_dim
        self.img_resolution = img_resolution
        self.img_channels = img_channels
        self.num_components = num_components
        self.style_mixing_prob = style_mixing_prob
        self.label_dim = label_dim
        self.init_kwargs = dict(
            z_dim=z_dim, w_dim=w_dim, img_resolution=img_resolution,
            img_channels=img_channels, num_components=num_components,
            transformer=transformer, integration=integration,
            attn_resolutions=attn_resolutions, use_pos=use_pos,
            channel_base=channel_base, channel_max=channel_max,
            bf16_res_count=bf16_res_count, mapping_layers=mapping_layers,
            mapping_lr_mul=mapping_lr_mul, w_avg_beta=w_avg_beta,
            conv_clamp=conv_clamp, style_mixing_prob=style_mixing_prob,
            label_dim=label_dim)
        self.synthesis = SynthesisNetwork(
            w_dim=w_dim, img_resolution=img_resolution,
            img_channels=img_channels, num_components=num_components,
            transformer=transformer, integration=integration,
            attn_resolutions=attn_resolutions, use_pos=use_pos,
            channel_base=channel_base, channel_max=channel_max,
            bf16_res_count=bf16_res_count, conv_clamp=conv_clamp)
        self.mapping = MappingNetwork(
            z_dim=z_dim, w_dim=w_dim, num_components=num_components,
            num_ws=self.synthesis.num_ws, num_layers=mapping_layers,
            lr_mul=mapping_lr_mul, w_avg_beta=w_avg_beta,
            label_dim=label_dim)

    @property
    def num_latents(self):
        return self.num_components + 1

    def sample_z(self, batch_size, device=None, generator=None):
        return torch.randn(batch_size, self.num_latents, self.z_dim,
                           device=device, generator=generator)

    def forward(self, z, label=None, truncation_psi=1.0,
                truncation_cutoff=None, noise_mode="random",
                style_mixing=False, update_w_avg=False, return_ws=False):
        ws = self.mapping(z, label=label, truncation_psi=truncation_psi,
                          truncation_cutoff=truncation_cutoff,
                          update_w_avg=update_w_avg)
        if style_mixing and self.style_mixing_prob > 0:
            if float(torch.rand(())) < self.style_mixing_prob:
                cutoff = int(torch.randint(1, ws.shape[1], ()))
                z2 = torch.randn_like(z)
                ws2 = self.mapping(z2, label=label)
                ws = torch.cat([ws[:, :cutoff], ws2[:, cutoff:]], dim=1)
        img = self.synthesis(ws, noise_mode=noise_mode)
        if return_ws:
            return img, ws
        return img


class DiscriminatorBlock(nn.Module):
    def __init__(self, in_channels, out_channels, resolution, use_bf16=False,
                 conv_clamp=256.0, resample_filter=(1, 3, 3, 1)):
        super().__init__()
        self.use_bf16 = use_bf16
        self.conv0 = Conv2dLayer(in_channels, in_channels, 3, act="lrelu",
                                 conv_clamp=conv_clamp)
        self.conv1 = Conv2dLayer(in_channels, out_channels, 3, act="lrelu",
                                 down=2, conv_clamp=conv_clamp,
                                 resample_filter=resample_filter)
        self.skip = Conv2dLayer(in_channels, out_channels, 1, bias=False,
                                down=2, resample_filter=resample_filter)

    def forward(self, x, force_fp32=False):
        dtype = torch.bfloat16 if (self.use_bf16 and not force_fp32 and x.is_cuda) \
            else torch.float32
        x = x.to(dtype)
        y = self.skip(x)
        x = self.conv0(x)
        x = self.conv1(x)
        return (x + y) * (1.0 / math.sqrt(2.0))


class Discriminator(nn.Module):
    def __init__(self, img_resolution=256, img_channels=3, channel_base=32768,
                 channel_max=512, mbstd_group_size=4, mbstd_num_channels=1,
                 bf16_res_count=16, conv_clamp=256.0, label_dim=0):
        super().__init__()
        self.img_resolution = img_resolution
        self.img_channels = img_channels
        self.mbstd_group_size = mbstd_group_size
        self.mbstd_num_channels = mbstd_num_channels
        self.label_dim = label_dim
        self.init_kwargs = dict(
            img_resolution=img_resolution, img_channels=img_channels,
            channel_base=channel_base, channel_max=channel_max,
            mbstd_group_size=mbstd_group_size,
            mbstd_num_channels=mbstd_num_channels,
            bf16_res_count=bf16_res_count, conv_clamp=conv_clamp,
            label_dim=label_dim)
        res_log2 = int(math.log2(img_resolution))
        self.block_resolutions = [2 ** i for i in range(res_log2, 2, -1)]
        bf16_start = img_resolution / (2 ** (bf16_res_count - 1)) \
            if bf16_res_count > 0 else float("inf")
        ch0 = channels_for(img_resolution, channel_base, channel_max)
        self.frgb = Conv2dLayer(img_channels, ch0, 1, act="lrelu",
                                conv_clamp=conv_clamp)
        blocks = []
        prev_ch = ch0
        for res in self.block_resolutions:
            out_ch = channels_for(res // 2, channel_base, channel_max)
            blocks.append(DiscriminatorBlock(prev_ch, out_ch, res,
                                             use_bf16=(res >= bf16_start),
                                             conv_clamp=conv_clamp))
            prev_ch = out_ch
        self.blocks = nn.ModuleList(blocks)
        self.conv_out = Conv2dLayer(prev_ch + mbstd_num_channels, prev_ch, 3,
                                    act="lrelu", conv_clamp=conv_clamp)
        self.fc = FullyConnected(prev_ch * 4 * 4, prev_ch, act="lrelu")
        self.out = FullyConnected(prev_ch, 1)
        if label_dim > 0:
            # projection-discriminator conditioning: the logit gets a
            # label-embedding dot product with the final features
            self.label_embed = FullyConnected(label_dim, prev_ch)
        self._proj_dim = prev_ch

    def forward(self, img, label=None, force_fp32=False):
        blk0_bf16 = (self.blocks and self.blocks[0].use_bf16
                     and not force_fp32 and img.is_cuda)
        x = self.frgb(img.to(torch.bfloat16 if blk0_bf16 else torch.float32))
        for blk in self.blocks:
            x = blk(x, force_fp32=force_fp32)
        x = x.to(torch.float32)
        x = minibatch_stddev(x, self.mbstd_group_size, self.mbstd_num_channels)
        x = self.conv_out(x)
        x = self.fc(x.flatten(1))
        logit = self.out(x)
        if self.label_dim > 0:
            assert label is not None and label.shape[-1] == self.label_dim, \
                "conditional D needs a [B, label_dim] label"
            emb = self.label_embed(label.to(x.dtype))
            logit = logit + (emb * x).sum(dim=-1, keepdim=True) \
                / math.sqrt(self._proj_dim)
        return logit
