"""VITS inference network: text encoder (relative-position attention),
stochastic duration predictor (neural spline flows), residual-coupling
normalizing flow, HiFi-GAN generator.

This is the graph the reference outsources to ONNX Runtime (SURVEY.md §2.2
inventories the ops; the model family is the canonical Piper export of VITS,
crates/sonata/models/piper/README.md).  Implemented here from first
principles in PyTorch modules whose hot ops dispatch to hand-written
CDNA4 HIP kernels via sonata_amd.ops when running on GPU.

Inference-only: no dropout, no posterior encoder, no discriminators.
"""

from __future__ import annotations

import math
import os
from typing import List, Optional, Tuple

import torch
import torch.nn.functional as F
from torch import nn

from ..ops import (
    conv_mod,
    row_ln_cl,
    depthwise_conv1d_cl,
    expand_states,
    fused_gate,
    fused_gate_cl,
    layer_norm_ct,
    leaky_conv1d,
    leaky_conv1d_cl,
    leaky_convtranspose1d,
    leaky_convtranspose1d_cl,
    mask_tail_,
    prior_sample,
)
from .config import VitsArchitecture

LRELU_SLOPE = 0.1


# --------------------------------------------------------------------------- #
# helpers
# --------------------------------------------------------------------------- #
def sequence_mask(lengths: torch.Tensor, max_len: Optional[int] = None) -> torch.Tensor:
    """[B] lengths -> [B, 1, T] float mask."""
    if max_len is None:
        max_len = int(lengths.max().item())
    pos = torch.arange(max_len, device=lengths.device)
    return (pos.unsqueeze(0) < lengths.unsqueeze(1)).unsqueeze(1).to(torch.float32)


class LayerNormCT(nn.Module):
    """LayerNorm across the channel dim of [B, C, T]."""

    def __init__(self, channels: int, eps: float = 1e-5):
        super().__init__()
        self.eps = eps
        self.gamma = nn.Parameter(torch.ones(channels))
        self.beta = nn.Parameter(torch.zeros(channels))

    def forward(self, x: torch.Tensor,
                residual: "Optional[torch.Tensor]" = None) -> torch.Tensor:
        return layer_norm_ct(x, self.gamma, self.beta, self.eps,
                             residual=residual)


# --------------------------------------------------------------------------- #
# Relative-position multi-head attention (window_size=4)
# --------------------------------------------------------------------------- #
class RelativeAttention(nn.Module):
    def __init__(self, channels: int, n_heads: int, window_size: int = 4):
        super().__init__()
        assert channels % n_heads == 0
        self.channels = channels
        self.n_heads = n_heads
        self.head_dim = channels // n_heads
        self.window_size = window_size
        self.conv_q = nn.Conv1d(channels, channels, 1)
        self.conv_k = nn.Conv1d(channels, channels, 1)
        self.conv_v = nn.Conv1d(channels, channels, 1)
        self.conv_o = nn.Conv1d(channels, channels, 1)
        rel_std = self.head_dim ** -0.5
        # heads share one table of 2w+1 relative embeddings
        self.emb_rel_k = nn.Parameter(
            torch.randn(1, 2 * window_size + 1, self.head_dim) * rel_std
        )
        self.emb_rel_v = nn.Parameter(
            torch.randn(1, 2 * window_size + 1, self.head_dim) * rel_std
        )

    # -- relative/absolute index plumbing (music-transformer style) -------- #
    @staticmethod
    def _rel_to_abs(x: torch.Tensor) -> torch.Tensor:
        """[b, h, l, 2l-1] -> [b, h, l, l]"""
        b, h, l, _ = x.shape
        x = F.pad(x, (0, 1))
        x_flat = x.reshape(b, h, l * 2 * l)
        x_flat = F.pad(x_flat, (0, l - 1))
        return x_flat.reshape(b, h, l + 1, 2 * l - 1)[:, :, :l, l - 1 :]

    @staticmethod
    def _abs_to_rel(x: torch.Tensor) -> torch.Tensor:
        """[b, h, l, l] -> [b, h, l, 2l-1]"""
        b, h, l, _ = x.shape
        x = F.pad(x, (0, l - 1))
        x_flat = x.reshape(b, h, l * l + l * (l - 1))
        x_flat = F.pad(x_flat, (l, 0))
        return x_flat.reshape(b, h, l, 2 * l)[:, :, :, 1:]

    def _rel_embeddings(self, emb: torch.Tensor, length: int) -> torch.Tensor:
        pad_len = max(length - (self.window_size + 1), 0)
        start = max((self.window_size + 1) - length, 0)
        if pad_len > 0:
            emb = F.pad(emb, (0, 0, pad_len, pad_len))
        return emb[:, start : start + 2 * length - 1]

    def forward(self, x: torch.Tensor, attn_mask: torch.Tensor) -> torch.Tensor:
        B, C, T = x.shape
        q = conv_mod(self.conv_q, x).view(B, self.n_heads, self.head_dim, T).transpose(2, 3)
        k = conv_mod(self.conv_k, x).view(B, self.n_heads, self.head_dim, T).transpose(2, 3)
        v = conv_mod(self.conv_v, x).view(B, self.n_heads, self.head_dim, T).transpose(2, 3)
        scale = self.head_dim ** -0.5
        scores = torch.matmul(q * scale, k.transpose(-2, -1))  # [B,h,T,T]
        rel_k = self._rel_embeddings(self.emb_rel_k, T)  # [1, 2T-1, d]
        rel_logits = torch.matmul(q * scale, rel_k.unsqueeze(0).transpose(-2, -1))
        scores = scores + self._rel_to_abs(rel_logits)
        scores = scores.masked_fill(attn_mask == 0, -1e4)
        p = torch.softmax(scores, dim=-1)
        out = torch.matmul(p, v)  # [B,h,T,d]
        rel_w = self._abs_to_rel(p)  # [B,h,T,2T-1]
        rel_v = self._rel_embeddings(self.emb_rel_v, T)
        out = out + torch.matmul(rel_w, rel_v.unsqueeze(0))
        out = out.transpose(2, 3).contiguous().view(B, C, T)
        return conv_mod(self.conv_o, out)

    def forward_cl(self, x: torch.Tensor,
                   attn_mask: torch.Tensor) -> torch.Tensor:
        """Channel-last attention: x [B,T,C]; the [B,h,T,d] head view is a
        free reshape of channel-last rows (no transpose of C against T),
        and all projections are hipBLASLt linears."""
        B, T, C = x.shape
        h, d = self.n_heads, self.head_dim

        def proj(conv, t):
            return F.linear(t, conv.weight.squeeze(-1), conv.bias)

        q = proj(self.conv_q, x).view(B, T, h, d).transpose(1, 2)
        k = proj(self.conv_k, x).view(B, T, h, d).transpose(1, 2)
        v = proj(self.conv_v, x).view(B, T, h, d).transpose(1, 2)
        scale = d ** -0.5
        scores = torch.matmul(q * scale, k.transpose(-2, -1))
        rel_k = self._rel_embeddings(self.emb_rel_k, T)
        rel_logits = torch.matmul(q * scale,
                                  rel_k.unsqueeze(0).transpose(-2, -1))
        scores = scores + self._rel_to_abs(rel_logits)
        scores = scores.masked_fill(attn_mask == 0, -1e4)
        p = torch.softmax(scores, dim=-1)
        out = torch.matmul(p, v)
        rel_w = self._abs_to_rel(p)
        rel_v = self._rel_embeddings(self.emb_rel_v, T)
        out = out + torch.matmul(rel_w, rel_v.unsqueeze(0))
        out = out.transpose(1, 2).reshape(B, T, C)
        return proj(self.conv_o, out)


class FFN(nn.Module):
    """Conv1d(k) -> ReLU -> Conv1d(k), masked."""

    def __init__(self, channels: int, filter_channels: int, kernel_size: int):
        super().__init__()
        self.conv1 = nn.Conv1d(channels, filter_channels, kernel_size,
                               padding=kernel_size // 2)
        self.conv2 = nn.Conv1d(filter_channels, channels, kernel_size,
                               padding=kernel_size // 2)

    def forward(self, x: torch.Tensor, x_mask: torch.Tensor) -> torch.Tensor:
        x = conv_mod(self.conv1, x * x_mask)
        x = torch.relu(x)
        x = conv_mod(self.conv2, x * x_mask)
        return x * x_mask


class TextEncoder(nn.Module):
    def __init__(self, n_vocab: int, out_channels: int, arch: VitsArchitecture):
        super().__init__()
        h = arch.hidden_channels
        self.hidden_channels = h
        self.emb = nn.Embedding(n_vocab, h)
        nn.init.normal_(self.emb.weight, 0.0, h ** -0.5)
        self.attn_layers = nn.ModuleList(
            [RelativeAttention(h, arch.n_heads, arch.window_size)
             for _ in range(arch.n_layers)]
        )
        self.norm1 = nn.ModuleList([LayerNormCT(h) for _ in range(arch.n_layers)])
        self.ffn_layers = nn.ModuleList(
            [FFN(h, arch.filter_channels, arch.kernel_size)
             for _ in range(arch.n_layers)]
        )
        self.norm2 = nn.ModuleList([LayerNormCT(h) for _ in range(arch.n_layers)])
        self.proj = nn.Conv1d(h, out_channels * 2, 1)

    def forward(
        self, ids: torch.Tensor, lengths: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
        from ..ops import use_hip

        if use_hip(ids):
            return self._forward_cl(ids, lengths)
        x = self.emb(ids) * math.sqrt(self.hidden_channels)  # [B,T,H]
        x = x.transpose(1, 2)  # [B,H,T]
        x_mask = sequence_mask(lengths, ids.shape[1]).to(x.dtype)
        attn_mask = (x_mask.unsqueeze(2) * x_mask.unsqueeze(-1)).squeeze(1)
        x = x * x_mask
        for attn, n1, ffn, n2 in zip(self.attn_layers, self.norm1,
                                     self.ffn_layers, self.norm2):
            y = attn(x * x_mask, attn_mask.unsqueeze(1))
            x = n1(x, residual=y)   # fused residual-add + LayerNorm
            y = ffn(x, x_mask)
            x = n2(x, residual=y)
        stats = conv_mod(self.proj, x) * x_mask
        m, logs = stats.chunk(2, dim=1)
        return x, m, logs, x_mask

    def _forward_cl(
        self, ids: torch.Tensor, lengths: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
        """GPU path in the native channel-last layout: the embedding is
        already [B,T,H]; attention/FFN/LN all run on contiguous rows and
        only the (small) encoder outputs transpose to [B,C,T] for the
        duration/prior stages.  Numerics match forward()."""
        import os

        from ..ops import attn_relpos_cl

        fused_attn = os.environ.get("SONATA_FUSED_ATTN", "1") == "1"
        x = self.emb(ids) * math.sqrt(self.hidden_channels)  # [B,T,H]
        x_mask_cl = sequence_mask(lengths, ids.shape[1]).to(
            x.dtype).transpose(1, 2)  # [B,T,1]
        attn_mask = None
        if not fused_attn:
            attn_mask = (x_mask_cl * x_mask_cl.transpose(1, 2))  # [B,T,T]
        x = x * x_mask_cl
        for attn, n1, ffn, n2 in zip(self.attn_layers, self.norm1,
                                     self.ffn_layers, self.norm2):
            if fused_attn:
                y = attn_relpos_cl(x * x_mask_cl, attn, lengths)
            else:
                y = attn.forward_cl(x * x_mask_cl, attn_mask.unsqueeze(1))
            x = row_ln_cl(x, n1.gamma, n1.beta, n1.eps, residual=y)
            f = leaky_conv1d_cl(x * x_mask_cl, ffn.conv1.weight,
                                ffn.conv1.bias,
                                padding=ffn.conv1.padding[0], post_relu=True,
                                out_lens=lengths)
            f = leaky_conv1d_cl(f, ffn.conv2.weight, ffn.conv2.bias,
                                padding=ffn.conv2.padding[0],
                                out_lens=lengths)
            x = row_ln_cl(x, n2.gamma, n2.beta, n2.eps, residual=f)
        stats = F.linear(x, self.proj.weight.squeeze(-1), self.proj.bias)
        stats = (stats * x_mask_cl).transpose(1, 2)  # [B,2C,T]
        m, logs = stats.chunk(2, dim=1)
        x_mask = x_mask_cl.transpose(1, 2)
        return x.transpose(1, 2) * x_mask, m.contiguous(), \
            logs.contiguous(), x_mask


# --------------------------------------------------------------------------- #
# WaveNet stack (residual coupling enc)
# --------------------------------------------------------------------------- #
class WN(nn.Module):
    def __init__(self, hidden: int, kernel_size: int, dilation_rate: int,
                 n_layers: int, gin_channels: int = 0):
        super().__init__()
        self.hidden = hidden
        self.n_layers = n_layers
        self.in_layers = nn.ModuleList()
        self.res_skip_layers = nn.ModuleList()
        if gin_channels:
            self.cond_layer = nn.Conv1d(gin_channels, 2 * hidden * n_layers, 1)
        else:
            self.cond_layer = None
        for i in range(n_layers):
            dilation = dilation_rate ** i
            pad = (kernel_size - 1) * dilation // 2
            self.in_layers.append(
                nn.Conv1d(hidden, 2 * hidden, kernel_size,
                          dilation=dilation, padding=pad)
            )
            res_skip_ch = 2 * hidden if i < n_layers - 1 else hidden
            self.res_skip_layers.append(nn.Conv1d(hidden, res_skip_ch, 1))

    def forward(self, x: torch.Tensor, x_mask: torch.Tensor,
                g: Optional[torch.Tensor] = None) -> torch.Tensor:
        output = torch.zeros_like(x)
        if g is not None and self.cond_layer is not None:
            g_all = conv_mod(self.cond_layer, g)
        else:
            g_all = None
        for i in range(self.n_layers):
            x_in = conv_mod(self.in_layers[i], x)
            g_l = (
                g_all[:, i * 2 * self.hidden : (i + 1) * 2 * self.hidden]
                if g_all is not None else None
            )
            acts = fused_gate(x_in, g_l, self.hidden)
            res_skip = conv_mod(self.res_skip_layers[i], acts)
            if i < self.n_layers - 1:
                x = (x + res_skip[:, : self.hidden]) * x_mask
                output = output + res_skip[:, self.hidden :]
            else:
                output = output + res_skip
        return output * x_mask

    def forward_cl(self, x: torch.Tensor, x_mask: torch.Tensor,
                   g: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Channel-last WN: x [B,F,H], x_mask [B,F,1]; 1x1 convs as
        hipBLASLt linears, k-tap convs as channel-last MFMA kernels,
        gate fused (fused_gate_cl)."""
        output = torch.zeros_like(x)
        H = self.hidden
        g_all = None
        if g is not None and self.cond_layer is not None:
            # g [B,gin,1] -> [B, 2H*n_layers] (per-utterance bias rows)
            g_all = F.linear(g.squeeze(-1), self.cond_layer.weight.squeeze(-1),
                             self.cond_layer.bias)
        for i in range(self.n_layers):
            conv = self.in_layers[i]
            x_in = leaky_conv1d_cl(x, conv.weight, conv.bias,
                                   padding=conv.padding[0],
                                   dilation=conv.dilation[0])
            g_l = (g_all[:, i * 2 * H:(i + 1) * 2 * H]
                   if g_all is not None else None)
            acts = fused_gate_cl(x_in, g_l, H)
            rs = self.res_skip_layers[i]
            res_skip = F.linear(acts, rs.weight.squeeze(-1), rs.bias)
            if i < self.n_layers - 1:
                x = (x + res_skip[..., :H]) * x_mask
                output = output + res_skip[..., H:]
            else:
                output = output + res_skip
        return output * x_mask


class ResidualCouplingLayer(nn.Module):
    """Mean-only affine coupling with a WN conditioner."""

    def __init__(self, channels: int, hidden: int, kernel_size: int,
                 dilation_rate: int, n_layers: int, gin_channels: int = 0):
        super().__init__()
        self.half = channels // 2
        self.pre = nn.Conv1d(self.half, hidden, 1)
        self.enc = WN(hidden, kernel_size, dilation_rate, n_layers, gin_channels)
        self.post = nn.Conv1d(hidden, self.half, 1)
        nn.init.zeros_(self.post.weight)
        nn.init.zeros_(self.post.bias)

    def forward(self, x: torch.Tensor, x_mask: torch.Tensor,
                g: Optional[torch.Tensor] = None,
                reverse: bool = False) -> torch.Tensor:
        x0, x1 = x[:, : self.half], x[:, self.half :]
        h = conv_mod(self.pre, x0) * x_mask
        h = self.enc(h, x_mask, g=g)
        m = conv_mod(self.post, h) * x_mask
        if not reverse:
            x1 = (m + x1) * x_mask
        else:
            x1 = (x1 - m) * x_mask
        return torch.cat([x0, x1], dim=1)

    def reverse_cl(self, x: torch.Tensor, x_mask: torch.Tensor,
                   g: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Channel-last inverse coupling: x [B,F,C], x_mask [B,F,1]."""
        x0 = x[..., : self.half].contiguous()
        x1 = x[..., self.half :]
        h = F.linear(x0, self.pre.weight.squeeze(-1), self.pre.bias) * x_mask
        h = self.enc.forward_cl(h, x_mask, g=g)
        m = F.linear(h, self.post.weight.squeeze(-1), self.post.bias) * x_mask
        x1 = (x1 - m) * x_mask
        return torch.cat([x0, x1], dim=-1)


class ResidualCouplingBlock(nn.Module):
    def __init__(self, channels: int, hidden: int, kernel_size: int,
                 dilation_rate: int, n_layers: int, n_flows: int = 4,
                 gin_channels: int = 0):
        super().__init__()
        self.flows = nn.ModuleList(
            [ResidualCouplingLayer(channels, hidden, kernel_size,
                                   dilation_rate, n_layers, gin_channels)
             for _ in range(n_flows)]
        )

    def forward(self, x: torch.Tensor, x_mask: torch.Tensor,
                g: Optional[torch.Tensor] = None,
                reverse: bool = False) -> torch.Tensor:
        if not reverse:
            for flow in self.flows:
                x = flow(x, x_mask, g=g, reverse=False)
                x = torch.flip(x, [1])
        else:
            for flow in reversed(self.flows):
                x = torch.flip(x, [1])
                x = flow(x, x_mask, g=g, reverse=True)
        return x

    def reverse_cl(self, x: torch.Tensor, x_mask: torch.Tensor,
                   g: Optional[torch.Tensor] = None) -> torch.Tensor:
        for flow in reversed(self.flows):
            x = torch.flip(x, [-1])
            x = flow.reverse_cl(x, x_mask, g=g)
        return x


# --------------------------------------------------------------------------- #
# Neural spline flow pieces (stochastic duration predictor)
# --------------------------------------------------------------------------- #
def _searchsorted(bin_locations: torch.Tensor, inputs: torch.Tensor) -> torch.Tensor:
    return torch.sum(inputs[..., None] >= bin_locations, dim=-1) - 1


def rational_quadratic_spline(
    inputs: torch.Tensor,
    unnormalized_widths: torch.Tensor,
    unnormalized_heights: torch.Tensor,
    unnormalized_derivatives: torch.Tensor,
    inverse: bool = False,
    tail_bound: float = 5.0,
    min_bin_width: float = 1e-3,
    min_bin_height: float = 1e-3,
    min_derivative: float = 1e-3,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Monotonic piecewise rational-quadratic spline with linear tails
    (Durkan et al., Neural Spline Flows).  Returns (outputs, logabsdet).

    Sync-free formulation: instead of boolean-mask compaction (whose
    `inside.any()` host-syncs on every flow call — 4 per utterance on
    the latency path — and whose index_put breaks hipGraph capture),
    the spline is evaluated on ALL elements with tail-clamped inputs
    and blended with the identity tails via torch.where.  Per-element
    math is identical to the compacted version."""
    inside = (inputs >= -tail_bound) & (inputs <= tail_bound)

    # pad derivatives so boundary derivative == 1
    constant = math.log(math.exp(1.0 - min_derivative) - 1.0)
    unnormalized_derivatives = F.pad(unnormalized_derivatives, (1, 1),
                                     value=constant)

    num_bins = unnormalized_widths.shape[-1]
    uw = unnormalized_widths
    uh = unnormalized_heights
    ud = unnormalized_derivatives
    x = torch.clamp(inputs, -tail_bound, tail_bound)

    widths = torch.softmax(uw, dim=-1)
    widths = min_bin_width + (1 - min_bin_width * num_bins) * widths
    cumwidths = torch.cumsum(widths, dim=-1)
    cumwidths = F.pad(cumwidths, (1, 0), value=0.0)
    cumwidths = (2 * tail_bound) * cumwidths - tail_bound
    cumwidths[..., 0] = -tail_bound
    cumwidths[..., -1] = tail_bound
    widths = cumwidths[..., 1:] - cumwidths[..., :-1]

    derivatives = min_derivative + F.softplus(ud)

    heights = torch.softmax(uh, dim=-1)
    heights = min_bin_height + (1 - min_bin_height * num_bins) * heights
    cumheights = torch.cumsum(heights, dim=-1)
    cumheights = F.pad(cumheights, (1, 0), value=0.0)
    cumheights = (2 * tail_bound) * cumheights - tail_bound
    cumheights[..., 0] = -tail_bound
    cumheights[..., -1] = tail_bound
    heights = cumheights[..., 1:] - cumheights[..., :-1]

    if inverse:
        bin_idx = _searchsorted(cumheights, x)[..., None]
    else:
        bin_idx = _searchsorted(cumwidths, x)[..., None]
    # x clamped exactly onto +tail_bound lands one past the last bin
    bin_idx = bin_idx.clamp(0, num_bins - 1)

    in_cumwidths = cumwidths.gather(-1, bin_idx)[..., 0]
    in_widths = widths.gather(-1, bin_idx)[..., 0]
    in_cumheights = cumheights.gather(-1, bin_idx)[..., 0]
    in_heights = heights.gather(-1, bin_idx)[..., 0]
    delta = in_heights / in_widths
    in_deriv = derivatives.gather(-1, bin_idx)[..., 0]
    in_deriv_p1 = derivatives[..., 1:].gather(-1, bin_idx)[..., 0]

    if inverse:
        a = (x - in_cumheights) * (in_deriv + in_deriv_p1 - 2 * delta) + \
            in_heights * (delta - in_deriv)
        b = in_heights * in_deriv - (x - in_cumheights) * \
            (in_deriv + in_deriv_p1 - 2 * delta)
        c = -delta * (x - in_cumheights)
        disc = b.pow(2) - 4 * a * c
        disc = torch.clamp(disc, min=0.0)
        root = (2 * c) / (-b - torch.sqrt(disc))
        out = root * in_widths + in_cumwidths
        theta_one_minus_theta = root * (1 - root)
        denom = delta + (in_deriv + in_deriv_p1 - 2 * delta) * theta_one_minus_theta
        deriv_num = delta.pow(2) * (
            in_deriv_p1 * root.pow(2)
            + 2 * delta * theta_one_minus_theta
            + in_deriv * (1 - root).pow(2)
        )
        lad = torch.log(deriv_num) - 2 * torch.log(denom)
        outputs = torch.where(inside, out, inputs)
        logabsdet = torch.where(inside, -lad, torch.zeros_like(lad))
    else:
        theta = (x - in_cumwidths) / in_widths
        theta_one_minus_theta = theta * (1 - theta)
        numerator = in_heights * (delta * theta.pow(2)
                                  + in_deriv * theta_one_minus_theta)
        denom = delta + (in_deriv + in_deriv_p1 - 2 * delta) * theta_one_minus_theta
        out = in_cumheights + numerator / denom
        deriv_num = delta.pow(2) * (
            in_deriv_p1 * theta.pow(2)
            + 2 * delta * theta_one_minus_theta
            + in_deriv * (1 - theta).pow(2)
        )
        lad = torch.log(deriv_num) - 2 * torch.log(denom)
        outputs = torch.where(inside, out, inputs)
        logabsdet = torch.where(inside, lad, torch.zeros_like(lad))
    return outputs, logabsdet


class DDSConv(nn.Module):
    """Dilated depth-separable conv stack with LayerNorm + GELU."""

    def __init__(self, channels: int, kernel_size: int, n_layers: int):
        super().__init__()
        self.convs_sep = nn.ModuleList()
        self.convs_1x1 = nn.ModuleList()
        self.norms_1 = nn.ModuleList()
        self.norms_2 = nn.ModuleList()
        for i in range(n_layers):
            dilation = kernel_size ** i
            pad = (kernel_size - 1) * dilation // 2
            self.convs_sep.append(
                nn.Conv1d(channels, channels, kernel_size, groups=channels,
                          dilation=dilation, padding=pad)
            )
            self.convs_1x1.append(nn.Conv1d(channels, channels, 1))
            self.norms_1.append(LayerNormCT(channels))
            self.norms_2.append(LayerNormCT(channels))

    def forward(self, x: torch.Tensor, x_mask: torch.Tensor,
                g: Optional[torch.Tensor] = None) -> torch.Tensor:
        if g is not None:
            x = x + g
        for sep, one, n1, n2 in zip(self.convs_sep, self.convs_1x1,
                                    self.norms_1, self.norms_2):
            y = conv_mod(sep, x * x_mask)
            y = n1(y)
            y = F.gelu(y)
            y = conv_mod(one, y)
            y = n2(y)
            y = F.gelu(y)
            x = x + y
        return x * x_mask

    def forward_cl(self, x: torch.Tensor, x_mask: torch.Tensor,
                   g: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Channel-last DDS stack: x [B,T,C], x_mask [B,T,1]; depthwise
        via depthwise_cl kernel, 1x1s as linears, LN over rows."""
        if g is not None:
            x = x + g
        C = x.shape[-1]
        for sep, one, n1, n2 in zip(self.convs_sep, self.convs_1x1,
                                    self.norms_1, self.norms_2):
            y = depthwise_conv1d_cl(x * x_mask, sep.weight, sep.bias,
                                    sep.dilation[0], sep.padding[0])
            y = row_ln_cl(y, n1.gamma, n1.beta, n1.eps)
            y = F.gelu(y)
            y = F.linear(y, one.weight.squeeze(-1), one.bias)
            y = row_ln_cl(y, n2.gamma, n2.beta, n2.eps)
            y = F.gelu(y)
            x = x + y
        return x * x_mask


class ElementwiseAffine(nn.Module):
    def __init__(self, channels: int):
        super().__init__()
        self.m = nn.Parameter(torch.zeros(channels, 1))
        self.logs = nn.Parameter(torch.zeros(channels, 1))

    def forward(self, x, x_mask, reverse=False, **kwargs):
        if not reverse:
            y = (self.m + torch.exp(self.logs) * x) * x_mask
            logdet = torch.sum(self.logs * x_mask, [1, 2])
            return y, logdet
        return (x - self.m) * torch.exp(-self.logs) * x_mask


class ConvFlow(nn.Module):
    def __init__(self, in_channels: int, filter_channels: int,
                 kernel_size: int, n_layers: int, num_bins: int = 10,
                 tail_bound: float = 5.0):
        super().__init__()
        self.half = in_channels // 2
        self.filter_channels = filter_channels
        self.num_bins = num_bins
        self.tail_bound = tail_bound
        self.pre = nn.Conv1d(self.half, filter_channels, 1)
        self.convs = DDSConv(filter_channels, kernel_size, n_layers)
        self.proj = nn.Conv1d(filter_channels, self.half * (num_bins * 3 - 1), 1)
        nn.init.zeros_(self.proj.weight)
        nn.init.zeros_(self.proj.bias)

    def forward(self, x, x_mask, g=None, reverse=False):
        x0, x1 = x[:, : self.half], x[:, self.half :]
        h = conv_mod(self.pre, x0)
        h = self.convs(h, x_mask, g=g)
        h = conv_mod(self.proj, h) * x_mask
        B, _, T = x0.shape
        h = h.reshape(B, self.half, 3 * self.num_bins - 1, T).permute(0, 1, 3, 2)
        scale = math.sqrt(self.filter_channels)
        uw = h[..., : self.num_bins] / scale
        uh = h[..., self.num_bins : 2 * self.num_bins] / scale
        ud = h[..., 2 * self.num_bins :]
        x1, logabsdet = rational_quadratic_spline(
            x1, uw, uh, ud, inverse=reverse, tail_bound=self.tail_bound
        )
        x = torch.cat([x0, x1], dim=1) * x_mask
        logdet = torch.sum(logabsdet * x_mask, [1, 2])
        if not reverse:
            return x, logdet
        return x

    def reverse_cl(self, x: torch.Tensor, x_mask_cf: torch.Tensor,
                   x_mask_cl: torch.Tensor,
                   g_cl: torch.Tensor) -> torch.Tensor:
        """Inverse spline coupling with the DDS conditioner running
        channel-last; z itself stays [B,2,T] (tiny)."""
        B, _, T = x.shape
        x0, x1 = x[:, : self.half], x[:, self.half :]
        h = F.linear(x0.transpose(1, 2), self.pre.weight.squeeze(-1),
                     self.pre.bias)  # [B,T,F]
        h = self.convs.forward_cl(h, x_mask_cl, g=g_cl)
        h = F.linear(h, self.proj.weight.squeeze(-1),
                     self.proj.bias) * x_mask_cl  # [B,T,half*(3b-1)]
        h = h.view(B, T, self.half, 3 * self.num_bins - 1).permute(0, 2, 1, 3)
        scale = math.sqrt(self.filter_channels)
        uw = h[..., : self.num_bins] / scale
        uh = h[..., self.num_bins : 2 * self.num_bins] / scale
        ud = h[..., 2 * self.num_bins :]
        x1, _ = rational_quadratic_spline(
            x1, uw, uh, ud, inverse=True, tail_bound=self.tail_bound
        )
        return torch.cat([x0, x1], dim=1) * x_mask_cf


class Flip(nn.Module):  # noqa: E302
    def forward(self, x, *args, reverse=False, **kwargs):
        x = torch.flip(x, [1])
        if not reverse:
            return x, torch.zeros(x.shape[0], device=x.device, dtype=x.dtype)
        return x


class StochasticDurationPredictor(nn.Module):
    def __init__(self, in_channels: int, filter_channels: int,
                 kernel_size: int = 3, n_flows: int = 4,
                 gin_channels: int = 0):
        super().__init__()
        self.pre = nn.Conv1d(in_channels, filter_channels, 1)
        self.convs = DDSConv(filter_channels, kernel_size, n_layers=3)
        self.proj = nn.Conv1d(filter_channels, filter_channels, 1)
        self.flows = nn.ModuleList([ElementwiseAffine(2)])
        for _ in range(n_flows):
            self.flows.append(ConvFlow(2, filter_channels, kernel_size, 3))
            self.flows.append(Flip())
        if gin_channels:
            self.cond = nn.Conv1d(gin_channels, filter_channels, 1)
        else:
            self.cond = None

    def infer(self, x: torch.Tensor, x_mask: torch.Tensor,
              g: Optional[torch.Tensor] = None, noise_scale: float = 0.8,
              noise: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Reverse pass: sample log-durations. x: [B, H, T] text states.
        `noise` is [B, 2, T] standard-normal (masked per utterance so that
        batch composition cannot change an utterance's durations)."""
        from ..ops import use_hip

        if use_hip(x):
            return self._infer_cl(x, x_mask, g, noise_scale, noise)
        x = conv_mod(self.pre, x.detach())
        if g is not None and self.cond is not None:
            x = x + conv_mod(self.cond, g.detach())
        x = self.convs(x, x_mask)
        x = conv_mod(self.proj, x) * x_mask
        flows = list(reversed(self.flows))
        flows = flows[:-2] + [flows[-1]]  # drop the final (unused) Flip pair
        if noise is None:
            noise = torch.randn((x.shape[0], 2, x.shape[2]),
                                device=x.device, dtype=x.dtype)
        z = noise * noise_scale * x_mask
        for flow in flows:
            z = flow(z, x_mask, g=x, reverse=True)
        z0, _ = z.chunk(2, dim=1)
        return z0  # logw [B, 1, T]

    def _infer_cl(self, x: torch.Tensor, x_mask: torch.Tensor,
                  g: Optional[torch.Tensor], noise_scale: float,
                  noise: Optional[torch.Tensor]) -> torch.Tensor:
        """GPU path: DDS/1x1 stages channel-last; the 2-channel flow state
        and the spline stay channel-first (tiny tensors)."""
        mc = x_mask.transpose(1, 2)  # [B,T,1]
        h = F.linear(x.detach().transpose(1, 2),
                     self.pre.weight.squeeze(-1), self.pre.bias)
        if g is not None and self.cond is not None:
            h = h + F.linear(g.detach().squeeze(-1),
                             self.cond.weight.squeeze(-1),
                             self.cond.bias).unsqueeze(1)
        h = self.convs.forward_cl(h, mc)
        h = F.linear(h, self.proj.weight.squeeze(-1), self.proj.bias) * mc
        if noise is None:
            noise = torch.randn((x.shape[0], 2, x.shape[2]),
                                device=x.device, dtype=x.dtype)
        z = noise * noise_scale * x_mask
        flows = list(reversed(self.flows))
        flows = flows[:-2] + [flows[-1]]
        for flow in flows:
            if isinstance(flow, ConvFlow):
                z = flow.reverse_cl(z, x_mask, mc, h)
            elif isinstance(flow, Flip):
                z = torch.flip(z, [1])
            else:  # ElementwiseAffine
                z = flow(z, x_mask, reverse=True)
        z0, _ = z.chunk(2, dim=1)
        return z0


# --------------------------------------------------------------------------- #
# HiFi-GAN generator
# --------------------------------------------------------------------------- #
class ResBlock1(nn.Module):
    """MRF resblock: pairs of dilated+plain convs with LeakyReLU."""

    def __init__(self, channels: int, kernel_size: int, dilations: List[int]):
        super().__init__()
        self.convs1 = nn.ModuleList()
        self.convs2 = nn.ModuleList()
        for d in dilations:
            self.convs1.append(
                nn.Conv1d(channels, channels, kernel_size, dilation=d,
                          padding=(kernel_size - 1) * d // 2)
            )
            self.convs2.append(
                nn.Conv1d(channels, channels, kernel_size, dilation=1,
                          padding=(kernel_size - 1) // 2)
            )

    def forward(self, x: torch.Tensor,
                lengths: Optional[torch.Tensor] = None) -> torch.Tensor:
        for c1, c2 in zip(self.convs1, self.convs2):
            xt = leaky_conv1d(
                x, c1.weight, c1.bias,
                padding=c1.padding[0], dilation=c1.dilation[0],
                pre_lrelu=LRELU_SLOPE,
            )
            mask_tail_(xt, lengths)
            # second conv fuses the residual add into its epilogue
            x = leaky_conv1d(
                xt, c2.weight, c2.bias, padding=c2.padding[0],
                pre_lrelu=LRELU_SLOPE, residual=x,
            )
            mask_tail_(x, lengths)
        return x

    def forward_cl(self, x: torch.Tensor,
                   lengths: Optional[torch.Tensor] = None,
                   accum: Optional[torch.Tensor] = None,
                   out_scale: float = 1.0) -> torch.Tensor:
        """Channel-last ([B,T,C]) path: each conv pair runs as ONE fused
        kernel with the intermediate tensor in LDS (csrc/resblock_cl.hip).
        `accum`/`out_scale` apply to the LAST pair (MRF sum fusion);
        numerically equivalent to forward() then +accum, *out_scale."""
        from ..ops import resblock_pair_cl
        from ..ops.functional import resblock_chain_cl

        import os

        if os.environ.get("SONATA_RB_CHAIN", "0") == "1":
            # whole-resblock fusion (opt-in): DOCUMENTED NEGATIVE RESULT
            # (profiles/r02_rbchain_ab.log, 2-3.4x slower) - the 6 GEMM
            # stages serialize on each block's critical path (~32 W-load/
            # barrier windows) and 2-3 blocks/CU of overlap cannot hide
            # it across ~10k blocks; the 3x HBM traffic cut never pays.
            # Kept for the kernel-structure record + parity tests.
            y = resblock_chain_cl(self, x, out_lens=lengths, accum=accum,
                                  out_scale=out_scale)
            if y is not None:
                return y
        n = len(self.convs1)
        for i, (c1, c2) in enumerate(zip(self.convs1, self.convs2)):
            last = i == n - 1
            x = resblock_pair_cl(x, c1.weight, c1.bias, c2.weight, c2.bias,
                                 dilation=c1.dilation[0], out_lens=lengths,
                                 accum=accum if last else None,
                                 out_scale=out_scale if last else 1.0)
        return x


class Generator(nn.Module):
    def __init__(self, arch: VitsArchitecture, gin_channels: int = 0):
        super().__init__()
        self.num_kernels = len(arch.resblock_kernel_sizes)
        self.num_upsamples = len(arch.upsample_rates)
        ch0 = arch.upsample_initial_channel
        self.conv_pre = nn.Conv1d(arch.inter_channels, ch0, 7, padding=3)
        self.ups = nn.ModuleList()
        self.resblocks = nn.ModuleList()
        for i, (r, k) in enumerate(zip(arch.upsample_rates,
                                       arch.upsample_kernel_sizes)):
            in_ch = ch0 // (2 ** i)
            out_ch = ch0 // (2 ** (i + 1))
            self.ups.append(
                nn.ConvTranspose1d(in_ch, out_ch, k, stride=r,
                                   padding=(k - r) // 2)
            )
            for kk, dd in zip(arch.resblock_kernel_sizes,
                              arch.resblock_dilation_sizes):
                self.resblocks.append(ResBlock1(out_ch, kk, dd))
        self.conv_post = nn.Conv1d(ch0 // (2 ** self.num_upsamples), 1, 7,
                                   padding=3, bias=False)
        if gin_channels:
            self.cond = nn.Conv1d(gin_channels, ch0, 1)
        else:
            self.cond = None

    def forward(self, x: torch.Tensor,
                g: Optional[torch.Tensor] = None,
                lengths: Optional[torch.Tensor] = None) -> torch.Tensor:
        # `lengths`: valid frame count per batch row.  Masking the padded
        # tail after every stage makes ragged-batch output bit-equal to
        # single-utterance output (the tail would otherwise leak into the
        # last receptive-field window of valid audio).  All-equal lengths
        # short-circuit inside mask_tail_.
        if lengths is not None and x.shape[0] == 1:
            lengths = None
        from ..ops import use_hip

        if use_hip(x):
            return self._forward_cl(x, g, lengths)
        x = conv_mod(self.conv_pre, x)
        if g is not None and self.cond is not None:
            x = x + conv_mod(self.cond, g)
        mask_tail_(x, lengths)
        for i, up in enumerate(self.ups):
            x = leaky_convtranspose1d(
                x, up.weight, up.bias, stride=up.stride[0],
                padding=up.padding[0], pre_lrelu=LRELU_SLOPE,
            )
            if lengths is not None:
                lengths = lengths * up.stride[0]
                mask_tail_(x, lengths)
            xs = None
            for j in range(self.num_kernels):
                out = self.resblocks[i * self.num_kernels + j](x, lengths)
                xs = out if xs is None else xs + out
            x = xs / self.num_kernels
        x = leaky_conv1d(x, self.conv_post.weight, None, padding=3,
                         pre_lrelu=LRELU_SLOPE)
        return torch.tanh(x)

    def _forward_cl(self, x: torch.Tensor,
                    g: Optional[torch.Tensor],
                    lengths: Optional[torch.Tensor]) -> torch.Tensor:
        """MI355X serving path: activations kept channel-last [B,T,C]
        end-to-end so every conv is a k-contiguous MFMA GEMM; ragged-batch
        masking fused into each conv epilogue."""
        x = x.transpose(1, 2).contiguous()  # [B, F, C]
        x = leaky_conv1d_cl(x, self.conv_pre.weight, self.conv_pre.bias,
                            padding=3, out_lens=lengths)
        if g is not None and self.cond is not None:
            x = x + conv_mod(self.cond, g).transpose(1, 2)
            if lengths is not None:  # re-mask: cond bias hit padded rows
                idx = torch.arange(x.shape[1], device=x.device)
                x = x.masked_fill(
                    (idx.unsqueeze(0) >= lengths.unsqueeze(1)).unsqueeze(-1),
                    0)
        for i, up in enumerate(self.ups):
            if lengths is not None:
                lengths = lengths * up.stride[0]
            x = leaky_convtranspose1d_cl(
                x, up.weight, up.bias, stride=up.stride[0],
                padding=up.padding[0], pre_lrelu=LRELU_SLOPE,
                out_lens=lengths,
            )
            xs = None
            for j in range(self.num_kernels):
                last = j == self.num_kernels - 1
                xs = self.resblocks[i * self.num_kernels + j].forward_cl(
                    x, lengths, accum=xs,
                    out_scale=1.0 / self.num_kernels if last else 1.0)
            x = xs
        x = leaky_conv1d_cl(x, self.conv_post.weight, None, padding=3,
                            pre_lrelu=LRELU_SLOPE, post_tanh=True,
                            out_lens=lengths)
        return x.transpose(1, 2)  # [B, 1, T]


# --------------------------------------------------------------------------- #
# Full model
# --------------------------------------------------------------------------- #
def masked_noise_rows(
    batch: int,
    channels: int,
    max_len: int,
    lengths,
    generators: Optional[List[torch.Generator]],
    device,
    dtype,
) -> torch.Tensor:
    """Standard-normal noise [B, C, T] where row b is drawn from
    generators[b] on its own [C, len_b] grid and zero beyond — so an
    utterance's noise stream is independent of batch padding/composition
    (SURVEY.md §7 hard part 7: seed per utterance, not per rank).

    `lengths` may be a tensor or a plain int list (pre-fetched to avoid
    per-row device syncs on the latency path)."""
    from ..ops import hip_ext, use_hip

    if (generators is not None and str(device).startswith("cuda")
            and os.environ.get("SONATA_FORCE_TORCH", "0") != "1"):
        # ONE launch (csrc/elementwise.hip seeded_noise): counter-based
        # normal noise keyed by (seed, c, t) — replaces B per-row randn
        # launches; the C++ engine uses the same kernel, so engine ==
        # python parity holds on GPU
        ext = hip_ext(required=True)
        seeds = torch.tensor([g.initial_seed() for g in generators],
                             dtype=torch.long, device=device)
        if torch.is_tensor(lengths):
            lens32 = lengths.to(device=device, dtype=torch.int32)
        else:
            lens32 = torch.tensor(lengths, dtype=torch.int32, device=device)
        return ext.seeded_noise(
            batch, channels, max_len, lens32.contiguous(),
            seeds.contiguous(),
            "bf16" if dtype == torch.bfloat16 else "f32")
    if torch.is_tensor(lengths):
        lengths = lengths.tolist()
    out = torch.zeros((batch, channels, max_len), device=device, dtype=dtype)
    for b in range(batch):
        lb = int(lengths[b])
        gen = generators[b] if generators is not None else None
        out[b, :, :lb] = torch.randn((channels, lb), device=device,
                                     dtype=dtype, generator=gen)
    return out


class VitsModel(nn.Module):
    """The complete Piper/VITS inference graph.

    `infer` = the reference's one-shot model (VitsModel session,
    piper/src/lib.rs:342-399); `infer_encoder` + `decode` = the streaming
    encoder.onnx/decoder.onnx pair (:671-763)."""

    def __init__(self, n_vocab: int, arch: VitsArchitecture,
                 n_speakers: int = 1):
        super().__init__()
        self.arch = arch
        self.n_speakers = n_speakers
        gin = arch.gin_channels if n_speakers > 1 else 0
        self.enc_p = TextEncoder(n_vocab, arch.inter_channels, arch)
        self.dp = StochasticDurationPredictor(
            arch.hidden_channels, 192, kernel_size=3, n_flows=4,
            gin_channels=gin,
        )
        self.flow = ResidualCouplingBlock(
            arch.inter_channels, arch.hidden_channels, 5, 1, 4,
            gin_channels=gin,
        )
        self.dec = Generator(arch, gin_channels=gin)
        if n_speakers > 1:
            self.emb_g = nn.Embedding(n_speakers, arch.gin_channels)
        else:
            self.emb_g = None

    # ------------------------------------------------------------------ #
    def infer_encoder(
        self,
        ids: torch.Tensor,
        lengths: torch.Tensor,
        sid: Optional[torch.Tensor] = None,
        noise_scale: float = 0.667,
        length_scale: float = 1.0,
        noise_w: float = 0.8,
        generators: Optional[List[torch.Generator]] = None,
    ) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
        """phoneme ids -> latent frames.  Returns (z, y_mask, g):
        z [B, C, F], y_mask [B, 1, F], g [B, gin, 1] or None.
        `generators`: one torch.Generator per utterance (deterministic,
        batch-composition-independent sampling)."""
        B = ids.shape[0]
        g = None
        if self.emb_g is not None:
            if sid is None:
                sid = torch.zeros(ids.shape[0], dtype=torch.long,
                                  device=ids.device)
            g = self.emb_g(sid).unsqueeze(-1)  # [B, gin, 1]
        dtype = self.enc_p.emb.weight.dtype
        sdp_noise = masked_noise_rows(B, 2, ids.shape[1], lengths, generators,
                                      ids.device, dtype)
        x, m_p, logs_p, x_mask, logw = self.encode_phase1(
            ids, lengths, g, noise_w, sdp_noise)
        return self.encode_phase2(m_p, logs_p, x_mask, logw, g,
                                  noise_scale, length_scale, generators)

    def encode_phase2(self, m_p, logs_p, x_mask, logw, g,
                      noise_scale, length_scale, generators):
        """Duration -> frames -> prior -> flow: the shape-DYNAMIC suffix
        of the encoder (frame count F is data-dependent; stays eager)."""
        B = m_p.shape[0]
        w = torch.exp(logw) * x_mask * length_scale
        w_ceil = torch.ceil(w)
        y_lengths = torch.clamp_min(torch.sum(w_ceil, [1, 2]), 1).long()
        # ONE host sync fetches all frame counts (the latency path was
        # paying several pipeline drains: sequence_mask .max().item(),
        # expand_states F_max, per-row noise .item()s)
        lens_list = y_lengths.tolist()
        F_max = max(lens_list)
        y_mask = sequence_mask(y_lengths, F_max).to(m_p.dtype)
        durations = w_ceil.squeeze(1).long()
        m_p_f = expand_states(m_p, durations, y_lengths, F_max)
        logs_p_f = expand_states(logs_p, durations, y_lengths, F_max)
        prior_noise = masked_noise_rows(
            B, m_p_f.shape[1], m_p_f.shape[2], lens_list, generators,
            m_p.device, m_p.dtype,
        )
        z_p = prior_sample(m_p_f, logs_p_f, y_mask, prior_noise, noise_scale)
        from ..ops import use_hip

        if use_hip(z_p):
            # channel-last flow: one transpose in, one out (the decoder's
            # cl path transposes again at entry; net cost ~zero, and all
            # WN convs/gates/1x1s run on contiguous channel rows)
            mask_cl = y_mask.transpose(1, 2)  # [B,F,1]
            z_cl = self.flow.reverse_cl(
                z_p.transpose(1, 2).contiguous(), mask_cl, g=g)
            return z_cl.transpose(1, 2), y_mask, g
        z = self.flow(z_p, y_mask, g=g, reverse=True)
        return z, y_mask, g

    def encode_phase1(self, ids, lengths, g, noise_w, sdp_noise):
        """Text encoder + duration predictor: the shape-static prefix of
        the encoder (hipGraph-capturable per padded-T bucket — everything
        after depends on the predicted frame count).  Padding invariance
        holds exactly (masked everywhere), so callers may pad T freely."""
        x, m_p, logs_p, x_mask = self.enc_p(ids, lengths)
        logw = self.dp.infer(x, x_mask, g=g, noise_scale=noise_w,
                             noise=sdp_noise)
        return x, m_p, logs_p, x_mask, logw

    def decode(self, z: torch.Tensor, y_mask: torch.Tensor,
               g: Optional[torch.Tensor] = None,
               lengths: Optional[torch.Tensor] = None) -> torch.Tensor:
        """latent frames -> waveform [B, 1, F*hop].  `lengths` (valid
        frames per row) makes ragged-batch decode match single-utterance
        decode exactly; omit for uniform batches / streaming chunks."""
        return self.dec(z * y_mask, g=g, lengths=lengths)

    @torch.no_grad()
    def infer(self, ids, lengths, sid=None, noise_scale=0.667,
              length_scale=1.0, noise_w=0.8, generators=None):
        z, y_mask, g = self.infer_encoder(
            ids, lengths, sid, noise_scale, length_scale, noise_w, generators
        )
        y_lengths = y_mask.squeeze(1).sum(-1).long()
        audio = self.decode(z, y_mask, g, lengths=y_lengths)
        return audio, y_lengths * self.arch.hop_length
