"""ViT-B/16 from scratch (BASELINE config #5: "transformer MFMA GEMM
path"). State-dict keys follow torchvision VisionTransformer naming
(class_token, conv_proj, encoder.pos_embedding,
encoder.layers.encoder_layer_i.{ln_1, self_attention.*, ln_2, mlp.0/3},
encoder.ln, heads.head) for checkpoint parity.

Compute design (MI355X-first): the patch embedding is a 16x16/s16 conv
through the same im2col+MFMA igemm path as the ResNet stem; the token
stream is PADDED from 197 to 256 rows so every GEMM K/M dim is a multiple
of 64 — padded rows never mix into real rows (softmax masks key columns
>= 197, padded V rows are zero, LN/MLP are row-local), so results for the
class token are exact. Attention runs as batched MFMA GEMMs (QK^T in
fp32) + a fused masked-softmax kernel (csrc/ln.hip).
"""
from __future__ import annotations

from collections import OrderedDict

import torch
import torch.nn as nn

from ..ops import functional as OF
from .layers import Conv2dNHWC, Linear

SEQ = 197      # 14*14 patches + class token
SEQ_PAD = 256  # padded to a GEMM-friendly length


class LayerNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x):
        on_direct = x.is_cuda and self.training
        return OF.layer_norm(x, self.weight, self.bias, self.eps,
                             gslot_w=(getattr(self, "_gslot_w", None)
                                      if on_direct else None),
                             gslot_b=(getattr(self, "_gslot_b", None)
                                      if on_direct else None))

    def forward_fused(self, a, b):
        """(ln(a+b), a+b) — fused residual join (AddLayerNormFn); the
        pair-threaded encoder keeps the sum single-consumer so no
        autograd fork add exists."""
        if b is None:
            return self.forward(a), a
        on_direct = a.is_cuda and self.training
        return OF.add_layer_norm(
            a, b, self.weight, self.bias, self.eps,
            gslot_w=(getattr(self, "_gslot_w", None)
                     if on_direct else None),
            gslot_b=(getattr(self, "_gslot_b", None)
                     if on_direct else None))


class MultiheadSelfAttention(nn.Module):
    """torchvision-compatible parameter layout: in_proj_weight [3d, d],
    in_proj_bias [3d], out_proj Linear."""

    def __init__(self, dim: int, num_heads: int):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.in_proj_weight = nn.Parameter(torch.empty(3 * dim, dim))
        self.in_proj_bias = nn.Parameter(torch.zeros(3 * dim))
        nn.init.xavier_uniform_(self.in_proj_weight)
        self.out_proj = Linear(dim, dim)

    def forward(self, x, nvalid: int | None = None):
        # x [B, s, d] — the encoder runs PACKED (s = 197): only this
        # module pads to the attention kernel's S=256 tile and slices
        # back, so every GEMM/LN/GELU in the block skips the ~30% dead
        # rows the old pad-once design carried through the whole encoder
        b, s, d = x.shape
        if nvalid is None:
            nvalid = s
        h, hd = self.num_heads, self.head_dim
        packed = getattr(self, "_pack_inproj", None)
        packed_t = getattr(self, "_pack_inproj_t", None)
        if packed is None or packed.dtype != x.dtype:
            packed = self.in_proj_weight.to(x.dtype)
            packed_t = None
        on_direct = x.is_cuda and self.training
        qkv = OF.linear(x.reshape(b * s, d), self.in_proj_weight, packed,
                        self.in_proj_bias, packed_t,
                        gslot_w=(getattr(self, "_gslot_inw", None)
                                 if on_direct else None),
                        gslot_b=(getattr(self, "_gslot_inb", None)
                                 if on_direct else None))
        scale = 1.0 / (hd ** 0.5)
        if x.is_cuda:
            # zero-copy path: q/k/v read in place, grads scattered in
            # place. The fused kernel consumes the PACKED [b, s=197, 3d]
            # qkv directly (rows past nvalid are clamped/zeroed inside;
            # attn.hip header) — no pad/slice copies anywhere.
            if hd == 64 and 64 <= s <= SEQ_PAD:
                out = OF.attention_qkv(qkv, b, s, h, hd, nvalid, scale)
                return self.out_proj(out.reshape(b * s, d)).view(b, s, d)
            # other shapes: pad to the decomposed path's 64-multiple
            import torch.nn.functional as TF
            sp = (s + 63) // 64 * 64
            q3 = qkv
            if sp != s:
                q3 = TF.pad(qkv.view(b, s, 3 * d),
                            (0, 0, 0, sp - s)).reshape(b * sp, 3 * d)
            out = OF.attention_qkv(q3, b, sp, h, hd, nvalid, scale)
            out = out.reshape(b, sp, d)
            if sp != s:
                out = out[:, :s].contiguous()
            return self.out_proj(out.reshape(b * s, d)).view(b, s, d)
        qkv = qkv.view(b, s, 3, h, hd).permute(2, 0, 3, 1, 4)
        q = qkv[0].reshape(b * h, s, hd).contiguous()
        k = qkv[1].reshape(b * h, s, hd).contiguous()
        v = qkv[2].reshape(b * h, s, hd).contiguous()
        out = OF.attention(q, k, v, nvalid, scale)
        out = out.view(b, h, s, hd).permute(0, 2, 1, 3).reshape(b * s, d)
        return self.out_proj(out.contiguous()).view(b, s, d)


class GELU(nn.Module):
    def forward(self, x):
        return OF.gelu(x)


class EncoderBlock(nn.Module):
    def __init__(self, dim: int, num_heads: int, mlp_dim: int):
        super().__init__()
        self.ln_1 = LayerNorm(dim)
        self.self_attention = MultiheadSelfAttention(dim, num_heads)
        self.ln_2 = LayerNorm(dim)
        self.mlp = nn.Sequential(
            Linear(dim, mlp_dim), GELU(), nn.Identity(), Linear(mlp_dim, dim))

    def forward(self, pair):
        # (residual, branch) pair threading: each block receives its
        # input UNSUMMED; ln_1 performs the previous join, ln_2 the
        # attention join, and the mlp branch is handed onward unsummed
        u, v = pair if isinstance(pair, tuple) else (pair, None)
        b, s, d = u.shape
        y1, x1 = self.ln_1.forward_fused(u, v)
        a = self.self_attention(y1)
        y2, x2 = self.ln_2.forward_fused(x1, a)
        h = y2.reshape(b * s, d)
        h = self.mlp[3](self.mlp[1](self.mlp[0](h))).view(b, s, d)
        return (x2, h)


class Encoder(nn.Module):
    def __init__(self, num_layers: int, dim: int, num_heads: int,
                 mlp_dim: int):
        super().__init__()
        self.pos_embedding = nn.Parameter(
            torch.empty(1, SEQ, dim).normal_(std=0.02))
        layers = OrderedDict()
        for i in range(num_layers):
            layers[f"encoder_layer_{i}"] = EncoderBlock(dim, num_heads,
                                                        mlp_dim)
        self.layers = nn.Sequential(layers)
        self.ln = LayerNorm(dim)

    def forward(self, x):  # x [B, SEQ, dim] — runs PACKED (no pre-pad)
        x = x + self.pos_embedding.to(x.dtype)
        u, v = self.layers((x, None))
        y, _ = self.ln.forward_fused(u, v)
        return y


class VisionTransformer(nn.Module):
    def __init__(self, image_size=224, patch_size=16, num_layers=12,
                 num_heads=12, hidden_dim=768, mlp_dim=3072,
                 num_classes=1000, compute_dtype=None):
        super().__init__()
        self.compute_dtype = compute_dtype or torch.float32
        self.patch_size = patch_size
        self.hidden_dim = hidden_dim
        self.class_token = nn.Parameter(torch.zeros(1, 1, hidden_dim))
        self.conv_proj = Conv2dNHWC(3, hidden_dim, patch_size,
                                    stride=patch_size, padding=0, bias=True)
        self.encoder = Encoder(num_layers, hidden_dim, num_heads, mlp_dim)
        self.heads = nn.Sequential(OrderedDict(
            [("head", Linear(hidden_dim, num_classes))]))
        nn.init.zeros_(self.heads.head.weight)
        nn.init.zeros_(self.heads.head.bias)

    def forward(self, x):  # NCHW input, like the reference's loader feeds
        b = x.shape[0]
        x = x.permute(0, 2, 3, 1).to(self.compute_dtype).contiguous()
        x = self.conv_proj(x)                      # [B, 14, 14, 768]
        x = x.reshape(b, -1, self.hidden_dim)      # [B, 196, 768]
        cls = self.class_token.to(x.dtype).expand(b, -1, -1)
        x = torch.cat([cls, x], dim=1)             # [B, 197, 768]
        x = self.encoder(x)                        # [B, 256, 768] (padded)
        return self.heads(x[:, 0].contiguous())    # class token -> logits


def vit_b_16(num_classes=1000, compute_dtype=None):
    return VisionTransformer(num_classes=num_classes,
                             compute_dtype=compute_dtype)
