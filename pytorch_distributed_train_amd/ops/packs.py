"""Per-step batched weight packing (GPU).

With flat master storage, every layer's compute-dtype weight packs (conv
wt/w2, linear cast + transposed cast) are slices of ONE pool refreshed by
ONE kernel per step — instead of ~4 torch permute/cast launches per layer.
Modules pick up their pool views via `_pack_*` attributes; without a
manager (CPU path, standalone tests) they fall back to ad-hoc packing.
"""
from __future__ import annotations

import torch

from .backend import ext


def _align8(n: int) -> int:
    return (n + 7) // 8 * 8


class PackManager:
    def __init__(self, model, flat, dtype, device):
        self.dtype = dtype
        self.device = device
        self.master = flat.master
        base_ptr = flat.master.data_ptr()

        segs = []   # (kind, src_off, dst_off, K, C, R, S, numel)
        tsegs = []  # kind-3 routed to the LDS-tiled transpose kernel:
                    # (src_off, dst_off, N, K) for [N][K] fp32 -> [K][N]
        assign = []  # (module, attr, dst_off, shape)
        dst = 0

        def add(kind, param, shape_dims, attr, mod, view_shape):
            nonlocal dst
            src_off = (param.data_ptr() - base_ptr) // 4
            numel = param.numel()
            if kind == 3 and shape_dims[1] % 8 == 0:
                # shape_dims = (K, N, 1, 1) with master [N][K]; the
                # gather form reads one 4-B element per cache line at
                # large K (16x amplification) — use the tiled kernel.
                # N % 8 keeps every dst row 16-B aligned for store8;
                # others (none today) stay on the gather form.
                tsegs.append((src_off, dst, shape_dims[1], shape_dims[0]))
            else:
                segs.append((kind, src_off, dst, *shape_dims, numel))
            assign.append((mod, attr, dst, view_shape))
            dst += _align8(numel)

        from ..models.layers import Conv2dNHWC, Linear
        from ..models import vit as vit_mod
        for mod in model.modules():
            if isinstance(mod, Conv2dNHWC):
                k, c, r, s = mod.weight.shape
                add(0, mod.weight, (k, c, r, s), "_pack_wt", mod,
                    (k, r, s, c))
                add(1, mod.weight, (k, c, r, s), "_pack_w2", mod,
                    (r, s, c, k))
            elif isinstance(mod, Linear):
                n, kk = mod.weight.shape
                add(2, mod.weight, (kk, n, 1, 1), "_pack_w", mod, (n, kk))
                add(3, mod.weight, (kk, n, 1, 1), "_pack_wt", mod, (kk, n))
            elif isinstance(mod, vit_mod.MultiheadSelfAttention):
                n, kk = mod.in_proj_weight.shape
                add(2, mod.in_proj_weight, (kk, n, 1, 1), "_pack_inproj",
                    mod, (n, kk))
                add(3, mod.in_proj_weight, (kk, n, 1, 1), "_pack_inproj_t",
                    mod, (kk, n))

        self.pool = torch.empty(dst, dtype=dtype, device=device)
        self.desc = torch.tensor([list(s) for s in segs], dtype=torch.long,
                                 device=device)
        chunks = [_align8(s[-1]) // 8 for s in segs]
        prefix = torch.tensor(chunks, dtype=torch.long,
                              device=device).cumsum(0)
        self.prefix = prefix.contiguous()
        self.total_chunks = int(prefix[-1]) if chunks else 0
        self.tdesc = torch.tensor([list(s) for s in tsegs],
                                  dtype=torch.long, device=device)
        tiles = [((n + 63) // 64) * ((k + 63) // 64)
                 for (_, _, n, k) in tsegs]
        tprefix = torch.tensor(tiles, dtype=torch.long,
                               device=device).cumsum(0)
        self.tprefix = tprefix.contiguous()
        self.total_tiles = int(tprefix[-1]) if tiles else 0

        for mod, attr, off, shape in assign:
            n = 1
            for d in shape:
                n *= d
            setattr(mod, attr, self.pool[off:off + n].view(shape))

    def refresh(self) -> None:
        if self.total_chunks:
            ext().pack_weights(self.master, self.pool, self.desc,
                               self.prefix, self.total_chunks)
        if self.total_tiles:
            ext().pack_transpose(self.master, self.pool, self.tdesc,
                                 self.tprefix, self.total_tiles)
