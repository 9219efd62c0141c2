"""PackManager descriptor construction (CPU: no kernels run — refresh()
is GPU-only, but segment routing, offsets, and pool views are pure
logic that the GPU numerics tests consume blindly)."""
import torch

from pytorch_distributed_train_amd.models import build_model
from pytorch_distributed_train_amd.parallel.flat import FlatParams
from pytorch_distributed_train_amd.ops.packs import PackManager, _align8


def _mgr(model_name):
    model = build_model(model_name, 10, torch.bfloat16)
    flat = FlatParams(model.named_parameters(), device="cpu")
    return model, PackManager(model, flat, torch.bfloat16, "cpu")


def test_vit_linear_weights_route_to_tiled_transpose():
    model, m = _mgr("vit_b_16")
    # every encoder Linear + in_proj transpose pack must use the tiled
    # kernel (N % 8 == 0 holds for all ViT-B shapes) — the gather form
    # is 16x read-amplified at these K (packs.hip)
    assert m.total_tiles > 0
    assert m.tdesc.shape[1] == 4
    for src, dst, n, k in m.tdesc.tolist():
        assert n % 8 == 0 and n > 0 and k > 0
        assert dst % 8 == 0  # _align8 keeps every dst slice 16B-aligned
    # kind-3 may remain in the gather-kernel desc ONLY for rows whose
    # dst-row length breaks store8 alignment (this test's 10-class head)
    for kind, src, dst, K, C, R, S, numel in m.desc.tolist():
        if kind == 3:
            assert C % 8 != 0, "aligned transpose left on gather path"
    # tile prefix is consistent with the per-seg tile counts
    tiles = [((n + 63) // 64) * ((k + 63) // 64)
             for (_, _, n, k) in m.tdesc.tolist()]
    assert m.total_tiles == sum(tiles)


def test_resnet_conv_weights_route_to_gather_kinds():
    model, m = _mgr("resnet18")
    kinds = [row[0] for row in m.desc.tolist()]
    assert 0 in kinds and 1 in kinds  # conv wt + w2 packs
    # conv segs carry (K, C, R, S) consistent with numel
    for kind, src, dst, K, C, R, S, numel in m.desc.tolist():
        if kind in (0, 1):
            assert K * C * R * S == numel
        assert dst % 8 == 0


def test_pool_views_match_module_attributes():
    model, m = _mgr("resnet18")
    from pytorch_distributed_train_amd.models.layers import Conv2dNHWC
    for mod in model.modules():
        if isinstance(mod, Conv2dNHWC):
            k, c, r, s = mod.weight.shape
            assert mod._pack_wt.shape == (k, r, s, c)
            assert mod._pack_w2.shape == (r, s, c, k)
            assert mod._pack_wt.dtype == torch.bfloat16
            # views alias the single pool tensor
            assert mod._pack_wt.data_ptr() >= m.pool.data_ptr()
