"""Bijection/coverage proof for the XCD-affinity block map used by the
attention kernels (csrc/common.h xcd_tile_map): every (tile, bh) pair
is produced exactly once, and all tiles of one bh land on one XCD."""

import pytest


def xcd_tile_map(block_id, nbx, grid):
    gy = grid // nbx
    if gy % 8 == 0:
        xcd, idx = block_id % 8, block_id // 8
        return idx % nbx, xcd + 8 * (idx // nbx)
    return block_id % nbx, block_id // nbx


@pytest.mark.parametrize("nbx,gy", [
    (16, 32),   # llama dkv: S=4096/256 tiles x B4*Hkv8
    (32, 64),   # llama dq 4w: S/128 x B2*H32
    (4, 192),   # gpt2 fwd: S=1024/256 x B16*H12
    (2, 4),     # tiny test shapes (gy % 8 != 0 -> identity)
    (8, 96),
    (1, 8),
])
def test_xcd_map_bijective_and_affine(nbx, gy):
    grid = nbx * gy
    seen = set()
    bh_to_xcd = {}
    for bid in range(grid):
        bx, bh = xcd_tile_map(bid, nbx, grid)
        assert 0 <= bx < nbx and 0 <= bh < gy, (bid, bx, bh)
        assert (bx, bh) not in seen, "duplicate mapping"
        seen.add((bx, bh))
        if gy % 8 == 0:
            # affinity: all tile-blocks of one bh on one XCD (id % 8)
            xcd = bid % 8
            assert bh_to_xcd.setdefault(bh, xcd) == xcd, bh
    assert len(seen) == grid  # every (tile, bh) covered exactly once
