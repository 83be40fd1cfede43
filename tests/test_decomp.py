"""Decomposition geometry unit tests (device-free index math)."""

import pytest

from parallel_convolution_amd.parallel.decomp import (
    ChunkPlan,
    factor_grid,
    make_blocks,
    max_chunk_iters,
    plan_chunks,
    split_extents,
)


def test_factor_grid_mandated_points():
    # BASELINE.json:7-11 scaling points: 1, 2, 4, 8 ranks
    assert factor_grid(1) == (1, 1)
    assert factor_grid(2) == (1, 2)
    assert factor_grid(4) == (2, 2)
    assert factor_grid(8) == (2, 4)
    assert factor_grid(6) == (2, 3)
    assert factor_grid(7) == (1, 7)


def test_split_extents_covers():
    for n in [10, 16384, 17, 1]:
        for p in [1, 2, 3, 4, 7]:
            if p > n:
                continue
            ext = split_extents(n, p)
            assert ext[0][0] == 0
            assert sum(s for _, s in ext) == n
            for i in range(1, p):
                assert ext[i][0] == ext[i - 1][0] + ext[i - 1][1]
            sizes = [s for _, s in ext]
            assert max(sizes) - min(sizes) <= 1


def test_blocks_tile_image_exactly():
    for p in [1, 2, 4, 8]:
        blocks = make_blocks(100, 237, p, channels=3)
        covered = set()
        for b in blocks:
            for y in range(b.y0, b.y0 + b.h):
                for x in range(b.x0, b.x0 + b.w):
                    assert (y, x) not in covered
                    covered.add((y, x))
        assert len(covered) == 100 * 237


def test_neighbors_symmetric():
    blocks = make_blocks(64, 64, 8)
    by_rank = {b.rank: b for b in blocks}
    for b in blocks:
        if b.north is not None:
            assert by_rank[b.north].south == b.rank
        if b.south is not None:
            assert by_rank[b.south].north == b.rank
        if b.east is not None:
            assert by_rank[b.east].west == b.rank
        if b.west is not None:
            assert by_rank[b.west].east == b.rank
    # 2x4 grid: rank 0 is NW corner
    b0 = by_rank[0]
    assert b0.north is None and b0.west is None
    assert b0.south == 4 and b0.east == 1


def test_same_axis_sizes_match():
    # N/S neighbors must share w; E/W neighbors must share h (exchange
    # strip shapes depend on it)
    blocks = make_blocks(101, 237, 8)
    by_rank = {b.rank: b for b in blocks}
    for b in blocks:
        if b.south is not None:
            assert by_rank[b.south].w == b.w and by_rank[b.south].x0 == b.x0
        if b.east is not None:
            assert by_rank[b.east].h == b.h and by_rank[b.east].y0 == b.y0


def test_chunk_plan_regions():
    blocks = make_blocks(64, 64, 4, halo=16)
    plans = plan_chunks(100, 1, 16)
    assert sum(p.iters for p in plans) == 100
    assert plans[0].iters == 16 and plans[0].depth == 16
    assert plans[-1].iters == 4 and plans[-1].depth == 4
    b = blocks[0]  # NW corner: neighbors S and E only
    p0 = plans[0]
    # iteration 0: ext = 15 on neighbor sides, 0 on physical sides
    y0, y1, x0, x1 = p0.region(b, 0)
    assert (y0, x0) == (16, 16)  # physical N/W edges not extended
    assert y1 == 16 + 32 + 15 and x1 == 16 + 32 + 15
    # final iteration: exactly the owned block
    y0, y1, x0, x1 = p0.region(b, 15)
    assert (y0, y1, x0, x1) == (16, 48, 16, 48)


def test_max_chunk_iters_small_blocks():
    blocks = make_blocks(8, 8, 4, halo=16)  # 4x4 blocks
    assert max_chunk_iters(blocks, radius=1, halo=16) == 4
    assert max_chunk_iters(blocks, radius=2, halo=16) == 2
    single = make_blocks(8, 8, 1, halo=16)
    assert max_chunk_iters(single, radius=2, halo=16) == 8


def test_pitch_alignment():
    blocks = make_blocks(1000, 1000, 8, channels=3, halo=16)
    for b in blocks:
        assert b.buf_w_elems % 64 == 0
        assert b.buf_w_elems >= (b.w + 2 * b.halo) * 3


def test_max_chunk_caps_at_block_dims():
    """Exchange depth d = r*T must fit the sender's OWNED region on every
    neighbored axis — the staged N/S corner patches index recv buffers at
    row h-d and require d <= h (csrc exchange_nccl); same for columns."""
    from parallel_convolution_amd.parallel.decomp import (
        halo_for_radius,
        make_blocks,
        max_chunk_iters,
    )

    for (H, W, p, r) in [(55, 662, 8, 1), (27, 500, 8, 2), (500, 30, 6, 2),
                         (64, 64, 4, 1), (40, 40, 8, 2)]:
        halo = halo_for_radius(r)
        blocks = make_blocks(H, W, p, 1, halo=halo)
        T = max_chunk_iters(blocks, r, halo)
        d = r * T
        assert d <= halo
        for b in blocks:
            if b.north is not None or b.south is not None:
                assert d <= b.h, (H, W, p, r, b)
            if b.west is not None or b.east is not None:
                assert d <= b.w, (H, W, p, r, b)
