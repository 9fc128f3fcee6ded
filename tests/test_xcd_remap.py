"""The XCD-aware blockIdx remap used by both GEMM kernels
(native/loadgen/gemm_bf16*.hip) must be a BIJECTION for every grid size —
a non-bijective remap silently drops/duplicates output tiles (the naive
`(b%8)*ceil(n/8)+b/8` form breaks when n%8 != 0). This replicates the
kernel formula bit-for-bit and checks it exhaustively, plus the 4x4
super-tile rasterization's coverage."""


def xcd_remap(wgid: int, nwg: int) -> int:
    q, r = nwg >> 3, nwg & 7
    xcd, pos = wgid & 7, wgid >> 3
    return (xcd * (q + 1) if xcd < r else r * (q + 1) + (xcd - r) * q) + pos


def super4(tile: int, n_tiles_n: int) -> tuple:
    sb, wi = tile >> 4, tile & 15
    sbn = n_tiles_n >> 2
    tm = (sb // sbn) * 4 + (wi >> 2)
    tn = (sb % sbn) * 4 + (wi & 3)
    return tm, tn


def test_remap_bijective_all_sizes():
    for nwg in list(range(1, 130)) + [256, 768, 1000, 1024, 2048]:
        seen = {xcd_remap(w, nwg) for w in range(nwg)}
        assert seen == set(range(nwg)), f"nwg={nwg} not bijective"


def test_remap_groups_consecutive_tiles_per_xcd():
    # block b runs on XCD b%8; the remap must give XCD k a CONTIGUOUS tile
    # range so its L2 sees contiguous panels
    nwg = 1024
    per_xcd = {}
    for w in range(nwg):
        per_xcd.setdefault(w & 7, []).append(xcd_remap(w, nwg))
    for k, tiles in per_xcd.items():
        tiles.sort()
        assert tiles == list(range(tiles[0], tiles[0] + len(tiles))), k


def test_super4_covers_grid():
    for tm_tiles, tn_tiles in [(8, 8), (32, 32), (64, 32), (4, 4)]:
        n = tm_tiles * tn_tiles
        seen = {super4(t, tn_tiles) for t in range(n)}
        assert len(seen) == n
        assert seen == {(i, j) for i in range(tm_tiles) for j in range(tn_tiles)}


def test_super4_blocks_are_4x4():
    # 16 consecutive tile indices = one 4x4 block of the tile grid
    tn_tiles = 32
    for blk in range(0, 8):
        coords = [super4(blk * 16 + i, tn_tiles) for i in range(16)]
        tms = {c[0] for c in coords}
        tns = {c[1] for c in coords}
        assert len(tms) == 4 and max(tms) - min(tms) == 3
        assert len(tns) == 4 and max(tns) - min(tns) == 3
