"""Python mirrors of the pure device-side index mappings, with exhaustive
bijectivity / involution checks — the bug class the guide's ERRATA #11
documents (a non-bijective XCD swizzle silently dropping tiles at
nwg % 8 != 0) and the swizzle-mismatch class (stage/read disagreeing).

Each mirror matches its HIP source line-for-line; if a kernel mapping
changes, change it here too (the point is to make that a conscious act).
"""
import pytest


def xcd_remap(wgid, nwg, nxcd=8):
    # device.hpp:164-170
    if nwg < nxcd:
        return wgid
    xcd = wgid % nxcd
    idx = wgid // nxcd
    q, r = nwg // nxcd, nwg % nxcd
    return (xcd * (q + 1) if xcd < r
            else r * (q + 1) + (xcd - r) * q) + idx


def tile_coords(wgid, tiles_m, tiles_n, GM=4):
    # gemm256.hip tile_coords (GROUP_M supertile)
    group = wgid // (GM * tiles_n)
    first_m = group * GM
    gsz = min(tiles_m - first_m, GM)
    pid_m = first_m + (wgid % (GM * tiles_n)) % gsz
    pid_n = (wgid % (GM * tiles_n)) // gsz
    return pid_m, pid_n


def swz(row, j):
    # gemm256.hip swz: chunk XOR within a slice row
    return j ^ ((row >> 1) & 3)


def swz_off(byte_off):
    # gemm256_v2.hip st_16x32-style: XOR bit 9 into bit 5
    return byte_off ^ (((byte_off >> 9) & 1) << 5)


@pytest.mark.parametrize("nwg", [1, 7, 8, 9, 64, 80, 96, 100, 255, 256,
                                 257, 768, 1024, 1376, 2048])
def test_xcd_remap_bijective(nwg):
    seen = {xcd_remap(w, nwg) for w in range(nwg)}
    assert seen == set(range(nwg)), nwg


@pytest.mark.parametrize("tm,tn", [(2, 40), (4, 12), (16, 16), (17, 5),
                                   (2, 200), (43, 32), (1, 8)])
def test_tile_coords_bijective(tm, tn):
    seen = {tile_coords(w, tm, tn) for w in range(tm * tn)}
    assert len(seen) == tm * tn
    assert all(0 <= m < tm and 0 <= n < tn for m, n in seen)


def test_swz_involution():
    for row in range(256):
        for j in range(4):
            assert swz(row, swz(row, j)) == j


def test_swz_off_involution_and_chunk_granularity():
    for off in range(0, 32768, 2):
        assert swz_off(swz_off(off)) == off
    # a 16-byte chunk maps contiguously (bit 5 >= chunk granularity)
    for base in range(0, 4096, 16):
        s0 = swz_off(base)
        assert all(swz_off(base + i) == s0 + i for i in range(16))


def test_gemm_dispatch_policies():
    """Lock in the measured routing decisions (profiles/README.md)."""
    from triton_dist_amd.ops.gemm import choose_splits, sk256_pick

    # decode down-proj routes to the sk tier (188 vs 262 us measured)
    assert sk256_pick(512, 5120, 25600) > 0
    # qkv / o / gate_up stay on hipBLASLt (sk tier measured slower)
    assert sk256_pick(512, 10240, 5120) == 0
    assert sk256_pick(512, 5120, 8192) == 0
    assert sk256_pick(512, 51200, 5120) == 0
    # big square shapes: no split-K
    assert choose_splits(4096, 4096, 4096) == 1
    # occupancy-starved decode shard: split-K on
    assert choose_splits(512, 1280, 5120) > 1
