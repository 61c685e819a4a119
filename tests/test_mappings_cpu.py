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


def test_gemm256_v2_stage_read_consistency():
    """Full addressing simulation of the experimental v2 kernel
    (csrc/kernels/gemm256_v2.hip): emulate stage_half's global->LDS
    placement and read_frag's swizzled reads, and check every fragment
    element equals the matrix element the MFMA layout requires. Catches
    swizzle/layout bugs before any GPU time is spent."""
    BM, BK, NTH = 256, 64, 512
    HALF_ELEMS = BM * BK // 2

    # A[row][col] encoded as row * 1000 + col
    def A(row, col):
        return row * 1000 + col

    lds = {}
    # stage both halves of one K-tile (k0 = 0, buf offset ignored)
    for h in range(2):
        for it in range(2):
            for tid in range(NTH):
                q = it * NTH + tid
                dst_byte = (h * HALF_ELEMS + q * 8) * 2
                un = swz_off(dst_byte)
                row = un // (BK * 2)
                col = (un % (BK * 2)) // 2
                # global_load_lds writes lane's 16B at the linear dest
                for e in range(8):
                    lds[h * HALF_ELEMS + q * 8 + e] = A(row, col + e)

    assert len(lds) == BM * BK  # every slot written exactly once

    def read_frag(row, ks, lane):
        byte_off = (row * BK + ks * 32 + (lane >> 4) * 8) * 2
        off = swz_off(byte_off) // 2
        return [lds[off + e] for e in range(8)]

    # every (row, ks, lane) fragment must deliver A[row][ks*32+(l>>4)*8+e]
    for lane in range(64):
        for base_row in range(0, BM, 16):
            row = base_row + (lane & 15)
            for ks in range(2):
                frag = read_frag(row, ks, lane)
                want = [A(row, ks * 32 + (lane >> 4) * 8 + e)
                        for e in range(8)]
                assert frag == want, (row, ks, lane, frag[:2], want[:2])


@pytest.mark.parametrize("seg,chunks", [(4096, 8), (1000 * 8, 8), (64, 8),
                                        (8, 8), (8192, 3), (24, 5)])
def test_rs_chunk_ranges_cover_segment(seg, chunks):
    """Mirror of the reduce_scatter/all_to_all chunk math
    (csrc/kernels/collectives.hip): 8-aligned chunk boundaries computed
    IDENTICALLY by pusher and reducer must tile [0, seg) exactly, with
    trailing chunks allowed to be empty."""
    per = ((seg + chunks - 1) // chunks + 7) & ~7
    covered = []
    for c in range(chunks):
        lo = c * per
        hi = min(lo + per, seg)
        if lo >= seg:
            continue  # empty trailing chunk (flag still published)
        # vectorized path only: the kernels assume seg % 8 == 0, so when
        # it holds every non-empty chunk is 8-aligned
        if seg % 8 == 0:
            assert lo % 8 == 0
        covered.append((lo, hi))
    # exact tiling, no overlap, no gap
    pos = 0
    for lo, hi in covered:
        assert lo == pos and hi > lo
        pos = hi
    assert pos == seg


def test_flash_decode_tr_read_bfrag_layout():
    """Executable mirror of the ds_read_b64_tr_b16 semantics measured in
    scripts/probe/probe_attn.hip (within a 16-lane group, lane k supplies
    an 8-byte word address; lane l receives element l&3 of the words
    fetched by lanes ((l&15)>>2)+4j, j=0..3) against the PV fragment
    addressing in csrc/kernels/attention.hip — proves every lane's
    B-fragment is V[k0+j][col] for its MFMA slot."""
    kD, kTile = 128, 32
    stride = kD + 8  # padded V row

    def V(k, c):
        return k * 1000 + c

    # LDS contents: v_lds[k][c] row-major with padding
    def lds_elem(addr_elems):
        k, c = divmod(addr_elems, stride)
        assert c < kD, "read touched the pad region"
        return V(k, c)

    def tr_read(addrs_bytes, lane):
        """addrs_bytes[l] = per-lane byte address within the wave."""
        grp = lane & ~15
        out = []
        for j in range(4):
            word_lane = grp + (((lane & 15) >> 2) + 4 * j)
            word_addr = addrs_bytes[word_lane]
            assert word_addr % 8 == 0
            out.append(lds_elem(word_addr // 2 + (lane & 3)))
        return out

    for wave in range(4):
        for lane in range(64):
            for h in range(2):
                cg = wave * 2 + h
                k0 = (lane >> 4) * 8
                addrs = [((k0_ + ((ll & 15) >> 2)) * stride
                          + cg * 16 + 4 * (ll & 3)) * 2
                         for ll in range(64)
                         for k0_ in [(ll >> 4) * 8]]
                lo = tr_read(addrs, lane)
                hi = tr_read([a + 4 * stride * 2 for a in addrs], lane)
                col = cg * 16 + (lane & 15)
                want = [V(k0 + j, col) for j in range(4)] + \
                       [V(k0 + 4 + j, col) for j in range(4)]
                assert lo + hi == want, (wave, lane, h)


def test_gemm256_ring_stage_read_consistency():
    """Addressing mirror of the PRODUCTION K-slice-ring kernel
    (csrc/kernels/gemm256.hip stage_slice + kloop fragment reads): the
    chunk-XOR swizzle applied to the global source at stage time must
    cancel against the read-side swizzle for every fragment element."""
    NTH, SLICE_K = 512, 32

    def A(row, col):
        return row * 1000 + col

    lds = {}
    # stage one slice (k0 = 0) of one matrix
    for it in range(2):
        for tid in range(NTH):
            q = it * NTH + tid
            row = q >> 2
            jp = q & 3
            jg = swz(row, jp)
            # global_load_lds lands the lane's 16B at linear chunk q
            for e in range(8):
                lds[q * 8 + e] = A(row, jg * 8 + e)
    assert len(lds) == 256 * SLICE_K

    # kloop read: af[i] = lds[row * SLICE_K + swz(row, jn) * 8 .. +8)
    for lane in range(64):
        jn = lane >> 4
        for base in range(0, 256, 16):
            row = base + (lane & 15)
            off = row * SLICE_K + swz(row, jn) * 8
            frag = [lds[off + e] for e in range(8)]
            want = [A(row, jn * 8 + e) for e in range(8)]
            assert frag == want, (row, jn)


def test_gemm256_accumulator_coverage():
    """The 8-wave (2x4) accumulator layout of the 256^2 kernels must
    cover every C-tile element exactly once: wave (wr, wc), frag (i, j),
    reg r, lane l -> row = wr*128 + i*16 + (l>>4)*4 + r,
    col = wc*64 + j*16 + (l&15). Also validates the v2 quadrant split
    (ih/jh gray order) covers all 32 fragment positions."""
    seen = set()
    for wave in range(8):
        wr, wc = wave >> 2, wave & 3
        for lane in range(64):
            for i in range(8):
                for j in range(4):
                    for r in range(4):
                        row = wr * 128 + i * 16 + ((lane >> 4) * 4 + r)
                        col = wc * 64 + j * 16 + (lane & 15)
                        key = (row, col)
                        assert key not in seen, key
                        seen.add(key)
    assert len(seen) == 256 * 256

    # v2 gray-coded quadrants: phases 0..3 cover each (i, j) frag once
    covered = set()
    for qq in range(4):
        ih = 1 if qq in (2, 3) else 0
        jh = 1 if qq in (1, 2) else 0
        for i in range(4):
            for j in range(2):
                covered.add((ih * 4 + i, jh * 2 + j))
    assert covered == {(i, j) for i in range(8) for j in range(4)}


def test_flash_decode_qk_partial_coverage():
    """QK^T partial mapping (attention.hip): per wave and pos-half, lanes
    write s_part[wave][(l>>4)*4+r][(l&15)+16h] — every (slot, pos) cell
    of each wave's partial plane must be written exactly once."""
    for wave in range(4):
        seen = set()
        for h in range(2):
            for lane in range(64):
                for r in range(4):
                    slot = (lane >> 4) * 4 + r
                    pos = (lane & 15) + 16 * h
                    key = (slot, pos)
                    assert key not in seen, key
                    seen.add(key)
        assert len(seen) == 16 * 32


def test_flash_decode_pv_output_coverage():
    """PV epilogue (attention.hip): wave w, halves h, regs r, lanes l ->
    out[row][w*32 + h*16 + (l&15)] with row = (l>>4)*4 + r; the 4 waves
    together must cover [16 rows] x [128 cols] exactly once."""
    seen = set()
    for wave in range(4):
        for lane in range(64):
            for h in range(2):
                for r in range(4):
                    row = (lane >> 4) * 4 + r
                    col = wave * 32 + h * 16 + (lane & 15)
                    key = (row, col)
                    assert key not in seen, key
                    seen.add(key)
    assert len(seen) == 16 * 128

def test_gemm256_v3_pipeline_ledger():
    """Discrete-event simulation of the PIPELINED v3 schedule
    (csrc/kernels/gemm256_v3.hip): stage order [A0,B1,A1,B0] per tile,
    7-half-tile prologue, one stage per phase TAIL, next-phase reads
    pre-issued in the tail, boundary reads at the top, vmcnt(6) drains at
    the tails of phases 3/7. Events are ordered (phase, seg) with seg
    0=top, 1=tail; a barrier closes every phase. Proves for every even
    ktiles up to 64:
      1. register model: every MFMA phase holds the (tile, half) it
         needs, loaded by a read that issued at or before its use,
      2. drain soundness: every read's target half-tile was staged AND
         retired by a drain whose certifying barrier precedes the read,
      3. overwrite safety: every stage lands >= 1 barrier after the
         replaced slot's last read issue — except the one latency-
         protected class (boundary-top read vs same-phase tail stage,
         separated by lgkmcnt(0) + 16 MFMAs in every wave), which must
         be the ONLY violation and only at qq == 0 phases,
      4. no stage targets a slot that a pre-read in the SAME tail reads.
    """
    for ktiles in [2, 4, 6, 8, 16, 64]:
        total = 4 * ktiles
        pairs = ktiles // 2

        def seq(s):
            # -> (tile, matrix, half); matrix 0 = A, 1 = B
            t = s >> 2
            return [(t, 0, 0), (t, 1, 1), (t, 0, 1), (t, 1, 0)][s & 3]

        # --- build event streams ---------------------------------------
        stage_ev = {}      # s -> (phase, seg)
        for s in range(min(7, total)):
            stage_ev[s] = (-1, 1)
        reads = []         # (phase, seg, tile, matrix, half, use_phase)
        drains = []        # (phase, seg, n_halftiles_allowed_outstanding)
        issued_at = {(-1, 1): min(7, total)}
        drains.append((-1, 1, 3 if total > 3 else 0))
        nxt = 7
        regs = {}
        reg_ok = True
        for p in range(pairs):
            last = (p == pairs - 1)
            for ph in range(8):
                g = p * 8 + ph
                tile = 2 * p + (ph >> 2)
                qq = ph & 3
                ih = 1 if qq >= 2 else 0
                jh = 1 if qq in (1, 2) else 0
                if qq == 0:  # boundary top reads
                    reads.append((g, 0, tile, 0, 0, g))
                    reads.append((g, 0, tile, 1, 0, g))
                    regs[0] = (tile, 0)
                    regs[1] = (tile, 0)
                # MFMA needs (tile, ih)/(tile, jh)
                reg_ok &= regs.get(0) == (tile, ih)
                reg_ok &= regs.get(1) == (tile, jh)
                # tail: stage, then pre-reads, then drain
                if (not last) or ph == 0:
                    if nxt < total:
                        stage_ev[nxt] = (g, 1)
                        nxt += 1
                if qq == 0:
                    reads.append((g, 1, tile, 1, 1, g + 1))
                    regs[1] = (tile, 1)
                elif qq == 1:
                    reads.append((g, 1, tile, 0, 1, g + 1))
                    regs[0] = (tile, 1)
                elif qq == 2:
                    reads.append((g, 1, tile, 1, 0, g + 1))
                    regs[1] = (tile, 0)
                if qq == 3 and (ph == 3 or not last):
                    n = 0 if nxt >= total else 3
                    drains.append((g, 1, n))
                issued_at[(g, 1)] = nxt
        assert reg_ok
        assert nxt == total

        slot_of = {}
        for s in range(total):
            slot_of[seq(s)] = s

        # --- 2. drain soundness ----------------------------------------
        def landed_before(ev):
            # certified-landed stages: drains whose phase closed (barrier)
            # strictly before ev's phase
            best = 0
            for dg, dseg, n in drains:
                if dg < ev[0]:
                    best = max(best, issued_at[(dg, dseg)] - n)
            return best

        for g, seg, tile, mat, half, use in reads:
            s = slot_of[(tile, mat, half)]
            sg = stage_ev.get(s)
            assert sg is not None and sg < (g, seg), (ktiles, g, seg)
            assert s < landed_before((g, seg)), \
                f"ktiles={ktiles} read@({g},{seg}) unretired stage {s}"

        # --- 3. overwrite safety + 4. same-tail conflicts ----------------
        last_read = {}
        for g, seg, tile, mat, half, use in reads:
            key = (tile, mat, half)
            last_read[key] = max(last_read.get(key, (-2, 0)), (g, seg))
        for s in range(8, total):
            prev = seq(s - 8)
            sg, sseg = stage_ev[s]
            lg, lseg = last_read[prev]
            strictly_after = (sg > lg)
            same_phase_protected = (sg == lg and lseg == 0 and sseg == 1
                                    and (sg % 4) == 0)
            assert strictly_after or same_phase_protected, \
                f"ktiles={ktiles} stage {s} vs live slot {prev}"
            # 4: same-tail pre-reads must not touch the staged slot
            cur = seq(s)
            for g2, seg2, tile2, mat2, half2, _ in reads:
                if (g2, seg2) == (sg, 1):
                    assert (tile2, mat2, half2) != cur, (ktiles, s)


@pytest.mark.parametrize("band", [64, 32])
def test_gemm256_v3_stage_quad_consistency(band):
    """Addressing mirror of gemm256_v3.hip stage_quad<BAND>: the staged
    quadrant-union (compact rows -> banded tile rows), the wave-uniform
    LDS base + lane*16B hardware placement, and the swizzled source must
    together land every LDS slot of the quad's rows exactly once with
    the element read_frag expects."""
    BM, BK, NTH = 256, 64, 512

    def A(row, col):
        return row * 1000 + col

    lds = {}
    for h in range(2):
        for it in range(2):
            for wave in range(8):
                cr0 = it * 64 + wave * 8
                row0 = (cr0 // band) * 2 * band + h * band + (cr0 % band)
                base_elem = row0 * BK
                for ln in range(64):
                    tid = wave * 64 + ln
                    q = it * NTH + tid
                    cr = q >> 3
                    row = (cr // band) * 2 * band + h * band + (cr % band)
                    p_byte = (row * BK + (q & 7) * 8) * 2
                    un = p_byte ^ (((p_byte >> 9) & 1) << 5)
                    assert un // (BK * 2) == row  # swizzle is row-local
                    col = (un % (BK * 2)) // 2
                    # HW places lane ln's 16B at base + ln*16
                    dst = base_elem + ln * 8
                    assert dst * 2 == p_byte, (band, h, it, wave, ln)
                    for e in range(8):
                        lds[dst + e] = A(row, col + e)
    # both halves staged -> the full tile is covered exactly once
    assert len(lds) == BM * BK

    def read_frag(row, ks, lane):
        byte_off = (row * BK + ks * 32 + (lane >> 4) * 8) * 2
        off = (byte_off ^ (((byte_off >> 9) & 1) << 5)) // 2
        return [lds[off + e] for e in range(8)]

    for lane in range(64):
        for base_row in range(0, BM, 16):
            row = base_row + (lane & 15)
            for ks in range(2):
                frag = read_frag(row, ks, lane)
                want = [A(row, ks * 32 + (lane >> 4) * 8 + e)
                        for e in range(8)]
                assert frag == want, (band, row, ks, lane)

    # quad-union row sets match exactly what the gray-walk reads consume:
    # A (band 64): wave wr reads rows wr*128 + ih*64 + [0,64)
    # B (band 32): wave wc reads rows wc*64 + jh*32 + [0,32)
    for h in range(2):
        staged = {(cr // band) * 2 * band + h * band + (cr % band)
                  for cr in range(128)}
        if band == 64:
            want = {wr * 128 + h * 64 + r for wr in range(2)
                    for r in range(64)}
        else:
            want = {wc * 64 + h * 32 + r for wc in range(4)
                    for r in range(32)}
        assert staged == want, (band, h)


def _gs_swz(row, j):
    return j ^ ((row >> 1) & 3)


def test_gemm_stream_stage_read_consistency():
    """gemm_stream.hip: each MFMA fragment read returns exactly the
    logical (row, k-chunk) elements the stage wrote (swizzle applied on
    both sides), for A (512x32) and B (64x32)."""
    BK = 32
    # A: instruction it in 0..3, wave, lane -> unit u; LDS elem off = u*8
    lds_src = {}          # lds unit -> (row, logical chunk)
    for it in range(4):
        for wave in range(8):
            for lane in range(64):
                u = it * 512 + wave * 64 + lane
                row, jp = u >> 2, u & 3
                lds_src[u] = (row, _gs_swz(row, jp))
    for wave in range(8):
        for i in range(4):
            for lane in range(64):
                row = wave * 64 + i * 16 + (lane & 15)
                kq = (lane >> 4) * 8
                u = row * 4 + _gs_swz(row, kq >> 3)
                srow, schunk = lds_src[u]
                assert srow == row and schunk * 8 == kq, (wave, i, lane)
    # B: one instruction per wave, lanes 0..31 -> unit wave*32+lane
    lds_src = {}
    for wave in range(8):
        for lane in range(32):
            u = wave * 32 + lane
            row, jp = u >> 2, u & 3
            lds_src[u] = (row, _gs_swz(row, jp))
    assert len(lds_src) == 256  # full 64x32 slice staged exactly once
    for wave in range(8):
        for j in range(4):
            for lane in range(64):
                row = j * 16 + (lane & 15)
                kq = (lane >> 4) * 8
                u = row * 4 + _gs_swz(row, kq >> 3)
                srow, schunk = lds_src[u]
                assert srow == row and schunk * 8 == kq, (wave, j, lane)


def test_gemm_stream_output_coverage():
    """Epilogue (wave, i, j, r, lane) -> (row, col) covers the 512x64
    tile exactly once; s-major grid layout groups one k-chunk per XCD."""
    seen = set()
    for wave in range(8):
        for i in range(4):
            for j in range(4):
                for r in range(4):
                    for lane in range(64):
                        row = wave * 64 + i * 16 + (lane >> 4) * 4 + r
                        col = j * 16 + (lane & 15)
                        key = (row, col)
                        assert key not in seen
                        seen.add(key)
    assert len(seen) == 512 * 64

    # grid layout: wgid = s * tiles_n + tn with xcd_remap means each
    # XCD's contiguous range stays within one s when tiles_n == grid/8
    def xcd_remap(wgid, nwg, nxcd=8):
        if nwg < nxcd:
            return wgid
        xcd, idx = wgid % nxcd, wgid // nxcd
        q, r2 = divmod(nwg, nxcd)
        return (xcd * (q + 1) if xcd < r2
                else r2 * (q + 1) + (xcd - r2) * q) + idx
    tiles_n, sk = 80, 8   # down-proj: n=5120, k=25600
    per_xcd_s = {}
    for b in range(tiles_n * sk):
        w = xcd_remap(b, tiles_n * sk)
        per_xcd_s.setdefault(b % 8, set()).add(w // tiles_n)
    for xcd, ss in per_xcd_s.items():
        assert len(ss) == 1, (xcd, ss)  # one A k-chunk per XCD


import pytest as _pytest


@_pytest.mark.parametrize("nbuf,ksteps", [(2, 1), (2, 5), (3, 1), (3, 2),
                                          (3, 7), (4, 1), (4, 2), (4, 3),
                                          (4, 9)])
def test_moe_pq_pipeline_ledger(nbuf, ksteps):
    """moe.hip k_moe_grouped_gemm_pq<NBUF>: simulate the stage/wait/
    compute schedule and check (a) every compute reads a fully-landed
    stage, (b) no stage overwrites a buffer before its last reader, for
    the generalized NBUF in {2,3,4} waits (0 / 5 / 10)."""
    LOADS = 5
    issued = []          # (step, buf) in issue order
    landed_upto = 0      # prefix of `issued` guaranteed landed

    def stage(step, buf):
        issued.append((step, buf))

    def wait_vmcnt(allowed):
        nonlocal landed_upto
        landed_upto = max(landed_upto, len(issued) * LOADS - allowed)

    stage(0, 0)
    if nbuf >= 3 and ksteps > 1:
        stage(1, 1)
    if nbuf >= 4 and ksteps > 2:
        stage(2, 2)
    buf_owner = {}
    for s, b in issued:
        buf_owner[b] = s
    last_read = {}
    for t in range(ksteps):
        buf = t % nbuf
        if nbuf >= 4 and t + 2 < ksteps:
            wait_vmcnt(10)
        elif nbuf >= 3 and t + 1 < ksteps:
            wait_vmcnt(5)
        else:
            wait_vmcnt(0)
        # barrier here; then stage t+ahead, then compute t
        ahead = nbuf - 1
        if t + ahead < ksteps:
            b2 = (t + ahead) % nbuf
            # overwrite safety: previous occupant's compute finished
            prev = buf_owner.get(b2)
            if prev is not None:
                assert last_read.get(prev, -1) == prev, (t, b2, prev)
            stage(t + ahead, b2)
            buf_owner[b2] = t + ahead
        # compute t: stage t must be fully landed
        idx = issued.index((t, buf))
        assert (idx + 1) * LOADS <= landed_upto, \
            f"stage {t} not landed: {(idx+1)*LOADS} > {landed_upto}"
        last_read[t] = t
    assert len(last_read) == ksteps


@pytest.mark.parametrize("ktiles,kt0", [(1, 0), (4, 0), (4, 3), (32, 7),
                                        (40, 39)])
def test_kloop_stagger_walk(ktiles, kt0):
    """gemm256 kloop StaggerU walk: tiles kt0, kt0+1, ... mod ktiles —
    every K tile consumed exactly once, every stage lands before its
    consume, and the staged (region, slice) sequence matches the ring's
    p-1 staging discipline."""
    consumed = []
    staged = {}  # region -> (tile, slice)
    # prologue: slices 0..2 of tile kt0 into regions 0..2
    for p in range(3):
        staged[p] = (kt0, p)
    for t in range(ktiles):
        tt = kt0 + t
        if tt >= ktiles:
            tt -= ktiles
        tn = 0 if tt + 1 == ktiles else tt + 1
        has_next = (t + 1) < ktiles
        for p in range(4):
            if p == 0:
                staged[3] = (tt, 3)  # current tile's last slice
            elif has_next:
                staged[p - 1] = (tn, p - 1)
            # consume (tile tt, slice p) from region p
            assert staged.get(p) == (tt, p), (t, p, staged.get(p))
        consumed.append(tt)
    assert sorted(consumed) == list(range(ktiles))
    assert consumed[0] == kt0 % ktiles


@pytest.mark.parametrize("m_total,m_per,chunks", [(4096, 2048, 4),
                                                  (1024, 512, 4),
                                                  (2048, 256, 2)])
def test_ag_consumer_chunk_wait_ranges(m_total, m_per, chunks):
    """k_ag_gemm256_consumer flag indexing: every 256-row tile waits on
    exactly the global chunk flags covering its rows, and all indices
    stay inside the world*chunks flag array."""
    world = m_total // m_per
    rows_per_chunk = m_per // chunks
    covered = set()
    for pid_m in range(m_total // 256):
        c_lo = (pid_m * 256) // rows_per_chunk
        c_hi = (pid_m * 256 + 255) // rows_per_chunk
        assert 0 <= c_lo <= c_hi < world * chunks, (pid_m, c_lo, c_hi)
        for c in range(c_lo, c_hi + 1):
            covered.add(c)
            # chunk c's rows overlap the tile's rows
            r0, r1 = c * rows_per_chunk, (c + 1) * rows_per_chunk
            t0, t1 = pid_m * 256, pid_m * 256 + 256
            assert r0 < t1 and t0 < r1
    assert covered == set(range(world * chunks))


def test_split_pick_divisibility_invariants():
    """Every split factor either tuner returns must divide the K-step
    grid it will be launched with (the launchers throw otherwise), over
    a broad shape sweep."""
    from triton_dist_amd.ops.gemm import choose_splits, sk256_pick

    for m in (128, 256, 512, 1024, 4096):
        for n in (256, 1280, 5120, 14336, 51200):
            for k in (512, 2048, 5120, 8192, 25600, 27648):
                s = choose_splits(m, n, k)
                assert s >= 1
                if s > 1:
                    assert (k // 32) % s == 0 and k // s >= 512, (m, n, k, s)
                sk = sk256_pick(m, n, k)
                if sk:
                    assert m % 256 == 0 and n % 256 == 0
                    assert k % (128 * sk) == 0, (m, n, k, sk)
                    assert (k // 128) // sk >= 4
