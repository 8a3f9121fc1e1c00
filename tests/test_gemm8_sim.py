"""CPU simulation of the experimental 8-phase GEMM's index arithmetic.

Emulates, in numpy, exactly what gemm8.hip computes: the staging address
permutation (global_load_lds chunks + optional st_16x32 swizzle), the
per-phase LDS fragment reads, and the MFMA 16x16x32 fragment contract
(layouts verified on hardware by mfma_probe). If this reproduces
A @ W^T bit-for-math, the kernel's indexing is correct and only
synchronization remains to validate on the GPU in round 2.
"""

import numpy as np
import pytest

BM = BN = 256
BK = 64
QM, QN = 4, 2


def swz(byte, mode):
    # mode 1 — T2 st_16x32: XOR three row bits into the 16-B chunk index
    # mode 2 — conflict-free 4-bit: XOR (row>>1)&15 into granule bits 4-7
    if mode == 1 or mode is True:
        return byte ^ (((byte >> 7) & 7) << 4)
    if mode == 2:
        return byte ^ (((byte >> 8) & 15) << 4)
    return byte


def stage_tile(src_tile, swizzle):
    """src_tile: [256, 64] fp32 (stands in for bf16). Returns the LDS
    array as the kernel's stage_half writes it: LDS[lin] = data[swz(lin)]
    with 2-byte elements."""
    lds = np.zeros(BM * BK, dtype=src_tile.dtype)
    flat = src_tile.reshape(-1)
    for row0 in (0, 128):
        for chunk in range(16):
            base = row0 * 128 + chunk * 1024  # bytes
            for lane in range(64):
                for e in range(8):  # 16 B = 8 elements
                    dst_byte = base + lane * 16 + e * 2
                    src_byte = swz(base + lane * 16, swizzle) + e * 2
                    lds[dst_byte // 2] = flat[src_byte // 2]
    return lds


def lds_frag(lds, row, col, swizzle):
    """8 consecutive elements at (row, col) through the swizzled view."""
    byte = swz(row * 128 + col * 2, swizzle)
    return lds[byte // 2 : byte // 2 + 8]


def mfma_16x16x32(a_frags, b_frags, acc):
    """Emulate v_mfma_f32_16x16x32_bf16 with the verified layouts:
    A: row=lane&15, k=(lane>>4)*8+j ; B: col=lane&15, same k ;
    C: col=lane&15, row=(lane>>4)*4+reg.
    a_frags/b_frags: [64][8] per-lane fragments. acc: [16,16] (row, col).
    """
    A = np.zeros((16, 32))
    B = np.zeros((32, 16))
    for lane in range(64):
        r = lane & 15
        k0 = (lane >> 4) * 8
        A[r, k0:k0 + 8] = a_frags[lane]
        B[k0:k0 + 8, r] = b_frags[lane]
    return acc + A @ B


def stage_unit(src_tile, lds, kind, row_add, swizzle):
    """Emulate gemm8.hip stage_unit: one 16 KiB phase-granular unit.
    kind 0 = A quarter-pair (rows r..r+63 and 128+r..128+r+63),
    kind 1 = B qn-strip (rows 64w+r..64w+r+31 for w<4)."""
    flat = src_tile.reshape(-1)
    for chunk in range(16):
        if kind == 0:
            row0 = ((chunk >> 3) << 7) + ((chunk & 7) << 3)
        else:
            row0 = ((chunk >> 2) << 6) + ((chunk & 3) << 3)
        row0 += row_add
        base = row0 * 128  # bytes
        for lane in range(64):
            for e in range(8):
                dst_byte = base + lane * 16 + e * 2
                src_byte = swz(base + lane * 16, swizzle) + e * 2
                lds[dst_byte // 2] = flat[src_byte // 2]


@pytest.mark.parametrize("swizzle", [0, 1, 2])
def test_stage_unit_union_equals_full_tile(swizzle):
    """The four half-granular units (A-qp0/A-qp1 or B-qn0/B-qn1 pairs)
    together write exactly what stage_half writes, and each unit touches
    only its liveness set (the rows its two phases read)."""
    rng = np.random.default_rng(1)
    tile = rng.standard_normal((BM, BK))
    ref = stage_tile(tile, swizzle)

    lds = np.full(BM * BK, np.nan)
    stage_unit(tile, lds, 0, 0, swizzle)    # A-qp0
    # rows covered so far: 0-63 and 128-191 only
    view = lds.reshape(BM, BK)
    assert not np.isnan(view[0:64]).any() and not np.isnan(view[128:192]).any()
    assert np.isnan(view[64:128]).all() and np.isnan(view[192:256]).all()
    stage_unit(tile, lds, 0, 64, swizzle)   # A-qp1
    np.testing.assert_array_equal(lds, ref)

    lds = np.full(BM * BK, np.nan)
    stage_unit(tile, lds, 1, 0, swizzle)    # B-qn0
    view = lds.reshape(BM, BK)
    for w in range(4):
        assert not np.isnan(view[w * 64 : w * 64 + 32]).any()
        assert np.isnan(view[w * 64 + 32 : w * 64 + 64]).all()
    stage_unit(tile, lds, 1, 32, swizzle)   # B-qn1
    np.testing.assert_array_equal(lds, ref)


def quarter_wave_bank_starts(swizzle, r0=0, col_bytes=0):
    """LDS bank-group start (addr>>4 mod 16) for each lane of a hardware
    quarter-wave reading rows r0..r0+15 at a fixed column. ds_read_b128 is
    conflict-free iff the 16 starts are a permutation of 0..15 (each 16-B
    read covers 4 banks; 16 x 4 = all 64 banks exactly once)."""
    return [
        (swz((r0 + i) * 128 + col_bytes, swizzle) >> 4) & 15
        for i in range(16)
    ]


def test_swizzle2_reads_are_bank_conflict_free():
    """Mode 2 (4-bit row>>1 XOR) makes every quarter-wave fragment read a
    bank permutation; modes 0/1 do not (the measured residual conflicts).
    Checked across all row offsets and fragment columns the kernel uses."""
    for r0 in range(0, 256, 16):
        for ks in (0, 1):
            for g in range(4):
                col_bytes = ks * 64 + g * 16
                starts = quarter_wave_bank_starts(2, r0, col_bytes)
                assert sorted(starts) == list(range(16)), (r0, col_bytes)
    # and the involution property that makes any mode correct
    for mode in (0, 1, 2):
        for b in range(0, 32768, 97):
            assert swz(swz(b, mode), mode) == b
    # mode 1 leaves a 2-way conflict (rows i and i+8 share a granule)
    starts1 = quarter_wave_bank_starts(1)
    assert sorted(starts1) != list(range(16))


@pytest.mark.parametrize("swizzle", [0, 1, 2])
def test_gemm8_index_math_reproduces_matmul(swizzle):
    rng = np.random.default_rng(0)
    M = N = 256
    K = 128  # 2 K-tiles exercises the double-buffer indexing
    A = rng.standard_normal((M, K)).astype(np.float64)
    W = rng.standard_normal((N, K)).astype(np.float64)
    ref = A @ W.T

    D = np.zeros((M, N))
    for kt in range(K // BK):
        a_lds = stage_tile(np.ascontiguousarray(A[:, kt * BK:(kt + 1) * BK]), swizzle)
        b_lds = stage_tile(np.ascontiguousarray(W[:, kt * BK:(kt + 1) * BK]), swizzle)
        for wave in range(8):
            wm, wn = wave >> 2, wave & 3
            wrow0, wcol0 = wm * 128, wn * 64
            for ph in range(4):
                qm = (ph >> 1) * QM
                qn = (ph & 1) * QN
                for m in range(QM):
                    for n in range(QN):
                        fr = wrow0 + (qm + m) * 16
                        fc = wcol0 + (qn + n) * 16
                        acc = D[fr:fr + 16, fc:fc + 16]
                        for ks in range(2):
                            # exact kernel addressing:
                            a_frags = [
                                lds_frag(
                                    a_lds,
                                    fr + (lane & 15),
                                    ks * 32 + ((lane >> 4) << 3),
                                    swizzle,
                                )
                                for lane in range(64)
                            ]
                            b_frags = [
                                lds_frag(
                                    b_lds,
                                    fc + (lane & 15),
                                    ks * 32 + ((lane >> 4) << 3),
                                    swizzle,
                                )
                                for lane in range(64)
                            ]
                            acc = mfma_16x16x32(a_frags, b_frags, acc)
                        D[fr:fr + 16, fc:fc + 16] = acc
    np.testing.assert_allclose(D, ref, rtol=1e-10, atol=1e-10)


def test_swz1_lane_hoist_equivalence():
    """gemm8.hip swz1_lane_low: for SW=1 the XOR key reduces to
    (lane&7)<<4 (fragment rows differ by 16-multiples), so
    swz(row*128+col*2) == wave_base + lane_low(ks) + quadrant*2048.
    Exhaustive over every lane/wave/quadrant/ks the kernel uses."""
    def lane_low(lane, ks):
        low = ks * 64 + ((lane >> 4) << 4) + (lane & 15) * 128
        return low ^ ((lane & 7) << 4)

    for lane in range(64):
        rc = lane & 15
        col = lambda ks: ks * 32 + ((lane >> 4) << 3)
        for wm in range(2):             # A side: wrow0 = wm*128
            for qmm in range(8):
                for ks in range(2):
                    row = wm * 128 + qmm * 16 + rc
                    assert swz(row * 128 + col(ks) * 2, 1) == (
                        wm * 16384 + lane_low(lane, ks) + qmm * 2048)
        for wn in range(4):             # B side: wcol0 = wn*64
            for qnn in range(4):
                for ks in range(2):
                    row = wn * 64 + qnn * 16 + rc
                    assert swz(row * 128 + col(ks) * 2, 1) == (
                        wn * 8192 + lane_low(lane, ks) + qnn * 2048)


def test_swz1_stage_hoist_equivalence():
    """gemm8.hip stage_half/stage_unit SW=1 source hoist: chunk bases
    are 8-row multiples, so swz(lin + lane*16) decomposes into
    lin + ((lane*16) ^ (((lane>>3)&7)<<4)) for every chunk base."""
    for lane in range(64):
        lo = (lane * 16) ^ (((lane >> 3) & 7) << 4)
        for row0 in range(0, 256, 8):
            lin = row0 * 128
            assert swz(lin + lane * 16, 1) == lin + lo
