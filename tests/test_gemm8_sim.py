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


def swz(byte, on):
    # T2 st_16x32: XOR three row bits into the 16-B chunk index
    return byte ^ (((((byte >> 7) & 7) << 4)) if on else 0)


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


@pytest.mark.parametrize("swizzle", [False, True])
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
