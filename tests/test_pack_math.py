"""CPU (numpy) model of the bit-packed XNOR convolution math — validates
the kernel's algebra (inverted weight bits, garbage-bit constant G, the
per-(k,tap) pad-correction table S) independently of any HIP code.

The GPU kernel (csrc/xnor_conv.hip) computes
    dot(sp,k) = 2*POP + BASE - sum_{t invalid(sp)} S[k][t]
with POP = popcount over ALL packed words of (a XOR b_inv),
BASE = -2*G - C*T, G = (32*CW - C)*T,
S[k][t] = 2*popc(inverted real bits of tap t) - C.
This test re-implements that bit-for-bit in numpy and checks it against
a direct +-1 convolution, including tail channel words and borders.
"""

import numpy as np
import pytest


def pack_activations(x):  # x: (N,H,W,C) float; bit=1 iff x>=0; tail bits 0
    N, H, W, C = x.shape
    CW = (C + 31) // 32
    out = np.zeros((N, H, W, CW), dtype=np.uint64)
    for c in range(C):
        bit = (x[..., c] >= 0).astype(np.uint64)
        out[..., c // 32] |= bit << np.uint64(c % 32)
    return out


def pack_weights(w):  # w: (K,C,KH,KW); INVERTED bits (1 iff w<0); tail bits 1
    K, C, KH, KW = w.shape
    CW = (C + 31) // 32
    wp = np.zeros((K, KH, KW, CW), dtype=np.uint64)
    for cw in range(CW):
        nbits = min(32, C - cw * 32)
        for c in range(nbits):
            bit = (w[:, cw * 32 + c, :, :] < 0).astype(np.uint64)
            wp[..., cw] |= bit << np.uint64(c)
        if nbits < 32:
            wp[..., cw] |= np.uint64(((1 << 32) - 1) ^ ((1 << nbits) - 1))
    return wp


def popc(a):
    return np.vectorize(lambda v: bin(int(v)).count("1"))(a)


@pytest.mark.parametrize("C,K,ks,stride,pad", [
    (32, 8, 3, 1, 1),
    (48, 8, 3, 1, 1),    # tail word
    (16, 16, 3, 2, 1),   # tail word + stride
    (64, 4, 1, 2, 0),    # 1x1 downsample
])
def test_packed_formula_equals_direct_pm1_conv(C, K, ks, stride, pad):
    rng = np.random.RandomState(0)
    N, H, W = 2, 7, 9
    x = rng.randn(N, H, W, C).astype(np.float32)
    w = rng.randn(K, C, ks, ks).astype(np.float32)

    xp = pack_activations(x)
    wp = pack_weights(w)
    T = ks * ks
    CW = (C + 31) // 32
    G = (32 * CW - C) * T
    BASE = -2 * G - C * T
    # S[k][t] = 2*popc(inverted real+garbage bits) - 2*garbage - C
    S = np.zeros((K, T))
    garbage = 32 * CW - C
    for k in range(K):
        for t in range(T):
            S[k, t] = 2 * (int(popc(wp[k, t // ks, t % ks]).sum()) - garbage) - C

    xb = np.where(x >= 0, 1.0, -1.0)
    wb = np.where(w >= 0, 1.0, -1.0)

    Ho = (H + 2 * pad - ks) // stride + 1
    Wo = (W + 2 * pad - ks) // stride + 1
    for n in range(N):
        for oy in range(Ho):
            for ox in range(Wo):
                for k in range(K):
                    pop = 0
                    corr = 0.0
                    for t in range(T):
                        iy = oy * stride - pad + t // ks
                        ix = ox * stride - pad + t % ks
                        if 0 <= iy < H and 0 <= ix < W:
                            a = xp[n, iy, ix]
                        else:
                            a = np.zeros(CW, dtype=np.uint64)
                            corr += S[k, t]
                        b = wp[k, t // ks, t % ks]
                        pop += int(popc(a ^ b).sum())
                    got = 2 * pop + BASE - corr
                    # direct +-1 conv with zero padding
                    want = 0.0
                    for t in range(T):
                        iy = oy * stride - pad + t // ks
                        ix = ox * stride - pad + t % ks
                        if 0 <= iy < H and 0 <= ix < W:
                            want += float(
                                (xb[n, iy, ix] * wb[k, :, t // ks, t % ks])
                                .sum())
                    assert got == pytest.approx(want), (n, oy, ox, k)
