"""Host-side simulation of the conv3x3_wrw kernel's LDS layout
(ops/csrc/conv_ops.hip): staging (three pre-shifted x copies + transposed dy,
XOR-block swizzle) and every MFMA fragment fetch are replayed in numpy and
checked against the mathematically-required x/dy elements. This is the
simulator that de-risked the kernel before GPU time; it pins the layout
contract (pitches, swizzle, halo/shift indexing) on CPU."""
import numpy as np
import pytest

SPITCH = 264
DPITCH = 136


def swz(c, sp):
    return c * SPITCH + ((((sp >> 3) ^ ((c >> 3) & 7)) << 3) | (sp & 7))


@pytest.mark.parametrize("W_,BH_,XC,c0,KC,k0", [
    (32, 4, 64, 0, 64, 0),
    (16, 8, 64, 0, 64, 0),
    (16, 8, 128, 64, 128, 64),  # sub-slice dispatch (C/K = 128)
])
def test_wrw_lds_layout_roundtrip(W_, BH_, XC, c0, KC, k0):
    H, N = W_, 1
    LP = W_ + 8
    rng = np.random.default_rng(0)
    x = rng.standard_normal((N, H, W_, XC)).astype(np.float32)
    dy = rng.standard_normal((N, H, W_, KC)).astype(np.float32)
    h_groups = H // BH_
    for tile in range(N * h_groups):
        n0, h0 = tile // h_groups, (tile % h_groups) * BH_
        s_xt = np.zeros((3, 64 * SPITCH), dtype=np.float32)
        s_dyt = np.zeros(64 * DPITCH, dtype=np.float32)
        for idx in range((BH_ + 2) * (W_ + 2) * 8):
            c8, sp = idx & 7, idx >> 3
            line, wx = sp // (W_ + 2), sp % (W_ + 2)
            hh, ww = h0 - 1 + line, wx - 1
            v = x[n0, hh, ww, c0:c0 + 64][c8 * 8:(c8 + 1) * 8] if (0 <= hh < H and 0 <= ww < W_) else np.zeros(8)
            for d in range(3):
                col = ww + 1 - d
                if 0 <= col < LP:
                    for j in range(8):
                        s_xt[d, swz(c8 * 8 + j, line * LP + col)] = v[j]
        for idx in range(BH_ * W_ * 8):
            k8, sp = idx & 7, idx >> 3
            line, ww = sp // W_, sp % W_
            v = dy[n0, h0 + line, ww, k0:k0 + 64][k8 * 8:(k8 + 1) * 8]
            for j in range(8):
                s_dyt[(k8 * 8 + j) * DPITCH + sp] = v[j]
        for kc in range(BH_ * W_ // 32):
            for km in range(4):
                r0 = kc * 32 + km * 8
                line, col = r0 // W_, r0 % W_
                for ln in range(0, 16, 5):  # sample lanes (full sweep is slow)
                    for t_k in (ln, 48 + ln):
                        got = s_dyt[t_k * DPITCH + r0: t_k * DPITCH + r0 + 8]
                        want = np.array([dy[n0, h0 + (r0 + i) // W_, (r0 + i) % W_, k0 + t_k] for i in range(8)])
                        assert np.array_equal(got, want)
                    for tap in range(9):
                        dyy, dxx = tap // 3, tap % 3
                        for c in (ln, 48 + ln):
                            base = swz(c, (line + dyy) * LP + col)
                            got = s_xt[dxx, base:base + 8]
                            want = []
                            for i in range(8):
                                r = r0 + i
                                hh = h0 + r // W_ + dyy - 1
                                ww = r % W_ + dxx - 1
                                want.append(x[n0, hh, ww, c0 + c] if (0 <= hh < H and 0 <= ww < W_) else 0.0)
                            assert np.allclose(got, np.array(want)), (W_, kc, km, tap, c)


def test_wrw_swizzle_is_bijective_in_bounds():
    """The XOR block swizzle must keep every (c, sp) inside its channel row
    and remain a bijection (no two stores collide)."""
    seen = set()
    LP, lines = 40, 6  # W=32 geometry
    for c in range(64):
        for sp in range(lines * LP):
            a = swz(c, sp)
            assert c * SPITCH <= a < (c + 1) * SPITCH
            assert a not in seen
            seen.add(a)
