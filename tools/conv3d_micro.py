"""3D conv: MIOpen NCDHW (Im3d2Col fallback) vs the depth-decomposed 2D NHWC
path, fwd+bwd, on the U-Net shapes. Run: PYTHONPATH=. python tools/conv3d_micro.py"""
import time

import torch
import torch.nn as nn

from fl4health_amd.ops.conv3d import _conv3x3x3_by_2d


def bench(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def run_shape(n, c, k, s):
    torch.manual_seed(0)
    conv = nn.Conv3d(c, k, 3, padding=1).cuda()
    x = torch.randn(n, c, s, s, s, device="cuda", requires_grad=True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        ref = conv(x)
        out = _conv3x3x3_by_2d(x, conv.weight, conv.bias)
    rel = (out.float() - ref.float()).abs().max() / ref.float().abs().max().clamp(min=1e-6)
    g = torch.randn_like(ref)

    def base():
        x.grad = None
        conv.weight.grad = None
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y = conv(x)
        y.backward(g)

    def ours():
        x.grad = None
        conv.weight.grad = None
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y = _conv3x3x3_by_2d(x, conv.weight, conv.bias)
        y.backward(g)

    tb = bench(base)
    to = bench(ours)
    flops = 2.0 * n * s**3 * k * c * 27 * 3  # fwd + dx + dw
    print(
        f"N{n} {s}^3 C{c}->K{k}: {'OK ' if rel < 3e-2 else 'FAIL rel=%.3f' % rel}"
        f" | MIOpen3d {tb:.2f} ms ({flops/tb/1e9:.0f} TF) | 2d-decomp {to:.2f} ms"
        f" ({flops/to/1e9:.0f} TF) | {tb/to:.2f}x"
    )


def main():
    assert torch.cuda.is_available()
    torch.backends.cudnn.benchmark = True
    for shape in [
        (2, 32, 32, 64),    # U-Net level 1 (128^3 cfg downscaled batch... s=64)
        (2, 32, 64, 64),
        (2, 64, 64, 32),    # level 2
        (2, 128, 128, 16),  # level 3
        (2, 256, 256, 8),   # level 4
        (2, 1, 32, 64),     # stem-ish
    ]:
        n, c, k, s = shape
        run_shape(n, c, k, s)


if __name__ == "__main__":
    main()
