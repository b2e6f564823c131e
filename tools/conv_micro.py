"""Direct 3x3 MFMA conv prototype vs MIOpen, on the ResNet-18 CIFAR shapes.

Run on the GPU box: PYTHONPATH=. python tools/conv_micro.py
Correctness vs torch/MIOpen (bf16 tolerances), then timed fwd at batch 128.
"""
import time

import torch

from fl4health_amd import _C


def pack_weight(w: torch.Tensor) -> torch.Tensor:
    """torch conv weight [K, C, 3, 3] -> [9, C, K] taps-major."""
    return w.permute(2, 3, 1, 0).reshape(9, w.shape[1], w.shape[0]).contiguous()


def run_shape(n, h, w, c, k, iters=100):
    x_nchw = torch.randn(n, c, h, w, device="cuda", dtype=torch.bfloat16)
    weight = torch.randn(k, c, 3, 3, device="cuda", dtype=torch.bfloat16) * 0.05
    ref = torch.nn.functional.conv2d(
        x_nchw.contiguous(memory_format=torch.channels_last), weight, padding=1
    )
    x_nhwc = x_nchw.permute(0, 2, 3, 1).contiguous()
    wp = pack_weight(weight)
    out = _C.conv3x3_fwd(x_nhwc, wp, None)  # [N, H, W, K]
    out_nchw = out.permute(0, 3, 1, 2)
    diff = (out_nchw.float() - ref.float()).abs()
    rel = diff.max() / ref.float().abs().max().clamp(min=1e-6)
    ok = rel < 2e-2
    torch.cuda.synchronize()

    def bench(fn):
        for _ in range(10):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e3

    x_cl = x_nchw.contiguous(memory_format=torch.channels_last)
    t_miopen = bench(lambda: torch.nn.functional.conv2d(x_cl, weight, padding=1))
    t_ours = bench(lambda: _C.conv3x3_fwd(x_nhwc, wp, None))
    flops = 2.0 * n * h * w * k * c * 9
    print(
        f"N{n} {h}x{w} C{c}->K{k}: {'OK ' if ok else 'FAIL'} maxrel={float(rel):.4f} | "
        f"MIOpen {t_miopen:.3f} ms ({flops / t_miopen / 1e9:.0f} TF) | "
        f"direct {t_ours:.3f} ms ({flops / t_ours / 1e9:.0f} TF) | ratio {t_miopen / t_ours:.2f}x"
    )
    return ok


def main():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    allok = True
    for shape in [
        (128, 32, 32, 64, 64),    # ResNet-18 CIFAR layer1
        (128, 16, 16, 128, 128),  # layer2
        (128, 8, 8, 256, 256),    # layer3
        (128, 4, 4, 512, 512),    # layer4
    ]:
        allok &= run_shape(*shape)
    print("ALL OK" if allok else "FAILURES PRESENT")


if __name__ == "__main__":
    main()
