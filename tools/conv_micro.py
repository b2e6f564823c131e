"""Direct 3x3 MFMA conv variants vs MIOpen, on the ResNet-18 CIFAR shapes.

Run on the GPU box: PYTHONPATH=. python tools/conv_micro.py
Correctness vs torch/MIOpen (bf16 tolerances), then timed fwd at batch 128.

Variants:
  dispatch — launch_conv3x3_fwd (auto: glds persistent-weight kernel B for
             C <= 64 halo-fitting shapes, else the round-1 kernel A)
  kb32     — conv3x3_fwd_kb32 (KB=32, both operands glds double-buffered,
             weights prepacked to the LDS image) for C % 64 == 0, K % 32 == 0
"""
import time

import torch

from fl4health_amd import _C


def pack_weight(w: torch.Tensor) -> torch.Tensor:
    """torch conv weight [K, C, 3, 3] -> [9, C, K] taps-major."""
    return w.permute(2, 3, 1, 0).reshape(9, w.shape[1], w.shape[0]).contiguous()


def pack_weight_kb32(w: torch.Tensor) -> torch.Tensor:
    """[K, C, 3, 3] -> [K/32, C/64, 9, 32, 64] LDS-image slabs, bank-swizzled:
    rows with kk bit 2 set get channel bit 4 XORed (matches conv_swz on the
    read side — glds stages the image verbatim, so the swizzle lives here)."""
    k, c = w.shape[0], w.shape[1]
    w9 = w.permute(2, 3, 1, 0).reshape(9, c, k)  # [tap, c, k]
    img = w9.reshape(9, c // 64, 64, k // 32, 32).permute(3, 1, 0, 4, 2).contiguous()
    kk_mask = (torch.arange(32, device=w.device) >> 2) & 1 == 1
    cc_swz = torch.arange(64, device=w.device) ^ 16
    img[:, :, :, kk_mask, :] = img[:, :, :, kk_mask, :][..., cc_swz]
    return img


def bench(fn, iters):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def run_shape(n, h, w, c, k, iters=100):
    x_nchw = torch.randn(n, c, h, w, device="cuda", dtype=torch.bfloat16)
    weight = torch.randn(k, c, 3, 3, device="cuda", dtype=torch.bfloat16) * 0.05
    ref = torch.nn.functional.conv2d(
        x_nchw.contiguous(memory_format=torch.channels_last), weight, padding=1
    )
    refmax = ref.float().abs().max().clamp(min=1e-6)
    x_nhwc = x_nchw.permute(0, 2, 3, 1).contiguous()
    flops = 2.0 * n * h * w * k * c * 9
    x_cl = x_nchw.contiguous(memory_format=torch.channels_last)
    t_miopen = bench(lambda: torch.nn.functional.conv2d(x_cl, weight, padding=1), iters)
    line = f"N{n} {h}x{w} C{c}->K{k}: MIOpen {t_miopen:.3f} ms ({flops / t_miopen / 1e9:.0f} TF)"
    ok = True

    wp = pack_weight(weight)
    out = _C.conv3x3_fwd(x_nhwc, wp, None).permute(0, 3, 1, 2)
    rel = float(((out.float() - ref.float()).abs().max() / refmax))
    okd = rel < 2e-2
    ok &= okd
    t = bench(lambda: _C.conv3x3_fwd(x_nhwc, wp, None), iters)
    line += f" | dispatch {t:.3f} ms ({flops / t / 1e9:.0f} TF) {t_miopen / t:.2f}x{'' if okd else ' FAIL rel=%.4f' % rel}"

    if c % 64 == 0 and k % 32 == 0:
        wimg = pack_weight_kb32(weight)
        out2 = _C.conv3x3_fwd_kb32(x_nhwc, wimg, None).permute(0, 3, 1, 2)
        rel2 = float(((out2.float() - ref.float()).abs().max() / refmax))
        ok2 = rel2 < 2e-2
        ok &= ok2
        t2 = bench(lambda: _C.conv3x3_fwd_kb32(x_nhwc, wimg, None), iters)
        line += f" | kb32 {t2:.3f} ms ({flops / t2 / 1e9:.0f} TF) {t_miopen / t2:.2f}x{'' if ok2 else ' FAIL rel=%.4f' % rel2}"
        # variant D gate: all input chunks + 2 weight slabs resident
        bh = min(max(128 // w, 1), h)
        sb = max(128 // (h * w), 1)
        in_chunks = sb * (bh + 2) * (w + 2) * 8
        if c // 64 <= 2 and in_chunks + 63 <= 1664 and k >= 64:
            out3 = _C.conv3x3_fwd_kzloop(x_nhwc, wimg, None).permute(0, 3, 1, 2)
            rel3 = float(((out3.float() - ref.float()).abs().max() / refmax))
            ok3 = rel3 < 2e-2
            ok &= ok3
            t3 = bench(lambda: _C.conv3x3_fwd_kzloop(x_nhwc, wimg, None), iters)
            line += f" | kzloop {t3:.3f} ms ({flops / t3 / 1e9:.0f} TF) {t_miopen / t3:.2f}x{'' if ok3 else ' FAIL rel=%.4f' % rel3}"

    print(line)
    return ok


def main():
    assert torch.cuda.is_available()
    torch.backends.cudnn.benchmark = True  # tuned MIOpen find (what bench.py runs)
    torch.manual_seed(0)
    allok = True
    for shape in [
        (128, 32, 32, 64, 64),    # ResNet-18 CIFAR layer1
        (128, 16, 16, 128, 128),  # layer2
        (128, 8, 8, 256, 256),    # layer3
        (128, 4, 4, 512, 512),    # layer4
        (128, 16, 16, 128, 64),   # bwd_data layer2 shape (K=C swap, downsample)
        (128, 8, 8, 128, 128),    # layer3 downsample input
    ]:
        allok &= run_shape(*shape)
    print("ALL OK" if allok else "FAILURES PRESENT")


if __name__ == "__main__":
    main()
