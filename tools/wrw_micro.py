"""Custom wrw (dW) kernel vs tuned MIOpen convolution_backward, layer-1 shape."""
import time

import torch

torch.backends.cudnn.benchmark = True
from fl4health_amd import _C


def t(fn, iters=100):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def run(n, h, w, c, k):
    torch.manual_seed(0)
    x = (torch.randn(n, h, w, c, device="cuda") * 0.5).to(torch.bfloat16)
    dy = (torch.randn(n, h, w, k, device="cuda") * 0.5).to(torch.bfloat16)
    dw = _C.conv3x3_wrw(x, dy)
    # reference: fp32 conv backward via autograd
    xf = x.permute(0, 3, 1, 2).float().requires_grad_(True)
    wf = torch.zeros(k, c, 3, 3, device="cuda", requires_grad=True)
    y = torch.nn.functional.conv2d(xf, wf, None, 1, 1)
    y.backward(dy.permute(0, 3, 1, 2).float())
    ref = wf.grad
    rel = (dw.float() - ref).norm() / ref.norm()
    print(f"{h}x{w} C{c}->K{k} N{n}: rel={rel:.2e}")
    assert rel < 2e-2, f"numerics FAIL {rel}"
    # perf: ours vs aten (tuned MIOpen wrw; first call runs find)
    x_ncl = x.permute(0, 3, 1, 2)
    dy_ncl = dy.permute(0, 3, 1, 2)
    wcl = torch.zeros(k, c, 3, 3, device="cuda", dtype=torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    def aten():
        return torch.ops.aten.convolution_backward(
            dy_ncl, x_ncl, wcl, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [False, True, False])[1]
    ms_a = t(aten)
    ms_o = t(lambda: _C.conv3x3_wrw(x, dy))
    gf = 2 * n * h * w * c * k * 9 / 1e9
    print(f"  aten/MIOpen {ms_a:.4f} ms ({gf/ms_a:.0f} TF)  ours {ms_o:.4f} ms "
          f"({gf/ms_o:.0f} TF)  speedup {ms_a/ms_o:.2f}x")


if __name__ == "__main__":
    run(128, 32, 32, 64, 64)
    run(64, 32, 32, 64, 64)
    run(128, 16, 16, 128, 128)
    run(128, 16, 16, 64, 128)
    run(128, 16, 16, 128, 64)
