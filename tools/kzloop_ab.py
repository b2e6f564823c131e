"""A/B the kzloop wave count (FL4_KZLOOP_WAVES env, read once per process)."""
import os, sys, time
import torch
from fl4health_amd import _C
from fl4health_amd.ops.conv import _pack_fwd, _image_kb32

def bench(n, h, w, c, k, iters=200):
    torch.manual_seed(0)
    x = torch.randn(n, h, w, c, device="cuda", dtype=torch.bfloat16)
    wt = (torch.randn(k, c, 3, 3, device="cuda") * 0.05).to(torch.bfloat16)
    wimg = _image_kb32(_pack_fwd(wt))
    y = _C.conv3x3_fwd_kzloop(x, wimg, None)
    # numerics vs fp32 conv
    ref = torch.nn.functional.conv2d(
        x.permute(0, 3, 1, 2).float(), wt.float(), None, 1, 1
    ).permute(0, 2, 3, 1)
    rel = (y.float() - ref).norm() / ref.norm()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        y = _C.conv3x3_fwd_kzloop(x, wimg, None)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / iters * 1e3
    tf = 2 * n * h * w * c * k * 9 / (ms * 1e-3) / 1e12
    print(f"waves={os.environ.get('FL4_KZLOOP_WAVES','8')} {h}x{w} C{c}->K{k}: "
          f"{ms:.4f} ms  {tf:.0f} TF  rel={rel:.2e}")
    assert rel < 0.02, f"numerics FAIL rel={rel}"

if __name__ == "__main__":
    for c, k in [(128, 128), (128, 64), (64, 128), (128, 256)]:
        bench(256, 16, 16, c, k)
