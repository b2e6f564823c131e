import time, sys, torch
sys.path.insert(0, "/root/repo")
from fl4health_amd.ops import functional as F

def t(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

for R, C in [(131072, 64), (32768, 128), (8192, 256), (2048, 512)]:
    x = torch.randn(R, C, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn_like(x)
    gamma = torch.ones(C, device="cuda"); beta = torch.zeros(C, device="cuda")
    rm = torch.zeros(C, device="cuda"); rv = torch.ones(C, device="cuda")
    ms_f = t(lambda: F._C.bn_fwd_train(x, gamma, beta, rm, rv, 0.1, 1e-5, False, None))
    y, mean, invstd = F._C.bn_fwd_train(x, gamma, beta, rm, rv, 0.1, 1e-5, False, None)
    ms_b = t(lambda: F._C.bn_bwd(x, dy, mean, invstd, gamma))
    bytes_f = R * C * 2 * 3  # read x twice + write y
    bytes_b = R * C * 2 * 5
    print(f"[{R}x{C}] fwd {ms_f*1e3:7.1f} us ({bytes_f/ms_f/1e6:6.1f} GB/s)   bwd {ms_b*1e3:7.1f} us ({bytes_b/ms_b/1e6:6.1f} GB/s)")
    # torch reference
    xf = x
    ms_ref = t(lambda: torch.nn.functional.batch_norm(
        xf.view(R, C, 1, 1), rm, rv, gamma, beta, True, 0.1, 1e-5))
    print(f"         torch batch_norm fwd {ms_ref*1e3:7.1f} us")
