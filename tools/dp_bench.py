"""Instance-level DP-SGD overhead on MI355X: per-sample clipping + Gaussian
noise (own GradSample engine + per_sample_sqnorm/clip_rowsum/gaussian_noise
HIP kernels) vs plain SGD on the same model/batch.

Run on the GPU box: PYTHONPATH=. python tools/dp_bench.py
"""
import time

import torch

from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.privacy.dp_sgd import DpSgdEngine
from fl4health_amd.privacy.grad_sample import GradSampleModule, convert_batchnorm_modules
from fl4health_amd.utils.random import set_all_random_seeds


def bench_step(step, iters=50, warmup=10):
    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        step()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    assert torch.cuda.is_available()
    set_all_random_seeds(0)
    batch = 128
    x = torch.randn(batch, 3, 32, 32, device="cuda")
    y = torch.randint(0, 10, (batch,), device="cuda")
    criterion = torch.nn.CrossEntropyLoss()

    # plain SGD
    model = convert_batchnorm_modules(SmallCnn()).cuda()
    opt = torch.optim.SGD(model.parameters(), lr=0.05)

    def plain():
        opt.zero_grad(set_to_none=True)
        criterion(model(x), y).backward()
        opt.step()

    t_plain = bench_step(plain)

    # DP-SGD: per-sample grads + clip + noise; ghost clipping avoids ever
    # materializing the [B, out, in] per-sample grads of the Linear layers
    results = {}
    for ghost in (False, True):
        print(f"smallcnn ghost={ghost}...", flush=True)
        model2 = convert_batchnorm_modules(SmallCnn()).cuda()
        gsm = GradSampleModule(model2, ghost_clipping=ghost)
        opt2 = torch.optim.SGD(model2.parameters(), lr=0.05)
        engine = DpSgdEngine(gsm, opt2, noise_multiplier=1.0, clipping_bound=1.0, seed=0)

        def dp():
            engine.zero_grad()
            criterion(gsm(x), y).backward()
            engine.step()

        results[ghost] = bench_step(dp)

    n_params = sum(p.numel() for p in model.parameters())
    print(
        f"SmallCnn ({n_params/1e6:.2f}M params) batch {batch}: plain SGD {t_plain:.3f} ms/step | "
        f"DP-SGD materialized {results[False]:.3f} ms ({results[False] / t_plain:.2f}x) | "
        f"DP-SGD ghost {results[True]:.3f} ms ({results[True] / t_plain:.2f}x, "
        f"{batch * 1000 / results[True]:.0f} private samples/s)"
    )

    # flagship-model DP: ResNet-18 (GN), where deep blocks hit the conv
    # ghost-norm path (L^2 << |W|)
    from fl4health_amd.models.resnet import ResNet18

    res = {}
    for ghost in (False, True):
        print(f"resnet ghost={ghost}...", flush=True)
        model3 = convert_batchnorm_modules(ResNet18(num_classes=10)).cuda()
        gsm = GradSampleModule(model3, ghost_clipping=ghost)
        opt3 = torch.optim.SGD(model3.parameters(), lr=0.05)
        engine = DpSgdEngine(gsm, opt3, noise_multiplier=1.0, clipping_bound=1.0, seed=0)

        def dp3():
            engine.zero_grad()
            criterion(gsm(x), y).backward()
            engine.step()

        res[ghost] = bench_step(dp3, iters=30, warmup=5)
    n3 = sum(p.numel() for p in model3.parameters())
    print(
        f"ResNet-18 GN ({n3/1e6:.2f}M params) batch {batch}: "
        f"DP-SGD materialized {res[False]:.3f} ms | ghost {res[True]:.3f} ms "
        f"({res[False] / res[True]:.2f}x faster, {batch * 1000 / res[True]:.0f} private samples/s)"
    )


if __name__ == "__main__":
    main()
