"""Measure the heavy workload families on one MI355X (BASELINE configs #4/#5):
- BERT-base (random init) + LoRA + MOON contrastive, AG-News-shaped synthetic
- 3D U-Net on synthetic 128^3 volumes, deep supervision
Prints per-step times + throughput; writes gpurun_out/workloads.md.
"""
import os
import sys
import time

import torch

torch.backends.cudnn.benchmark = True  # MIOpen find BEFORE first conv: the
# 3D U-Net's default solver picks hit the Im3d2Col fallback (168 ms/step);
# find mode reaches 54 ms/step. Setting this after any conv ran is a no-op
# for already-selected algos (the round-1 trap).

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

OUT = []


def log(msg):
    print(msg, flush=True)
    OUT.append(msg)


def bench_bert():
    from fl4health_amd.models.bert import BertMoonModel, synthetic_agnews_batch
    from fl4health_amd.models.lora import apply_lora
    from fl4health_amd.losses.contrastive_loss import MoonContrastiveLoss

    torch.manual_seed(0)
    model = apply_lora(BertMoonModel(num_classes=4, small=False), ("query", "value"), r=8).cuda()
    model.train()
    opt = torch.optim.AdamW([p for p in model.parameters() if p.requires_grad], lr=2e-4)
    contrastive = MoonContrastiveLoss(temperature=0.5)
    batch, seq = 32, 128
    ids, mask, y = synthetic_agnews_batch(batch, seq_len=seq, vocab=30522, seed=0)
    ids, mask, y = ids.cuda(), mask.cuda(), y.cuda()
    old_feats = torch.randn(1, batch, 768, device="cuda")
    glob_feats = torch.randn(1, batch, 768, device="cuda")

    def step():
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            preds, feats = model(ids, mask)
            loss = torch.nn.functional.cross_entropy(preds["prediction"], y)
            loss = loss + contrastive(feats["features"].float(), glob_feats, old_feats)
        loss.backward()
        opt.step()

    for _ in range(5):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    n = 20
    for _ in range(n):
        step()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / n * 1e3
    toks = batch * seq / (ms / 1e3)
    log(f"BERT-base + LoRA + MOON (batch {batch}, seq {seq}, bf16): {ms:.1f} ms/step = {toks/1e3:.1f}k tokens/s/GPU")


def bench_unet(patch=128, base=32, levels=5, batch=2, channels_last=False, fused_in=True):
    from fl4health_amd.models.unet3d import DeepSupervisionLoss, UNet3D

    torch.manual_seed(0)
    model = UNet3D(1, 3, base_channels=base, num_levels=levels, deep_supervision=True).cuda()
    if fused_in:
        from fl4health_amd.ops.instancenorm import fuse_unet3d_norm_relu

        model = fuse_unet3d_norm_relu(model)
    if channels_last:
        model = model.to(memory_format=torch.channels_last_3d)
    model.train()
    opt = torch.optim.SGD(model.parameters(), lr=1e-2, momentum=0.99, nesterov=True)
    crit = DeepSupervisionLoss(3)
    x = torch.randn(batch, 1, patch, patch, patch, device="cuda")
    if channels_last:
        x = x.contiguous(memory_format=torch.channels_last_3d)
    y = torch.randint(0, 3, (batch, patch, patch, patch), device="cuda")

    def step():
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(x)
            loss = crit(out, y)
        loss.backward()
        torch.nn.utils.clip_grad_norm_(model.parameters(), 12.0)
        opt.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    n = 10
    for _ in range(n):
        step()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / n * 1e3
    vox = batch * patch**3 / (ms / 1e3)
    log(f"3D U-Net {patch}^3 (base {base}, {levels} levels, batch {batch}, bf16, ds, cl={channels_last}, fusedIN={fused_in}): "
        f"{ms:.1f} ms/step = {vox/1e6:.1f}M voxels/s/GPU; peak mem {torch.cuda.max_memory_allocated()/2**30:.1f} GiB")


if __name__ == "__main__":
    import os
    if os.environ.get("SKIP_BERT") != "1":
        bench_bert()
    torch.cuda.reset_peak_memory_stats()
    # NCDHW is the right layout here: channels_last_3d measured 17x SLOWER
    # (MIOpen lacks direct NDHWC 3D kernels and falls back to naive conv)
    bench_unet(channels_last=False)  # find mode ON above: 54 ms/step vs 168
    with open("gpurun_out/workloads.md", "w") as f:
        f.write("# Heavy-workload single-GPU measurements (MI355X)\n\n")
        for line in OUT:
            f.write(f"- {line}\n")
