import os, sys
import torch, torch.nn as nn
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from fl4health_amd.ops.batchnorm import convert_batchnorm_to_cdna
from fl4health_amd.utils.random import set_all_random_seeds

def make(seed):
    set_all_random_seeds(seed)
    m = nn.Sequential(
        nn.Conv2d(3, 16, 3, padding=1, bias=False), nn.BatchNorm2d(16), nn.ReLU(),
        nn.Conv2d(16, 16, 3, padding=1, bias=False), nn.BatchNorm2d(16), nn.ReLU(),
        nn.Flatten(), nn.Linear(16 * 16 * 16, 10),
    ).cuda().to(memory_format=torch.channels_last)
    return m

def train(m, steps=50):
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
    g = torch.Generator().manual_seed(7)
    losses = []
    for i in range(steps):
        x = torch.randn(64, 3, 16, 16, generator=g).cuda().contiguous(memory_format=torch.channels_last)
        y = torch.randint(0, 10, (64,), generator=g).cuda()
        opt.zero_grad()
        loss = nn.functional.cross_entropy(m(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    return losses

ref = make(0)
ours = make(0)
convert_batchnorm_to_cdna(ours)
lr_ref = train(ref)
lr_ours = train(ours)
print("ref  losses:", [round(v,3) for v in lr_ref[::10]])
print("ours losses:", [round(v,3) for v in lr_ours[::10]])
for (n1,p1),(n2,p2) in zip(ref.state_dict().items(), ours.state_dict().items()):
    d = (p1.float()-p2.float()).abs().max()
    if d > 1e-2:
        print(f"DIVERGED {n1}: max diff {d:.4f}")
print("running_mean diff:", (ref[1].running_mean - ours[1].running_mean).abs().max().item())
print("running_var  diff:", (ref[1].running_var - ours[1].running_var).abs().max().item())
