import time, torch, sys
sys.path.insert(0, "/root/repo")
from fl4health_amd.ops.batchnorm import CdnaBatchNorm2d

def bench_bn(n, c, h, w, dtype):
    x = torch.randn(n, c, h, w, device="cuda").to(dtype).contiguous(memory_format=torch.channels_last)
    ref = torch.nn.BatchNorm2d(c).cuda().train()
    ours = CdnaBatchNorm2d(c).cuda().train()
    g = torch.randn(n, c, h, w, device="cuda").to(dtype).contiguous(memory_format=torch.channels_last)
    def run(mod, xx):
        xx = xx.clone().requires_grad_(True)
        for _ in range(3):
            y = mod(xx); y.backward(g)
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(20):
            y = mod(xx); y.backward(g)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / 20 * 1e3
    t_ref = run(ref, x.float() if dtype==torch.bfloat16 else x)
    t_ours = run(ours, x)
    print(f"BN {n}x{c}x{h}x{w} {dtype}: torch {t_ref:.3f} ms  ours {t_ours:.3f} ms")

for shape in [(128,64,32,32),(128,128,16,16),(128,256,8,8),(128,512,4,4)]:
    bench_bn(*shape, torch.bfloat16)

# graphed step with custom BN: does capture work and how fast is replay?
from fl4health_amd.models.resnet import ResNet18
from fl4health_amd.ops.batchnorm import convert_batchnorm_to_cdna
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.optimizers import FlatProxSGD
torch.backends.cudnn.benchmark = True
for use_cdna in (False, True):
    m = ResNet18().cuda().to(memory_format=torch.channels_last)
    if use_cdna: convert_batchnorm_to_cdna(m)
    view = FlatParameterView(m, bind=True)
    opt = FlatProxSGD(view, lr=0.05, momentum=0.9)
    xb = torch.randn(128,3,32,32,device="cuda").contiguous(memory_format=torch.channels_last)
    yb = torch.randint(0,10,(128,),device="cuda")
    def step():
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(m(xb), yb)
        loss.backward(); opt.step(); return loss
    s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3): step()
    torch.cuda.current_stream().wait_stream(s)
    gph = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(gph):
            step()
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(50): gph.replay()
        torch.cuda.synchronize()
        print(f"cdna_bn={use_cdna}: graphed replay {(time.perf_counter()-t0)/50*1e3:.3f} ms/step")
    except Exception as e:
        print(f"cdna_bn={use_cdna}: CAPTURE FAILED: {e}")
