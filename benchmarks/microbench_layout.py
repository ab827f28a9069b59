import torch, time, sys, os
sys.path.insert(0, "/root/repo")
def t(f, n=5, w=8):
    for _ in range(w): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1000

from blades_amd.models import resnet18
torch.backends.cudnn.benchmark = True
C, B = 100, 32
x = torch.randn(C*B, 3, 32, 32, device='cuda')
y = torch.randint(0, 10, (C*B,), device='cuda')

m = resnet18(norm="batch-local").cuda()
def f_std():
    loss = torch.nn.functional.cross_entropy(m(x), y)
    m.zero_grad(set_to_none=True)
    loss.backward()
print("NCHW benchmark-mode:", round(t(f_std),2), "ms", flush=True)

m2 = resnet18(norm="batch-local").cuda().to(memory_format=torch.channels_last)
xcl = x.to(memory_format=torch.channels_last)
def f_cl():
    loss = torch.nn.functional.cross_entropy(m2(xcl), y)
    m2.zero_grad(set_to_none=True)
    loss.backward()
print("NHWC benchmark-mode:", round(t(f_cl),2), "ms", flush=True)
