import torch, time, sys
sys.path.insert(0, "/root/repo")
from blades_amd import _hip_popconv as ext

def t(f, n=10, w=3):
    for _ in range(w): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1000

C, B = 100, 32
for ci, co, hw in [(64,64,32),(128,128,16),(256,256,8),(512,512,4),(3,64,32)]:
    Hp = Wp = hw + 2
    Np = B*Hp*Wp
    X = torch.randn(C, ci, Np, device='cuda')
    W = torch.randn(C, co, ci, 3, 3, device='cuda')
    def f(): return ext.popconv_fwd(X, W, B, Hp, Wp)
    ms = t(f)
    tf = 2*co*ci*9*Np*C/(ms/1e3)/1e12
    print(f"fwd ci={ci} co={co} hw={hw}: {ms:.2f} ms  {tf:.1f} TF", flush=True)
    # shared-weight broadcast
    Wsh = W[0:1].expand(C, co, ci, 3, 3)
    def g(): return ext.popconv_fwd(X, Wsh, B, Hp, Wp)
    print(f"   shared: {t(g):.2f} ms", flush=True)
