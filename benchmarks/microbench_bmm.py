import torch, time
def t(f, n=10, w=3):
    for _ in range(w): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1000

C, B = 100, 32
# fwd shift-GEMM per offset: W [C,co,ci] @ X [C,ci,L]
for ci, hw in [(64,34),(128,18),(256,10),(512,6)]:
    L = B*hw*hw
    W = torch.randn(C, ci, ci, device='cuda')
    X = torch.randn(C, ci, L, device='cuda')
    Y = torch.zeros(C, ci, L, device='cuda')
    def f(): Y.baddbmm_(W, X)
    ms = t(f)
    tf = 2*ci*ci*L*C/ (ms/1e3) / 1e12
    print(f"fwd bmm ci={ci} L={L}: {ms:.2f} ms  {tf:.1f} TF", flush=True)
    # wrw: dW = dY @ X^T : [C,ci,L]@[C,L,ci]
    def g(): torch.baddbmm(W, Y, X.transpose(1,2))
    ms = t(g)
    tf = 2*ci*ci*L*C/(ms/1e3)/1e12
    print(f"wrw bmm ci={ci}: {ms:.2f} ms  {tf:.1f} TF", flush=True)
