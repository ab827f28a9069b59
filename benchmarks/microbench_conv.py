import torch, time, sys
def t(f, n=5):
    for _ in range(2): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1000

C, B = 100, 32
for ch, hw in [(64,32),(128,16),(256,8),(512,4)]:
    x = torch.randn(C*B, ch, hw, hw, device='cuda', requires_grad=True)
    w = torch.randn(ch, ch, 3, 3, device='cuda', requires_grad=True)
    def fa():
        y = torch.nn.functional.conv2d(x, w, padding=1)
        y.backward(torch.ones_like(y))
    print(f'std_{ch}x{hw}', round(t(fa),2), flush=True)
    del x, w
    xg = torch.randn(B, C*ch, hw, hw, device='cuda', requires_grad=True)
    wg = torch.randn(C*ch, ch, 3, 3, device='cuda', requires_grad=True)
    def fb():
        y = torch.nn.functional.conv2d(xg, wg, padding=1, groups=C)
        y.backward(torch.ones_like(y))
    print(f'grp_{ch}x{hw}', round(t(fb),2), flush=True)
    del xg, wg
