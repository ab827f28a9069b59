import torch, time, sys
sys.path.insert(0, "/root/repo")
def t(f, n=5):
    for _ in range(2): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1000

from blades_amd.models import resnet18
C, B = 100, 32
m = resnet18(norm="batch-local").cuda()
x = torch.randn(C*B, 3, 32, 32, device='cuda')
y = torch.randint(0, 10, (C*B,), device='cuda')
def f_std():
    loss = torch.nn.functional.cross_entropy(m(x), y)
    m.zero_grad(set_to_none=True)
    loss.backward()
print("std resnet18 fwd+bwd b3200:", round(t(f_std),2), "ms", flush=True)

# per-client wrw as strided-batched GEMM: dW_c = dy_c @ unfold(x_c)^T
for ci, hw in [(64,32),(128,16),(256,8),(512,4)]:
    xx = torch.randn(C*B, ci, hw, hw, device='cuda')
    dy = torch.randn(C, ci, B*hw*hw, device='cuda')
    def f_wrw():
        u = torch.nn.functional.unfold(xx, 3, padding=1)      # [C*B, ci*9, hw*hw]
        u = u.view(C, B, ci*9, hw*hw).permute(0, 2, 1, 3).reshape(C, ci*9, B*hw*hw)
        dW = torch.bmm(dy, u.transpose(1, 2))                  # [C, ci, ci*9]
        return dW
    print(f"wrw_bmm_{ci}x{hw}:", round(t(f_wrw),2), "ms", flush=True)

# fwd as bmm conv (option C full): unfold + bmm with per-client weights
ci, hw = 64, 32
xx = torch.randn(C*B, ci, hw, hw, device='cuda')
W = torch.randn(C, ci, ci*9, device='cuda')
def f_fwdbmm():
    u = torch.nn.functional.unfold(xx, 3, padding=1)
    u = u.view(C, B, ci*9, hw*hw).permute(0, 2, 1, 3).reshape(C, ci*9, B*hw*hw)
    return torch.bmm(W, u)
print("fwd_bmm_64x32:", round(t(f_fwdbmm),2), "ms", flush=True)
