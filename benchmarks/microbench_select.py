import torch, time, sys
sys.path.insert(0, "/root/repo")
from blades_amd import _hip_ops as ext

def t(f, n=20, w=3):
    for _ in range(w): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1000

for K, b, d in [(100, 20, 11173962), (100, 49, 11173962), (1000, 499, 11173962),
                (1000, 20, 11173962), (10000, 20, 1400000),
                (1000, 300, 1400000), (10000, 4999, 1400000)]:
    # padded row stride, as the runtime allocates (float4 kernel paths)
    d_pad = (d + 3) // 4 * 4
    buf = torch.randn(K, d_pad, device='cuda')
    U = buf[:, :d]
    r1 = t(lambda: ext.trimmed_mean(U, b), n=10)
    r2 = t(lambda: ext.trimmed_mean_radix(U, b), n=10)
    r3 = None
    if K <= 1280 and b > 0:
        r3 = t(lambda: ext.trimmed_mean_radix_lds(U, b), n=10)
    gb = K*d*4/1e9
    lds = f", radix-lds {r3:.2f} ms" if r3 is not None else ""
    print(f"K={K} b={b} d={d}: auto {r1:.2f} ms, radix {r2:.2f} ms{lds}  (slab {gb:.1f} GB)", flush=True)
    ok = torch.allclose(ext.trimmed_mean(U, b), ext.trimmed_mean_radix(U, b), atol=1e-5)
    print("   agree:", ok, flush=True)
    del U, buf
