import sys, torch
sys.path.insert(0, "/root/repo")
from blades_amd import _hip_ops as ext
K, d = 100, 11173964
U = torch.randn(K, d, device='cuda')
for _ in range(5):
    ext.trimmed_mean(U, 20)
    ext.gram(U)
    ext.col_mean(U)
torch.cuda.synchronize()
print("done")
