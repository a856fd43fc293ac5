import sys, os, torch
sys.path.insert(0, "/root/repo")
from mpi4dl_amd.ops import backend
ge = backend.ext()
torch.manual_seed(0)
# C1664 -> K1664 @ 64^2 b2 (a losing fat shape, L3-resident)
x = torch.randn(2, 1664, 64, 64, device="cuda", dtype=torch.bfloat16)
w = torch.randn(1664, 1664, 1, 1, device="cuda", dtype=torch.bfloat16) * 0.05
for _ in range(3):
    y = ge.pw_fwd(x, w, None, 1, 1)
torch.cuda.synchronize()
for _ in range(10):
    y = ge.pw_fwd(x, w, None, 1, 1)
torch.cuda.synchronize()
print("done", y.shape)
