import sys, os, struct
sys.path.insert(0, "/root/repo")
import torch
from min_tfs_client_amd import ops
found = 0
for it in range(300):
    x = torch.randn(1 << 18, device="cuda:0") * 10
    q = ops.quantize_q8(x, 0.1, 3.0)
    inv = float(1.0 / 0.1)
    ref = torch.clamp(torch.round(x * inv + 3.0), -128, 127).to(torch.int8)
    bad = (q != ref).nonzero().flatten()
    for i in bad[:10].tolist():
        xv = x[i].item()
        mul = (x[i] * inv).item()
        add = (x[i] * inv + 3.0).item()
        rnd = torch.round(x[i] * inv + 3.0).item()
        print("x bits=%s mul=%r add=%r round=%r kernel=%d torch=%d" %
              (struct.pack("<f", xv).hex(), mul, add, rnd,
               q[i].item(), ref[i].item()))
        found += 1
    if found >= 10:
        break
print("done, found", found)
