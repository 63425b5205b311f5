python - <<'PYEOF'
import torch
from min_tfs_client_amd import ops
bad_total = 0
for it in range(200):
    x = torch.randn(1 << 18, device="cuda:0") * 10
    q = ops.quantize_q8(x, 0.1, 3.0)
    inv = float(1.0 / 0.1)
    ref = torch.clamp(torch.round(x * inv + 3.0), -128, 127).to(torch.int8)
    bad_total += int((q != ref).sum())
print("quantize soak: 200 iters, mismatches =", bad_total)
assert bad_total == 0
PYEOF
python -m pytest tests -m gpu -q 2>&1 | tail -1
