"""Dump v2 conv error structure vs fp32 reference (debug aid)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch as t
import torch.nn.functional as F
from machin_amd.ops import _require_ext

ext = _require_ext()
t.manual_seed(0)
B = 1
frames = t.randint(0, 256, (B, 84, 84, 4), dtype=t.uint8, device="cuda")
w = (t.randn(32, 4, 8, 8, device="cuda") * 0.1)
# repack to [256, 32]: patch-major (r, c, ci) rows
wp = w.permute(2, 3, 1, 0).reshape(256, 32).contiguous().to(t.bfloat16)
bias = t.zeros(32, device="cuda")
out = ext.conv1_fwd(frames.contiguous(), wp, bias, 1.0 / 255.0)  # [K,32]
x = frames.permute(0, 3, 1, 2).float() / 255.0
ref = F.conv2d(x, w, stride=4).permute(0, 2, 3, 1).reshape(-1, 32)
err = (out.float() - ref).abs()
print("fwd: max err", err.max().item(), "mean", err.mean().item())
# error by output column (channel) and by row mod patterns
print("err by col:", err.mean(0).round(decimals=3).tolist())
rows_bad = (err.max(1).values > 0.1).nonzero().flatten()
print("bad rows:", rows_bad[:20].tolist(), "count", len(rows_bad))
# per row group (k%4, which lane cluster row)
for m in range(4):
    print(f"err rows k%4=={m}:", err[m::4].mean().item())
em = err.mean(0)
for c in range(4):
    print(f"err cols n%4=={c}:", em[c::4].mean().item())
