"""Minimal conv kernel exerciser for rocprofv3 --pmc runs."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch as t
from machin_amd.ops import _machin_hip as ext

B = 8192
frames = t.randint(0, 256, (B, 84, 84, 4), dtype=t.uint8, device="cuda")
gy = (t.randn(B * 400, 32, device="cuda") * 0.1).to(t.bfloat16).contiguous()
w = (t.randn(256, 32, device="cuda") * 0.1).to(t.bfloat16).contiguous()
bias = t.zeros(32, device="cuda")
for v in ("1", "3"):
    os.environ["MACHIN_CONV1_V"] = v
    for _ in range(5):
        ext.conv1_fwd(frames, w, bias, 1.0 / 255.0)
        ext.conv1_wrw(gy, frames, 1.0 / 255.0)
t.cuda.synchronize()
print("done")
