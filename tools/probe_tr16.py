import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch as t
from machin_amd.ops import _machin_hip as ext

for mode in range(5):
    out = ext.tr16_probe(mode).cpu().to(t.int32)
    print(f"mode {mode}:")
    for lane in range(0, 64, 1 if mode in (0, 1) else 4):
        if lane < 20 or lane in (31, 32, 33, 47, 48, 63):
            print(f"  lane {lane:2d}: {out[lane].tolist()}")
