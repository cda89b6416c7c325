import torch as t
from machin_amd.ops import _machin_hip as ext
for mode in range(5):
    out = ext.tr16_probe(mode).cpu().to(t.int32)
    print(f"mode {mode}:")
    for lane in (0, 1, 2, 15, 16, 17, 31, 32, 48, 63):
        print(f"  lane {lane:2d}: {out[lane].tolist()}")
