import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch as t
from machin_amd.ops import _machin_hip as ext

kimg = ext.tr16_diag(0).cpu().to(t.int32)
nimg = ext.tr16_diag(1).cpu().to(t.int32)
bad = 0
for i in range(8192):
    nc, rem = divmod(i, 1024)
    kq, rem2 = divmod(rem, 16)
    krel, nrel = divmod(rem2, 4)
    ek, en = kq * 4 + krel, nc * 4 + nrel
    if kimg[i] != ek or nimg[i] != en:
        if bad < 10:
            print(f"image[{i}] = (k={int(kimg[i])}, n={int(nimg[i])}) expect (k={ek}, n={en})")
        bad += 1
print(f"staging image: {bad} wrong of 8192")

lo = ext.tr16_diag(2).cpu().to(t.int32).view(2, 256, 8)
hi = ext.tr16_diag(3).cpu().to(t.int32).view(2, 256, 8)
E = lo + 256 * hi
print("read element map (LDS element index actually read):")
for tid in (0, 1, 2, 3, 4, 5, 16, 17, 20, 64, 128, 192):
    print(f"  pr0 tid{tid:3d} (lane {tid % 64:2d} wave {tid // 64}): {E[0, tid].tolist()}")
print(f"  pr1 tid  0: {E[1, 0].tolist()}")
