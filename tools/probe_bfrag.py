import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch as t
from machin_amd.ops import _machin_hip as ext

for which, name in ((0, "k"), (1, "n")):
    out = ext.fwd_bfrag_probe(which).cpu().to(t.int32)
    bad = 0
    for pr in (0, 1):
        for tid in range(256):
            wave, lane = tid // 64, tid % 64
            wn = wave >> 1
            expect = [
                (pr * 32 + (lane >> 4) * 8 + j) if which == 0
                else (wn * 16 + (lane & 15))
                for j in range(8)
            ]
            got = out[pr, tid].tolist()
            if got != expect:
                if bad < 8:
                    print(f"{name}: pr{pr} tid{tid} (wave{wave} lane{lane}) got {got} expect {expect}")
                bad += 1
    print(f"{name}-map: {bad} mismatching lanes of 512")
