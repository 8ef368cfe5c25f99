import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import vescale_amd.ops as ops
C = ops.require_ext()
def logical(kv, d):
    return kv * 128 + d
out = C.tr_probe().cpu().numpy().astype("uint16")
bad = 0
for ds in range(4):
    for ks in range(4):
        for lane in range(64):
            l31, g16, half = lane & 31, lane >> 4, lane >> 5
            d = ds * 32 + l31
            for e in range(8):
                kv = ks * 16 + half * 8 + e
                want = logical(kv, d)
                got = int(out[ds, ks, lane, e])
                if got != want:
                    if bad < 12:
                        print(f"ds{ds} ks{ks} lane{lane} e{e}: got {got} want {want} (kv{kv} d{d})")
                    bad += 1
print("mismatches:", bad, "/", 4*4*64*8)

print("\nfull map ds0 ks0: lane -> [(kv,d) x8]")
for lane in range(32):
    row = out[0, 0, lane]
    dec = [(int(v) >> 7, int(v) & 127) for v in row]
    print(f"lane {lane:2d}: {dec}")
