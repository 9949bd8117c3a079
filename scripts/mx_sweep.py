"""Decode the MX scale lane/byte -> k-coverage map (quarter-marked data)."""
import ctypes, os
import numpy as np
import torch  # noqa

HERE = os.path.dirname(os.path.abspath(__file__))
lib = ctypes.CDLL(os.path.join(HERE, "mx_probe.so"))

# quarter markers: k in [0,16)=1.0, [16,32)=2.0, [32,48)=4.0, [48,64)=8.0
QV = {0: 0x38, 1: 0x40, 2: 0x48, 3: 0x50}  # e4m3 1,2,4,8
A = np.zeros((32, 64), dtype=np.uint8)
for q in range(4):
    A[:, q*16:(q+1)*16] = QV[q]
B = np.full((32, 64), 0x38, dtype=np.uint8)  # 1.0

for name, fn in (("byte0", lib.run_mx_sweep), ("byte1", lib.run_mx_sweep_b1)):
    D = np.zeros((64, 32, 32), dtype=np.float32)
    rc = fn(A.ctypes.data_as(ctypes.c_void_p), B.ctypes.data_as(ctypes.c_void_p),
            D.ctypes.data_as(ctypes.c_void_p))
    assert rc == 0, rc
    base = np.median(D[:, :, 0])
    print(f"== {name}: baseline D[.,0] = {base} (full sum = 16*(1+2+4+8) = 240)")
    # delta identifies covered quarters: subset sum of {16,32,64,128}
    for L in (0, 1, 31, 32, 33, 63):
        d = D[L, :, 0] - base
        rows = np.nonzero(np.abs(d) > 1e-3)[0]
        for r in rows:
            v = d[r]
            quarters = [q for q in range(4) if (int(round(v)) >> 4) & ... ] if False else None
            print(f"  lane {L:2d} -> row {r}: delta {v:g} "
                  f"(covered-quarter sum = {v:g}; quarters by bits of {int(round(v))//16} over weights 1,2,4,8)")
