"""TEST INFRASTRUCTURE / data extraction: dump the published LG4M / LG4X
rate matrices and frequencies (Le, Dang & Gascuel 2012) from the
reference's initProtMat (models.c:225, exposed by oracle/_ref/libref.so's
-Dstatic= build) into lg4_models.npz.  Run in the dev container where
/root/reference is present:

    python examl_amd/data/tools/gen_lg4_data.py
"""

import ctypes
import os

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(os.path.dirname(os.path.dirname(HERE)))
LIB = os.path.join(REPO, "oracle", "_ref", "libref.so")

LG4M, LG4X = 20, 21


def main():
    lib = ctypes.CDLL(LIB)
    out = {}
    for name, mid in (("lg4m", LG4M), ("lg4x", LG4X)):
        rates = np.zeros((4, 190))
        freqs = np.zeros((4, 20))
        for k in range(4):
            f = np.zeros(20)
            r = np.zeros(190)
            lib.initProtMat(
                f.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
                ctypes.c_int(mid),
                r.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
                ctypes.c_int(k))
            rates[k] = r
            freqs[k] = f
        out[name + "_rates190"] = rates
        out[name + "_frequencies"] = freqs
    np.savez(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "lg4_models.npz"), **out)
    print("wrote lg4_models.npz", {k: v.shape for k, v in out.items()})


if __name__ == "__main__":
    main()
