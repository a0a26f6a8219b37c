#!/usr/bin/env python3
"""Regenerate examl_amd/data/lg_model.npz from the reference's initProtMat
(oracle/_ref must be built; dev container only)."""
import ctypes, os, sys
import numpy as np
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import oracle as O
f = np.zeros(20); rates = np.zeros(190)
O._ref.initProtMat(f.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
                   ctypes.c_int(10),  # LG, axml.h:252
                   rates.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
                   ctypes.c_int(0))
np.savez(os.path.join(os.path.dirname(__file__), "..", "examl_amd", "data",
                      "lg_model.npz"), frequencies=f, rates190=rates)
print("ok")
