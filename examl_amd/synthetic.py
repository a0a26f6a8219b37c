"""Seeded synthetic DNA alignments (the BASELINE.json §8d generator:
random base sequence, per-taxon mutations, small ambiguity fraction)."""

import numpy as np


def make_alignment(ntips, width, seed=42, mutation=0.10, ambiguity=0.01):
    """uint8 tip matrix [ntips+1, width] (row 0 unused, codes 1..15:
    A=1 C=2 G=4 T=8 + IUPAC bitmasks) and unit pattern weights."""
    rng = np.random.default_rng(seed)
    tips = np.zeros((ntips + 1, width), dtype=np.uint8)
    pure = np.array([1, 2, 4, 8], dtype=np.uint8)
    base = pure[rng.integers(0, 4, width)]
    for t in range(1, ntips + 1):
        row = base.copy()
        mut = rng.random(width) < mutation
        row[mut] = pure[rng.integers(0, 4, int(mut.sum()))]
        if ambiguity > 0:
            amb = rng.random(width) < ambiguity
            row[amb] = rng.integers(1, 16, int(amb.sum())).astype(np.uint8)
        tips[t] = row
    wgt = np.ones(width, dtype=np.int32)
    return tips, wgt


def make_alignment_aa(ntips, width, seed=42, mutation=0.15, ambiguity=0.01):
    """uint8 protein tip matrix [ntips+1, width] (codes 1..22: 20 residues +
    B/Z ambiguity + X) and unit pattern weights."""
    rng = np.random.default_rng(seed)
    tips = np.zeros((ntips + 1, width), dtype=np.uint8)
    base = rng.integers(1, 21, width).astype(np.uint8)
    for t in range(1, ntips + 1):
        row = base.copy()
        mut = rng.random(width) < mutation
        row[mut] = rng.integers(1, 21, int(mut.sum())).astype(np.uint8)
        if ambiguity > 0:
            amb = rng.random(width) < ambiguity
            row[amb] = rng.integers(1, 23, int(amb.sum())).astype(np.uint8)
        tips[t] = row
    wgt = np.ones(width, dtype=np.int32)
    return tips, wgt
