"""Parity of the fused multi-partition executors (mseg) against the
per-partition engines on the same data.

The fused path computes P matrices on device (k_make_p_mseg), so the bar
is <=1e-11 relative (device exp vs host libm in the last ulp), not
bit-exact — stated in include/examl_hip.h."""

import math

import numpy as np
import pytest

import examl_amd as ea
from tests.helpers import make_synthetic

pytestmark = pytest.mark.gpu


def _mk(widths=(1500, 700, 64, 2300), seed=5, states=4):
    import os
    rng = np.random.default_rng(seed)
    engines = []
    if states == 20:
        aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "examl_amd", "data",
            "aa_models.npz"))
    for i, w in enumerate(widths):
        if states == 4:
            tips, wgt = make_synthetic(12, w, seed=seed + i)
            freqs = rng.uniform(0.1, 0.4, 4)
            freqs /= freqs.sum()
            rates = list(rng.uniform(0.3, 3.0, 5)) + [1.0]
            m = ea.DnaGtrModel(list(freqs), rates,
                               alpha=float(rng.uniform(0.2, 1.5)))
        else:
            tips = np.zeros((13, w), dtype=np.uint8)
            for t in range(1, 13):
                tips[t] = rng.integers(1, 23, w).astype(np.uint8)
            wgt = rng.integers(1, 4, w).astype(np.int32)
            m = ea.ProtGtrModel(aa["frequencies"][4 + i % 3],
                                aa["rates190"][4 + i % 3],
                                float(rng.uniform(0.4, 1.2)))
        engines.append(ea.DnaGammaEngine(tips, wgt, m, device="cuda:0"))
    tree = ea.PhyloTree.random(12, seed=31, rng_z=True)
    return engines, ea.MultiDnaEngine(engines), tree


@pytest.mark.parametrize("states", [4, 20])
def test_full_lnl_matches_per_partition(states):
    engines, multi, tree = _mk(states=states)
    entries, (p, q, z) = tree.full_traversal()
    single = []
    for e in engines:
        e.newview_traversal(entries)
        single.append(float(e.evaluate_root(tree, p, q, z).cpu()))
    vec = multi.full_lnl(tree).cpu().numpy()
    for s, f in zip(single, vec):
        assert abs(f - s) <= 1e-11 * abs(s), (s, f)


@pytest.mark.parametrize("states", [4, 20])
def test_masked_partitions_stay_stale(states):
    engines, multi, tree = _mk(states=states)
    entries, (p, q, z) = tree.full_traversal()
    # first an all-partition traversal to give every CLV a value
    multi.newview_traversal(entries)
    base = multi.evaluate_root(tree, p, q, z).cpu().numpy().copy()
    # perturb branch lengths and redo the traversal with partitions 1,3
    # masked: their CLVs (and lnL at the OLD z) must be untouched
    entries2 = [ea.TravEntry(e.tipCase, e.pNumber, e.qNumber, e.rNumber,
                             e.x1Slot, e.x2Slot, e.x3Slot,
                             min(e.qz * 0.8, 0.99), min(e.rz * 0.9, 0.99))
                for e in entries]
    active = [1, 0, 1, 0]
    multi.newview_traversal(entries2, active=active)
    after = multi.evaluate_root(tree, p, q, z).cpu().numpy()
    for i, a in enumerate(active):
        if not a:
            assert after[i] == base[i], (i, base[i], after[i])
        else:
            assert after[i] != base[i]


@pytest.mark.parametrize("states", [4, 20])
def test_makenewz_matches_per_partition(states):
    engines, multi, tree = _mk(states=states)
    entries, (p, q, z) = tree.full_traversal()
    for e in engines:
        e.newview_traversal(entries)
    multi.newview_traversal(entries)
    # joint NR over all partitions: single path sums per-engine derivs
    for e in engines:
        e.sum_root(tree, p, q)
    multi.sum_root(tree, p, q)
    lz = math.log(max(z, ea.ZMIN))
    d1 = d2 = 0.0
    for e in engines:
        a, b = e.core_derivs(lz)
        d1 += a
        d2 += b
    f1, f2 = multi.core_derivs(lz)
    assert abs(f1 - d1) <= 1e-9 * max(1.0, abs(d1)), (d1, f1)
    assert abs(f2 - d2) <= 1e-9 * max(1.0, abs(d2)), (d2, f2)


def test_per_partition_branch_lengths():
    """-M shape: per-(op, partition) z overrides and per-partition root z."""
    engines, multi, tree = _mk(widths=(900, 1100))
    entries, (p, q, z) = tree.full_traversal()
    NP = len(engines)
    rng = np.random.default_rng(9)
    qz = np.empty((len(entries), NP))
    rz = np.empty((len(entries), NP))
    for i, e in enumerate(entries):
        for m in range(NP):
            qz[i, m] = min(max(e.qz * rng.uniform(0.7, 1.3), 1e-6), 0.999)
            rz[i, m] = min(max(e.rz * rng.uniform(0.7, 1.3), 1e-6), 0.999)
    zroot = np.array([min(max(z * 0.9, 1e-6), 0.999),
                      min(max(z * 1.1, 1e-6), 0.999)])
    single = []
    for m, e in enumerate(engines):
        ent = [ea.TravEntry(t.tipCase, t.pNumber, t.qNumber, t.rNumber,
                            t.x1Slot, t.x2Slot, t.x3Slot, qz[i, m],
                            rz[i, m]) for i, t in enumerate(entries)]
        e.newview_traversal(ent)
        single.append(float(e.evaluate_root(tree, p, q,
                                            float(zroot[m])).cpu()))
    multi.newview_traversal(entries, qz_ov=qz, rz_ov=rz)
    vec = multi.evaluate_root(tree, p, q, zroot).cpu().numpy()
    for s, f in zip(single, vec):
        assert abs(f - s) <= 1e-11 * abs(s), (s, f)


def test_treesearch_fused_matches_unfused():
    """TreeSearch auto-enables the fused path for homogeneous dense GAMMA
    engines; full -f E-style treeEvaluate agrees with the per-partition
    loop to 1e-9 (device-exp P ulps compound through Brent probes)."""
    from examl_amd.search import TreeSearch
    eng_a, multi, _ = _mk(widths=(800, 1200, 96))
    eng_b, _, _ = _mk(widths=(800, 1200, 96))
    t_a = ea.PhyloTree.random(12, seed=31, rng_z=True)
    t_b = ea.PhyloTree.random(12, seed=31, rng_z=True)
    ts_a = TreeSearch(t_a, eng_a)
    assert ts_a.fused is not None
    ts_b = TreeSearch(t_b, eng_b)
    ts_b.fused = None  # force the per-partition loop
    la = ts_a.evaluate_generic(full=True)
    lb = ts_b.evaluate_generic(full=True)
    assert abs(la - lb) <= 1e-9 * abs(lb), (la, lb)
    ta = ts_a.tree_evaluate(1.0)
    tb = ts_b.tree_evaluate(1.0)
    assert abs(ta - tb) <= 1e-8 * abs(tb), (ta, tb)
