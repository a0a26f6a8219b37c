"""The literal C drop-in (SURVEY §8b / VERDICT r01 #1): the UNMODIFIED
reference search (searchAlgo.c/optimizeModel.c/axml.c compiled in place
from /root/reference) linked against libexaml_hip.so through
shim/axml_shim.c.  The hybrid binary shim/_build/examl-HIP is prebuilt by
__graft_entry__.build() in the dev container and travels to the GPU box
with the snapshot; nothing here reads /root/reference at run time.

Goldens: testData/49 -f E final lnL -16205.671990 and -f d final lnL
-16194.095475 (reference examl-AVX, bit-identical at 1 and 2
MPI ranks)."""

import os
import shutil
import subprocess

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
HYBRID = os.path.join(REPO, "shim", "_build", "examl-HIP")
GOLDEN = os.path.join(HERE, "golden")

GOLDEN_F_E = -16205.671990   # reference -f E on testData/49
GOLDEN_F_D = -16194.095475   # reference -f d (full SPR search)


def _run(tmp_path, args, timeout=900):
    shutil.copy(os.path.join(GOLDEN, "49.binary"), str(tmp_path))
    shutil.copy(os.path.join(GOLDEN, "49.tree"), str(tmp_path))
    r = subprocess.run([HYBRID] + args, cwd=str(tmp_path),
                       capture_output=True, text=True, timeout=timeout)
    return r


@pytest.mark.skipif(not os.path.exists(HYBRID),
                    reason="hybrid binary not built")
def test_hybrid_fails_loudly_without_gpu_or_runs(tmp_path):
    """CPU-side: the binary exists, links (exercising every exported
    boundary symbol at load time) and starts up through byte-file read,
    tree parse and mode dispatch; on a GPU-less box it must abort loudly
    at the first likelihood call (no silent CPU fallback)."""
    r = _run(tmp_path, ["-s", "49.binary", "-t", "49.tree", "-m", "GAMMA",
                        "-f", "E", "-n", "CPU"], timeout=300)
    out = r.stdout + r.stderr
    if "no ROCm-capable device" in out or "no HIP device" in out:
        assert r.returncode != 0
        assert "Found 1 trees to evaluate" in out  # got past startup
    else:  # on a GPU box this test simply sees the real run succeed
        assert "Likelihood tree 0" in out, out[-2000:]


@pytest.mark.gpu
@pytest.mark.skipif(not os.path.exists(HYBRID),
                    reason="hybrid binary not built")
def test_hybrid_tree_evaluation(tmp_path):
    """-f E through searchAlgo.c's own treeEvaluate/modOpt driving the HIP
    kernels: final lnL vs the reference golden to 1e-6 relative (the
    north-star parity bar)."""
    r = _run(tmp_path, ["-s", "49.binary", "-t", "49.tree", "-m", "GAMMA",
                        "-f", "E", "-n", "HYB"])
    out = r.stdout + r.stderr
    assert r.returncode == 0, out[-3000:]
    lines = [ln for ln in out.splitlines() if "Likelihood tree 0" in ln]
    assert lines, out[-3000:]
    lnl = float(lines[0].split(":")[1])
    rel = abs(lnl - GOLDEN_F_E) / abs(GOLDEN_F_E)
    assert rel < 1e-6, f"hybrid -f E lnL {lnl} vs {GOLDEN_F_E} rel {rel}"


@pytest.mark.gpu
@pytest.mark.skipif(not os.path.exists(HYBRID),
                    reason="hybrid binary not built")
def test_hybrid_full_search(tmp_path):
    """-f d: computeBIGRAPID (searchAlgo.c:1914) — the reference's
    default full SPR hill-climb — end to end on the HIP kernels."""
    r = _run(tmp_path, ["-s", "49.binary", "-t", "49.tree", "-m", "GAMMA",
                        "-f", "d", "-n", "HYBD"], timeout=1200)
    out = r.stdout + r.stderr
    assert r.returncode == 0, out[-3000:]
    lines = [ln for ln in out.splitlines()
             if "Likelihood of best tree" in ln]
    assert lines, out[-3000:]
    lnl = float(lines[0].split(":")[1])
    rel = abs(lnl - GOLDEN_F_D) / abs(GOLDEN_F_D)
    assert rel < 1e-6, f"hybrid -f d lnL {lnl} vs {GOLDEN_F_D} rel {rel}"


@pytest.mark.gpu
@pytest.mark.skipif(not os.path.exists(HYBRID),
                    reason="hybrid binary not built")
def test_hybrid_140_partitioned_protein(tmp_path):
    """testData/140 (BASELINE.json configs[4]): WAG + two AUTO protein
    partitions through the unmodified reference -f E — exercises the
    protein kernels, AUTO model selection (optimizeModel.c:2669) and the
    per-partition dispatch under the C shim.  Golden: reference examl-AVX
    -f E final lnL -121288.814123 (858 s on the dev box CPU)."""
    shutil.copy(os.path.join(GOLDEN, "140.binary"), str(tmp_path))
    shutil.copy(os.path.join(GOLDEN, "140.tree"), str(tmp_path))
    r = subprocess.run(
        [HYBRID, "-s", "140.binary", "-t", "140.tree", "-m", "GAMMA",
         "-f", "E", "-n", "H140"], cwd=str(tmp_path), capture_output=True,
        text=True, timeout=1800)
    out = r.stdout + r.stderr
    assert r.returncode == 0, out[-3000:]
    lines = [ln for ln in out.splitlines() if "Likelihood tree 0" in ln]
    assert lines, out[-3000:]
    lnl = float(lines[0].split(":")[1])
    golden = -121288.814123
    rel = abs(lnl - golden) / abs(golden)
    assert rel < 1e-6, f"hybrid 140 lnL {lnl} vs {golden} rel {rel}"


@pytest.mark.gpu
@pytest.mark.skipif(not os.path.exists(HYBRID),
                    reason="hybrid binary not built")
def test_hybrid_psr(tmp_path):
    """-m PSR -f E: the reference's CAT per-site-rate machinery
    (optimizeRateCategories, optimizeModel.c:2403) driving the CAT HIP
    kernels through the shim, incl. evaluatePartialGeneric as host math.
    Golden: reference examl-AVX -m PSR -f E on testData/49 =
    -14702.970620."""
    r = _run(tmp_path, ["-s", "49.binary", "-t", "49.tree", "-m", "PSR",
                        "-f", "E", "-n", "HYBPSR"], timeout=1200)
    out = r.stdout + r.stderr
    assert r.returncode == 0, out[-3000:]
    lines = [ln for ln in out.splitlines() if "Likelihood tree 0" in ln]
    assert lines, out[-3000:]
    lnl = float(lines[0].split(":")[1])
    golden = -14702.970620
    rel = abs(lnl - golden) / abs(golden)
    assert rel < 1e-6, f"hybrid PSR lnL {lnl} vs {golden} rel {rel}"


@pytest.mark.gpu
@pytest.mark.skipif(not os.path.exists(HYBRID),
                    reason="hybrid binary not built")
def test_hybrid_two_ranks(tmp_path):
    """2 MPI ranks (the reference's own data distribution,
    partitionAssignment.c:398, + the C1/C2 MPI_Allreduce) with both
    ranks' HIP contexts on one GPU: final -f E lnL must match the 1-rank
    golden — the reference's own determinism anchor (SURVEY §8c:
    bit-identical at 1 and 2 ranks)."""
    shutil.copy(os.path.join(GOLDEN, "49.binary"), str(tmp_path))
    shutil.copy(os.path.join(GOLDEN, "49.tree"), str(tmp_path))
    mpiexec = "/opt/conda/bin/mpiexec"
    if not os.path.exists(mpiexec):
        pytest.skip("mpiexec not present")
    r = subprocess.run(
        [mpiexec, "-n", "2", HYBRID, "-s", "49.binary", "-t", "49.tree",
         "-m", "GAMMA", "-f", "E", "-n", "HYB2"], cwd=str(tmp_path),
        capture_output=True, text=True, timeout=900)
    out = r.stdout + r.stderr
    assert r.returncode == 0, out[-3000:]
    lines = [ln for ln in out.splitlines() if "Likelihood tree 0" in ln]
    assert lines, out[-3000:]
    lnl = float(lines[0].split(":")[1])
    rel = abs(lnl - GOLDEN_F_E) / abs(GOLDEN_F_E)
    assert rel < 1e-6, f"2-rank hybrid lnL {lnl} vs {GOLDEN_F_E} rel {rel}"


@pytest.mark.gpu
@pytest.mark.skipif(not os.path.exists(HYBRID),
                    reason="hybrid binary not built")
def test_hybrid_lg4(tmp_path):
    """LG4M + LG4X + WAG partitions (tests/golden/140lg4.binary) through
    the unmodified reference -f E: the per-gamma-category eigensystem
    path (makeP_FlexLG4, newviewGenericSpecial.c:170; optLG4X,
    optimizeModel.c:1116) on the LG4 HIP executors.  Golden: reference
    examl-AVX -f E final lnL -120844.546570 (tests/test_lg4.py)."""
    shutil.copy(os.path.join(GOLDEN, "140lg4.binary"), str(tmp_path))
    shutil.copy(os.path.join(GOLDEN, "140.tree"), str(tmp_path))
    r = subprocess.run(
        [HYBRID, "-s", "140lg4.binary", "-t", "140.tree", "-m", "GAMMA",
         "-f", "E", "-n", "HLG4"], cwd=str(tmp_path), capture_output=True,
        text=True, timeout=1800)
    out = r.stdout + r.stderr
    assert r.returncode == 0, out[-3000:]
    lines = [ln for ln in out.splitlines() if "Likelihood tree 0" in ln]
    assert lines, out[-3000:]
    lnl = float(lines[0].split(":")[1])
    golden = -120844.546570
    rel = abs(lnl - golden) / abs(golden)
    assert rel < 1e-6, f"hybrid LG4 lnL {lnl} vs {golden} rel {rel}"


@pytest.mark.gpu
@pytest.mark.skipif(not os.path.exists(HYBRID),
                    reason="hybrid binary not built")
def test_hybrid_quartets(tmp_path):
    """-f q -r 30 -p 12345: the reference's quartet machinery
    (computeQuartets, quartets.c:349) driving the HIP kernels through
    the boundary on 4-taxon trees — the launch-latency stress shape.
    The quartet file must match the reference's own line for line
    (tests/golden/49.quartets.txt)."""
    shutil.copy(os.path.join(GOLDEN, "49.binary"), str(tmp_path))
    shutil.copy(os.path.join(GOLDEN, "49.tree"), str(tmp_path))
    r = subprocess.run(
        [HYBRID, "-s", "49.binary", "-t", "49.tree", "-m", "GAMMA",
         "-f", "q", "-r", "30", "-p", "12345", "-n", "HQ"],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=900)
    out = r.stdout + r.stderr
    assert r.returncode == 0, out[-3000:]
    qf = [f for f in os.listdir(tmp_path) if f.startswith("ExaML_quartets")]
    assert qf, out[-2000:]
    got = open(os.path.join(tmp_path, qf[0])).read().strip().splitlines()
    ref = open(os.path.join(GOLDEN, "49.quartets.txt")
               ).read().strip().splitlines()

    def parse(lines):
        # topology triple + lnLs: compare taxa + lnL to 1e-6 rel
        rows = []
        for ln in lines:
            if not ln.strip() or ":" not in ln:
                continue
            rows.append(ln.split())
        return rows

    g, f = parse(got), parse(ref)
    assert len(g) == len(f), (len(g), len(f))
    for a, b in zip(g, f):
        assert len(a) == len(b)
        for xa, xb in zip(a, b):
            try:
                va, vb = float(xa), float(xb)
                assert abs(va - vb) <= 1e-6 * max(1.0, abs(vb)), (a, b)
            except ValueError:
                assert xa == xb, (a, b)
