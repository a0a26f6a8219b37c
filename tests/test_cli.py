"""The `python -m examl_amd` command-line drop-in: the reference's
option surface (axml.c get_args:941) over the MI355X engines.

CPU tests cover argument validation and the PRODUCT-path guarantee that
the CLI fails loudly without a GPU (no CPU/oracle fallback); the GPU
test runs a whole -f E job and checks the reference golden."""

import os

import pytest

from examl_amd.__main__ import _parse_args, _parse_groups, main


def _run(args):
    with pytest.raises(SystemExit) as e:
        main(args)
    return e.value.code


def test_cli_argument_validation(golden_dir, tmp_path):
    assert _run([]) == "-s and -n are required"
    code = _run(["-s", "x.bin", "-n", "R"])
    assert "starting tree" in code
    code = _run(["-s", "x.bin", "-n", "R", "-g", "c.tree"])
    assert "-p" in code
    code = _run(["-s", "x.bin", "-n", "R", "-t", "t.tree", "-f", "q"])
    assert "-r" in code and "-Y" in code
    code = _run(["-Z"])
    assert "unknown option" in code


def test_cli_fails_loudly_without_gpu(golden_dir, tmp_path):
    """PRODUCT path: no silent CPU fallback — without torch.cuda the CLI
    refuses to run (this container has no GPU, so this is the real
    behavior, not a mock)."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("host has a GPU")
    code = _run(["-s", os.path.join(golden_dir, "12.binary"),
                 "-t", os.path.join(golden_dir, "12.tree"),
                 "-n", "R", "-f", "E", "-w", str(tmp_path)])
    assert "requires an AMD GPU" in code


def test_cli_defaults_and_parsing():
    o = _parse_args(["-s", "a", "-t", "b", "-n", "c"])
    assert o["m"] == "GAMMA" and o["f"] == "d" and o["c"] == 25
    o = _parse_args(["-s", "a", "-t", "b", "-n", "c", "-m", "PSR",
                     "-f", "E", "-B", "5", "-c", "10", "-e", "0.01",
                     "-i", "10", "-D", "-M", "-S", "-a"])
    assert o["m"] == "PSR" and o["B"] == 5 and o["c"] == 10
    assert o["e"] == 0.01 and o["i"] == 10
    assert o["D"] and o["M"] and o["S"] and o["a"]


def test_cli_group_parser(golden_dir):
    taxa = [f"T{i + 1:02d}" for i in range(12)]
    groups = _parse_groups(os.path.join(golden_dir, "12.groups"), taxa)
    assert groups == [[1, 2, 3], [4, 5, 6], [7, 8, 9], [10, 11, 12]]


@pytest.mark.gpu
def test_cli_f_E_12_gpu(golden_dir, tmp_path):
    """Whole-job CLI run on the GPU engines: -f E on the 12-taxon golden
    lands on the reference's -3650.993621 and writes the info/result
    files."""
    import torch
    assert torch.cuda.is_available()
    rc = main(["-s", os.path.join(golden_dir, "12.binary"),
               "-t", os.path.join(golden_dir, "12.tree"),
               "-n", "CLI12", "-f", "E", "-w", str(tmp_path)])
    assert rc == 0
    info = open(os.path.join(tmp_path, "ExaML_info.CLI12")).read()
    line = [ln for ln in info.splitlines()
            if ln.startswith("Likelihood tree 0:")][0]
    lnl = float(line.split(":")[1])
    assert abs(lnl - (-3650.993621)) < abs(3650.993621) * 1e-6
    assert os.path.exists(os.path.join(tmp_path, "ExaML_result.CLI12"))


def _oracle_build_engines(parts, opts, device):
    """Test-only stand-in for the GPU engine factory: same per-partition
    dispatch over the CPU oracle engines (the product path never does
    this — oracle/ is test infrastructure)."""
    import numpy as np

    import examl_amd as ea
    from tests.helpers import OracleCatEngine, OracleEngine
    psr = opts["m"] == "PSR"
    engines = []
    auto_flags = []
    empirical = []
    for p in parts:
        auto_flags.append(False)
        empirical.append(p.frequencies)
        m = ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0,
                           use_median=opts["a"])
        if psr:
            w = p.upper - p.lower
            engines.append(OracleCatEngine(p.tips, p.wgt, m,
                                           np.zeros(w, dtype=np.int32),
                                           np.array([1.0])))
        else:
            engines.append(OracleEngine(p.tips, p.wgt, m))
    return engines, auto_flags, empirical


def test_cli_full_flow_cpu(golden_dir, tmp_path, monkeypatch):
    """The whole CLI flow (parse -> engines -> mode -> output files) on
    CPU by swapping only the engine factory for the oracle engines; the
    numbers are the same reference goldens the engine-level tests pin."""
    import examl_amd.__main__ as cli
    monkeypatch.setattr(cli, "_build_engines", _oracle_build_engines)
    # -f E
    rc = cli.main(["-s", os.path.join(golden_dir, "12.binary"),
                   "-t", os.path.join(golden_dir, "12.tree"),
                   "-n", "E", "-f", "E", "-w", str(tmp_path)],
                  device="cpu")
    assert rc == 0
    info = open(os.path.join(tmp_path, "ExaML_info.E")).read()
    lnl = float([ln for ln in info.splitlines()
                 if ln.startswith("Likelihood tree 0:")][0].split(":")[1])
    assert abs(lnl - (-3650.993621)) < 1e-2
    assert os.path.exists(os.path.join(tmp_path, "ExaML_result.E"))
    # -f d with -B and -D
    rc = cli.main(["-s", os.path.join(golden_dir, "12.binary"),
                   "-t", os.path.join(golden_dir, "12.tree"),
                   "-n", "D", "-f", "d", "-B", "3", "-D",
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    info = open(os.path.join(tmp_path, "ExaML_info.D")).read()
    lnl = float([ln for ln in info.splitlines()
                 if ln.startswith("Likelihood of best tree:")]
                [0].split(":")[1])
    assert abs(lnl - (-2741.473102)) < 1e-2
    assert os.path.exists(os.path.join(tmp_path,
                                       "RAxML_3_goodTrees.D"))
    # -f q with -Y
    rc = cli.main(["-s", os.path.join(golden_dir, "12.binary"),
                   "-t", os.path.join(golden_dir, "12.tree"),
                   "-n", "Q", "-f", "q",
                   "-Y", os.path.join(golden_dir, "12.groups"),
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    qlines = [ln for ln in
              open(os.path.join(tmp_path, "ExaML_quartets.Q"))
              if "|" in ln]
    assert len(qlines) == 243
    # -g constraint path
    rc = cli.main(["-s", os.path.join(golden_dir, "12.binary"),
                   "-g", os.path.join(golden_dir, "12.constraint.tree"),
                   "-p", "12345", "-n", "G", "-f", "d",
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    info = open(os.path.join(tmp_path, "ExaML_info.G")).read()
    lnl = float([ln for ln in info.splitlines()
                 if ln.startswith("Likelihood of best tree:")]
                [0].split(":")[1])
    assert abs(lnl - (-3435.016697)) < 1e-2


def test_cli_psr_search_writes_checkpoints(golden_dir, tmp_path,
                                           monkeypatch):
    """-m PSR -f d now writes per-cycle binary checkpoints like the
    reference (searchAlgo.c:1188-1201: rateCategory/patrat + per-model
    category state under CAT); the file parses back with the CAT layout
    and, when the reference binary is present, the unmodified reference
    restarts from it."""
    import subprocess

    import examl_amd.__main__ as cli
    from examl_amd.checkpoint import read_checkpoint
    monkeypatch.setattr(cli, "_build_engines", _oracle_build_engines)
    rc = cli.main(["-s", os.path.join(golden_dir, "12.binary"),
                   "-t", os.path.join(golden_dir, "12.tree"),
                   "-n", "P", "-f", "d", "-m", "PSR",
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    cks = sorted(f for f in os.listdir(tmp_path)
                 if f.startswith("ExaML_binaryCheckpoint.P_"))
    assert cks, os.listdir(tmp_path)
    last = os.path.join(tmp_path, cks[-1])
    from examl_amd.examl_io import read_byte_file
    _, parts = read_byte_file(os.path.join(golden_dir, "12.binary"))
    width_total = sum(p.upper - p.lower for p in parts)
    ck = read_checkpoint(last, 12, [4] * len(parts), rate_het="CAT",
                         crunched_length=width_total)
    assert ck.rate_category.size == ck.patrat.size
    assert ck.rate_category.size == width_total
    assert all(m["num_cats"] >= 1 for m in ck.models)
    # cmd block says CAT (rateHetModel, axml.h:672)
    import struct as _s
    d = open(last, "rb").read()
    assert _s.unpack_from("<i", d, 1248 + 52)[0] == 0  # CAT
    ref_bin = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "oracle", "_ref", "examl-AVX")
    if os.path.exists(ref_bin):
        import shutil as _sh
        _sh.copy(os.path.join(golden_dir, "12.binary"),
                 str(tmp_path / "r.binary"))
        _sh.copy(os.path.join(golden_dir, "12.tree"),
                 str(tmp_path / "12.tree"))
        r = subprocess.run(
            [ref_bin, "-s", "r.binary", "-t", "12.tree", "-m", "PSR",
             "-f", "d", "-R", cks[-1], "-n", "RP"], cwd=str(tmp_path),
            capture_output=True, text=True, timeout=600)
        out = r.stdout + r.stderr
        assert "Restart with likelihood" in out, out[-2000:]
        assert r.returncode == 0, out[-2000:]


def test_cli_m_search_writes_checkpoints(golden_dir, tmp_path,
                                         monkeypatch):
    """-M -f d writes checkpoints with per-partition branch vectors in
    the node image (writeTree z[numBranches]); the reference restarts
    from them when present."""
    import subprocess

    import examl_amd.__main__ as cli
    monkeypatch.setattr(cli, "_build_engines", _oracle_build_engines)
    rc = cli.main(["-s", os.path.join(golden_dir, "12m.binary"),
                   "-t", os.path.join(golden_dir, "12.tree"),
                   "-n", "M", "-f", "d", "-M",
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    cks = sorted(f for f in os.listdir(tmp_path)
                 if f.startswith("ExaML_binaryCheckpoint.M_"))
    assert cks, os.listdir(tmp_path)
    d = open(os.path.join(tmp_path, cks[-1]), "rb").read()
    import struct as _s
    assert _s.unpack_from("<i", d, 1248 + 16)[0] == 1  # perGeneBL
    ref_bin = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "oracle", "_ref", "examl-AVX")
    if os.path.exists(ref_bin):
        import shutil as _sh
        _sh.copy(os.path.join(golden_dir, "12m.binary"),
                 str(tmp_path / "r.binary"))
        _sh.copy(os.path.join(golden_dir, "12.tree"),
                 str(tmp_path / "12.tree"))
        r = subprocess.run(
            [ref_bin, "-s", "r.binary", "-t", "12.tree", "-M",
             "-m", "GAMMA", "-f", "d", "-R", cks[-1], "-n", "RM"],
            cwd=str(tmp_path),
            capture_output=True, text=True, timeout=600)
        out = r.stdout + r.stderr
        assert "Restart with likelihood" in out, out[-2000:]
        assert r.returncode == 0, out[-2000:]


def test_cli_d_checkpoint_carries_tree_strings(golden_dir, tmp_path,
                                               monkeypatch):
    """-f d -D checkpoints carry the RF-convergence tree0/tree1 topology
    strings (searchAlgo.c:2178-2185), so a reference -D restart can
    re-populate its hash table (readCheckpoint:1545-1580)."""
    import subprocess

    import examl_amd.__main__ as cli
    monkeypatch.setattr(cli, "_build_engines", _oracle_build_engines)
    rc = cli.main(["-s", os.path.join(golden_dir, "12.binary"),
                   "-t", os.path.join(golden_dir, "12.tree"),
                   "-n", "DD", "-f", "d", "-D",
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    cks = sorted(f for f in os.listdir(tmp_path)
                 if f.startswith("ExaML_binaryCheckpoint.DD_"))
    assert cks
    # a FAST_SPRS checkpoint written after the first store has tree0
    from examl_amd.checkpoint import read_checkpoint
    got_string = False
    for f in cks:
        ck = read_checkpoint(os.path.join(tmp_path, f), 12, [4])
        t0 = ck.tree0.split(b"\0", 1)[0]
        if t0.startswith(b"(") and b"T" in t0:
            got_string = True
            break
    assert got_string, "no checkpoint carried a tree0 topology string"
    ref_bin = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "oracle", "_ref", "examl-AVX")
    if os.path.exists(ref_bin):
        import shutil as _sh
        _sh.copy(os.path.join(golden_dir, "12.binary"),
                 str(tmp_path / "r.binary"))
        _sh.copy(os.path.join(golden_dir, "12.tree"),
                 str(tmp_path / "12.tree"))
        r = subprocess.run(
            [ref_bin, "-s", "r.binary", "-t", "12.tree", "-D",
             "-m", "GAMMA", "-f", "d", "-R", cks[-1], "-n", "RD"],
            cwd=str(tmp_path), capture_output=True, text=True,
            timeout=600)
        out = r.stdout + r.stderr
        assert "Restart with likelihood" in out, out[-2000:]
        assert r.returncode == 0, out[-2000:]


def test_cli_psr_resume_from_own_checkpoint(golden_dir, tmp_path,
                                            monkeypatch):
    """Our own -R restart from a checkpoint WE wrote mid-PSR-search lands
    on the same final lnL as the uninterrupted run (restart determinism,
    searchAlgo.c:1726)."""
    import examl_amd.__main__ as cli
    monkeypatch.setattr(cli, "_build_engines", _oracle_build_engines)
    rc = cli.main(["-s", os.path.join(golden_dir, "12.binary"),
                   "-t", os.path.join(golden_dir, "12.tree"),
                   "-n", "PA", "-f", "d", "-m", "PSR",
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    info = open(os.path.join(tmp_path, "ExaML_info.PA")).read()
    final_a = float([ln for ln in info.splitlines()
                     if ln.startswith("Likelihood of best tree:")]
                    [0].split(":")[1])
    cks = sorted(f for f in os.listdir(tmp_path)
                 if f.startswith("ExaML_binaryCheckpoint.PA_"))
    assert len(cks) >= 2
    mid = os.path.join(tmp_path, cks[len(cks) // 2])
    rc = cli.main(["-s", os.path.join(golden_dir, "12.binary"),
                   "-n", "PB", "-f", "d", "-m", "PSR", "-R", mid,
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    info = open(os.path.join(tmp_path, "ExaML_info.PB")).read()
    final_b = float([ln for ln in info.splitlines()
                     if ln.startswith("Likelihood of best tree:")]
                    [0].split(":")[1])
    assert abs(final_a - final_b) <= 1e-6 * abs(final_a), \
        (final_a, final_b)


def test_cli_m_resume_from_own_checkpoint(golden_dir, tmp_path,
                                          monkeypatch):
    """-M -f d: our own -R restart from our mid-search checkpoint lands
    on the uninterrupted run's final lnL (per-partition branch vectors
    restored from the node image)."""
    import examl_amd.__main__ as cli
    monkeypatch.setattr(cli, "_build_engines", _oracle_build_engines)
    rc = cli.main(["-s", os.path.join(golden_dir, "12m.binary"),
                   "-t", os.path.join(golden_dir, "12.tree"),
                   "-n", "MA", "-f", "d", "-M",
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    info = open(os.path.join(tmp_path, "ExaML_info.MA")).read()
    final_a = float([ln for ln in info.splitlines()
                     if ln.startswith("Likelihood of best tree:")]
                    [0].split(":")[1])
    cks = sorted(f for f in os.listdir(tmp_path)
                 if f.startswith("ExaML_binaryCheckpoint.MA_"))
    assert len(cks) >= 2
    mid = os.path.join(tmp_path, cks[len(cks) // 2])
    rc = cli.main(["-s", os.path.join(golden_dir, "12m.binary"),
                   "-n", "MB", "-f", "d", "-M", "-R", mid,
                   "-w", str(tmp_path)], device="cpu")
    assert rc == 0
    info = open(os.path.join(tmp_path, "ExaML_info.MB")).read()
    final_b = float([ln for ln in info.splitlines()
                     if ln.startswith("Likelihood of best tree:")]
                    [0].split(":")[1])
    assert abs(final_a - final_b) <= 1e-6 * abs(final_a), \
        (final_a, final_b)
