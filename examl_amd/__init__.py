"""examl_amd — MI355X-native (gfx950/CDNA4) implementation of ExaML's
per-site conditional-likelihood hot path.

The compute path is libexaml_hip.so (hand-written HIP kernels behind the
C-ABI in include/examl_hip.h).  This package is the host-side mirror of the
reference's likelihood entry points (newviewGeneric / evaluateGeneric /
makenewzGeneric shapes); PyTorch is used only for device memory, streams and
the RCCL all-reduce.

The HIP extension is REQUIRED on a GPU machine: there is no CPU fallback in
the product path (the CPU restatement under oracle/ is test infrastructure
and must never be imported from here).
"""

import ctypes
import os

# Load torch (and thus its bundled HIP runtime, soname libamdhip64.so.7)
# BEFORE dlopen'ing our C-ABI library: our DT_NEEDED "libamdhip64.so.7" then
# resolves to the already-loaded instance and the process has ONE HIP
# runtime.  In the other order torch's DT_NEEDED "libamdhip64.so" loads a
# second runtime whose HSA instance sees no device (observed: rc=100
# hipErrorNoDevice from every call of ours while torch works).
import torch  # noqa: F401

__version__ = "0.1"

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libexaml_hip.so")

TIP_TIP, TIP_INNER, INNER_INNER = 0, 1, 2
ZMIN, ZMAX = 1.0e-15, 1.0 - 1.0e-6


class TravEntry(ctypes.Structure):
    """examl_hip_trav_entry (include/examl_hip.h) — one post-order CLV
    update, mirroring the reference's traversalInfo (examl/axml.h:434)."""
    _fields_ = [
        ("tipCase", ctypes.c_int),
        ("pNumber", ctypes.c_int),
        ("qNumber", ctypes.c_int),
        ("rNumber", ctypes.c_int),
        ("x1Slot", ctypes.c_int),
        ("x2Slot", ctypes.c_int),
        ("x3Slot", ctypes.c_int),
        ("qz", ctypes.c_double),
        ("rz", ctypes.c_double),
    ]


def _bind(lib):
    d = ctypes.c_double
    i = ctypes.c_int
    l = ctypes.c_long
    p = ctypes.c_void_p
    lib.examl_hip_version.restype = ctypes.c_char_p
    lib.examl_hip_last_error_string.restype = ctypes.c_char_p
    lib.examl_host_make_p.argtypes = [d, d, p, p, p, i, p, p, i]
    lib.examl_host_calc_diagptable.argtypes = [d, i, i, p, p, p]
    lib.examl_host_core_dtables_dna.argtypes = [p, p, d, p]
    lib.examl_host_init_gtr_dna.argtypes = [p, p, p, p, p, p]
    lib.examl_host_make_gamma_cats.argtypes = [d, p, i]
    lib.examl_hip_newview_dna_gamma.argtypes = \
        [i, p, p, p, p, p, p, p, l, p, p, p, p, p]
    lib.examl_hip_evaluate_dna_gamma.argtypes = \
        [p, p, p, p, p, l, p, p, p, d, p, p, p]
    lib.examl_hip_sum_dna_gamma.argtypes = [i, p, p, p, p, p, p, l, p]
    lib.examl_hip_core_dna_gamma.argtypes = [l, p, p, p, p, p, p]
    lib.examl_hip_newview_traversal_dna_gamma.argtypes = \
        [p, i, p, p, p, p, p, p, l, p, l, p, l, p, p, p, p]
    lib.examl_hip_evaluate_root_dna_gamma.argtypes = \
        [i, i, i, i, i, i, d, p, p, p, p, l, p, l, p, l, p, p, p, p, p]
    lib.examl_hip_sum_root_dna_gamma.argtypes = \
        [i, i, i, i, i, p, p, l, p, l, p, l, p]
    lib.examl_hip_core_root_dna_gamma.argtypes = \
        [l, p, p, p, d, p, p, p, p, p]
    # protein surface (same shapes, states=20)
    lib.examl_host_init_gtr_aa.argtypes = [p, p, p, p, p, p]
    lib.examl_host_core_dtables_prot.argtypes = [p, p, d, p]
    lib.examl_hip_newview_prot_gamma.argtypes = \
        [i, p, p, p, p, p, p, p, l, p, p, p, p, p]
    lib.examl_hip_evaluate_prot_gamma.argtypes = \
        [p, p, p, p, p, l, p, p, p, d, p, p, p]
    lib.examl_hip_sum_prot_gamma.argtypes = [i, p, p, p, p, p, p, l, p]
    lib.examl_hip_core_prot_gamma.argtypes = [l, p, p, p, p, p, p]
    lib.examl_hip_newview_traversal_prot_gamma.argtypes = \
        [p, i, p, p, p, p, p, p, l, p, l, p, l, p, p, p, p]
    lib.examl_hip_evaluate_root_prot_gamma.argtypes = \
        [i, i, i, i, i, i, d, p, p, p, p, l, p, l, p, l, p, p, p, p, p]
    lib.examl_hip_sum_root_prot_gamma.argtypes = \
        [i, i, i, i, i, p, p, l, p, l, p, l, p]
    lib.examl_hip_core_root_prot_gamma.argtypes = \
        [l, p, p, p, d, p, p, p, p, p]
    # LG4 (per-category matrices)
    lib.examl_host_make_gamma_cats_median.argtypes = [d, p, i]
    lib.examl_host_make_p_lg4.argtypes = [d, d, p, p, p, p, p]
    lib.examl_host_calc_diag_lg4.argtypes = [d, p, p, p]
    lib.examl_host_core_dtables_prot_lg4.argtypes = [p, p, d, p]
    lib.examl_hip_newview_traversal_prot_lg4.argtypes = \
        [p, i, p, p, p, p, p, p, l, p, l, p, l, p, p, p, p]
    lib.examl_hip_evaluate_root_prot_lg4.argtypes = \
        [i, i, i, i, i, i, d, p, p, p, p, p, l, p, l, p, l, p, p, p, p, p]
    lib.examl_hip_sum_root_prot_lg4.argtypes = \
        [i, i, i, i, i, p, p, l, p, l, p, l, p]
    lib.examl_hip_core_root_prot_lg4.argtypes = \
        [l, p, p, p, p, d, p, p, p, p, p]
    # protein CAT (-m PSR on AA)
    lib.examl_host_evaluate_partial_prot_cat.restype = ctypes.c_double
    lib.examl_host_evaluate_partial_prot_cat.argtypes = \
        lib.examl_host_evaluate_partial_dna_cat.argtypes
    lib.examl_host_core_dtables_prot_cat.argtypes = [p, p, i, d, p]
    lib.examl_hip_newview_traversal_prot_cat.argtypes = \
        [p, i, p, p, p, i, p, p, p, p, l, p, l, p, l, p, p, p, p]
    lib.examl_hip_evaluate_root_prot_cat.argtypes = \
        [i, i, i, i, i, i, d, p, p, i, p, p, p, l, p, l, p, l, p, p, p, p, p]
    lib.examl_hip_sum_root_prot_cat.argtypes = \
        [i, i, i, i, i, p, p, l, p, l, p, l, p]
    lib.examl_hip_core_root_prot_cat.argtypes = \
        [l, p, p, p, i, d, p, p, p, p, p, p]
    # -S (saveMemory) DNA kernels
    lib.examl_hip_gap_and_prefix.argtypes = [p, p, p, p, i, l, p]
    lib.examl_hip_newview_dna_save.argtypes = \
        [i, p, p, p, p, p, p, p, p, p, l, p, p, p, p, p, p, p, p, p, p, p, p]
    lib.examl_hip_evaluate_dna_save.argtypes = \
        [i, p, p, p, p, p, p, l, p, p, p, p, p, p, i, i, p, p, p, p]
    lib.examl_hip_sum_dna_save.argtypes = \
        [i, p, p, p, p, p, p, l, p, p, p, p, p, p, p]
    lib.examl_hip_profile_enable.argtypes = [i]
    lib.examl_hip_profile_reset.argtypes = []
    lib.examl_hip_profile_get.argtypes = [p, p]
    lib.examl_host_core_dtables_dna_cat.argtypes = [p, p, i, d, p]
    lib.examl_hip_newview_dna_cat.argtypes = \
        [i, p, p, p, p, p, p, p, p, l, p, i, p, p, p]
    lib.examl_hip_evaluate_dna_cat.argtypes = \
        [p, p, p, p, p, p, l, p, i, p, p, d, p, p, p]
    lib.examl_hip_sum_dna_cat.argtypes = [i, p, p, p, p, p, p, l, p]
    lib.examl_hip_core_root_dna_cat.argtypes = \
        [l, p, p, p, i, d, p, p, p, p, p, p]
    lib.examl_hip_newview_traversal_dna_cat.argtypes = \
        [p, i, p, p, p, i, p, p, p, p, l, p, l, p, l, p, p, p, p]
    lib.examl_hip_evaluate_root_dna_cat_x.argtypes = \
        [i, i, i, i, i, i, d, p, p, i, p, p, p, l, p, l, p, l, p, p, p, p, p]
    lib.examl_hip_sum_root_dna_cat.argtypes = \
        [i, i, i, i, i, p, p, l, p, l, p, l, p]
    lib.examl_host_evaluate_partial_dna_cat.restype = ctypes.c_double
    lib.examl_host_evaluate_partial_dna_cat.argtypes = \
        [p, i, i, i, d, l, d, i, p, p, p, p, p, l, i]
    lib.examl_hip_use_graphs.argtypes = [i]
    lib.examl_hip_graphs_clear.argtypes = []
    lib.examl_hip_fast_math.argtypes = [i]
    # -S prot GAMMA + CAT families
    lib.examl_host_make_p_save.argtypes = [d, d, p, p, p, i, p, p, i, i]
    lib.examl_hip_newview_prot_save.argtypes = \
        [i, p, p, p, p, p, p, p, p, p, l, p, p, p, p, p, p, p, p, p, p, p,
         p]
    lib.examl_hip_evaluate_prot_save.argtypes = \
        [i, p, p, p, p, p, p, l, p, p, p, p, p, p, i, i, p, p, p, p]
    lib.examl_hip_sum_prot_save.argtypes = \
        [i, p, p, p, p, p, p, l, p, p, p, p, p, p, p]
    lib.examl_hip_newview_cat_save.argtypes = \
        [i, i, p, p, p, p, p, p, p, p, p, l, p, i, p, p, p, p, p, p, p, p,
         p, p, p, p]
    lib.examl_hip_evaluate_cat_save.argtypes = \
        [i, p, p, p, p, p, p, l, p, p, p, p, p, p, p, i, i, p, p, p, p]
    lib.examl_hip_sum_cat_save.argtypes = \
        [i, i, p, p, p, p, p, p, l, p, p, p, p, p, p, p]
    # multi-partition fused executors
    lib.examl_hip_multi_create.argtypes = \
        [i, i, p, p, p, p, p, p, p, p, p, i, p]
    lib.examl_hip_multi_destroy.argtypes = [p]
    lib.examl_hip_newview_traversal_multi.argtypes = \
        [p, p, i, p, p, p, p, p, p, p]
    lib.examl_hip_evaluate_root_multi.argtypes = \
        [p, i, i, i, i, i, i, p, i, p, p, p, p, p]
    lib.examl_hip_sum_root_multi.argtypes = [p, i, i, i, i, i, p, p]
    lib.examl_hip_core_root_multi.argtypes = [p, p, i, p, p, p, p, p]
    return lib


_lib = None
_load_error = None
try:
    _lib = _bind(ctypes.CDLL(_LIB_PATH))
except OSError as e:  # pragma: no cover
    _load_error = e


def lib():
    """The HIP extension.  Fails loudly if it is missing — no fallback."""
    if _lib is None:
        raise RuntimeError(
            f"examl_amd: HIP extension not available at {_LIB_PATH} "
            f"({_load_error}); build it with __graft_entry__.build()")
    return _lib


def check(rc, what):
    if rc != 0:
        raise RuntimeError(
            f"examl_amd: {what} failed (rc={rc}): "
            f"{lib().examl_hip_last_error_string().decode()}")


from .model import DnaGtrModel, Lg4Model, ProtGtrModel  # noqa: E402
from .tree import PhyloTree             # noqa: E402
from .spr import SprSearch, SprTree     # noqa: E402
from .engine import (DnaCatEngine, DnaGammaEngine, Lg4Engine,  # noqa: E402
                     MultiDnaEngine, MultiEngine, ProtCatEngine, SaveCatEngine,
                     SaveDnaEngine, SaveProtEngine)

__all__ = [
    "lib", "check", "TravEntry", "DnaGtrModel", "ProtGtrModel",
    "Lg4Model", "PhyloTree", "SprSearch", "SprTree",
    "DnaGammaEngine", "DnaCatEngine", "Lg4Engine", "SaveDnaEngine",
    "ProtCatEngine", "MultiDnaEngine", "MultiEngine", "SaveProtEngine",
    "SaveCatEngine",
    "TIP_TIP", "TIP_INNER",
    "INNER_INNER", "ZMIN", "ZMAX",
]
