"""DNA GTR+GAMMA model state, computed by the product host math in
libexaml_hip.so (examl_host_init_gtr_dna / examl_host_make_gamma_cats —
replacing examl/models.c:3462/3795)."""

import ctypes

import numpy as np

from . import lib


def _dp(a):
    return a.ctypes.data_as(ctypes.c_void_p)


class DnaGtrModel:
    """EIGN / EV / EI / tipVector / gammaRates for one DNA GTRGAMMA
    partition (the pInfo model block, examl/axml.h:533-629)."""

    def __init__(self, frequencies, rates6, alpha):
        self.frequencies = np.ascontiguousarray(frequencies, dtype=np.float64)
        self.rates6 = np.ascontiguousarray(rates6, dtype=np.float64)
        self.alpha = float(alpha)
        assert self.frequencies.shape == (4,)
        assert self.rates6.shape == (6,)
        self.EIGN = np.zeros(4)
        self.EV = np.zeros(16)
        self.EI = np.zeros(16)
        self.tipVector = np.zeros(64)
        self.gammaRates = np.zeros(4)
        L = lib()
        L.examl_host_init_gtr_dna(_dp(self.frequencies), _dp(self.rates6),
                                  _dp(self.EIGN), _dp(self.EV), _dp(self.EI),
                                  _dp(self.tipVector))
        L.examl_host_make_gamma_cats(self.alpha, _dp(self.gammaRates), 4)

    def set_alpha(self, alpha):
        self.alpha = float(alpha)
        lib().examl_host_make_gamma_cats(self.alpha, _dp(self.gammaRates), 4)

    @staticmethod
    def jukes_cantor(alpha=1.0):
        return DnaGtrModel([0.25] * 4, [1.0] * 6, alpha)
