/* ============================================================================
 * oracle/oracle.c — TEST INFRASTRUCTURE ONLY.
 *
 * CPU restatement of ExaML's per-site conditional-likelihood hot path
 * (the Felsenstein pruning core), used exclusively as the parity oracle for
 * the MI355X HIP kernels in examl_amd/csrc/.  Nothing in the product path
 * may import, link or call this library; only tests/, __graft_entry__.smoke()
 * and bench.py's cpu_baseline leg use it (and there only as the checker).
 *
 * Parity is PINNED: every function cites the reference implementation it
 * restates (file:line into /root/reference/), and tests/golden/ holds vectors
 * generated from the reference's own compiled kernels (oracle/_ref, see
 * Makefile) that this restatement must reproduce bit-for-bit.
 *
 * All arithmetic is fp64 and the summation ORDER deliberately mirrors the
 * reference's AVX/SSE3 pairwise (hadd) order so results are bit-identical,
 * not merely close.
 * ==========================================================================*/

#include <math.h>
#include <stdlib.h>
#include <string.h>
#include <assert.h>

/* Constants — reference: examl/axml.h:88,94,110-117 */
#define ORC_TWOTOTHE256 \
  115792089237316195423570985008687907853269984665640564039457584007913129639936.0
#define ORC_MINLIKELIHOOD (1.0 / ORC_TWOTOTHE256)
#define ORC_ZMIN 1.0E-15
#define ORC_ZMAX (1.0 - 1.0E-6)
#define ORC_MAX_TIP_EV 0.999999999

/* tipCase values — reference: examl/axml.h:302-304 */
#define ORC_TIP_TIP 0
#define ORC_TIP_INNER 1
#define ORC_INNER_INNER 2

#define EXPORT __attribute__((visibility("default")))

/* --------------------------------------------------------------------------
 * makeP — P(z) = exp(rate_c * EIGN_k * log z) * EI, column 0 == 1.
 * Restates examl/newviewGenericSpecial.c:78 (makeP), saveMem path omitted
 * (out of scope).  z1/z2 here are ALREADY log-transformed branch
 * lengths (the caller applies the zmin clamp + log, as
 * newviewGenericSpecial.c:982-983 does).
 * ------------------------------------------------------------------------*/
EXPORT void oracle_make_p(double z1, double z2, const double *rptr,
                          const double *EI, const double *EIGN,
                          int numberOfCategories, double *left, double *right,
                          int states) {
  int i, j, k;
  int statesSquare = states * states;
  double d1[64], d2[64];
  assert(states <= 64);
  for (i = 0; i < numberOfCategories; i++) {
    for (j = 1; j < states; j++) {
      d1[j] = exp(rptr[i] * (EIGN[j] * z1));
      d2[j] = exp(rptr[i] * (EIGN[j] * z2));
    }
    for (j = 0; j < states; j++) {
      left[statesSquare * i + states * j] = 1.0;
      right[statesSquare * i + states * j] = 1.0;
      for (k = 1; k < states; k++) {
        left[statesSquare * i + states * j + k] = d1[k] * EI[states * j + k];
        right[statesSquare * i + states * j + k] = d2[k] * EI[states * j + k];
      }
    }
  }
}

/* --------------------------------------------------------------------------
 * calcDiagptable — diag[c*states+l] = exp(rate_c * EIGN_l * log z), col 0 = 1.
 * Restates examl/evaluateGenericSpecial.c:80.  Takes the RAW branch length z
 * (clamp + log happen here, as in the reference).
 * ------------------------------------------------------------------------*/
EXPORT void oracle_calc_diagptable(double z, int states,
                                   int numberOfCategories, const double *rptr,
                                   const double *EIGN, double *diagptable) {
  int i, l;
  double lz = (z < ORC_ZMIN) ? log(ORC_ZMIN) : log(z);
  for (i = 0; i < numberOfCategories; i++) {
    diagptable[i * states] = 1.0;
    for (l = 1; l < states; l++)
      diagptable[i * states + l] = exp(rptr[i] * (EIGN[l] * lz));
  }
}

/* pairwise 4-sum matching _mm256_hadd_pd+permute (avxLikelihood.c:33-61):
 * (a0+a1) + (a2+a3) */
static inline double hadd4d(const double *a) {
  return (a[0] + a[1]) + (a[2] + a[3]);
}

/* --------------------------------------------------------------------------
 * newview, DNA GTRGAMMA (states=4, 4 gamma cats, span=16).
 * Restates examl/avxLikelihood.c:64 (newviewGTRGAMMA_AVX) including its
 * exact scaling rule: a site is rescaled by 2^256 iff ALL 16 span entries
 * have |x| < 2^-256 (checked on the unscaled values, avxLikelihood.c:223-242),
 * and the TIP_TIP case performs NO scaling check (avxLikelihood.c:85-157).
 * Layout: x[site*16 + cat*4 + state]; left/right[cat*16 + row*4 + col];
 * extEV[row*4 + state] (row-major eigenvector matrix, models.c:3372-3376);
 * tipVector[code*4 + state], codes 1..15.
 * ------------------------------------------------------------------------*/
EXPORT void oracle_newview_dna_gamma(int tipCase, const double *x1,
                                     const double *x2, double *x3,
                                     const double *extEV,
                                     const double *tipVector,
                                     const unsigned char *tipX1,
                                     const unsigned char *tipX2, int n,
                                     const double *left, const double *right,
                                     const int *wgt, int *scalerIncrement) {
  int i, k, l, s;
  int addScale = 0;
  /* ump[code][cat*16 + row*4 + lane] in the AVX code stores the SAME scalar
   * in all 4 lanes (hadd3 broadcast, avxLikelihood.c:89-124); we store one. */
  double umpX1[16 * 16], umpX2[16 * 16];

  switch (tipCase) {
  case ORC_TIP_TIP: {
    for (i = 1; i < 16; i++) {
      const double *tv = &tipVector[i * 4];
      for (k = 0; k < 4; k++) /* cat (j in ref) */
        for (l = 0; l < 4; l++) { /* row (k in ref) */
          double p[4];
          for (s = 0; s < 4; s++) p[s] = left[k * 16 + l * 4 + s] * tv[s];
          umpX1[i * 16 + k * 4 + l] = hadd4d(p);
          for (s = 0; s < 4; s++) p[s] = right[k * 16 + l * 4 + s] * tv[s];
          umpX2[i * 16 + k * 4 + l] = hadd4d(p);
        }
    }
    for (i = 0; i < n; i++) {
      const double *uX1 = &umpX1[16 * tipX1[i]];
      const double *uX2 = &umpX2[16 * tipX2[i]];
      for (k = 0; k < 4; k++) {
        double xv[4] = {0, 0, 0, 0};
        for (l = 0; l < 4; l++) {
          double t = uX1[k * 4 + l] * uX2[k * 4 + l];
          for (s = 0; s < 4; s++) xv[s] += t * extEV[l * 4 + s];
        }
        for (s = 0; s < 4; s++) x3[16 * i + 4 * k + s] = xv[s];
      }
      /* NO scaling in TIP_TIP (matches avxLikelihood.c:85-157) */
    }
  } break;
  case ORC_TIP_INNER: {
    for (i = 1; i < 16; i++) {
      const double *tv = &tipVector[i * 4];
      for (k = 0; k < 4; k++)
        for (l = 0; l < 4; l++) {
          double p[4];
          for (s = 0; s < 4; s++) p[s] = left[k * 16 + l * 4 + s] * tv[s];
          umpX1[i * 16 + k * 4 + l] = hadd4d(p);
        }
    }
    for (i = 0; i < n; i++) {
      const double *uX1 = &umpX1[16 * tipX1[i]];
      double xv[16];
      int scale = 1;
      for (k = 0; k < 4; k++) {
        const double *xvr = &x2[i * 16 + k * 4];
        double acc[4] = {0, 0, 0, 0};
        for (l = 0; l < 4; l++) {
          double p[4];
          for (s = 0; s < 4; s++) p[s] = xvr[s] * right[k * 16 + l * 4 + s];
          double t = uX1[k * 4 + l] * hadd4d(p);
          for (s = 0; s < 4; s++) acc[s] += t * extEV[l * 4 + s];
        }
        for (s = 0; s < 4; s++) xv[k * 4 + s] = acc[s];
        if (scale) {
          for (s = 0; s < 4; s++)
            if (!(fabs(acc[s]) < ORC_MINLIKELIHOOD)) { scale = 0; break; }
        }
      }
      if (scale) {
        for (s = 0; s < 16; s++) xv[s] *= ORC_TWOTOTHE256;
        addScale += wgt[i];
      }
      for (s = 0; s < 16; s++) x3[16 * i + s] = xv[s];
    }
  } break;
  case ORC_INNER_INNER: {
    for (i = 0; i < n; i++) {
      double xv[16];
      int scale = 1;
      for (k = 0; k < 4; k++) {
        const double *xvl = &x1[i * 16 + k * 4];
        const double *xvr = &x2[i * 16 + k * 4];
        double acc[4] = {0, 0, 0, 0};
        for (l = 0; l < 4; l++) {
          double pl[4], pr[4];
          for (s = 0; s < 4; s++) {
            pl[s] = xvl[s] * left[k * 16 + l * 4 + s];
            pr[s] = xvr[s] * right[k * 16 + l * 4 + s];
          }
          /* hadd4 (avxLikelihood.c:33) multiplies the two pairwise sums */
          double t = hadd4d(pl) * hadd4d(pr);
          for (s = 0; s < 4; s++) acc[s] += t * extEV[l * 4 + s];
        }
        for (s = 0; s < 4; s++) xv[k * 4 + s] = acc[s];
        if (scale) {
          for (s = 0; s < 4; s++)
            if (!(fabs(acc[s]) < ORC_MINLIKELIHOOD)) { scale = 0; break; }
        }
      }
      if (scale) {
        for (s = 0; s < 16; s++) xv[s] *= ORC_TWOTOTHE256;
        addScale += wgt[i];
      }
      for (s = 0; s < 16; s++) x3[16 * i + s] = xv[s];
    }
  } break;
  default:
    assert(0);
  }
  *scalerIncrement = addScale;
}

/* --------------------------------------------------------------------------
 * evaluate, DNA GTRGAMMA.  Restates examl/evaluateGenericSpecial.c:1879
 * (evaluateGTRGAMMA, SSE3): term_i = sum_{c,k} x1*x2*diag, accumulated in
 * the SSE even/odd lane split; lnL += wgt_i * log(0.25*|term_i|).
 * tipX1 == NULL means INNER_INNER at the root branch.
 * Returns the partition log likelihood WITHOUT the scaler-undo term (the
 * caller adds (gs_p+gs_q)*log(minlikelihood), evaluateGenericSpecial.c:830).
 * ------------------------------------------------------------------------*/
EXPORT double oracle_evaluate_dna_gamma(const int *wptr, const double *x1_start,
                                        const double *x2_start,
                                        const double *tipVector,
                                        const unsigned char *tipX1, int n,
                                        const double *diagptable) {
  double sum = 0.0;
  int i, j;
  if (tipX1) {
    for (i = 0; i < n; i++) {
      const double *x1 = &tipVector[4 * tipX1[i]];
      const double *x2 = &x2_start[16 * i];
      double t0 = 0.0, t1 = 0.0; /* SSE lane 0 / lane 1 accumulators */
      for (j = 0; j < 4; j++) {
        t0 += x1[0] * x2[j * 4 + 0] * diagptable[j * 4 + 0];
        t1 += x1[1] * x2[j * 4 + 1] * diagptable[j * 4 + 1];
        t0 += x1[2] * x2[j * 4 + 2] * diagptable[j * 4 + 2];
        t1 += x1[3] * x2[j * 4 + 3] * diagptable[j * 4 + 3];
      }
      sum += wptr[i] * log(0.25 * fabs(t0 + t1));
    }
  } else {
    for (i = 0; i < n; i++) {
      const double *x1 = &x1_start[16 * i];
      const double *x2 = &x2_start[16 * i];
      double t0 = 0.0, t1 = 0.0;
      for (j = 0; j < 4; j++) {
        t0 += x1[j * 4 + 0] * x2[j * 4 + 0] * diagptable[j * 4 + 0];
        t1 += x1[j * 4 + 1] * x2[j * 4 + 1] * diagptable[j * 4 + 1];
        t0 += x1[j * 4 + 2] * x2[j * 4 + 2] * diagptable[j * 4 + 2];
        t1 += x1[j * 4 + 3] * x2[j * 4 + 3] * diagptable[j * 4 + 3];
      }
      sum += wptr[i] * log(0.25 * fabs(t0 + t1));
    }
  }
  return sum;
}

/* --------------------------------------------------------------------------
 * sumGAMMA — sum[i,c,k] = x1'[i,c,k] * x2'[i,c,k] with tip expansion.
 * Restates examl/makenewzGenericSpecial.c:1798 (sumGAMMA, SSE3).
 * ------------------------------------------------------------------------*/
EXPORT void oracle_sum_dna_gamma(int tipCase, double *sumtable,
                                 const double *x1_start, const double *x2_start,
                                 const double *tipVector,
                                 const unsigned char *tipX1,
                                 const unsigned char *tipX2, int n) {
  int i, j, k;
  switch (tipCase) {
  case ORC_TIP_TIP:
    for (i = 0; i < n; i++) {
      const double *x1 = &tipVector[4 * tipX1[i]];
      const double *x2 = &tipVector[4 * tipX2[i]];
      for (j = 0; j < 4; j++)
        for (k = 0; k < 4; k++)
          sumtable[i * 16 + j * 4 + k] = x1[k] * x2[k];
    }
    break;
  case ORC_TIP_INNER:
    for (i = 0; i < n; i++) {
      const double *x1 = &tipVector[4 * tipX1[i]];
      const double *x2 = &x2_start[16 * i];
      for (j = 0; j < 4; j++)
        for (k = 0; k < 4; k++)
          sumtable[i * 16 + j * 4 + k] = x1[k] * x2[j * 4 + k];
    }
    break;
  case ORC_INNER_INNER:
    for (i = 0; i < n; i++) {
      const double *x1 = &x1_start[16 * i];
      const double *x2 = &x2_start[16 * i];
      for (j = 0; j < 4; j++)
        for (k = 0; k < 4; k++)
          sumtable[i * 16 + j * 4 + k] = x1[j * 4 + k] * x2[j * 4 + k];
    }
    break;
  default:
    assert(0);
  }
}

/* --------------------------------------------------------------------------
 * coreGTRGAMMA — per-site 1st/2nd log-likelihood derivatives wrt the
 * log branch length lz.  Restates examl/makenewzGenericSpecial.c:2309
 * (SSE3 even/odd lane accumulation preserved).
 * ------------------------------------------------------------------------*/
EXPORT void oracle_core_dna_gamma(int upper, const double *sumtable,
                                  double *ext_dlnLdlz, double *ext_d2lnLdlz2,
                                  const double *EIGN, const double *gammaRates,
                                  double lz, const int *wgt) {
  double dlnLdlz = 0.0, d2lnLdlz2 = 0.0;
  double d0[16], d1[16], d2[16];
  int i, j, l;
  for (i = 0; i < 4; i++) {
    double ki = gammaRates[i], kisqr = ki * ki;
    d0[i * 4] = 1.0;
    d1[i * 4] = 0.0;
    d2[i * 4] = 0.0;
    for (l = 1; l < 4; l++) {
      d0[i * 4 + l] = exp(EIGN[l] * ki * lz);
      d1[i * 4 + l] = EIGN[l] * ki;
      d2[i * 4 + l] = EIGN[l] * EIGN[l] * kisqr;
    }
  }
  for (i = 0; i < upper; i++) {
    const double *sum = &sumtable[i * 16];
    double a0e = 0, a0o = 0, a1e = 0, a1o = 0, a2e = 0, a2o = 0;
    for (j = 0; j < 4; j++) {
      for (l = 0; l < 4; l += 2) {
        double te = d0[j * 4 + l] * sum[j * 4 + l];
        double to = d0[j * 4 + l + 1] * sum[j * 4 + l + 1];
        a0e += te;            a0o += to;
        a1e += te * d1[j * 4 + l];     a1o += to * d1[j * 4 + l + 1];
        a2e += te * d2[j * 4 + l];     a2o += to * d2[j * 4 + l + 1];
      }
    }
    double inv_Li = 1.0 / fabs(a0e + a0o);
    double dlnLidlz = (a1e + a1o) * inv_Li;
    double d2lnLidlz2 = (a2e + a2o) * inv_Li;
    dlnLdlz += wgt[i] * dlnLidlz;
    d2lnLdlz2 += wgt[i] * (d2lnLidlz2 - dlnLidlz * dlnLidlz);
  }
  *ext_dlnLdlz = dlnLdlz;
  *ext_d2lnLdlz2 = d2lnLdlz2;
}

/* ==========================================================================
 * DNA CAT (PSR) kernels — span 4, per-site rate category cptr[i].
 * ==========================================================================*/

/* --------------------------------------------------------------------------
 * newview, DNA CAT.  Restates avxLikelihood.c:326 (newviewGTRCAT_AVX):
 * per site, P rows at left/right[cptr[i]*16]; vv = EV . ((L x1) o (R x2));
 * rescale when all 4 |vv| < 2^-256 (TIP_INNER / INNER_INNER only).
 * ------------------------------------------------------------------------*/
EXPORT void oracle_newview_dna_cat(int tipCase, const double *EV,
                                   const int *cptr, const double *x1_start,
                                   const double *x2_start, double *x3_start,
                                   const double *tipVector,
                                   const unsigned char *tipX1,
                                   const unsigned char *tipX2, int n,
                                   const double *left, const double *right,
                                   const int *wgt, int *scalerIncrement) {
  int i, l, s;
  int addScale = 0;
  for (i = 0; i < n; i++) {
    const double *le = &left[cptr[i] * 16];
    const double *ri = &right[cptr[i] * 16];
    const double *x1, *x2;
    switch (tipCase) {
    case ORC_TIP_TIP:
      x1 = &tipVector[4 * tipX1[i]];
      x2 = &tipVector[4 * tipX2[i]];
      break;
    case ORC_TIP_INNER:
      x1 = &tipVector[4 * tipX1[i]];
      x2 = &x2_start[4 * i];
      break;
    default:
      x1 = &x1_start[4 * i];
      x2 = &x2_start[4 * i];
    }
    double vv[4] = {0, 0, 0, 0};
    for (l = 0; l < 4; l++) {
      double pl[4], pr[4];
      for (s = 0; s < 4; s++) {
        pl[s] = x1[s] * le[l * 4 + s];
        pr[s] = x2[s] * ri[l * 4 + s];
      }
      const double t = hadd4d(pl) * hadd4d(pr);
      for (s = 0; s < 4; s++) vv[s] += t * EV[l * 4 + s];
    }
    if (tipCase != ORC_TIP_TIP) {
      int scale = 1;
      for (s = 0; s < 4; s++)
        if (!(fabs(vv[s]) < ORC_MINLIKELIHOOD)) { scale = 0; break; }
      if (scale) {
        for (s = 0; s < 4; s++) vv[s] *= ORC_TWOTOTHE256;
        addScale += wgt[i];
      }
    }
    for (s = 0; s < 4; s++) x3_start[4 * i + s] = vv[s];
  }
  *scalerIncrement = addScale;
}

/* evaluateGTRCAT (evaluateGenericSpecial.c:1988): per-site diag row at
 * diagptable[4*cptr[i]]; NO 0.25 factor (single category per site). */
EXPORT double oracle_evaluate_dna_cat(const int *cptr, const int *wptr,
                                      const double *x1_start,
                                      const double *x2_start,
                                      const double *tipVector,
                                      const unsigned char *tipX1, int n,
                                      const double *diagptable) {
  double sum = 0.0;
  int i;
  for (i = 0; i < n; i++) {
    const double *x1 = tipX1 ? &tipVector[4 * tipX1[i]] : &x1_start[4 * i];
    const double *x2 = &x2_start[4 * i];
    const double *d = &diagptable[4 * cptr[i]];
    const double t0 = x1[0] * x2[0] * d[0] + x1[2] * x2[2] * d[2];
    const double t1 = x1[1] * x2[1] * d[1] + x1[3] * x2[3] * d[3];
    sum += wptr[i] * log(fabs(t0 + t1));
  }
  return sum;
}

/* sumCAT (makenewzGenericSpecial.c:1850) */
EXPORT void oracle_sum_dna_cat(int tipCase, double *sumtable,
                               const double *x1_start, const double *x2_start,
                               const double *tipVector,
                               const unsigned char *tipX1,
                               const unsigned char *tipX2, int n) {
  int i, k;
  for (i = 0; i < n; i++) {
    const double *x1, *x2;
    switch (tipCase) {
    case ORC_TIP_TIP:
      x1 = &tipVector[4 * tipX1[i]];
      x2 = &tipVector[4 * tipX2[i]];
      break;
    case ORC_TIP_INNER:
      x1 = &tipVector[4 * tipX1[i]];
      x2 = &x2_start[4 * i];
      break;
    default:
      x1 = &x1_start[4 * i];
      x2 = &x2_start[4 * i];
    }
    for (k = 0; k < 4; k++) sumtable[i * 4 + k] = x1[k] * x2[k];
  }
}

/* coreGTRCAT (makenewzGenericSpecial.c:2402): per-site rate r=rptr[cptr],
 * weights wr1=r*wgt, wr2=r^2*wgt; e1=EIGN, e2=EIGN^2 (NOT rate-scaled). */
EXPORT void oracle_core_dna_cat(int upper, int numberOfCategories,
                                const double *sumtable, double *ext_dlnLdlz,
                                double *ext_d2lnLdlz2, const int *wgt,
                                const double *rptr, const double *EIGN,
                                const int *cptr, double lz) {
  double e1[4], e2[4], d[25 * 4];
  double dlnLdlz = 0.0, d2lnLdlz2 = 0.0;
  int i, l;
  e1[0] = 0.0;
  e2[0] = 0.0;
  for (l = 1; l < 4; l++) {
    e1[l] = EIGN[l];
    e2[l] = EIGN[l] * EIGN[l];
  }
  {
    const double dd1 = EIGN[1] * lz, dd2 = EIGN[2] * lz, dd3 = EIGN[3] * lz;
    for (i = 0; i < numberOfCategories; i++) {
      d[i * 4 + 0] = 1.0;
      d[i * 4 + 1] = exp(dd1 * rptr[i]);
      d[i * 4 + 2] = exp(dd2 * rptr[i]);
      d[i * 4 + 3] = exp(dd3 * rptr[i]);
    }
  }
  for (i = 0; i < upper; i++) {
    const double *s = &sumtable[4 * i];
    const double *d1 = &d[4 * cptr[i]];
    const double r = rptr[cptr[i]];
    const double wr1 = r * wgt[i], wr2 = r * r * wgt[i];
    double a0e = 0, a0o = 0, a1e = 0, a1o = 0, a2e = 0, a2o = 0;
    for (l = 0; l < 4; l += 2) {
      const double te = d1[l] * s[l];
      const double to = d1[l + 1] * s[l + 1];
      a0e += te;
      a0o += to;
      a1e += te * e1[l];
      a1o += to * e1[l + 1];
      a2e += te * e2[l];
      a2o += to * e2[l + 1];
    }
    const double inv_Li = 1.0 / fabs(a0e + a0o);
    const double dlnLidlz = (a1e + a1o) * inv_Li;
    const double d2lnLidlz2 = (a2e + a2o) * inv_Li;
    dlnLdlz += wr1 * dlnLidlz;
    d2lnLdlz2 += wr2 * (d2lnLidlz2 - dlnLidlz * dlnLidlz);
  }
  *ext_dlnLdlz = dlnLdlz;
  *ext_d2lnLdlz2 = d2lnLdlz2;
}

/* ==========================================================================
 * Protein (20-state) GTRGAMMA kernels — span 80, tip codes 1..22.
 * ==========================================================================*/

/* 20-wide dot in the AVX lane order (avxLikelihood.c:1366-1383 etc.):
 * four lane accumulators over five 4-chunks, then (t0+t1)+(t2+t3). */
static inline double dot20_avx(const double *a, const double *b) {
  double t0 = 0, t1 = 0, t2 = 0, t3 = 0;
  int c;
  for (c = 0; c < 20; c += 4) {
    t0 += a[c] * b[c];
    t1 += a[c + 1] * b[c + 1];
    t2 += a[c + 2] * b[c + 2];
    t3 += a[c + 3] * b[c + 3];
  }
  return (t0 + t1) + (t2 + t3);
}

/* --------------------------------------------------------------------------
 * newview, protein GTRGAMMA.  Restates avxLikelihood.c:1312
 * (newviewGTRGAMMAPROT_AVX): per (site, cat):
 *   u1[l] = dot20(P_L[cat,l,:], x1'), u2[l] = dot20(P_R[cat,l,:], x2')
 *   x3[s] = sum_l (u1[l]*u2[l]) * EV[l,s]   (sequential l)
 * Tip operands via 23x80 ump tables; scaling over all 80 span entries,
 * no scaling in TIP_TIP (avxLikelihood.c:1349-1456/1603-1630/1784-1797).
 * left/right layout: [cat*400 + row*20 + col]; tipVector[code*20+s].
 * ------------------------------------------------------------------------*/
EXPORT void oracle_newview_prot_gamma(int tipCase, const double *x1,
                                      const double *x2, double *x3,
                                      const double *extEV,
                                      const double *tipVector,
                                      const unsigned char *tipX1,
                                      const unsigned char *tipX2, int n,
                                      const double *left, const double *right,
                                      const int *wgt, int *scalerIncrement) {
  int i, k, l, s;
  int addScale = 0;
  static double umpX1[23 * 80], umpX2[23 * 80];

  if (tipCase != ORC_INNER_INNER) {
    for (i = 0; i < 23; i++) {
      const double *v = &tipVector[20 * i];
      for (k = 0; k < 80; k++) {
        umpX1[80 * i + k] = dot20_avx(v, &left[k * 20]);
        if (tipCase == ORC_TIP_TIP)
          umpX2[80 * i + k] = dot20_avx(v, &right[k * 20]);
      }
    }
  }

  for (i = 0; i < n; i++) {
    double xv[80];
    for (k = 0; k < 4; k++) { /* cat */
      double acc[20];
      for (s = 0; s < 20; s++) acc[s] = 0.0;
      for (l = 0; l < 20; l++) {
        double u1, u2;
        if (tipCase == ORC_TIP_TIP) {
          u1 = umpX1[80 * tipX1[i] + k * 20 + l];
          u2 = umpX2[80 * tipX2[i] + k * 20 + l];
        } else if (tipCase == ORC_TIP_INNER) {
          u1 = umpX1[80 * tipX1[i] + k * 20 + l];
          u2 = dot20_avx(&x2[80 * i + 20 * k], &right[k * 400 + l * 20]);
        } else {
          u1 = dot20_avx(&x1[80 * i + 20 * k], &left[k * 400 + l * 20]);
          u2 = dot20_avx(&x2[80 * i + 20 * k], &right[k * 400 + l * 20]);
        }
        const double t = u1 * u2;
        for (s = 0; s < 20; s++) acc[s] += t * extEV[20 * l + s];
      }
      for (s = 0; s < 20; s++) xv[k * 20 + s] = acc[s];
    }
    if (tipCase != ORC_TIP_TIP) {
      int scale = 1;
      for (l = 0; scale && l < 80; l++)
        if (!(fabs(xv[l]) < ORC_MINLIKELIHOOD)) scale = 0;
      if (scale) {
        for (l = 0; l < 80; l++) xv[l] *= ORC_TWOTOTHE256;
        addScale += wgt[i];
      }
    }
    for (l = 0; l < 80; l++) x3[80 * i + l] = xv[l];
  }
  *scalerIncrement = addScale;
}

/* --------------------------------------------------------------------------
 * evaluate, protein GTRGAMMA.  Restates evaluateGenericSpecial.c:1393
 * (evaluateGTRGAMMAPROT, SSE even/odd lane split over all (cat,state)).
 * ------------------------------------------------------------------------*/
EXPORT double oracle_evaluate_prot_gamma(const int *wptr,
                                         const double *x1_start,
                                         const double *x2_start,
                                         const double *tipVector,
                                         const unsigned char *tipX1, int n,
                                         const double *diagptable) {
  double sum = 0.0;
  int i, j, l;
  for (i = 0; i < n; i++) {
    double t0 = 0.0, t1 = 0.0;
    for (j = 0; j < 4; j++) {
      const double *d = &diagptable[j * 20];
      const double *le = tipX1 ? &tipVector[20 * tipX1[i]]
                               : &x1_start[80 * i + 20 * j];
      const double *ri = &x2_start[80 * i + 20 * j];
      for (l = 0; l < 20; l += 2) {
        t0 += le[l] * ri[l] * d[l];
        t1 += le[l + 1] * ri[l + 1] * d[l + 1];
      }
    }
    sum += wptr[i] * log(0.25 * fabs(t0 + t1));
  }
  return sum;
}

/* sumGAMMAPROT — makenewzGenericSpecial.c:2083 (pure elementwise) */
EXPORT void oracle_sum_prot_gamma(int tipCase, double *sumtable,
                                  const double *x1_start,
                                  const double *x2_start,
                                  const double *tipVector,
                                  const unsigned char *tipX1,
                                  const unsigned char *tipX2, int n) {
  int i, l, k;
  for (i = 0; i < n; i++) {
    for (l = 0; l < 4; l++) {
      const double *le, *ri;
      switch (tipCase) {
      case ORC_TIP_TIP:
        le = &tipVector[20 * tipX1[i]];
        ri = &tipVector[20 * tipX2[i]];
        break;
      case ORC_TIP_INNER:
        le = &tipVector[20 * tipX1[i]];
        ri = &x2_start[80 * i + l * 20];
        break;
      default:
        le = &x1_start[80 * i + l * 20];
        ri = &x2_start[80 * i + l * 20];
      }
      for (k = 0; k < 20; k++)
        sumtable[i * 80 + l * 20 + k] = le[k] * ri[k];
    }
  }
}

/* coreGTRGAMMAPROT — makenewzGenericSpecial.c:2581 (SSE even/odd order) */
EXPORT void oracle_core_prot_gamma(int upper, const double *sumtable,
                                   double *ext_dlnLdlz, double *ext_d2lnLdlz2,
                                   const double *EIGN,
                                   const double *gammaRates, double lz,
                                   const int *wgt) {
  double dlnLdlz = 0.0, d2lnLdlz2 = 0.0;
  double d0[80], d1[80], d2[80];
  int i, j, l;
  for (i = 0; i < 4; i++) {
    double ki = gammaRates[i], kisqr = ki * ki;
    d0[i * 20] = 1.0;
    d1[i * 20] = 0.0;
    d2[i * 20] = 0.0;
    for (l = 1; l < 20; l++) {
      d0[i * 20 + l] = exp(EIGN[l] * ki * lz);
      d1[i * 20 + l] = EIGN[l] * ki;
      d2[i * 20 + l] = EIGN[l] * EIGN[l] * kisqr;
    }
  }
  for (i = 0; i < upper; i++) {
    const double *sum = &sumtable[i * 80];
    double a0e = 0, a0o = 0, a1e = 0, a1o = 0, a2e = 0, a2o = 0;
    for (j = 0; j < 4; j++) {
      for (l = 0; l < 20; l += 2) {
        const double te = d0[j * 20 + l] * sum[j * 20 + l];
        const double to = d0[j * 20 + l + 1] * sum[j * 20 + l + 1];
        a0e += te;
        a0o += to;
        a1e += te * d1[j * 20 + l];
        a1o += to * d1[j * 20 + l + 1];
        a2e += te * d2[j * 20 + l];
        a2o += to * d2[j * 20 + l + 1];
      }
    }
    const double inv_Li = 1.0 / fabs(a0e + a0o);
    const double dlnLidlz = (a1e + a1o) * inv_Li;
    const double d2lnLidlz2 = (a2e + a2o) * inv_Li;
    dlnLdlz += wgt[i] * dlnLidlz;
    d2lnLdlz2 += wgt[i] * (d2lnLidlz2 - dlnLidlz * dlnLidlz);
  }
  *ext_dlnLdlz = dlnLdlz;
  *ext_d2lnLdlz2 = d2lnLdlz2;
}

/* ==========================================================================
 * Model preparation (host math, runs once per model-parameter change).
 * ==========================================================================*/

/* LnGamma — Pike & Hill (1966) Algorithm 291, as used at models.c:3589 */
static double orc_LnGamma(double alpha) {
  double x = alpha, f = 0.0, z, result;
  if (x < 7.0) {
    f = 1.0;
    z = alpha - 1.0;
    while ((z = z + 1.0) < 7.0) f *= z;
    x = z;
    f = -log(f);
  }
  z = 1 / (x * x);
  result = f + (x - 0.5) * log(x) - x + .918938533204673 +
           (((-.000595238095238 * z + .000793650793651) * z -
             .002777777777778) * z + .083333333333333) / x;
  return result;
}

/* IncompleteGamma — Bhattacharjee (1970) AS32, as used at models.c:3627 */
static double orc_IncompleteGamma(double x, double alpha,
                                  double ln_gamma_alpha) {
  int i;
  double p = alpha, g = ln_gamma_alpha;
  double accurate = 1e-8, overflow = 1e30;
  double factor, gin = 0, rn = 0, a = 0, b = 0, an = 0, dif = 0, term = 0,
         pn[6];
  if (x == 0) return 0;
  if (x < 0 || p <= 0) return -1;
  factor = exp(p * log(x) - x - g);
  if (!(x > 1 && x >= p)) {
    /* series expansion */
    gin = 1; term = 1; rn = p;
    do { rn++; term *= x / rn; gin += term; } while (term > accurate);
    gin *= factor / p;
    return gin;
  }
  /* continued fraction */
  a = 1 - p; b = a + x + 1; term = 0;
  pn[0] = 1; pn[1] = x; pn[2] = x + 1; pn[3] = x * b;
  gin = pn[2] / pn[3];
  for (;;) {
    a++; b += 2; term++;
    an = a * term;
    for (i = 0; i < 2; i++) pn[i + 4] = b * pn[i + 2] - an * pn[i];
    if (pn[5] != 0) {
      rn = pn[4] / pn[5];
      dif = fabs(gin - rn);
      /* NOTE: on convergence the reference keeps the PREVIOUS gin
       * (models.c:3648-3651: the l42 jump precedes the l34 update) */
      if (dif <= accurate && dif <= accurate * rn) break;
      gin = rn;
    }
    for (i = 0; i < 4; i++) pn[i] = pn[i + 2];
    if (fabs(pn[4]) >= overflow)
      for (i = 0; i < 4; i++) pn[i] /= overflow;
  }
  return 1 - factor * gin;
}

/* PointNormal — Odeh & Evans (1974) AS70, as used at models.c:3700 */
static double orc_PointNormal(double prob) {
  double a0 = -.322232431088, a1 = -1, a2 = -.342242088547,
         a3 = -.0204231210245, a4 = -.453642210148e-4, b0 = .0993484626060,
         b1 = .588581570495, b2 = .531103462366, b3 = .103537752850,
         b4 = .0038560700634;
  double y, z = 0, p = prob, p1;
  p1 = (p < 0.5 ? p : 1 - p);
  if (p1 < 1e-20) return -9999;
  y = sqrt(log(1 / (p1 * p1)));
  z = y + ((((y * a4 + a3) * y + a2) * y + a1) * y + a0) /
          ((((y * b4 + b3) * y + b2) * y + b1) * y + b0);
  return (p < 0.5 ? -z : z);
}

/* PointChi2 — Best & Roberts (1975) AS91, as used at models.c:3725 */
static double orc_PointChi2(double prob, double v) {
  double e = .5e-6, aa = .6931471805, p = prob, g;
  double xx, c, ch, a = 0, q = 0, p1 = 0, p2 = 0, t = 0, x = 0, b = 0, s1, s2,
             s3, s4, s5, s6;
  if (p < .000002 || p > .999998 || v <= 0) return -1;
  g = orc_LnGamma(v / 2);
  xx = v / 2;
  c = xx - 1;
  if (v < -1.24 * log(p)) {
    ch = pow((p * xx * exp(g + xx * aa)), 1 / xx);
    if (ch - e < 0) return ch;
  } else if (v <= .32) {
    ch = 0.4;
    a = log(1 - p);
    for (;;) {
      q = ch;
      p1 = 1 + ch * (4.67 + ch);
      p2 = ch * (6.73 + ch * (6.66 + ch));
      t = -0.5 + (4.67 + 2 * ch) / p1 - (6.73 + ch * (13.32 + 3 * ch)) / p2;
      ch -= (1 - exp(a + g + .5 * ch + c * aa) * p2 / p1) / t;
      if (fabs(q / ch - 1) - .01 <= 0) break;
    }
  } else {
    x = orc_PointNormal(p);
    p1 = 0.222222 / v;
    ch = v * pow((x * sqrt(p1) + 1 - p1), 3.0);
    if (ch > 2.2 * v + 6) ch = -2 * (log(1 - p) - c * log(.5 * ch) + g);
  }
  do {
    q = ch;
    p1 = .5 * ch;
    if ((t = orc_IncompleteGamma(p1, xx, g)) < 0.0) return -1;
    p2 = p - t;
    t = p2 * exp(xx * aa + g + p1 - c * log(ch));
    b = t / ch;
    a = 0.5 * t - b * c;
    s1 = (210 + a * (140 + a * (105 + a * (84 + a * (70 + 60 * a))))) / 420;
    s2 = (420 + a * (735 + a * (966 + a * (1141 + 1278 * a)))) / 2520;
    s3 = (210 + a * (462 + a * (707 + 932 * a))) / 2520;
    s4 = (252 + a * (672 + 1182 * a) + c * (294 + a * (889 + 1740 * a))) / 5040;
    s5 = (84 + 264 * a + c * (175 + 606 * a)) / 2520;
    s6 = (120 + c * (346 + 127 * c)) / 5040;
    ch += t * (1 + 0.5 * t * s1 -
               b * c * (s1 - b * (s2 - b * (s3 - b * (s4 - b * (s5 - b * s6))))));
  } while (fabs(q / ch - 1) > e);
  return ch;
}

/* makeGammaCats (mean rates, useMedian=FALSE default) — models.c:3795 */
EXPORT void oracle_make_gamma_cats(double alpha, double *gammaRates, int K) {
  double factor = alpha / alpha * K, lnga1, alfa = alpha, beta = alpha;
  double gammaProbs[32];
  int i;
  lnga1 = orc_LnGamma(alfa + 1);
  for (i = 0; i < K - 1; i++)
    gammaProbs[i] = orc_PointChi2((i + 1.0) / K, 2.0 * alfa) / (2.0 * beta);
  for (i = 0; i < K - 1; i++)
    gammaProbs[i] = orc_IncompleteGamma(gammaProbs[i] * beta, alfa + 1, lnga1);
  gammaRates[0] = gammaProbs[0] * factor;
  gammaRates[K - 1] = (1 - gammaProbs[K - 2]) * factor;
  for (i = 1; i < K - 1; i++)
    gammaRates[i] = (gammaProbs[i] - gammaProbs[i - 1]) * factor;
}

/* --------------------------------------------------------------------------
 * Symmetric eigensolver: Householder tridiagonalization + QL.
 * Index-for-index restatement of the reference's mytred2 / mytqli
 * (models.c:3068 / models.c:3151) so that eigenvalue ORDER, eigenvector
 * SIGNS and every intermediate rounding match bit-for-bit (the reference
 * uses a column-major working convention and the classic pythag-free
 * rotation form; a textbook tred2/tqli differs in both).  After the pair,
 * the ROWS of `a` are the eigenvectors of the input (validated against
 * oracle/_ref in tests/test_oracle_cpu.py).
 * a: n*n row-major flat array (a[i*n+j] == the reference's a[i][j]).
 * ------------------------------------------------------------------------*/
static void orc_tred2(double *a, const int n, double *d, double *e) {
  int l, k, j, i;
  double scale, hh, h, g, f;
  for (i = n; i > 1; i--) {
    l = i - 1;
    h = 0.0;
    scale = 0.0;
    if (l > 1) {
      for (k = 1; k <= l; k++) scale += fabs(a[(k - 1) * n + (i - 1)]);
      if (scale == 0.0)
        e[i - 1] = a[(l - 1) * n + (i - 1)];
      else {
        for (k = 1; k <= l; k++) {
          a[(k - 1) * n + (i - 1)] /= scale;
          h += a[(k - 1) * n + (i - 1)] * a[(k - 1) * n + (i - 1)];
        }
        f = a[(l - 1) * n + (i - 1)];
        g = ((f > 0) ? -sqrt(h) : sqrt(h));
        e[i - 1] = scale * g;
        h -= f * g;
        a[(l - 1) * n + (i - 1)] = f - g;
        f = 0.0;
        for (j = 1; j <= l; j++) {
          a[(i - 1) * n + (j - 1)] = a[(j - 1) * n + (i - 1)] / h;
          g = 0.0;
          for (k = 1; k <= j; k++)
            g += a[(k - 1) * n + (j - 1)] * a[(k - 1) * n + (i - 1)];
          for (k = j + 1; k <= l; k++)
            g += a[(j - 1) * n + (k - 1)] * a[(k - 1) * n + (i - 1)];
          e[j - 1] = g / h;
          f += e[j - 1] * a[(j - 1) * n + (i - 1)];
        }
        hh = f / (h + h);
        for (j = 1; j <= l; j++) {
          f = a[(j - 1) * n + (i - 1)];
          g = e[j - 1] - hh * f;
          e[j - 1] = g;
          for (k = 1; k <= j; k++)
            a[(k - 1) * n + (j - 1)] -=
                (f * e[k - 1] + g * a[(k - 1) * n + (i - 1)]);
        }
      }
    } else
      e[i - 1] = a[(l - 1) * n + (i - 1)];
    d[i - 1] = h;
  }
  d[0] = 0.0;
  e[0] = 0.0;
  for (i = 1; i <= n; i++) {
    l = i - 1;
    if (d[i - 1] != 0.0) {
      for (j = 1; j <= l; j++) {
        g = 0.0;
        for (k = 1; k <= l; k++)
          g += a[(k - 1) * n + (i - 1)] * a[(j - 1) * n + (k - 1)];
        for (k = 1; k <= l; k++)
          a[(j - 1) * n + (k - 1)] -= g * a[(i - 1) * n + (k - 1)];
      }
    }
    d[i - 1] = a[(i - 1) * n + (i - 1)];
    a[(i - 1) * n + (i - 1)] = 1.0;
    for (j = 1; j <= l; j++)
      a[(i - 1) * n + (j - 1)] = a[(j - 1) * n + (i - 1)] = 0.0;
  }
}

static void orc_tqli(double *d, double *e, const int n, double *z) {
  int m, l, iter, i, k;
  double s, r, p, g, f, dd, c, b;
  for (i = 2; i <= n; i++) e[i - 2] = e[i - 1];
  e[n - 1] = 0.0;
  for (l = 1; l <= n; l++) {
    iter = 0;
    do {
      for (m = l; m <= n - 1; m++) {
        dd = fabs(d[m - 1]) + fabs(d[m]);
        if (fabs(e[m - 1]) + dd == dd) break;
      }
      if (m != l) {
        assert(iter < 30);
        iter++;
        g = (d[l] - d[l - 1]) / (2.0 * e[l - 1]);
        r = sqrt((g * g) + 1.0);
        g = d[m - 1] - d[l - 1] + e[l - 1] / (g + ((g < 0) ? -fabs(r) : fabs(r)));
        s = c = 1.0;
        p = 0.0;
        for (i = m - 1; i >= l; i--) {
          f = s * e[i - 1];
          b = c * e[i - 1];
          if (fabs(f) >= fabs(g)) {
            c = g / f;
            r = sqrt((c * c) + 1.0);
            e[i] = f * r;
            c *= (s = 1.0 / r);
          } else {
            s = f / g;
            r = sqrt((s * s) + 1.0);
            e[i] = g * r;
            s *= (c = 1.0 / r);
          }
          g = d[i] - p;
          r = (d[i - 1] - g) * s + 2.0 * c * b;
          p = s * r;
          d[i] = g + p;
          g = c * r - b;
          for (k = 1; k <= n; k++) {
            f = z[i * n + (k - 1)];
            z[i * n + (k - 1)] = s * z[(i - 1) * n + (k - 1)] + c * f;
            z[(i - 1) * n + (k - 1)] = c * z[(i - 1) * n + (k - 1)] - s * f;
          }
        }
        d[l - 1] = d[l - 1] - p;
        e[l - 1] = g;
        e[m - 1] = 0.0;
      }
    } while (m != l);
  }
}

/* --------------------------------------------------------------------------
 * GTR model initialization: symmetrized rate matrix -> eigendecomposition ->
 * EIGN / EV / EI / tipVector.  Restates examl/models.c:3234 (initGeneric):
 * a[i][j] = rate_ij*sqrt(f_i f_j), a[i][i] = -sum_j rate_ij f_j;
 * fracchange normalization; the near-zero eigenvalue moved to index 0 with
 * its eigenvector normalized to sum 1 (models.c:3336-3352); EIGN negated and
 * scaled by 1/fracchange; EV[i*n+j] = EIGV[i][j]; EI row scaling by
 * invfreq (models.c:3381-3388); tipVector = per-ambiguity-code sums of
 * eigenvector rows clamped to MAX_TIP_EV (models.c:3410-3436).
 *
 * valueVector: length vlen ambiguity-code bitmasks (identity 0..15 for DNA,
 * per getBitVector).  rates: upper-triangle initial rates (6 for DNA).
 * ------------------------------------------------------------------------*/
EXPORT void oracle_init_gtr(int n, const unsigned int *valueVector, int vlen,
                            double *ext_EIGN, double *EV, double *EI,
                            const double *frequencies, const double *rates,
                            double *tipVector) {
  double a[64 * 64], d[64], e[64], EIGV[64 * 64], invfreq[64], EIGN[64];
  double fracchange = 0.0;
  double r[64 * 64];
  int i, j, k, m, l;
  assert(n <= 64);

  memset(r, 0, sizeof(double) * n * n);
  i = 0;
  for (j = 0; j < n - 1; j++)
    for (k = j + 1; k < n; k++) r[j * n + k] = rates[i++];
  for (j = 0; j < n; j++) {
    r[j * n + j] = 0.0;
    for (k = 0; k < j; k++) r[j * n + k] = r[k * n + j];
  }
  for (j = 0; j < n; j++)
    for (k = 0; k < n; k++)
      fracchange += frequencies[j] * r[j * n + k] * frequencies[k];

  memset(a, 0, sizeof(double) * n * n);
  m = 0;
  for (i = 0; i < n; i++)
    for (j = i + 1; j < n; j++) {
      double factor = rates[m++];
      a[i * n + j] = a[j * n + i] =
          factor * sqrt(frequencies[i] * frequencies[j]);
      a[i * n + i] -= factor * frequencies[j];
      a[j * n + j] -= factor * frequencies[i];
    }

  orc_tred2(a, n, d, e);
  orc_tqli(d, e, n, a);

  /* columns of a are eigenvectors; postprocess exactly as models.c:3327+ */
  for (i = 0; i < n; i++)
    for (j = 0; j < n; j++) a[i * n + j] *= sqrt(frequencies[j]);

  for (i = 0; i < n; i++) {
    if (d[i] > -1e-8) {
      if (i != 0) {
        double tmp = d[i], sum = 0;
        d[i] = d[0];
        d[0] = tmp;
        for (j = 0; j < n; j++) {
          tmp = a[i * n + j];
          a[i * n + j] = a[0 * n + j];
          sum += (a[0 * n + j] = tmp);
        }
        for (j = 0; j < n; j++) a[0 * n + j] /= sum;
      }
      break;
    }
  }
  for (i = 0; i < n; i++) {
    EIGN[i] = -d[i];
    for (j = 0; j < n; j++) EIGV[i * n + j] = a[j * n + i];
    invfreq[i] = 1 / EIGV[i * n + 0];
  }
  ext_EIGN[0] = 0.0;
  for (l = 1; l < n; l++) {
    ext_EIGN[l] = EIGN[l] * (1.0 / fracchange);
    assert(ext_EIGN[l] > 0.0);
  }
  for (i = 0; i < n; i++)
    for (j = 0; j < n; j++) EV[i * n + j] = EIGV[i * n + j];
  for (i = 0; i < n; i++)
    for (j = 0; j < n; j++)
      EI[i * n + j] = (j == 0) ? 1.0 : EV[i * n + j] * invfreq[i];

  for (i = 0; i < vlen; i++) {
    unsigned int value = valueVector[i];
    for (j = 0; j < n; j++) tipVector[i * n + j] = 0;
    if (value > 0)
      for (j = 0; j < n; j++)
        if ((value >> j) & 1)
          for (l = 0; l < n; l++) tipVector[i * n + l] += EIGV[j * n + l];
  }
  for (i = 0; i < vlen; i++)
    for (j = 0; j < n; j++)
      if (tipVector[i * n + j] > ORC_MAX_TIP_EV)
        tipVector[i * n + j] = ORC_MAX_TIP_EV;
}

/* ==========================================================================
 * Protein LG4 (LG4M/LG4X) kernels — one matrix per gamma category
 * (Le/Dang/Gascuel 2012).  The AVX build uses newviewGTRGAMMAPROT_AVX_LG4
 * (avxLikelihood.c:814, 4-lane hadd3 order) for newview and the
 * __SIM_SSE3 generic kernels for the rest: evaluateGTRGAMMAPROT_LG4
 * (evaluateGenericSpecial.c:1164), sumGAMMAPROT_LG4
 * (makenewzGenericSpecial.c:1999), coreGTRGAMMAPROT_LG4
 * (makenewzGenericSpecial.c:2489), makeP_FlexLG4
 * (newviewGenericSpecial.c:170), calcDiagptableFlex_LG4
 * (evaluateGenericSpecial.c:122).  SSE 2-lane even/odd accumulation with
 * a final hadd; per-category buffers passed with strides 460 (tipVector),
 * 400 (EV/EI), 20 (EIGN).
 * ==========================================================================*/

/* 20-dot in the SSE even/odd order: acc_e/acc_o over l += 2, then sum */
static inline double dot20_sse(const double *a, const double *b) {
  double e = 0.0, o = 0.0;
  int l;
  for (l = 0; l < 20; l += 2) {
    e += a[l] * b[l];
    o += a[l + 1] * b[l + 1];
  }
  return e + o;
}

EXPORT void oracle_make_p_lg4(double z1, double z2, const double *rptr,
                              const double *EI4, const double *EIGN4,
                              int numCats, double *left, double *right) {
  int i, j, k;
  double d1[20], d2[20];
  for (i = 0; i < numCats; i++) {
    const double *EI = EI4 + i * 400;
    const double *EIGN = EIGN4 + i * 20;
    for (j = 1; j < 20; j++) {
      d1[j] = exp(rptr[i] * EIGN[j] * z1);
      d2[j] = exp(rptr[i] * EIGN[j] * z2);
    }
    for (j = 0; j < 20; j++) {
      left[400 * i + 20 * j] = 1.0;
      right[400 * i + 20 * j] = 1.0;
      for (k = 1; k < 20; k++) {
        left[400 * i + 20 * j + k] = d1[k] * EI[20 * j + k];
        right[400 * i + 20 * j + k] = d2[k] * EI[20 * j + k];
      }
    }
  }
}

EXPORT void oracle_calc_diagptable_lg4(double z, const double *rptr,
                                       const double *EIGN4, double *diag) {
  /* takes the RAW branch length; the zmin clamp + log happen here
   * (evaluateGenericSpecial.c:133-136) */
  int i, l;
  const double lz = (z < ORC_ZMIN) ? log(ORC_ZMIN) : log(z);
  for (i = 0; i < 4; i++) {
    const double ki = rptr[i];
    diag[i * 20] = 1.0;
    for (l = 1; l < 20; l++)
      diag[i * 20 + l] = exp(rptr[i] * EIGN4[i * 20 + l] * lz);
  }
}

EXPORT void oracle_newview_prot_lg4(int tipCase, const double *x1,
                                    const double *x2, double *x3,
                                    const double *EV4, const double *tv4,
                                    const unsigned char *tipX1,
                                    const unsigned char *tipX2, int n,
                                    const double *left, const double *right,
                                    const int *wgt, int *scalerIncrement) {
  int i, k, l, s;
  int addScale = 0;
  static double umpX1[23 * 80], umpX2[23 * 80];

  if (tipCase != ORC_INNER_INNER) {
    /* tip precompute: tipVector[k/20] (per-category) rows */
    for (i = 0; i < 23; i++) {
      for (k = 0; k < 80; k++) {
        const double *v = &tv4[(k / 20) * 460 + 20 * i];
        umpX1[80 * i + k] = dot20_avx(v, &left[k * 20]);
        if (tipCase == ORC_TIP_TIP)
          umpX2[80 * i + k] = dot20_avx(v, &right[k * 20]);
      }
    }
  }

  for (i = 0; i < n; i++) {
    double xv[80];
    for (k = 0; k < 4; k++) {
      const double *EV = EV4 + k * 400;
      double acc[20];
      for (s = 0; s < 20; s++) acc[s] = 0.0;
      for (l = 0; l < 20; l++) {
        double u1, u2;
        if (tipCase == ORC_TIP_TIP) {
          u1 = umpX1[80 * tipX1[i] + k * 20 + l];
          u2 = umpX2[80 * tipX2[i] + k * 20 + l];
        } else if (tipCase == ORC_TIP_INNER) {
          u1 = umpX1[80 * tipX1[i] + k * 20 + l];
          u2 = dot20_avx(&x2[80 * i + 20 * k], &right[k * 400 + l * 20]);
        } else {
          u1 = dot20_avx(&x1[80 * i + 20 * k], &left[k * 400 + l * 20]);
          u2 = dot20_avx(&x2[80 * i + 20 * k], &right[k * 400 + l * 20]);
        }
        const double t = u1 * u2;
        for (s = 0; s < 20; s++) acc[s] += t * EV[20 * l + s];
      }
      for (s = 0; s < 20; s++) xv[k * 20 + s] = acc[s];
    }
    if (tipCase != ORC_TIP_TIP) {
      int scale = 1;
      for (l = 0; scale && l < 80; l++)
        if (!(fabs(xv[l]) < ORC_MINLIKELIHOOD)) scale = 0;
      if (scale) {
        for (l = 0; l < 80; l++) xv[l] *= ORC_TWOTOTHE256;
        addScale += wgt[i];
      }
    }
    for (l = 0; l < 80; l++) x3[80 * i + l] = xv[l];
  }
  *scalerIncrement = addScale;
}

EXPORT double oracle_evaluate_prot_lg4(const int *wptr, const double *x1_start,
                                       const double *x2_start,
                                       const double *tv4,
                                       const unsigned char *tipX1, int n,
                                       const double *diagptable,
                                       const double *weights) {
  double sum = 0.0;
  int i, j, l;
  for (i = 0; i < n; i++) {
    /* tv 2-lane accumulator across categories: per category a weighted
     * even/odd partial pair, final hadd */
    double tv_e = 0.0, tv_o = 0.0;
    for (j = 0; j < 4; j++) {
      const double *d = &diagptable[j * 20];
      const double *le = tipX1 ? &tv4[j * 460 + 20 * tipX1[i]]
                               : &x1_start[80 * i + 20 * j];
      const double *ri = &x2_start[80 * i + 20 * j];
      double t_e = 0.0, t_o = 0.0;
      for (l = 0; l < 20; l += 2) {
        t_e += le[l] * ri[l] * d[l];
        t_o += le[l + 1] * ri[l + 1] * d[l + 1];
      }
      tv_e += weights[j] * t_e;
      tv_o += weights[j] * t_o;
    }
    sum += wptr[i] * log(fabs(tv_e + tv_o));
  }
  return sum;
}

EXPORT void oracle_sum_prot_lg4(int tipCase, double *sumtable,
                                const double *x1_start, const double *x2_start,
                                const double *tv4, const unsigned char *tipX1,
                                const unsigned char *tipX2, int n) {
  int i, l, k;
  for (i = 0; i < n; i++) {
    for (l = 0; l < 4; l++) {
      const double *le, *ri;
      switch (tipCase) {
      case ORC_TIP_TIP:
        le = &tv4[l * 460 + 20 * tipX1[i]];
        ri = &tv4[l * 460 + 20 * tipX2[i]];
        break;
      case ORC_TIP_INNER:
        le = &tv4[l * 460 + 20 * tipX1[i]];
        ri = &x2_start[80 * i + l * 20];
        break;
      default:
        le = &x1_start[80 * i + l * 20];
        ri = &x2_start[80 * i + l * 20];
      }
      for (k = 0; k < 20; k++)
        sumtable[i * 80 + l * 20 + k] = le[k] * ri[k];
    }
  }
}

EXPORT void oracle_core_prot_lg4(int upper, const double *sumtable,
                                 double *ext_dlnLdlz, double *ext_d2lnLdlz2,
                                 const double *EIGN4,
                                 const double *gammaRates,
                                 const double *weights, double lz,
                                 const int *wgt) {
  double dlnLdlz = 0.0, d2lnLdlz2 = 0.0;
  double d0[80], d1[80], d2[80];
  int i, j, l;
  for (i = 0; i < 4; i++) {
    const double ki = gammaRates[i], kisqr = ki * ki;
    const double *EIGN = EIGN4 + i * 20;
    d0[i * 20] = 1.0;
    d1[i * 20] = 0.0;
    d2[i * 20] = 0.0;
    for (l = 1; l < 20; l++) {
      d0[i * 20 + l] = exp(EIGN[l] * ki * lz);
      d1[i * 20 + l] = EIGN[l] * ki;
      d2[i * 20 + l] = EIGN[l] * EIGN[l] * kisqr;
    }
  }
  for (i = 0; i < upper; i++) {
    const double *sum = &sumtable[i * 80];
    /* per-category 2-lane accumulators hadd'ed per category, then
     * weighted into the per-site terms (:2528-2563) */
    double inv_Li = 0.0, dlnLidlz = 0.0, d2lnLidlz2 = 0.0;
    for (j = 0; j < 4; j++) {
      double a0e = 0, a0o = 0, a1e = 0, a1o = 0, a2e = 0, a2o = 0;
      for (l = 0; l < 20; l += 2) {
        const double te = d0[j * 20 + l] * sum[j * 20 + l];
        const double to = d0[j * 20 + l + 1] * sum[j * 20 + l + 1];
        a0e += te;
        a0o += to;
        a1e += te * d1[j * 20 + l];
        a1o += to * d1[j * 20 + l + 1];
        a2e += te * d2[j * 20 + l];
        a2o += to * d2[j * 20 + l + 1];
      }
      inv_Li += weights[j] * (a0e + a0o);
      dlnLidlz += weights[j] * (a1e + a1o);
      d2lnLidlz2 += weights[j] * (a2e + a2o);
    }
    inv_Li = 1.0 / fabs(inv_Li);
    dlnLidlz *= inv_Li;
    d2lnLidlz2 *= inv_Li;
    dlnLdlz += wgt[i] * dlnLidlz;
    d2lnLdlz2 += wgt[i] * (d2lnLidlz2 - dlnLidlz * dlnLidlz);
  }
  *ext_dlnLdlz = dlnLdlz;
  *ext_d2lnLdlz2 = d2lnLdlz2;
}

/* ==========================================================================
 * -S (saveMemory / SEV) DNA GTRGAMMA kernels: gap-bit-compacted CLVs plus a
 * per-node "gap column" holding the CLV of an all-undetermined site.
 * Restate newviewGTRGAMMA_AVX_GAPPED_SAVE (avxLikelihood.c:1806),
 * evaluateGTRGAMMA_GAPPED_SAVE (evaluateGenericSpecial.c:1750) and
 * sumGAMMA_GAPPED_SAVE (makenewzGenericSpecial.c:1716).
 * gap vectors: u32 words, bit (site%32) of word (site/32); compacted CLV
 * arrays hold only no-gap sites in site order.
 * ==========================================================================*/

static inline int orc_is_gap(const unsigned int *g, int i) {
  return (g[i / 32] >> (i % 32)) & 1u;
}

/* per-(site,cat) core of the AVX DNA newview, shared by the SAVE variant */
static void orc_nv_dna_site(int tipCase, const double *uX1, const double *uX2,
                            const double *xvl, const double *xvr,
                            const double *left, const double *right,
                            const double *extEV, double *xv, int *scale) {
  int k, l, s;
  *scale = 1;
  for (k = 0; k < 4; k++) {
    double acc[4] = {0, 0, 0, 0};
    for (l = 0; l < 4; l++) {
      double t;
      if (tipCase == ORC_TIP_TIP) {
        t = uX1[k * 4 + l] * uX2[k * 4 + l];
      } else if (tipCase == ORC_TIP_INNER) {
        double p[4];
        for (s = 0; s < 4; s++)
          p[s] = xvr[k * 4 + s] * right[k * 16 + l * 4 + s];
        t = uX1[k * 4 + l] * hadd4d(p);
      } else {
        double pl[4], pr[4];
        for (s = 0; s < 4; s++) {
          pl[s] = xvl[k * 4 + s] * left[k * 16 + l * 4 + s];
          pr[s] = xvr[k * 4 + s] * right[k * 16 + l * 4 + s];
        }
        t = hadd4d(pl) * hadd4d(pr);
      }
      for (s = 0; s < 4; s++) acc[s] += t * extEV[l * 4 + s];
    }
    for (s = 0; s < 4; s++) xv[k * 4 + s] = acc[s];
    if (*scale) {
      for (s = 0; s < 4; s++)
        if (!(fabs(acc[s]) < ORC_MINLIKELIHOOD)) { *scale = 0; break; }
    }
  }
}

EXPORT void oracle_newview_dna_gamma_save(
    int tipCase, const double *x1, const double *x2, double *x3,
    const double *extEV, const double *tipVector,
    const unsigned char *tipX1, const unsigned char *tipX2, int n,
    const double *left, const double *right, const int *wgt,
    int *scalerIncrement, const unsigned int *x1_gap,
    const unsigned int *x2_gap, unsigned int *x3_gap,
    const double *x1_gapColumn, const double *x2_gapColumn,
    double *x3_gapColumn) {
  int i, k, l, s;
  int addScale = 0, scale, scaleGap = 0;
  double umpX1[16 * 16], umpX2[16 * 16];
  const double *x1_ptr = x1, *x2_ptr = x2;
  double *x3_ptr = x3;

  if (tipCase != ORC_INNER_INNER) {
    for (i = 1; i < 16; i++) {
      const double *tv = &tipVector[i * 4];
      for (k = 0; k < 4; k++)
        for (l = 0; l < 4; l++) {
          double p[4];
          for (s = 0; s < 4; s++) p[s] = left[k * 16 + l * 4 + s] * tv[s];
          umpX1[i * 16 + k * 4 + l] = hadd4d(p);
          if (tipCase == ORC_TIP_TIP) {
            for (s = 0; s < 4; s++) p[s] = right[k * 16 + l * 4 + s] * tv[s];
            umpX2[i * 16 + k * 4 + l] = hadd4d(p);
          }
        }
    }
  }

  /* gap column first (scaleGap; TT never scales, avx:1879-1908) */
  {
    double xv[16];
    if (tipCase == ORC_TIP_TIP) {
      orc_nv_dna_site(ORC_TIP_TIP, &umpX1[240], &umpX2[240], NULL, NULL,
                      left, right, extEV, xv, &scale);
      scaleGap = 0;
    } else if (tipCase == ORC_TIP_INNER) {
      orc_nv_dna_site(ORC_TIP_INNER, &umpX1[240], NULL, NULL, x2_gapColumn,
                      left, right, extEV, xv, &scaleGap);
    } else {
      orc_nv_dna_site(ORC_INNER_INNER, NULL, NULL, x1_gapColumn,
                      x2_gapColumn, left, right, extEV, xv, &scaleGap);
    }
    if (scaleGap)
      for (s = 0; s < 16; s++) xv[s] *= ORC_TWOTOTHE256;
    for (s = 0; s < 16; s++) x3_gapColumn[s] = xv[s];
  }

  for (i = 0; i < n; i++) {
    if (tipCase != ORC_TIP_TIP && orc_is_gap(x3_gap, i)) {
      if (scaleGap) addScale += wgt[i];
      continue;
    }
    if (tipCase == ORC_TIP_TIP && orc_is_gap(x3_gap, i))
      continue; /* TT gap sites: nothing stored, never scaled */
    {
      double xv[16];
      const double *xl = NULL, *xr = NULL;
      const double *uX1 = NULL, *uX2 = NULL;
      if (tipCase == ORC_TIP_TIP) {
        uX1 = &umpX1[16 * tipX1[i]];
        uX2 = &umpX2[16 * tipX2[i]];
      } else if (tipCase == ORC_TIP_INNER) {
        uX1 = &umpX1[16 * tipX1[i]];
        if (orc_is_gap(x2_gap, i))
          xr = x2_gapColumn;
        else {
          xr = x2_ptr;
          x2_ptr += 16;
        }
      } else {
        if (orc_is_gap(x1_gap, i))
          xl = x1_gapColumn;
        else {
          xl = x1_ptr;
          x1_ptr += 16;
        }
        if (orc_is_gap(x2_gap, i))
          xr = x2_gapColumn;
        else {
          xr = x2_ptr;
          x2_ptr += 16;
        }
      }
      orc_nv_dna_site(tipCase, uX1, uX2, xl, xr, left, right, extEV, xv,
                      &scale);
      if (tipCase != ORC_TIP_TIP && scale) {
        for (s = 0; s < 16; s++) xv[s] *= ORC_TWOTOTHE256;
        addScale += wgt[i];
      }
      for (s = 0; s < 16; s++) x3_ptr[s] = xv[s];
      x3_ptr += 16;
    }
  }
  *scalerIncrement = addScale;
}

EXPORT double oracle_evaluate_dna_gamma_save(
    const int *wptr, const double *x1_start, const double *x2_start,
    const double *tipVector, const unsigned char *tipX1, int n,
    const double *diagptable, const double *x1_gapColumn,
    const double *x2_gapColumn, const unsigned int *x1_gap,
    const unsigned int *x2_gap) {
  double sum = 0.0;
  int i, j;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  for (i = 0; i < n; i++) {
    const double *x1, *x2;
    if (tipX1) {
      x1 = &tipVector[4 * tipX1[i]];
    } else if (orc_is_gap(x1_gap, i)) {
      x1 = x1_gapColumn;
    } else {
      x1 = x1_ptr;
      x1_ptr += 16;
    }
    if (orc_is_gap(x2_gap, i)) {
      x2 = x2_gapColumn;
    } else {
      x2 = x2_ptr;
      x2_ptr += 16;
    }
    double t0 = 0.0, t1 = 0.0;
    for (j = 0; j < 4; j++) {
      const double *l = tipX1 ? x1 : &x1[j * 4];
      t0 += l[0] * x2[j * 4 + 0] * diagptable[j * 4 + 0];
      t1 += l[1] * x2[j * 4 + 1] * diagptable[j * 4 + 1];
      t0 += l[2] * x2[j * 4 + 2] * diagptable[j * 4 + 2];
      t1 += l[3] * x2[j * 4 + 3] * diagptable[j * 4 + 3];
    }
    sum += wptr[i] * log(0.25 * fabs(t0 + t1));
  }
  return sum;
}

EXPORT void oracle_sum_dna_gamma_save(
    int tipCase, double *sumtable, const double *x1_start,
    const double *x2_start, const double *tipVector,
    const unsigned char *tipX1, const unsigned char *tipX2, int n,
    const double *x1_gapColumn, const double *x2_gapColumn,
    const unsigned int *x1_gap, const unsigned int *x2_gap) {
  int i, j, k;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  for (i = 0; i < n; i++) {
    const double *x1 = NULL, *x2 = NULL;
    switch (tipCase) {
    case ORC_TIP_TIP:
      x1 = &tipVector[4 * tipX1[i]];
      x2 = &tipVector[4 * tipX2[i]];
      break;
    case ORC_TIP_INNER:
      x1 = &tipVector[4 * tipX1[i]];
      if (orc_is_gap(x2_gap, i))
        x2 = x2_gapColumn;
      else {
        x2 = x2_ptr;
        x2_ptr += 16;
      }
      break;
    default:
      if (orc_is_gap(x1_gap, i))
        x1 = x1_gapColumn;
      else {
        x1 = x1_ptr;
        x1_ptr += 16;
      }
      if (orc_is_gap(x2_gap, i))
        x2 = x2_gapColumn;
      else {
        x2 = x2_ptr;
        x2_ptr += 16;
      }
    }
    for (j = 0; j < 4; j++)
      for (k = 0; k < 4; k++)
        sumtable[i * 16 + j * 4 + k] =
            (tipCase == ORC_INNER_INNER ? x1[j * 4 + k] : x1[k]) *
            (tipCase == ORC_TIP_TIP ? x2[k] : x2[j * 4 + k]);
  }
}

/* ==========================================================================
 * Protein CAT (PSR) kernels — span 20, per-site rate category.
 * Restate newviewGTRCATPROT_AVX (avxLikelihood.c:487, 4-lane dot order),
 * evaluateGTRCATPROT (evaluateGenericSpecial.c:1464, SSE even/odd,
 * log|term| without 0.25), sumGTRCATPROT (makenewzGenericSpecial.c:2156)
 * and coreGTRCATPROT (:2659, wr1/wr2-weighted derivatives).
 * left/right: [cat*400 + row*20 + col] from the generic makeP.
 * ==========================================================================*/

EXPORT void oracle_newview_prot_cat(int tipCase, const double *extEV,
                                    const int *cptr, const double *x1_start,
                                    const double *x2_start, double *x3_start,
                                    const double *tipVector,
                                    const unsigned char *tipX1,
                                    const unsigned char *tipX2, int n,
                                    const double *left, const double *right,
                                    const int *wgt, int *scalerIncrement) {
  int i, l, s;
  int addScale = 0;
  for (i = 0; i < n; i++) {
    const double *le = &left[cptr[i] * 400];
    const double *ri = &right[cptr[i] * 400];
    const double *vl, *vr;
    double xv[20];
    int scale;
    if (tipCase == ORC_TIP_TIP) {
      vl = &tipVector[20 * tipX1[i]];
      vr = &tipVector[20 * tipX2[i]];
    } else if (tipCase == ORC_TIP_INNER) {
      vl = &tipVector[20 * tipX1[i]];
      vr = &x2_start[20 * i];
    } else {
      vl = &x1_start[20 * i];
      vr = &x2_start[20 * i];
    }
    for (s = 0; s < 20; s++) xv[s] = 0.0;
    for (l = 0; l < 20; l++) {
      const double t =
          dot20_avx(vl, &le[l * 20]) * dot20_avx(vr, &ri[l * 20]);
      for (s = 0; s < 20; s++) xv[s] += t * extEV[l * 20 + s];
    }
    if (tipCase != ORC_TIP_TIP) {
      scale = 1;
      for (s = 0; scale && s < 20; s++)
        scale = (fabs(xv[s]) < ORC_MINLIKELIHOOD);
      if (scale) {
        for (s = 0; s < 20; s++) xv[s] *= ORC_TWOTOTHE256;
        addScale += wgt[i];
      }
    }
    for (s = 0; s < 20; s++) x3_start[20 * i + s] = xv[s];
  }
  *scalerIncrement = addScale;
}

EXPORT double oracle_evaluate_prot_cat(const int *cptr, const int *wptr,
                                       const double *x1, const double *x2,
                                       const double *tipVector,
                                       const unsigned char *tipX1, int n,
                                       const double *diagptable) {
  double sum = 0.0;
  int i, l;
  for (i = 0; i < n; i++) {
    const double *le = tipX1 ? &tipVector[20 * tipX1[i]] : &x1[20 * i];
    const double *ri = &x2[20 * i];
    const double *d = &diagptable[20 * cptr[i]];
    double t0 = 0.0, t1 = 0.0;
    for (l = 0; l < 20; l += 2) {
      t0 += le[l] * ri[l] * d[l];
      t1 += le[l + 1] * ri[l + 1] * d[l + 1];
    }
    sum += wptr[i] * log(fabs(t0 + t1));
  }
  return sum;
}

EXPORT void oracle_sum_prot_cat(int tipCase, double *sumtable,
                                const double *x1, const double *x2,
                                const double *tipVector,
                                const unsigned char *tipX1,
                                const unsigned char *tipX2, int n) {
  int i, l;
  for (i = 0; i < n; i++) {
    const double *le, *ri;
    switch (tipCase) {
    case ORC_TIP_TIP:
      le = &tipVector[20 * tipX1[i]];
      ri = &tipVector[20 * tipX2[i]];
      break;
    case ORC_TIP_INNER:
      le = &tipVector[20 * tipX1[i]];
      ri = &x2[20 * i];
      break;
    default:
      le = &x1[20 * i];
      ri = &x2[20 * i];
    }
    for (l = 0; l < 20; l++) sumtable[20 * i + l] = le[l] * ri[l];
  }
}

EXPORT void oracle_core_prot_cat(int upper, int numberOfCategories,
                                 const double *sumtable, const int *wgt,
                                 const double *rptr, const double *EIGN,
                                 const int *cptr, double lz,
                                 double *ext_dlnLdlz, double *ext_d2lnLdlz2) {
  double d_start[25 * 20], e[20], s_[20], dd[20];
  double dlnLdlz = 0.0, d2lnLdlz2 = 0.0;
  int i, l;
  e[0] = s_[0] = dd[0] = 0.0;
  for (l = 1; l < 20; l++) {
    e[l] = EIGN[l] * EIGN[l];
    s_[l] = EIGN[l];
    dd[l] = s_[l] * lz;
  }
  for (i = 0; i < numberOfCategories; i++) {
    d_start[20 * i] = 1.0;
    for (l = 1; l < 20; l++) d_start[20 * i + l] = exp(dd[l] * rptr[i]);
  }
  for (i = 0; i < upper; i++) {
    const double r = rptr[cptr[i]];
    const double wr1 = r * wgt[i], wr2 = r * r * wgt[i];
    const double *d = &d_start[20 * cptr[i]];
    const double *sum = &sumtable[20 * i];
    double a0e = 0, a0o = 0, a1e = 0, a1o = 0, a2e = 0, a2o = 0;
    for (l = 0; l < 20; l += 2) {
      const double te = d[l] * sum[l];
      const double to = d[l + 1] * sum[l + 1];
      a0e += te;
      a0o += to;
      a1e += te * s_[l];
      a1o += to * s_[l + 1];
      a2e += te * e[l];
      a2o += to * e[l + 1];
    }
    const double inv_Li = 1.0 / fabs(a0e + a0o);
    const double dlnLidlz = (a1e + a1o) * inv_Li;
    const double d2lnLidlz2 = (a2e + a2o) * inv_Li;
    dlnLdlz += wr1 * dlnLidlz;
    d2lnLdlz2 += wr2 * (d2lnLidlz2 - dlnLidlz * dlnLidlz);
  }
  *ext_dlnLdlz = dlnLdlz;
  *ext_d2lnLdlz2 = d2lnLdlz2;
}

/* ==========================================================================
 * -S (saveMemory) protein GTRGAMMA kernels: span-80 gap-compacted CLVs +
 * per-node gap columns; undetermined AA code 22 (gapOffset 440).
 * Restate newviewGTRGAMMAPROT_AVX_GAPPED_SAVE (avxLikelihood.c:3125),
 * evaluateGTRGAMMAPROT_GAPPED_SAVE (evaluateGenericSpecial.c:1291),
 * sumGAMMAPROT_GAPPED_SAVE (makenewzGenericSpecial.c:1896).
 * ==========================================================================*/

/* per-(site,cat) body of the dense AVX prot kernel, operand-pointer form */
static void orc_nv_prot_site(int tipCase, const double *uX1,
                             const double *uX2, const double *xl,
                             const double *xr, const double *left,
                             const double *right, const double *extEV,
                             double *xv, int *scale) {
  int k, l, s;
  *scale = 1;
  for (k = 0; k < 4; k++) {
    double acc[20];
    for (s = 0; s < 20; s++) acc[s] = 0.0;
    for (l = 0; l < 20; l++) {
      double u1, u2;
      if (tipCase == ORC_TIP_TIP) {
        u1 = uX1[k * 20 + l];
        u2 = uX2[k * 20 + l];
      } else if (tipCase == ORC_TIP_INNER) {
        u1 = uX1[k * 20 + l];
        u2 = dot20_avx(&xr[20 * k], &right[k * 400 + l * 20]);
      } else {
        u1 = dot20_avx(&xl[20 * k], &left[k * 400 + l * 20]);
        u2 = dot20_avx(&xr[20 * k], &right[k * 400 + l * 20]);
      }
      const double t = u1 * u2;
      for (s = 0; s < 20; s++) acc[s] += t * extEV[20 * l + s];
    }
    for (s = 0; s < 20; s++) xv[k * 20 + s] = acc[s];
  }
  for (s = 0; s < 80 && *scale; s++)
    if (!(fabs(xv[s]) < ORC_MINLIKELIHOOD)) *scale = 0;
}

EXPORT void oracle_newview_prot_gamma_save(
    int tipCase, const double *x1, const double *x2, double *x3,
    const double *extEV, const double *tipVector,
    const unsigned char *tipX1, const unsigned char *tipX2, int n,
    const double *left, const double *right, const int *wgt,
    int *scalerIncrement, const unsigned int *x1_gap,
    const unsigned int *x2_gap, unsigned int *x3_gap,
    const double *x1_gapColumn, const double *x2_gapColumn,
    double *x3_gapColumn) {
  int i, k, s;
  int addScale = 0, scale, scaleGap = 0;
  static double umpX1[23 * 80], umpX2[23 * 80];
  const double *x1_ptr = x1, *x2_ptr = x2;
  double *x3_ptr = x3;

  if (tipCase != ORC_INNER_INNER) {
    for (i = 0; i < 23; i++) {
      const double *v = &tipVector[20 * i];
      for (k = 0; k < 80; k++) {
        umpX1[80 * i + k] = dot20_avx(v, &left[k * 20]);
        if (tipCase == ORC_TIP_TIP)
          umpX2[80 * i + k] = dot20_avx(v, &right[k * 20]);
      }
    }
  }

  {
    double xv[80];
    if (tipCase == ORC_TIP_TIP) {
      orc_nv_prot_site(ORC_TIP_TIP, &umpX1[1760], &umpX2[1760], NULL, NULL,
                       left, right, extEV, xv, &scale);
      scaleGap = 0;
    } else if (tipCase == ORC_TIP_INNER) {
      orc_nv_prot_site(ORC_TIP_INNER, &umpX1[1760], NULL, NULL,
                       x2_gapColumn, left, right, extEV, xv, &scaleGap);
    } else {
      orc_nv_prot_site(ORC_INNER_INNER, NULL, NULL, x1_gapColumn,
                       x2_gapColumn, left, right, extEV, xv, &scaleGap);
    }
    if (scaleGap)
      for (s = 0; s < 80; s++) xv[s] *= ORC_TWOTOTHE256;
    for (s = 0; s < 80; s++) x3_gapColumn[s] = xv[s];
  }

  for (i = 0; i < n; i++) {
    if (orc_is_gap(x3_gap, i)) {
      if (tipCase != ORC_TIP_TIP && scaleGap) addScale += wgt[i];
      continue;
    }
    {
      double xv[80];
      const double *xl = NULL, *xr = NULL;
      const double *uX1 = NULL, *uX2 = NULL;
      if (tipCase == ORC_TIP_TIP) {
        uX1 = &umpX1[80 * tipX1[i]];
        uX2 = &umpX2[80 * tipX2[i]];
      } else if (tipCase == ORC_TIP_INNER) {
        uX1 = &umpX1[80 * tipX1[i]];
        if (orc_is_gap(x2_gap, i))
          xr = x2_gapColumn;
        else {
          xr = x2_ptr;
          x2_ptr += 80;
        }
      } else {
        if (orc_is_gap(x1_gap, i))
          xl = x1_gapColumn;
        else {
          xl = x1_ptr;
          x1_ptr += 80;
        }
        if (orc_is_gap(x2_gap, i))
          xr = x2_gapColumn;
        else {
          xr = x2_ptr;
          x2_ptr += 80;
        }
      }
      orc_nv_prot_site(tipCase, uX1, uX2, xl, xr, left, right, extEV, xv,
                       &scale);
      if (tipCase != ORC_TIP_TIP && scale) {
        for (s = 0; s < 80; s++) xv[s] *= ORC_TWOTOTHE256;
        addScale += wgt[i];
      }
      for (s = 0; s < 80; s++) x3_ptr[s] = xv[s];
      x3_ptr += 80;
    }
  }
  *scalerIncrement = addScale;
}

EXPORT double oracle_evaluate_prot_gamma_save(
    const int *wptr, const double *x1_start, const double *x2_start,
    const double *tipVector, const unsigned char *tipX1, int n,
    const double *diagptable, const double *x1_gapColumn,
    const double *x2_gapColumn, const unsigned int *x1_gap,
    const unsigned int *x2_gap) {
  double sum = 0.0;
  int i, j, l;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  for (i = 0; i < n; i++) {
    const double *le, *ri;
    if (tipX1) {
      le = &tipVector[20 * tipX1[i]];
    } else if (orc_is_gap(x1_gap, i)) {
      le = x1_gapColumn;
    } else {
      le = x1_ptr;
      x1_ptr += 80;
    }
    if (orc_is_gap(x2_gap, i)) {
      ri = x2_gapColumn;
    } else {
      ri = x2_ptr;
      x2_ptr += 80;
    }
    double t0 = 0.0, t1 = 0.0;
    for (j = 0; j < 4; j++) {
      const double *lrow = tipX1 ? le : &le[20 * j];
      const double *d = &diagptable[j * 20];
      const double *r = &ri[20 * j];
      for (l = 0; l < 20; l += 2) {
        t0 += lrow[l] * r[l] * d[l];
        t1 += lrow[l + 1] * r[l + 1] * d[l + 1];
      }
    }
    sum += wptr[i] * log(0.25 * fabs(t0 + t1));
  }
  return sum;
}

EXPORT void oracle_sum_prot_gamma_save(
    int tipCase, double *sumtable, const double *x1_start,
    const double *x2_start, const double *tipVector,
    const unsigned char *tipX1, const unsigned char *tipX2, int n,
    const double *x1_gapColumn, const double *x2_gapColumn,
    const unsigned int *x1_gap, const unsigned int *x2_gap) {
  int i, l, k;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  for (i = 0; i < n; i++) {
    const double *le = NULL, *ri = NULL;
    switch (tipCase) {
    case ORC_TIP_TIP:
      le = &tipVector[20 * tipX1[i]];
      ri = &tipVector[20 * tipX2[i]];
      break;
    case ORC_TIP_INNER:
      le = &tipVector[20 * tipX1[i]];
      if (orc_is_gap(x2_gap, i))
        ri = x2_gapColumn;
      else {
        ri = x2_ptr;
        x2_ptr += 80;
      }
      break;
    default:
      if (orc_is_gap(x1_gap, i))
        le = x1_gapColumn;
      else {
        le = x1_ptr;
        x1_ptr += 80;
      }
      if (orc_is_gap(x2_gap, i))
        ri = x2_gapColumn;
      else {
        ri = x2_ptr;
        x2_ptr += 80;
      }
    }
    for (l = 0; l < 4; l++)
      for (k = 0; k < 20; k++)
        sumtable[i * 80 + l * 20 + k] =
            (tipCase == ORC_INNER_INNER ? le[l * 20 + k] : le[k]) *
            (tipCase == ORC_TIP_TIP ? ri[k] : ri[l * 20 + k]);
  }
}

/* ==========================================================================
 * -S CAT kernels (DNA span 4 and protein span 20): gap-compacted CLVs with
 * per-site rate categories plus an EXTRA P-matrix pair at rate 1.0 in slot
 * maxCats for the gap column / gap operands (makeP's saveMem branch,
 * newviewGenericSpecial.c:140-165).  Restate
 * newviewGTRCAT_AVX_GAPPED_SAVE (avxLikelihood.c:2306),
 * newviewGTRCATPROT_AVX_GAPPED_SAVE (:2607), evaluateGTRCAT_SAVE
 * (evaluateGenericSpecial.c:1537-area), evaluateGTRCATPROT_SAVE (:1537),
 * sumCAT_SAVE / sumGTRCATPROT_SAVE (makenewzGenericSpecial.c:1648/...).
 * ==========================================================================*/

EXPORT void oracle_make_p_save(double z1, double z2, const double *rptr,
                               const double *EI, const double *EIGN,
                               int numCats, double *left, double *right,
                               int maxCats, int states) {
  int i, j, k;
  double d1[64], d2[64], lz1[64], lz2[64];
  for (j = 1; j < states; j++) {
    lz1[j] = EIGN[j] * z1;
    lz2[j] = EIGN[j] * z2;
  }
  const int sq = states * states;
  for (i = 0; i < numCats; i++) {
    for (j = 1; j < states; j++) {
      d1[j] = exp(rptr[i] * lz1[j]);
      d2[j] = exp(rptr[i] * lz2[j]);
    }
    for (j = 0; j < states; j++) {
      left[sq * i + states * j] = 1.0;
      right[sq * i + states * j] = 1.0;
      for (k = 1; k < states; k++) {
        left[sq * i + states * j + k] = d1[k] * EI[states * j + k];
        right[sq * i + states * j + k] = d2[k] * EI[states * j + k];
      }
    }
  }
  /* saveMem extra pair at rate 1.0 (slot maxCats) */
  i = maxCats;
  for (j = 1; j < states; j++) {
    d1[j] = exp(lz1[j]);
    d2[j] = exp(lz2[j]);
  }
  for (j = 0; j < states; j++) {
    left[sq * i + states * j] = 1.0;
    right[sq * i + states * j] = 1.0;
    for (k = 1; k < states; k++) {
      left[sq * i + states * j + k] = d1[k] * EI[states * j + k];
      right[sq * i + states * j + k] = d2[k] * EI[states * j + k];
    }
  }
}

/* per-site body shared with the dense CAT kernel: hadd4 pairwise dots */
static void orc_nv_cat_site(const double *x1, const double *x2,
                            const double *le, const double *ri,
                            const double *EV, double *xv) {
  int l, s;
  for (s = 0; s < 4; s++) xv[s] = 0.0;
  for (l = 0; l < 4; l++) {
    const double a = (x1[0] * le[l * 4] + x1[1] * le[l * 4 + 1]) +
                     (x1[2] * le[l * 4 + 2] + x1[3] * le[l * 4 + 3]);
    const double b = (x2[0] * ri[l * 4] + x2[1] * ri[l * 4 + 1]) +
                     (x2[2] * ri[l * 4 + 2] + x2[3] * ri[l * 4 + 3]);
    const double t = a * b;
    for (s = 0; s < 4; s++) xv[s] += t * EV[l * 4 + s];
  }
}

EXPORT void oracle_newview_dna_cat_save(
    int tipCase, const double *EV, const int *cptr, const double *x1_start,
    const double *x2_start, double *x3_start, const double *tipVector,
    const unsigned char *tipX1, const unsigned char *tipX2, int n,
    const double *left, const double *right, const int *wgt,
    int *scalerIncrement, const unsigned int *x1_gap,
    const unsigned int *x2_gap, const unsigned int *x3_gap,
    const double *x1_gapColumn, const double *x2_gapColumn,
    double *x3_gapColumn, int maxCats) {
  int i, s, scale;
  int addScale = 0, scaleGap = 0;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  double *x3_ptr = x3_start;

  /* gap column with the rate-1.0 P pair (avx:2332-2378) */
  {
    double xv[4];
    orc_nv_cat_site(x1_gapColumn, x2_gapColumn, &left[maxCats * 16],
                    &right[maxCats * 16], EV, xv);
    if (tipCase != ORC_TIP_TIP) {
      scale = 1;
      for (s = 0; s < 4; s++)
        if (!(fabs(xv[s]) < ORC_MINLIKELIHOOD)) { scale = 0; break; }
      if (scale) {
        for (s = 0; s < 4; s++) xv[s] *= ORC_TWOTOTHE256;
        scaleGap = 1;
      }
    }
    for (s = 0; s < 4; s++) x3_gapColumn[s] = xv[s];
  }

  for (i = 0; i < n; i++) {
    if (orc_is_gap(x3_gap, i)) {
      if (tipCase != ORC_TIP_TIP && scaleGap) addScale += wgt[i];
      continue;
    }
    {
      const double *x1, *x2, *le, *ri;
      double xv[4];
      if (tipCase == ORC_TIP_TIP) {
        x1 = &tipVector[4 * tipX1[i]];
        x2 = &tipVector[4 * tipX2[i]];
        le = orc_is_gap(x1_gap, i) ? &left[maxCats * 16]
                                   : &left[cptr[i] * 16];
        ri = orc_is_gap(x2_gap, i) ? &right[maxCats * 16]
                                   : &right[cptr[i] * 16];
      } else if (tipCase == ORC_TIP_INNER) {
        x1 = &tipVector[4 * tipX1[i]];
        le = orc_is_gap(x1_gap, i) ? &left[maxCats * 16]
                                   : &left[cptr[i] * 16];
        if (orc_is_gap(x2_gap, i)) {
          ri = &right[maxCats * 16];
          x2 = x2_gapColumn;
        } else {
          ri = &right[cptr[i] * 16];
          x2 = x2_ptr;
          x2_ptr += 4;
        }
      } else {
        if (orc_is_gap(x1_gap, i)) {
          x1 = x1_gapColumn;
          le = &left[maxCats * 16];
        } else {
          le = &left[cptr[i] * 16];
          x1 = x1_ptr;
          x1_ptr += 4;
        }
        if (orc_is_gap(x2_gap, i)) {
          x2 = x2_gapColumn;
          ri = &right[maxCats * 16];
        } else {
          ri = &right[cptr[i] * 16];
          x2 = x2_ptr;
          x2_ptr += 4;
        }
      }
      orc_nv_cat_site(x1, x2, le, ri, EV, xv);
      if (tipCase != ORC_TIP_TIP) {
        scale = 1;
        for (s = 0; s < 4; s++)
          if (!(fabs(xv[s]) < ORC_MINLIKELIHOOD)) { scale = 0; break; }
        if (scale) {
          for (s = 0; s < 4; s++) xv[s] *= ORC_TWOTOTHE256;
          addScale += wgt[i];
        }
      }
      for (s = 0; s < 4; s++) x3_ptr[s] = xv[s];
      x3_ptr += 4;
    }
  }
  *scalerIncrement = addScale;
}

EXPORT double oracle_evaluate_dna_cat_save(
    const int *cptr, const int *wptr, const double *x1_start,
    const double *x2_start, const double *tipVector,
    const unsigned char *tipX1, int n, const double *diagptable,
    const double *x1_gapColumn, const double *x2_gapColumn,
    const unsigned int *x1_gap, const unsigned int *x2_gap) {
  double sum = 0.0;
  int i;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  for (i = 0; i < n; i++) {
    const double *x1, *x2;
    if (tipX1) {
      x1 = &tipVector[4 * tipX1[i]];
    } else if (orc_is_gap(x1_gap, i)) {
      x1 = x1_gapColumn;
    } else {
      x1 = x1_ptr;
      x1_ptr += 4;
    }
    if (orc_is_gap(x2_gap, i)) {
      x2 = x2_gapColumn;
    } else {
      x2 = x2_ptr;
      x2_ptr += 4;
    }
    const double *d = &diagptable[4 * cptr[i]];
    const double t0 = x1[0] * x2[0] * d[0] + x1[2] * x2[2] * d[2];
    const double t1 = x1[1] * x2[1] * d[1] + x1[3] * x2[3] * d[3];
    sum += wptr[i] * log(fabs(t0 + t1));
  }
  return sum;
}

EXPORT void oracle_sum_dna_cat_save(
    int tipCase, double *sumtable, const double *x1_start,
    const double *x2_start, const double *tipVector,
    const unsigned char *tipX1, const unsigned char *tipX2, int n,
    const double *x1_gapColumn, const double *x2_gapColumn,
    const unsigned int *x1_gap, const unsigned int *x2_gap) {
  int i, j;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  for (i = 0; i < n; i++) {
    const double *x1, *x2;
    switch (tipCase) {
    case ORC_TIP_TIP:
      x1 = &tipVector[4 * tipX1[i]];
      x2 = &tipVector[4 * tipX2[i]];
      break;
    case ORC_TIP_INNER:
      x1 = &tipVector[4 * tipX1[i]];
      if (orc_is_gap(x2_gap, i))
        x2 = x2_gapColumn;
      else {
        x2 = x2_ptr;
        x2_ptr += 4;
      }
      break;
    default:
      if (orc_is_gap(x1_gap, i)) {
        x1 = x1_gapColumn;
      } else {
        x1 = x1_ptr;
        x1_ptr += 4;
      }
      if (orc_is_gap(x2_gap, i)) {
        x2 = x2_gapColumn;
      } else {
        x2 = x2_ptr;
        x2_ptr += 4;
      }
    }
    for (j = 0; j < 4; j++) sumtable[i * 4 + j] = x1[j] * x2[j];
  }
}

/* ==========================================================================
 * -S protein PSR (CAT) GAPPED_SAVE kernels: span-20 compacted CLVs + gap
 * columns, per-site P by rate category with the saveMem rate-1.0 pair at
 * slot maxCats.  Restate newviewGTRCATPROT_AVX_GAPPED_SAVE
 * (avxLikelihood.c:2607), evaluateGTRCATPROT_SAVE
 * (evaluateGenericSpecial.c:1537), sumGTRCATPROT_SAVE
 * (makenewzGenericSpecial.c:2218).  The AVX per-site body is
 * hadd4(dot20, dot20) per row l — the dot20_avx lane order.
 * ==========================================================================*/

static void orc_nv_prot_cat_site(const double *vl, const double *vr,
                                 const double *le, const double *ri,
                                 const double *extEV, double *xv) {
  int l, s;
  for (s = 0; s < 20; s++) xv[s] = 0.0;
  for (l = 0; l < 20; l++) {
    const double t =
        dot20_avx(vl, &le[l * 20]) * dot20_avx(vr, &ri[l * 20]);
    for (s = 0; s < 20; s++) xv[s] += t * extEV[l * 20 + s];
  }
}

EXPORT void oracle_newview_prot_cat_save(
    int tipCase, const double *extEV, const int *cptr,
    const double *x1_start, const double *x2_start, double *x3_start,
    const double *tipVector, const unsigned char *tipX1,
    const unsigned char *tipX2, int n, const double *left,
    const double *right, const int *wgt, int *scalerIncrement,
    const unsigned int *x1_gap, const unsigned int *x2_gap,
    const unsigned int *x3_gap, const double *x1_gapColumn,
    const double *x2_gapColumn, double *x3_gapColumn, int maxCats) {
  int i, s, scale;
  int addScale = 0, scaleGap = 0;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  double *x3_ptr = x3_start;

  /* gap column with the rate-1.0 P pair (avx:2636-2733) */
  {
    double xv[20];
    orc_nv_prot_cat_site(x1_gapColumn, x2_gapColumn, &left[maxCats * 400],
                         &right[maxCats * 400], extEV, xv);
    if (tipCase != ORC_TIP_TIP) {
      scale = 1;
      for (s = 0; s < 20; s++)
        if (!(fabs(xv[s]) < ORC_MINLIKELIHOOD)) { scale = 0; break; }
      if (scale) {
        for (s = 0; s < 20; s++) xv[s] *= ORC_TWOTOTHE256;
        scaleGap = 1;
      }
    }
    for (s = 0; s < 20; s++) x3_gapColumn[s] = xv[s];
  }

  for (i = 0; i < n; i++) {
    if (orc_is_gap(x3_gap, i)) {
      if (tipCase != ORC_TIP_TIP && scaleGap) addScale += wgt[i];
      continue;
    }
    {
      const double *vl, *vr, *le, *ri;
      double xv[20];
      if (tipCase == ORC_TIP_TIP) {
        vl = &tipVector[20 * tipX1[i]];
        vr = &tipVector[20 * tipX2[i]];
        le = orc_is_gap(x1_gap, i) ? &left[maxCats * 400]
                                   : &left[cptr[i] * 400];
        ri = orc_is_gap(x2_gap, i) ? &right[maxCats * 400]
                                   : &right[cptr[i] * 400];
      } else if (tipCase == ORC_TIP_INNER) {
        vl = &tipVector[20 * tipX1[i]];
        le = orc_is_gap(x1_gap, i) ? &left[maxCats * 400]
                                   : &left[cptr[i] * 400];
        if (orc_is_gap(x2_gap, i)) {
          ri = &right[maxCats * 400];
          vr = x2_gapColumn;
        } else {
          ri = &right[cptr[i] * 400];
          vr = x2_ptr;
          x2_ptr += 20;
        }
      } else {
        if (orc_is_gap(x1_gap, i)) {
          vl = x1_gapColumn;
          le = &left[maxCats * 400];
        } else {
          le = &left[cptr[i] * 400];
          vl = x1_ptr;
          x1_ptr += 20;
        }
        if (orc_is_gap(x2_gap, i)) {
          vr = x2_gapColumn;
          ri = &right[maxCats * 400];
        } else {
          ri = &right[cptr[i] * 400];
          vr = x2_ptr;
          x2_ptr += 20;
        }
      }
      orc_nv_prot_cat_site(vl, vr, le, ri, extEV, xv);
      if (tipCase != ORC_TIP_TIP) {
        scale = 1;
        for (s = 0; s < 20; s++)
          if (!(fabs(xv[s]) < ORC_MINLIKELIHOOD)) { scale = 0; break; }
        if (scale) {
          for (s = 0; s < 20; s++) xv[s] *= ORC_TWOTOTHE256;
          addScale += wgt[i];
        }
      }
      for (s = 0; s < 20; s++) x3_ptr[s] = xv[s];
      x3_ptr += 20;
    }
  }
  *scalerIncrement = addScale;
}

EXPORT double oracle_evaluate_prot_cat_save(
    const int *cptr, const int *wptr, const double *x1_start,
    const double *x2_start, const double *tipVector,
    const unsigned char *tipX1, int n, const double *diagptable,
    const double *x1_gapColumn, const double *x2_gapColumn,
    const unsigned int *x1_gap, const unsigned int *x2_gap) {
  double sum = 0.0;
  int i, l;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  for (i = 0; i < n; i++) {
    const double *le, *ri;
    if (tipX1) {
      le = &tipVector[20 * tipX1[i]];
    } else if (orc_is_gap(x1_gap, i)) {
      le = x1_gapColumn;
    } else {
      le = x1_ptr;
      x1_ptr += 20;
    }
    if (orc_is_gap(x2_gap, i)) {
      ri = x2_gapColumn;
    } else {
      ri = x2_ptr;
      x2_ptr += 20;
    }
    const double *d = &diagptable[20 * cptr[i]];
    double t0 = 0.0, t1 = 0.0;
    for (l = 0; l < 20; l += 2) {
      t0 += le[l] * ri[l] * d[l];
      t1 += le[l + 1] * ri[l + 1] * d[l + 1];
    }
    sum += wptr[i] * log(fabs(t0 + t1));
  }
  return sum;
}

EXPORT void oracle_sum_prot_cat_save(
    int tipCase, double *sumtable, const double *x1_start,
    const double *x2_start, const double *tipVector,
    const unsigned char *tipX1, const unsigned char *tipX2, int n,
    const double *x1_gapColumn, const double *x2_gapColumn,
    const unsigned int *x1_gap, const unsigned int *x2_gap) {
  int i, j;
  const double *x1_ptr = x1_start, *x2_ptr = x2_start;
  for (i = 0; i < n; i++) {
    const double *le, *ri;
    switch (tipCase) {
    case ORC_TIP_TIP:
      le = &tipVector[20 * tipX1[i]];
      ri = &tipVector[20 * tipX2[i]];
      break;
    case ORC_TIP_INNER:
      le = &tipVector[20 * tipX1[i]];
      if (orc_is_gap(x2_gap, i))
        ri = x2_gapColumn;
      else {
        ri = x2_ptr;
        x2_ptr += 20;
      }
      break;
    default:
      if (orc_is_gap(x1_gap, i)) {
        le = x1_gapColumn;
      } else {
        le = x1_ptr;
        x1_ptr += 20;
      }
      if (orc_is_gap(x2_gap, i)) {
        ri = x2_gapColumn;
      } else {
        ri = x2_ptr;
        x2_ptr += 20;
      }
    }
    for (j = 0; j < 20; j++) sumtable[i * 20 + j] = le[j] * ri[j];
  }
}
