/* ============================================================================
 * model_prep.cpp — host-side model initialization for the MI355X likelihood
 * core (product code; compiled into libexaml_hip.so).
 *
 * Replaces the reference's initReversibleGTR -> initGeneric eigendecomposition
 * pipeline (examl/models.c:3462/3234: symmetrized GTR matrix, Householder +
 * QL eigensolver in the reference's storage convention, fracchange
 * normalization, zero-eigenvalue-first ordering, EV/EI/tipVector emission)
 * and makeGammaCats (models.c:3795, via the published AS91/AS32/AS70
 * algorithms).  Arithmetic matches the reference bit-for-bit; pinned by
 * tests/test_oracle_cpu.py + tests/test_product_model_cpu.py against the
 * golden vectors.
 * ==========================================================================*/

#include <assert.h>
#include <math.h>
#include <string.h>

#include "../../include/examl_hip.h"

#define MAX_TIP_EV 0.999999999 /* examl/axml.h:88 */

/* ---- discrete gamma (models.c:3589-3795) -------------------------------- */

static double LnGamma_(double alpha) {
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

static double IncompleteGamma_(double x, double alpha, double ln_gamma_alpha) {
  int i;
  double p = alpha, g = ln_gamma_alpha;
  double accurate = 1e-8, overflow = 1e30;
  double factor, gin = 0, rn = 0, a = 0, b = 0, an = 0, dif = 0, term = 0,
         pn[6];
  if (x == 0) return 0;
  if (x < 0 || p <= 0) return -1;
  factor = exp(p * log(x) - x - g);
  if (!(x > 1 && x >= p)) {
    gin = 1;
    term = 1;
    rn = p;
    do {
      rn++;
      term *= x / rn;
      gin += term;
    } while (term > accurate);
    gin *= factor / p;
    return gin;
  }
  a = 1 - p;
  b = a + x + 1;
  term = 0;
  pn[0] = 1;
  pn[1] = x;
  pn[2] = x + 1;
  pn[3] = x * b;
  gin = pn[2] / pn[3];
  for (;;) {
    a++;
    b += 2;
    term++;
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

static double PointNormal_(double prob) {
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

static double PointChi2_(double prob, double v) {
  double e = .5e-6, aa = .6931471805, p = prob, g;
  double xx, c, ch, a = 0, q = 0, p1 = 0, p2 = 0, t = 0, x = 0, b = 0, s1, s2,
             s3, s4, s5, s6;
  if (p < .000002 || p > .999998 || v <= 0) return -1;
  g = LnGamma_(v / 2);
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
    x = PointNormal_(p);
    p1 = 0.222222 / v;
    ch = v * pow((x * sqrt(p1) + 1 - p1), 3.0);
    if (ch > 2.2 * v + 6) ch = -2 * (log(1 - p) - c * log(.5 * ch) + g);
  }
  do {
    q = ch;
    p1 = .5 * ch;
    if ((t = IncompleteGamma_(p1, xx, g)) < 0.0) return -1;
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
               b * c *
                   (s1 - b * (s2 - b * (s3 - b * (s4 - b * (s5 - b * s6))))));
  } while (fabs(q / ch - 1) > e);
  return ch;
}

/* the useMedian=TRUE branch of makeGammaCats (models.c:3795) */
extern "C" void examl_host_make_gamma_cats_median(double alpha,
                                                  double *gammaRates,
                                                  int K) {
  const double factor = alpha / alpha * K, alfa = alpha, beta = alpha;
  const double middle = 1.0 / (2.0 * K);
  double t = 0.0;
  for (int i = 0; i < K; i++)
    gammaRates[i] =
        PointChi2_((double)(i * 2 + 1) * middle, 2.0 * alfa) / (2.0 * beta);
  for (int i = 0; i < K; i++) t += gammaRates[i];
  for (int i = 0; i < K; i++) gammaRates[i] *= factor / t;
}

extern "C" void examl_host_make_gamma_cats(double alpha, double *gammaRates,
                                           int K) {
  const double factor = alpha / alpha * K, alfa = alpha, beta = alpha;
  double gammaProbs[32];
  const double lnga1 = LnGamma_(alfa + 1);
  for (int i = 0; i < K - 1; i++)
    gammaProbs[i] = PointChi2_((i + 1.0) / K, 2.0 * alfa) / (2.0 * beta);
  for (int i = 0; i < K - 1; i++)
    gammaProbs[i] = IncompleteGamma_(gammaProbs[i] * beta, alfa + 1, lnga1);
  gammaRates[0] = gammaProbs[0] * factor;
  gammaRates[K - 1] = (1 - gammaProbs[K - 2]) * factor;
  for (int i = 1; i < K - 1; i++)
    gammaRates[i] = (gammaProbs[i] - gammaProbs[i - 1]) * factor;
}

/* ---- eigensolver in the reference's convention (models.c:3068/3151) ----- */

static void tred2_(double *a, const int n, double *d, double *e) {
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

static void tqli_(double *d, double *e, const int n, double *z) {
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
        g = d[m - 1] - d[l - 1] +
            e[l - 1] / (g + ((g < 0) ? -fabs(r) : fabs(r)));
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

static void init_gtr_generic(int n, const unsigned int *valueVector, int vlen,
                             const double *frequencies, const double *rates,
                             double *EIGN_out, double *EV, double *EI,
                             double *tipVector) {
  double a[400], d[20], e[20], EIGV[400], invfreq[20], EIGN[20], r[400];
  double fracchange = 0.0;
  int i, j, k, m, l;

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
      const double factor = rates[m++];
      a[i * n + j] = a[j * n + i] =
          factor * sqrt(frequencies[i] * frequencies[j]);
      a[i * n + i] -= factor * frequencies[j];
      a[j * n + j] -= factor * frequencies[i];
    }

  tred2_(a, n, d, e);
  tqli_(d, e, n, a);

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
  EIGN_out[0] = 0.0;
  for (l = 1; l < n; l++) {
    EIGN_out[l] = EIGN[l] * (1.0 / fracchange);
    assert(EIGN_out[l] > 0.0);
  }
  for (i = 0; i < n; i++)
    for (j = 0; j < n; j++) EV[i * n + j] = EIGV[i * n + j];
  for (i = 0; i < n; i++)
    for (j = 0; j < n; j++)
      EI[i * n + j] = (j == 0) ? 1.0 : EV[i * n + j] * invfreq[i];

  for (i = 0; i < vlen; i++) {
    const unsigned int value = valueVector[i];
    for (j = 0; j < n; j++) tipVector[i * n + j] = 0;
    if (value > 0)
      for (j = 0; j < n; j++)
        if ((value >> j) & 1)
          for (l = 0; l < n; l++) tipVector[i * n + l] += EIGV[j * n + l];
  }
  for (i = 0; i < vlen; i++)
    for (j = 0; j < n; j++)
      if (tipVector[i * n + j] > MAX_TIP_EV) tipVector[i * n + j] = MAX_TIP_EV;
}

/* DNA: bitVectorIdentity[0..15] (globalVariables.h:80), getUndetermined=15 */
extern "C" void examl_host_init_gtr_dna(const double *frequencies,
                                        const double *rates6, double *EIGN,
                                        double *EV, double *EI,
                                        double *tipVector) {
  unsigned int vv[16];
  for (int i = 0; i < 16; i++) vv[i] = (unsigned int)i;
  init_gtr_generic(4, vv, 16, frequencies, rates6, EIGN, EV, EI, tipVector);
}

/* AA: bitVectorAA[23] (globalVariables.h:95) — 20 single-residue bits plus
 * B = D|N, Z = E|Q, X = all */
extern "C" void examl_host_init_gtr_aa(const double *frequencies,
                                       const double *rates190, double *EIGN,
                                       double *EV, double *EI,
                                       double *tipVector) {
  unsigned int vv[23];
  for (int i = 0; i < 20; i++) vv[i] = 1u << i;
  vv[20] = (1u << 2) | (1u << 3);  /* B: D or N */
  vv[21] = (1u << 5) | (1u << 6);  /* Z: E or Q */
  vv[22] = 0xFFFFFu;               /* X */
  init_gtr_generic(20, vv, 23, frequencies, rates190, EIGN, EV, EI,
                   tipVector);
}

/* ---------------------------------------------------------------------------
 * evaluatePartialGeneric for DNA CAT — the single-site re-evaluation at an
 * arbitrary rate used by optimizeRateCategories
 * (examl/evaluatePartialGenericSpecial.c:259/924/998, restated
 * index-for-index, including the reference's own quirks: the overwritten
 * zmin clamp on the root branch and the per-node (not weight-scaled)
 * rescale counter).  Pure host math: the reference runs this on the CPU
 * too (one site, the whole tree).
 *
 * ops: the post-order entries of the LAST full traversal (root entry
 * excluded, as tr->td[0].ti[1..]); root = (rootTip, rootQ, root_z).
 * tips: host yVector block [node row][site], row index = tip node id.
 * ------------------------------------------------------------------------ */
extern "C" double examl_host_evaluate_partial_dna_cat(
    const void *ops_, int numOps, int rootTipNumber, int rootQNumber,
    double root_z, long site, double ki, int w, const double *EIGN,
    const double *EI, const double *EV, const double *tipVector,
    const unsigned char *tips, long tipStride, int mxtips) {
  /* examl_hip_trav_entry layout without including the HIP header here */
  struct TE {
    int tipCase, pNumber, qNumber, rNumber;
    int x1Slot, x2Slot, x3Slot;
    double qz, rz;
  };
  const TE *ops = (const TE *)ops_;
  const double ZMIN_ = 1.0E-15;
  const double TWO256 =
      115792089237316195423570985008687907853269984665640564039457584007913129639936.0;
  const double MINLIK = 1.0 / TWO256;

  double *lVector = (double *)malloc(sizeof(double) * 4 * (size_t)mxtips);
  int scale = 0;

  for (int k = 0; k < numOps; k++) {
    const TE *t = &ops[k];
    double qz = t->qz, rz = t->rz;
    qz = (qz > ZMIN_) ? log(qz) : log(ZMIN_);
    rz = (rz > ZMIN_) ? log(rz) : log(ZMIN_);
    /* computeVectorGTRCAT (evaluatePartialGenericSpecial.c:924) */
    const double *x1, *x2;
    double *x3 = &lVector[4 * (t->pNumber - mxtips)];
    switch (t->tipCase) {
    case 0: /* TIP_TIP */
      x1 = &tipVector[4 * tips[(long)t->qNumber * tipStride + site]];
      x2 = &tipVector[4 * tips[(long)t->rNumber * tipStride + site]];
      break;
    case 1: /* TIP_INNER */
      x1 = &tipVector[4 * tips[(long)t->qNumber * tipStride + site]];
      x2 = &lVector[4 * (t->rNumber - mxtips)];
      break;
    default: /* INNER_INNER */
      x1 = &lVector[4 * (t->qNumber - mxtips)];
      x2 = &lVector[4 * (t->rNumber - mxtips)];
    }
    const double lz1 = qz * ki, lz2 = rz * ki;
    double d1[3], d2[3], x1px2[4];
    for (int j = 0; j < 3; j++) {
      d1[j] = x1[j + 1] * exp(EIGN[j + 1] * lz1);
      d2[j] = x2[j + 1] * exp(EIGN[j + 1] * lz2);
    }
    for (int j = 0; j < 4; j++) {
      double u1 = x1[0], u2 = x2[0];
      for (int kk = 0; kk < 3; kk++) {
        u1 += d1[kk] * EI[j * 4 + kk + 1];
        u2 += d2[kk] * EI[j * 4 + kk + 1];
      }
      x1px2[j] = u1 * u2;
    }
    for (int j = 0; j < 4; j++) x3[j] = 0.0;
    for (int j = 0; j < 4; j++)
      for (int kk = 0; kk < 4; kk++) x3[kk] += x1px2[j] * EV[4 * j + kk];
    if (x3[0] < MINLIK && x3[0] > -MINLIK && x3[1] < MINLIK &&
        x3[1] > -MINLIK && x3[2] < MINLIK && x3[2] > -MINLIK &&
        x3[3] < MINLIK && x3[3] > -MINLIK) {
      x3[0] *= TWO256;
      x3[1] *= TWO256;
      x3[2] *= TWO256;
      x3[3] *= TWO256;
      scale++;
    }
  }

  /* evaluatePartialGTRCAT tail (:998); NOTE the reference's dead zmin
   * clamp — lz is log(qz) regardless — restated as-is */
  const double *x1 = &tipVector[4 * tips[(long)rootTipNumber * tipStride +
                                         site]];
  const double *x2 = &lVector[4 * (rootQNumber - mxtips)];
  double lz = log(root_z);
  lz *= ki;
  const double d0 = exp(EIGN[1] * lz), dd1 = exp(EIGN[2] * lz),
               dd2 = exp(EIGN[3] * lz);
  double term = x1[0] * x2[0];
  term += x1[1] * x2[1] * d0;
  term += x1[2] * x2[2] * dd1;
  term += x1[3] * x2[3] * dd2;
  term = log(fabs(term)) + (scale * log(MINLIK));
  term = term * w;
  free(lVector);
  return term;
}

extern "C" double examl_host_evaluate_partial_prot_cat(
    const void *ops_, int numOps, int rootTipNumber, int rootQNumber,
    double root_z, long site, double ki, int w, const double *EIGN,
    const double *EI, const double *EV, const double *tipVector,
    const unsigned char *tips, long tipStride, int mxtips) {
  /* computeVectorGTRCATPROT / evaluatePartialGTRCATPROT
   * (evaluatePartialGenericSpecial.c:493/618), span 20, SSE even/odd ump
   * sums; the reference's dead zmin clamp restated as-is */
  struct TE {
    int tipCase, pNumber, qNumber, rNumber;
    int x1Slot, x2Slot, x3Slot;
    double qz, rz;
  };
  const TE *ops = (const TE *)ops_;
  const double ZMIN_ = 1.0E-15;
  const double TWO256 =
      115792089237316195423570985008687907853269984665640564039457584007913129639936.0;
  const double MINLIK = 1.0 / TWO256;

  double *lVector = (double *)malloc(sizeof(double) * 20 * (size_t)mxtips);
  int scale = 0;

  for (int k = 0; k < numOps; k++) {
    const TE *t = &ops[k];
    double qz = t->qz, rz = t->rz;
    qz = (qz > ZMIN_) ? log(qz) : log(ZMIN_);
    rz = (rz > ZMIN_) ? log(rz) : log(ZMIN_);
    const double *x1, *x2;
    double *x3 = &lVector[20 * (t->pNumber - mxtips)];
    switch (t->tipCase) {
    case 0:
      x1 = &tipVector[20 * tips[(long)t->qNumber * tipStride + site]];
      x2 = &tipVector[20 * tips[(long)t->rNumber * tipStride + site]];
      break;
    case 1:
      x1 = &tipVector[20 * tips[(long)t->qNumber * tipStride + site]];
      x2 = &lVector[20 * (t->rNumber - mxtips)];
      break;
    default:
      x1 = &lVector[20 * (t->qNumber - mxtips)];
      x2 = &lVector[20 * (t->rNumber - mxtips)];
    }
    const double lz1 = qz * ki, lz2 = rz * ki;
    double e1[20], e2[20], d1[20], d2[20];
    e1[0] = 1.0;
    e2[0] = 1.0;
    for (int l = 1; l < 20; l++) {
      e1[l] = exp(EIGN[l] * lz1);
      e2[l] = exp(EIGN[l] * lz2);
    }
    for (int l = 0; l < 20; l++) {
      d1[l] = x1[l] * e1[l];
      d2[l] = x2[l] * e2[l];
    }
    for (int l = 0; l < 20; l++) x3[l] = 0.0;
    for (int l = 0; l < 20; l++) {
      const double *ev = &EV[l * 20];
      double u1e = 0, u1o = 0, u2e = 0, u2o = 0;
      for (int kk = 0; kk < 20; kk += 2) {
        u1e += d1[kk] * EI[20 * l + kk];
        u1o += d1[kk + 1] * EI[20 * l + kk + 1];
        u2e += d2[kk] * EI[20 * l + kk];
        u2o += d2[kk + 1] * EI[20 * l + kk + 1];
      }
      const double x1px2 = (u1e + u1o) * (u2e + u2o);
      for (int kk = 0; kk < 20; kk++) x3[kk] += x1px2 * ev[kk];
    }
    int sc = 1;
    for (int l = 0; sc && l < 20; l++)
      sc = (x3[l] < MINLIK && x3[l] > -MINLIK);
    if (sc) {
      for (int l = 0; l < 20; l++) x3[l] *= TWO256;
      scale++;
    }
  }

  const double *x1 =
      &tipVector[20 * tips[(long)rootTipNumber * tipStride + site]];
  const double *x2 = &lVector[20 * (rootQNumber - mxtips)];
  double lz = log(root_z); /* the reference's zmin clamp is dead code */
  lz *= ki;
  double d[20];
  d[0] = 1.0;
  for (int l = 1; l < 20; l++) d[l] = exp(EIGN[l] * lz);
  double term = 0.0;
  for (int l = 0; l < 20; l++) term += x1[l] * x2[l] * d[l];
  term = log(fabs(term)) + (scale * log(MINLIK));
  term = term * w;
  free(lVector);
  return term;
}

/* =========================================================================
 * LG4 (LG4M/LG4X) host model math: per-category matrices.
 * EIGN4 stride 20 (SCALED, scaleLG4X_EIGN), EI4 stride 400.
 * =========================================================================*/

/* makeP_FlexLG4 (newviewGenericSpecial.c:170), numStates=20, 4 cats */
extern "C" void examl_host_make_p_lg4(double z1, double z2,
                                      const double *gammaRates,
                                      const double *EI4, const double *EIGN4,
                                      double *left, double *right) {
  double d1[20], d2[20];
  for (int i = 0; i < 4; i++) {
    const double *EI = EI4 + i * 400;
    const double *EIGN = EIGN4 + i * 20;
    for (int j = 1; j < 20; j++) {
      d1[j] = exp(gammaRates[i] * EIGN[j] * z1);
      d2[j] = exp(gammaRates[i] * EIGN[j] * z2);
    }
    for (int j = 0; j < 20; j++) {
      left[400 * i + 20 * j] = 1.0;
      right[400 * i + 20 * j] = 1.0;
      for (int k = 1; k < 20; k++) {
        left[400 * i + 20 * j + k] = d1[k] * EI[20 * j + k];
        right[400 * i + 20 * j + k] = d2[k] * EI[20 * j + k];
      }
    }
  }
}

/* calcDiagptableFlex_LG4 (evaluateGenericSpecial.c:122): takes RAW z */
extern "C" void examl_host_calc_diag_lg4(double z, const double *gammaRates,
                                         const double *EIGN4, double *diag) {
  const double kZMIN = 1.0E-15; /* axml.h zmin */
  const double lz = (z < kZMIN) ? log(kZMIN) : log(z);
  for (int i = 0; i < 4; i++) {
    diag[i * 20] = 1.0;
    for (int l = 1; l < 20; l++)
      diag[i * 20 + l] = exp(gammaRates[i] * EIGN4[i * 20 + l] * lz);
  }
}

/* coreGTRGAMMAPROT_LG4 d-tables (makenewzGenericSpecial.c:2501-2517) */
extern "C" void examl_host_core_dtables_prot_lg4(const double *EIGN4,
                                                 const double *gammaRates,
                                                 double lz, double *dtab) {
  double *d0 = dtab, *d1 = dtab + 80, *d2 = dtab + 160;
  for (int i = 0; i < 4; i++) {
    const double ki = gammaRates[i], kisqr = ki * ki;
    const double *EIGN = EIGN4 + i * 20;
    d0[i * 20] = 1.0;
    d1[i * 20] = 0.0;
    d2[i * 20] = 0.0;
    for (int l = 1; l < 20; l++) {
      d0[i * 20 + l] = exp(EIGN[l] * ki * lz);
      d1[i * 20 + l] = EIGN[l] * ki;
      d2[i * 20 + l] = EIGN[l] * EIGN[l] * kisqr;
    }
  }
}

/* coreGTRCATPROT d-tables (makenewzGenericSpecial.c:2659):
 * dtab = d[numCats*20] | s[20] | e[20] | rates[numCats] */
extern "C" void examl_host_core_dtables_prot_cat(const double *EIGN,
                                                 const double *rptr,
                                                 int numCats, double lz,
                                                 double *dtab) {
  double *d = dtab, *s_ = dtab + numCats * 20, *e = dtab + numCats * 20 + 20;
  double *rw = dtab + numCats * 20 + 40;
  double dd[20];
  e[0] = s_[0] = dd[0] = 0.0;
  for (int l = 1; l < 20; l++) {
    e[l] = EIGN[l] * EIGN[l];
    s_[l] = EIGN[l];
    dd[l] = s_[l] * lz;
  }
  for (int i = 0; i < numCats; i++) {
    d[20 * i] = 1.0;
    for (int l = 1; l < 20; l++) d[20 * i + l] = exp(dd[l] * rptr[i]);
    rw[i] = rptr[i];
  }
}
