/* ============================================================================
 * shim/axml_shim.c — the literal drop-in boundary.
 *
 * This TU exports the likelihood entry points declared in the reference's
 * axml.h (axml.h:1223-1253: evaluateGeneric, newviewGeneric,
 * makenewzGeneric, evaluatePartialGeneric, plus the iterative bodies
 * newviewIterative / evaluateIterative / makenewzIterative / execCore and
 * computeTraversalInfo), so the UNMODIFIED reference tree search
 * (searchAlgo.c / optimizeModel.c / axml.c, compiled in place from the
 * upstream sources) links against libexaml_hip.so and drives the CDNA4 HIP
 * kernels instead of the SSE3/AVX CPU kernels.
 *
 * It is a from-scratch restatement of the L1 orchestration semantics of
 *   newviewGenericSpecial.c:917  (newviewIterative dispatch)
 *   evaluateGenericSpecial.c:403 (evaluateIterative + scaler undo + C1)
 *   makenewzGenericSpecial.c:628/849/1133 (makenewz + NR loop + C2)
 * over the C-ABI executors of include/examl_hip.h: all CLVs/tips/weights
 * are device-resident for the life of the run (one context per partition),
 * and only traversal descriptors, P-matrix inputs and scalar results cross
 * the PCIe boundary.  The MPI collectives keep the reference's shapes
 * (evaluateGenericSpecial.c:969, makenewzGenericSpecial.c:1244) so the
 * multi-rank data distribution of partitionAssignment.c works unchanged,
 * one GPU per rank.
 *
 * Build: shim/Makefile (compiles the reference's own L3/L4 TUs in place
 * from /root/reference — never copied — and links them against this TU +
 * libexaml_hip.so into shim/_build/examl-HIP).
 * ==========================================================================*/

#define __HIP_PLATFORM_AMD__ 1
#include <hip/hip_runtime_api.h>

#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "axml.h" /* the reference ABI, included in place at build time */
#include "examl_hip.h"

/* ---------------------------------------------------------------------------
 * Error plumbing: the reference fails via assert/MPI_Abort (SURVEY §8b
 * "Errors are assert/MPI_Abort — no error codes"), so do the same.
 * ------------------------------------------------------------------------ */

static void shim_die(const char *what, int code)
{
  fprintf(stderr, "examl-HIP shim: %s failed (%d): %s\n", what, code,
          examl_hip_last_error_string());
  MPI_Abort(MPI_COMM_WORLD, 1);
  exit(1);
}

#define CK(call)                                                             \
  do {                                                                       \
    int _e = (call);                                                         \
    if (_e != 0) shim_die(#call, _e);                                        \
  } while (0)

#define HIP_OK(call)                                                         \
  do {                                                                       \
    hipError_t _e = (call);                                                  \
    if (_e != hipSuccess) {                                                  \
      fprintf(stderr, "examl-HIP shim: %s failed: %s\n", #call,              \
              hipGetErrorString(_e));                                        \
      MPI_Abort(MPI_COMM_WORLD, 1);                                          \
    }                                                                        \
  } while (0)

/* ---------------------------------------------------------------------------
 * Per-partition device context — the GPU mirror of pInfo (axml.h:533-629).
 * CLVs are allocated up front, one slot per inner node (the device answer
 * to the lazy xVector reallocation of newviewGenericSpecial.c:1200-1215).
 * ------------------------------------------------------------------------ */

typedef struct {
  int states;   /* 4 DNA, 20 AA */
  int span;     /* 4*states under GAMMA */
  long width;   /* this rank's site count for the partition */
  int maxOps;
  double *d_clv;          /* (mxtips-2) slots x width*span */
  double *d_EV;           /* states^2 */
  double *d_tipVector;    /* 64 DNA / 460 AA */
  unsigned char *d_tips;  /* (mxtips+1) rows x width, row 0 unused */
  int *d_wgt;             /* width */
  unsigned int *d_scalers; /* 2*mxtips, tips stay 0 (globalScaler) */
  unsigned int *d_inc;     /* maxOps per-op scaler increments */
  double *d_pbuf;          /* maxOps * 8*states^2 P blocks */
  double *d_diag;          /* 4*states */
  double *d_dtab;          /* 12*states */
  double *d_partials;      /* 2*8192 reduction scratch */
  double *d_lnl;           /* 1 */
  double *d_out2;          /* 2 */
  double *d_sum;           /* width*span sumBuffer (lazy) */
  examl_hip_trav_entry *ops; /* host scratch */
  double *h_lnl;             /* pinned readbacks */
  double *h_out2;
  int executed; /* scratch flag per call */
  int *d_cptr;               /* CAT: per-site rate category (refreshed per
                                traversal from pd->rateCategory) */
  unsigned char *h_tips;     /* host tips copy (evaluatePartialGeneric) */
  /* LG4M/LG4X: per-gamma-category eigensystems, packed contiguous from
   * the reference's *_LG4[4] row pointers per call */
  int lg4;
  double *d_EV4, *d_tipVec4; /* 1600 / 1840 */
  double *h_eign4, *h_ei4;   /* 80 / 1600 host staging */
  double *h_pack;            /* 3440 host staging for EV4|tipVec4 */
} ShimPart;

#define SHIM_MAXC 25 /* maxCategories default, axml.h */

static ShimPart *S = NULL;
static int S_n = 0;
static int S_mxtips = 0;

/* Fused multi-partition path (examl_hip_multi_*): one launch per
 * (traversal level x tipCase) covering all partitions, replacing the
 * per-partition loop below when every partition is DNA GAMMA.  This is
 * the C-side batching of newviewIterative's partition loop
 * (newviewGenericSpecial.c:1064) and of execCore's
 * (makenewzGenericSpecial.c:885). */
static void *g_multi = NULL;
static double *g_d_lnl_vec = NULL, *g_h_lnl_vec = NULL;
static double *g_d_out2_vec = NULL, *g_h_out2_vec = NULL;
static examl_hip_trav_entry *g_mops = NULL;
static double *g_qzov = NULL, *g_rzov = NULL;
static unsigned char *g_amask = NULL;
static const double **g_eign = NULL, **g_ei = NULL, **g_rates = NULL;

static void *dmalloc(size_t bytes)
{
  void *p = NULL;
  HIP_OK(hipMalloc(&p, bytes ? bytes : 8));
  return p;
}

static void init_part(tree *tr, int m)
{
  pInfo *pd = &tr->partitionData[m];
  ShimPart *p = &S[m];
  long w = (long)pd->width;
  int states = pd->states;
  int mxtips = tr->mxtips;

  if (states != 4 && states != 20) {
    fprintf(stderr,
            "examl-HIP shim: %d-state partitions are not supported\n",
            states);
    MPI_Abort(MPI_COMM_WORLD, 1);
  }

  p->states = states;
  p->span = (tr->rateHetModel == CAT) ? states : 4 * states;
  p->width = w;
  p->maxOps = mxtips + 8;
  p->executed = 0;
  if (w == 0) return;

  p->d_clv = dmalloc((size_t)(mxtips - 2) * w * p->span * sizeof(double));
  p->d_EV = dmalloc((size_t)states * states * sizeof(double));
  p->d_tipVector =
      dmalloc((size_t)(states == 4 ? 64 : 460) * sizeof(double));
  p->d_tips = dmalloc((size_t)(mxtips + 1) * w);
  p->d_wgt = dmalloc((size_t)w * sizeof(int));
  p->d_scalers = dmalloc((size_t)2 * mxtips * sizeof(unsigned int));
  p->d_inc = dmalloc((size_t)p->maxOps * sizeof(unsigned int));
  /* GAMMA: 2 P blocks of 4 cats per op; CAT: numCats<=25 pairs per op,
   * diag/dtab sized for maxCategories up front (axml.c:1936) */
  p->d_pbuf = dmalloc((size_t)p->maxOps *
                      ((tr->rateHetModel == CAT) ? (size_t)2 * SHIM_MAXC
                                                 : 8) *
                      states * states * sizeof(double));
  p->d_diag = dmalloc((size_t)(SHIM_MAXC + 4) * states * sizeof(double));
  p->d_dtab = dmalloc(((size_t)SHIM_MAXC * states + 2 * states +
                       SHIM_MAXC + 16) * sizeof(double));
  p->d_cptr =
      (tr->rateHetModel == CAT) ? (int *)dmalloc((size_t)w * sizeof(int))
                                : NULL;
  p->d_partials = dmalloc((size_t)2 * 8192 * sizeof(double));
  p->d_lnl = dmalloc(sizeof(double));
  p->d_out2 = dmalloc(2 * sizeof(double));
  p->d_sum = NULL;

  HIP_OK(hipMemset(p->d_scalers, 0, 2 * mxtips * sizeof(unsigned int)));

  /* tips: yVector rows 1..mxtips (axml.h:599), contiguous at stride w */
  {
    unsigned char *h = (unsigned char *)malloc((size_t)(mxtips + 1) * w);
    int t;
    memset(h, 0, w);
    for (t = 1; t <= mxtips; t++)
      memcpy(h + (size_t)t * w, pd->yVector[t], (size_t)w);
    HIP_OK(hipMemcpy(p->d_tips, h, (size_t)(mxtips + 1) * w,
                     hipMemcpyHostToDevice));
    p->h_tips = h; /* kept for evaluatePartialGeneric (CAT rate search) */
  }
  HIP_OK(hipMemcpy(p->d_wgt, pd->wgt, (size_t)w * sizeof(int),
                   hipMemcpyHostToDevice));

  p->lg4 = (pd->protModels == LG4M || pd->protModels == LG4X);
  if (p->lg4) {
    p->d_EV4 = (double *)dmalloc(1600 * sizeof(double));
    p->d_tipVec4 = (double *)dmalloc(1840 * sizeof(double));
    p->h_eign4 = (double *)malloc(80 * sizeof(double));
    p->h_ei4 = (double *)malloc(1600 * sizeof(double));
    HIP_OK(hipHostMalloc((void **)&p->h_pack, 3440 * sizeof(double), 0));
  }
  p->ops = (examl_hip_trav_entry *)malloc(p->maxOps *
                                          sizeof(examl_hip_trav_entry));
  HIP_OK(hipHostMalloc((void **)&p->h_lnl, sizeof(double), 0));
  HIP_OK(hipHostMalloc((void **)&p->h_out2, 2 * sizeof(double), 0));
}

static void shim_init(tree *tr)
{
  int rank, ndev, m;
  if (S != NULL) return;
  MPI_Comm_rank(MPI_COMM_WORLD, &rank);
  HIP_OK(hipGetDeviceCount(&ndev));
  if (ndev < 1) {
    fprintf(stderr, "examl-HIP shim: no HIP device visible — this build "
                    "has no CPU fallback\n");
    MPI_Abort(MPI_COMM_WORLD, 1);
  }
  HIP_OK(hipSetDevice(rank % ndev));
  S_n = tr->NumberOfModels;
  S_mxtips = tr->mxtips;
  S = (ShimPart *)calloc(S_n, sizeof(ShimPart));
  for (m = 0; m < S_n; m++) init_part(tr, m);
  if (tr->rateHetModel != GAMMA && tr->rateHetModel != CAT) {
    fprintf(stderr, "examl-HIP shim: unsupported rate-het model %d\n",
            tr->rateHetModel);
    MPI_Abort(MPI_COMM_WORLD, 1);
  }
  if (tr->saveMemory) {
    fprintf(stderr, "examl-HIP shim: -S not wired in the C shim\n");
    MPI_Abort(MPI_COMM_WORLD, 1);
  }

  /* all-DNA GAMMA runs go through the fused multi-partition executors */
  {
    int allDna = (tr->rateHetModel == GAMMA);
    for (m = 0; m < S_n; m++)
      if (S[m].states != 4) allDna = 0;
    if (allDna && S_n >= 1) {
      long widths[S_n], clvStrides[S_n], tipStrides[S_n];
      double *clvs[S_n];
      const unsigned char *tips[S_n];
      const int *wgts[S_n];
      unsigned int *scalers[S_n];
      const double *EVs[S_n], *tipVecs[S_n];
      int maxOps = 0;
      for (m = 0; m < S_n; m++) {
        widths[m] = S[m].width;
        clvs[m] = S[m].d_clv;
        clvStrides[m] = S[m].width * S[m].span;
        tips[m] = S[m].d_tips;
        tipStrides[m] = S[m].width;
        wgts[m] = S[m].d_wgt;
        scalers[m] = S[m].d_scalers;
        EVs[m] = S[m].d_EV;
        tipVecs[m] = S[m].d_tipVector;
        if (S[m].maxOps > maxOps) maxOps = S[m].maxOps;
      }
      if (examl_hip_multi_create(4, S_n, widths, clvs, clvStrides, tips,
                                 tipStrides, wgts, scalers, EVs, tipVecs,
                                 maxOps, &g_multi) != 0) {
        fprintf(stderr, "examl-HIP shim: multi_create failed (%s); using "
                        "per-partition path\n",
                examl_hip_last_error_string());
        g_multi = NULL;
      } else {
        g_d_lnl_vec = (double *)dmalloc(S_n * sizeof(double));
        g_d_out2_vec = (double *)dmalloc(2 * S_n * sizeof(double));
        HIP_OK(hipHostMalloc((void **)&g_h_lnl_vec, S_n * sizeof(double),
                             0));
        HIP_OK(hipHostMalloc((void **)&g_h_out2_vec,
                             2 * S_n * sizeof(double), 0));
        g_mops = (examl_hip_trav_entry *)malloc(
            maxOps * sizeof(examl_hip_trav_entry));
        g_qzov = (double *)malloc((size_t)maxOps * S_n * sizeof(double));
        g_rzov = (double *)malloc((size_t)maxOps * S_n * sizeof(double));
        g_amask = (unsigned char *)malloc(S_n);
        g_eign = (const double **)malloc(S_n * sizeof(double *));
        g_ei = (const double **)malloc(S_n * sizeof(double *));
        g_rates = (const double **)malloc(S_n * sizeof(double *));
      }
    }
  }
}

/* Re-upload EV/tipVector before each traversal: initReversibleGTR
 * (models.c:3462, kept host code) rewrites them on every model-parameter
 * probe and there is no hook to observe that, so refresh the tiny device
 * copies per call (0.6 KB DNA / 6.9 KB AA, stream-ordered). */
static void upload_model(ShimPart *p, pInfo *pd)
{
  HIP_OK(hipMemcpyAsync(p->d_EV, pd->EV,
                        (size_t)p->states * p->states * sizeof(double),
                        hipMemcpyHostToDevice, 0));
  HIP_OK(hipMemcpyAsync(p->d_tipVector, pd->tipVector,
                        (size_t)(p->states == 4 ? 64 : 460) * sizeof(double),
                        hipMemcpyHostToDevice, 0));
}

/* ---------------------------------------------------------------------------
 * computeTraversalInfo — post-order traversal descriptor
 * (newviewGenericSpecial.c:691): emits {tipCase, p,q,r, qz[],rz[]} for
 * every CLV that needs recomputation, flipping operands so tip data is
 * always in q, re-orienting x flags via getxnode (axml.c:461, kept).
 * ------------------------------------------------------------------------ */

void computeTraversalInfo(nodeptr p, traversalInfo *ti, int *counter,
                          int maxTips, int numBranches,
                          boolean partialTraversal)
{
  nodeptr q, r;
  int i;

  if (isTip(p->number, maxTips)) return;

  q = p->next->back;
  r = p->next->next->back;

  if (isTip(q->number, maxTips) && isTip(r->number, maxTips)) {
    if (!p->x) getxnode(p);

    ti[*counter].tipCase = TIP_TIP;
    ti[*counter].pNumber = p->number;
    ti[*counter].qNumber = q->number;
    ti[*counter].rNumber = r->number;
    for (i = 0; i < numBranches; i++) {
      ti[*counter].qz[i] = q->z[i];
      ti[*counter].rz[i] = r->z[i];
    }
    *counter = *counter + 1;
  } else if (isTip(q->number, maxTips) || isTip(r->number, maxTips)) {
    /* one tip: make q the tip side */
    if (isTip(r->number, maxTips)) {
      nodeptr tmp = r;
      r = q;
      q = tmp;
    }
    if (!r->x || !partialTraversal)
      computeTraversalInfo(r, ti, counter, maxTips, numBranches,
                           partialTraversal);
    if (!p->x) getxnode(p);

    ti[*counter].tipCase = TIP_INNER;
    ti[*counter].pNumber = p->number;
    ti[*counter].qNumber = q->number;
    ti[*counter].rNumber = r->number;
    for (i = 0; i < numBranches; i++) {
      ti[*counter].qz[i] = q->z[i];
      ti[*counter].rz[i] = r->z[i];
    }
    *counter = *counter + 1;
  } else {
    if (!q->x || !partialTraversal)
      computeTraversalInfo(q, ti, counter, maxTips, numBranches,
                           partialTraversal);
    if (!r->x || !partialTraversal)
      computeTraversalInfo(r, ti, counter, maxTips, numBranches,
                           partialTraversal);
    if (!p->x) getxnode(p);

    ti[*counter].tipCase = INNER_INNER;
    ti[*counter].pNumber = p->number;
    ti[*counter].qNumber = q->number;
    ti[*counter].rNumber = r->number;
    for (i = 0; i < numBranches; i++) {
      ti[*counter].qz[i] = q->z[i];
      ti[*counter].rz[i] = r->z[i];
    }
    *counter = *counter + 1;
  }
}

/* LG4: pack the four per-category eigensystem rows (pInfo *_LG4[4],
 * axml.h:566-575) into the contiguous stride layout the LG4 executors
 * take, and refresh the device EV4/tipVector4 copies (optLG4X /
 * initReversibleGTR rewrite them between calls). */
static void upload_lg4(ShimPart *p, pInfo *pd)
{
  int k;
  for (k = 0; k < 4; k++) {
    memcpy(p->h_eign4 + 20 * k, pd->EIGN_LG4[k], 20 * sizeof(double));
    memcpy(p->h_ei4 + 400 * k, pd->EI_LG4[k], 400 * sizeof(double));
    memcpy(p->h_pack + 400 * k, pd->EV_LG4[k], 400 * sizeof(double));
    memcpy(p->h_pack + 1600 + 460 * k, pd->tipVector_LG4[k],
           460 * sizeof(double));
  }
  HIP_OK(hipMemcpyAsync(p->d_EV4, p->h_pack, 1600 * sizeof(double),
                        hipMemcpyHostToDevice, 0));
  HIP_OK(hipMemcpyAsync(p->d_tipVec4, p->h_pack + 1600,
                        1840 * sizeof(double), hipMemcpyHostToDevice, 0));
}

/* ti entries -> executor ops: resolve the CLV-slot / tip-row bindings the
 * way newviewIterative does (newviewGenericSpecial.c:1221-1261): inner
 * node n -> slot n - mxtips - 1; tip operands use the tip row number. */
static int build_ops(tree *tr, int m, int startIndex,
                     examl_hip_trav_entry *ops)
{
  int brIdx = (tr->numBranches > 1) ? m : 0;
  int n = 0, i;
  for (i = startIndex; i < tr->td[0].count; i++) {
    traversalInfo *ti = &tr->td[0].ti[i];
    examl_hip_trav_entry *e = &ops[n++];
    e->tipCase = ti->tipCase;
    e->pNumber = ti->pNumber;
    e->qNumber = ti->qNumber;
    e->rNumber = ti->rNumber;
    e->x3Slot = ti->pNumber - tr->mxtips - 1;
    switch (ti->tipCase) {
      case TIP_TIP:
        e->x1Slot = ti->qNumber;
        e->x2Slot = ti->rNumber;
        break;
      case TIP_INNER:
        e->x1Slot = ti->qNumber; /* q carries the tip data */
        e->x2Slot = ti->rNumber - tr->mxtips - 1;
        break;
      default:
        e->x1Slot = ti->qNumber - tr->mxtips - 1;
        e->x2Slot = ti->rNumber - tr->mxtips - 1;
    }
    e->qz = ti->qz[brIdx];
    e->rz = ti->rz[brIdx];
  }
  return n;
}

/* ---------------------------------------------------------------------------
 * newviewIterative (newviewGenericSpecial.c:917): one executor call per
 * (partition, traversal) — P matrices on the host, one upload, one kernel
 * launch per post-order entry, device-side recursive scaler accumulation.
 * ------------------------------------------------------------------------ */

/* gather per-partition host model pointers + the executeModel mask */
static void multi_prep(tree *tr, int needWidth)
{
  int m;
  for (m = 0; m < tr->NumberOfModels; m++) {
    pInfo *pd = &tr->partitionData[m];
    g_eign[m] = pd->EIGN;
    g_ei[m] = pd->EI;
    g_rates[m] = pd->gammaRates;
    g_amask[m] =
        (tr->td[0].executeModel[m] && (!needWidth || S[m].width > 0)) ? 1
                                                                      : 0;
    if (S[m].width > 0) upload_model(&S[m], pd);
  }
}

void newviewIterative(tree *tr, int startIndex)
{
  int m;
  shim_init(tr);
  if (tr->td[0].count - startIndex <= 0) return;

  if (g_multi) {
    /* fused path: one launch per (level x tipCase) over all partitions */
    int n = 0, i;
    for (i = startIndex; i < tr->td[0].count; i++) {
      traversalInfo *ti = &tr->td[0].ti[i];
      examl_hip_trav_entry *e = &g_mops[n];
      e->tipCase = ti->tipCase;
      e->pNumber = ti->pNumber;
      e->qNumber = ti->qNumber;
      e->rNumber = ti->rNumber;
      e->x3Slot = ti->pNumber - tr->mxtips - 1;
      switch (ti->tipCase) {
        case TIP_TIP:
          e->x1Slot = ti->qNumber;
          e->x2Slot = ti->rNumber;
          break;
        case TIP_INNER:
          e->x1Slot = ti->qNumber;
          e->x2Slot = ti->rNumber - tr->mxtips - 1;
          break;
        default:
          e->x1Slot = ti->qNumber - tr->mxtips - 1;
          e->x2Slot = ti->rNumber - tr->mxtips - 1;
      }
      e->qz = ti->qz[0];
      e->rz = ti->rz[0];
      if (tr->numBranches > 1) {
        for (m = 0; m < S_n; m++) {
          g_qzov[(size_t)n * S_n + m] = ti->qz[m];
          g_rzov[(size_t)n * S_n + m] = ti->rz[m];
        }
      }
      n++;
    }
    multi_prep(tr, 0);
    CK(examl_hip_newview_traversal_multi(
        g_multi, g_mops, n, g_eign, g_ei, g_rates, g_amask,
        tr->numBranches > 1 ? g_qzov : NULL,
        tr->numBranches > 1 ? g_rzov : NULL, 0));
    return;
  }

  for (m = 0; m < tr->NumberOfModels; m++) {
    ShimPart *p = &S[m];
    pInfo *pd = &tr->partitionData[m];
    int n;
    if (!tr->td[0].executeModel[m] || p->width == 0) continue;
    n = build_ops(tr, m, startIndex, p->ops);
    if (n > p->maxOps) shim_die("traversal exceeds maxOps", n);
    upload_model(p, pd);
    if (tr->rateHetModel == CAT) {
      /* the per-site categorization can change between calls
       * (optimizeRateCategories writes pd->rateCategory in place) */
      HIP_OK(hipMemcpyAsync(p->d_cptr, pd->rateCategory,
                            (size_t)p->width * sizeof(int),
                            hipMemcpyHostToDevice, 0));
      if (p->states == 4)
        CK(examl_hip_newview_traversal_dna_cat(
            p->ops, n, pd->EIGN, pd->EI, pd->perSiteRates,
            pd->numberOfCategories, p->d_EV, p->d_tipVector, p->d_cptr,
            p->d_clv, p->width * p->span, p->d_tips, p->width, p->d_wgt,
            p->width, p->d_scalers, p->d_inc, p->d_pbuf, 0));
      else
        CK(examl_hip_newview_traversal_prot_cat(
            p->ops, n, pd->EIGN, pd->EI, pd->perSiteRates,
            pd->numberOfCategories, p->d_EV, p->d_tipVector, p->d_cptr,
            p->d_clv, p->width * p->span, p->d_tips, p->width, p->d_wgt,
            p->width, p->d_scalers, p->d_inc, p->d_pbuf, 0));
    } else if (p->states == 4)
      CK(examl_hip_newview_traversal_dna_gamma(
          p->ops, n, pd->EIGN, pd->EI, pd->gammaRates, p->d_EV,
          p->d_tipVector, p->d_clv, p->width * p->span, p->d_tips, p->width,
          p->d_wgt, p->width, p->d_scalers, p->d_inc, p->d_pbuf, 0));
    else if (p->lg4) {
      upload_lg4(p, pd);
      CK(examl_hip_newview_traversal_prot_lg4(
          p->ops, n, p->h_eign4, p->h_ei4, pd->gammaRates, p->d_EV4,
          p->d_tipVec4, p->d_clv, p->width * p->span, p->d_tips, p->width,
          p->d_wgt, p->width, p->d_scalers, p->d_inc, p->d_pbuf, 0));
    } else
      CK(examl_hip_newview_traversal_prot_gamma(
          p->ops, n, pd->EIGN, pd->EI, pd->gammaRates, p->d_EV,
          p->d_tipVector, p->d_clv, p->width * p->span, p->d_tips, p->width,
          p->d_wgt, p->width, p->d_scalers, p->d_inc, p->d_pbuf, 0));
  }
}

/* Root-branch operand binding shared by evaluate and makenewz
 * (evaluateIterative:612-668 / getVects, makenewzGenericSpecial.c:71). */
static void root_case(int pNumber, int qNumber, int mxtips, int *tipCase,
                      int *x1Slot, int *x2Slot, int *tipSlot, int *tipSlot2)
{
  *tipSlot2 = -1;
  if (isTip(pNumber, mxtips) || isTip(qNumber, mxtips)) {
    if (isTip(pNumber, mxtips) && isTip(qNumber, mxtips)) {
      *tipCase = TIP_TIP;
      *x1Slot = -1;
      *x2Slot = -1;
      *tipSlot = pNumber;
      *tipSlot2 = qNumber;
    } else {
      *tipCase = TIP_INNER;
      *x1Slot = -1;
      if (isTip(qNumber, mxtips)) {
        *tipSlot = qNumber;
        *x2Slot = pNumber - mxtips - 1;
      } else {
        *tipSlot = pNumber;
        *x2Slot = qNumber - mxtips - 1;
      }
    }
  } else {
    *tipCase = INNER_INNER;
    *x1Slot = pNumber - mxtips - 1;
    *x2Slot = qNumber - mxtips - 1;
    *tipSlot = -1;
  }
}

/* ---------------------------------------------------------------------------
 * evaluateIterative (evaluateGenericSpecial.c:403): partial/full traversal,
 * then per-partition root evaluation (diag tables on host, evaluate kernel
 * incl. the 2^-256 scaler undo of :830), perPartitionLH semantics of
 * :844-860 (masked partitions keep their stale value, width==0 -> 0.0).
 * ------------------------------------------------------------------------ */

void evaluateIterative(tree *tr)
{
  double *pz = tr->td[0].ti[0].qz;
  int pNumber = tr->td[0].ti[0].pNumber;
  int qNumber = tr->td[0].ti[0].qNumber;
  int m;

  shim_init(tr);
  newviewIterative(tr, 1);

  if (g_multi) {
    int tc, x1s, x2s, ts, ts2;
    double zs[NUM_BRANCHES];
    root_case(pNumber, qNumber, tr->mxtips, &tc, &x1s, &x2s, &ts, &ts2);
    if (tc == TIP_TIP) shim_die("evaluate at a tip-tip branch", 0);
    for (m = 0; m < tr->numBranches; m++) zs[m] = pz[m];
    multi_prep(tr, 0);
    HIP_OK(hipMemsetAsync(g_d_lnl_vec, 0, S_n * sizeof(double), 0));
    CK(examl_hip_evaluate_root_multi(
        g_multi, tc, pNumber, qNumber, x1s, x2s, ts, zs,
        tr->numBranches > 1 ? 1 : 0, g_eign, g_rates, g_amask, g_d_lnl_vec,
        0));
    HIP_OK(hipMemcpyAsync(g_h_lnl_vec, g_d_lnl_vec,
                          S_n * sizeof(double), hipMemcpyDeviceToHost, 0));
    HIP_OK(hipStreamSynchronize(0));
    for (m = 0; m < tr->NumberOfModels; m++) {
      if (tr->td[0].executeModel[m] && S[m].width > 0)
        tr->perPartitionLH[m] = g_h_lnl_vec[m];
      else if (S[m].width == 0)
        tr->perPartitionLH[m] = 0.0;
      /* masked + width > 0: keep stale (evaluateGenericSpecial.c:855) */
    }
    return;
  }

  for (m = 0; m < tr->NumberOfModels; m++) {
    ShimPart *p = &S[m];
    pInfo *pd = &tr->partitionData[m];
    int tc, x1s, x2s, ts, ts2;
    double z;
    p->executed = 0;
    if (!tr->td[0].executeModel[m] || p->width == 0) continue;
    z = (tr->numBranches > 1) ? pz[m] : pz[0];
    root_case(pNumber, qNumber, tr->mxtips, &tc, &x1s, &x2s, &ts, &ts2);
    if (tc == TIP_TIP) shim_die("evaluate at a tip-tip branch", 0);
    HIP_OK(hipMemsetAsync(p->d_lnl, 0, sizeof(double), 0));
    upload_model(p, pd);
    if (tr->rateHetModel == CAT) {
      if (p->states == 4)
        CK(examl_hip_evaluate_root_dna_cat_x(
            tc, pNumber, qNumber, x1s, x2s, ts, z, pd->EIGN,
            pd->perSiteRates, pd->numberOfCategories, p->d_tipVector,
            p->d_cptr, p->d_clv, p->width * p->span, p->d_tips, p->width,
            p->d_wgt, p->width, p->d_scalers, p->d_diag, p->d_partials,
            p->d_lnl, 0));
      else
        CK(examl_hip_evaluate_root_prot_cat(
            tc, pNumber, qNumber, x1s, x2s, ts, z, pd->EIGN,
            pd->perSiteRates, pd->numberOfCategories, p->d_tipVector,
            p->d_cptr, p->d_clv, p->width * p->span, p->d_tips, p->width,
            p->d_wgt, p->width, p->d_scalers, p->d_diag, p->d_partials,
            p->d_lnl, 0));
    } else if (p->states == 4)
      CK(examl_hip_evaluate_root_dna_gamma(
          tc, pNumber, qNumber, x1s, x2s, ts, z, pd->EIGN, pd->gammaRates,
          p->d_tipVector, p->d_clv, p->width * p->span, p->d_tips, p->width,
          p->d_wgt, p->width, p->d_scalers, p->d_diag, p->d_partials,
          p->d_lnl, 0));
    else if (p->lg4) {
      upload_lg4(p, pd);
      CK(examl_hip_evaluate_root_prot_lg4(
          tc, pNumber, qNumber, x1s, x2s, ts, z, p->h_eign4,
          pd->gammaRates, pd->weights, p->d_tipVec4, p->d_clv,
          p->width * p->span, p->d_tips, p->width, p->d_wgt, p->width,
          p->d_scalers, p->d_diag, p->d_partials, p->d_lnl, 0));
    } else
      CK(examl_hip_evaluate_root_prot_gamma(
          tc, pNumber, qNumber, x1s, x2s, ts, z, pd->EIGN, pd->gammaRates,
          p->d_tipVector, p->d_clv, p->width * p->span, p->d_tips, p->width,
          p->d_wgt, p->width, p->d_scalers, p->d_diag, p->d_partials,
          p->d_lnl, 0));
    HIP_OK(hipMemcpyAsync(p->h_lnl, p->d_lnl, sizeof(double),
                          hipMemcpyDeviceToHost, 0));
    p->executed = 1;
  }
  HIP_OK(hipStreamSynchronize(0));

  for (m = 0; m < tr->NumberOfModels; m++) {
    if (S[m].executed)
      tr->perPartitionLH[m] = *S[m].h_lnl;
    else if (S[m].width == 0)
      tr->perPartitionLH[m] = 0.0;
    /* else: masked with width > 0 — keep the stale value, exactly as
       evaluateGenericSpecial.c:855-859 */
  }
}

/* evaluateGeneric (evaluateGenericSpecial.c:897): traversal descriptor
 * setup, evaluateIterative, then the C1 all-reduce of perPartitionLH
 * (:969) and tr->likelihood. */
void evaluateGeneric(tree *tr, nodeptr p, boolean fullTraversal)
{
  volatile double result = 0.0;
  nodeptr q = p->back;
  int i, model;

  tr->td[0].ti[0].pNumber = p->number;
  tr->td[0].ti[0].qNumber = q->number;
  for (i = 0; i < tr->numBranches; i++)
    tr->td[0].ti[0].qz[i] = q->z[i];

  tr->td[0].count = 1;

  if (fullTraversal) {
    assert(isTip(p->number, tr->mxtips));
    computeTraversalInfo(q, &(tr->td[0].ti[0]), &(tr->td[0].count),
                         tr->mxtips, tr->numBranches, FALSE);
  } else {
    if (!p->x)
      computeTraversalInfo(p, &(tr->td[0].ti[0]), &(tr->td[0].count),
                           tr->mxtips, tr->numBranches, TRUE);
    if (!q->x)
      computeTraversalInfo(q, &(tr->td[0].ti[0]), &(tr->td[0].count),
                           tr->mxtips, tr->numBranches, TRUE);
  }

  storeExecuteMaskInTraversalDescriptor(tr);
  tr->td[0].traversalHasChanged = TRUE;

  evaluateIterative(tr);

  {
    double *recv = (double *)malloc(sizeof(double) * tr->NumberOfModels);
    MPI_Allreduce(tr->perPartitionLH, recv, tr->NumberOfModels, MPI_DOUBLE,
                  MPI_SUM, MPI_COMM_WORLD);
    memcpy(tr->perPartitionLH, recv,
           tr->NumberOfModels * sizeof(double));
    for (model = 0; model < tr->NumberOfModels; model++)
      result += tr->perPartitionLH[model];
    free(recv);
  }

  tr->likelihood = result;
  tr->td[0].traversalHasChanged = FALSE;
}

/* newviewGeneric (newviewGenericSpecial.c:1523) with the masked
 * (-M partitionConverged) executeModel juggling of :1559-1590. */
void newviewGeneric(tree *tr, nodeptr p, boolean masked)
{
  if (isTip(p->number, tr->mxtips)) return;

  tr->td[0].count = 0;
  computeTraversalInfo(p, &(tr->td[0].ti[0]), &(tr->td[0].count),
                       tr->mxtips, tr->numBranches, TRUE);
  tr->td[0].traversalHasChanged = TRUE;

  if (masked) {
    int model;
    for (model = 0; model < tr->NumberOfModels; model++)
      tr->executeModel[model] =
          tr->partitionConverged[model] ? FALSE : TRUE;
  }

  if (tr->td[0].count > 0) {
    storeExecuteMaskInTraversalDescriptor(tr);
    newviewIterative(tr, 0);
  }

  if (masked) {
    int model;
    for (model = 0; model < tr->NumberOfModels; model++)
      tr->executeModel[model] = TRUE;
  }
  tr->td[0].traversalHasChanged = FALSE;
}

/* ---------------------------------------------------------------------------
 * makenewz: sumBuffer precompute + NR loop.
 * ------------------------------------------------------------------------ */

/* makenewzIterative (makenewzGenericSpecial.c:628): traversal to the
 * branch, then per-partition sumBuffer = x1' o x2' on the device. */
void makenewzIterative(tree *tr)
{
  int pNumber = tr->td[0].ti[0].pNumber;
  int qNumber = tr->td[0].ti[0].qNumber;
  int m;

  shim_init(tr);
  newviewIterative(tr, 1);

  if (g_multi) {
    int tc, x1s, x2s, ts, ts2;
    root_case(pNumber, qNumber, tr->mxtips, &tc, &x1s, &x2s, &ts, &ts2);
    multi_prep(tr, 0);
    CK(examl_hip_sum_root_multi(g_multi, tc, x1s, x2s, ts, ts2, g_amask,
                                0));
    return;
  }

  for (m = 0; m < tr->NumberOfModels; m++) {
    ShimPart *p = &S[m];
    pInfo *pd = &tr->partitionData[m];
    int tc, x1s, x2s, ts, ts2;
    if (!tr->td[0].executeModel[m] || p->width == 0) continue;
    if (p->d_sum == NULL)
      p->d_sum = dmalloc((size_t)p->width * p->span * sizeof(double));
    root_case(pNumber, qNumber, tr->mxtips, &tc, &x1s, &x2s, &ts, &ts2);
    upload_model(p, pd);
    if (tr->rateHetModel == CAT) {
      if (p->states == 4)
        CK(examl_hip_sum_root_dna_cat(tc, x1s, x2s, ts, ts2,
                                      p->d_tipVector, p->d_clv,
                                      p->width * p->span, p->d_tips,
                                      p->width, p->d_sum, p->width, 0));
      else
        CK(examl_hip_sum_root_prot_cat(tc, x1s, x2s, ts, ts2,
                                       p->d_tipVector, p->d_clv,
                                       p->width * p->span, p->d_tips,
                                       p->width, p->d_sum, p->width, 0));
    } else if (p->states == 4)
      CK(examl_hip_sum_root_dna_gamma(tc, x1s, x2s, ts, ts2,
                                      p->d_tipVector, p->d_clv,
                                      p->width * p->span, p->d_tips,
                                      p->width, p->d_sum, p->width, 0));
    else if (p->lg4) {
      upload_lg4(p, pd);
      CK(examl_hip_sum_root_prot_lg4(tc, x1s, x2s, ts, ts2, p->d_tipVec4,
                                     p->d_clv, p->width * p->span,
                                     p->d_tips, p->width, p->d_sum,
                                     p->width, 0));
    } else
      CK(examl_hip_sum_root_prot_gamma(tc, x1s, x2s, ts, ts2,
                                       p->d_tipVector, p->d_clv,
                                       p->width * p->span, p->d_tips,
                                       p->width, p->d_sum, p->width, 0));
  }
}

/* execCore (makenewzGenericSpecial.c:849): per-partition first/second
 * derivatives from the sumBuffer at lz = td[0].parameterValues, summed
 * into the branchIndex slots with the reference's reset rule (:940-947).
 */
void execCore(tree *tr, volatile double *_dlnLdlz, volatile double *_d2lnLdlz2)
{
  int m;
  shim_init(tr);

  if (g_multi) {
    double lzs[NUM_BRANCHES];
    for (m = 0; m < tr->numBranches; m++)
      lzs[m] = tr->td[0].parameterValues[m];
    multi_prep(tr, 1);
    HIP_OK(hipMemsetAsync(g_d_out2_vec, 0, 2 * S_n * sizeof(double), 0));
    CK(examl_hip_core_root_multi(g_multi, lzs,
                                 tr->numBranches > 1 ? 1 : 0, g_eign,
                                 g_rates, g_amask, g_d_out2_vec, 0));
    HIP_OK(hipMemcpyAsync(g_h_out2_vec, g_d_out2_vec,
                          2 * S_n * sizeof(double), hipMemcpyDeviceToHost,
                          0));
    HIP_OK(hipStreamSynchronize(0));
    for (m = 0; m < tr->NumberOfModels; m++) {
      const int brIdx = (tr->numBranches > 1) ? m : 0;
      if (brIdx == m) {
        _dlnLdlz[brIdx] = 0.0;
        _d2lnLdlz2[brIdx] = 0.0;
      }
      if (!(tr->td[0].executeModel[m] && S[m].width > 0)) continue;
      _dlnLdlz[brIdx] += g_h_out2_vec[2 * m];
      _d2lnLdlz2[brIdx] += g_h_out2_vec[2 * m + 1];
    }
    return;
  }

  for (m = 0; m < tr->NumberOfModels; m++) {
    ShimPart *p = &S[m];
    pInfo *pd = &tr->partitionData[m];
    int brIdx = (tr->numBranches > 1) ? m : 0;
    double lz = tr->td[0].parameterValues[brIdx];
    p->executed = 0;
    if (brIdx == m) {
      _dlnLdlz[brIdx] = 0.0;
      _d2lnLdlz2[brIdx] = 0.0;
    }
    if (!(tr->td[0].executeModel[m] && p->width > 0)) continue;
    HIP_OK(hipMemsetAsync(p->d_out2, 0, 2 * sizeof(double), 0));
    if (tr->rateHetModel == CAT) {
      if (p->states == 4)
        CK(examl_hip_core_root_dna_cat(p->width, p->d_sum, pd->EIGN,
                                       pd->perSiteRates,
                                       pd->numberOfCategories, lz,
                                       p->d_wgt, p->d_cptr, p->d_dtab,
                                       p->d_partials, p->d_out2, 0));
      else
        CK(examl_hip_core_root_prot_cat(p->width, p->d_sum, pd->EIGN,
                                        pd->perSiteRates,
                                        pd->numberOfCategories, lz,
                                        p->d_wgt, p->d_cptr, p->d_dtab,
                                        p->d_partials, p->d_out2, 0));
    } else if (p->states == 4)
      CK(examl_hip_core_root_dna_gamma(p->width, p->d_sum, pd->EIGN,
                                       pd->gammaRates, lz, p->d_wgt,
                                       p->d_dtab, p->d_partials, p->d_out2,
                                       0));
    else if (p->lg4)
      CK(examl_hip_core_root_prot_lg4(p->width, p->d_sum, p->h_eign4,
                                      pd->gammaRates, pd->weights, lz,
                                      p->d_wgt, p->d_dtab, p->d_partials,
                                      p->d_out2, 0));
    else
      CK(examl_hip_core_root_prot_gamma(p->width, p->d_sum, pd->EIGN,
                                        pd->gammaRates, lz, p->d_wgt,
                                        p->d_dtab, p->d_partials, p->d_out2,
                                        0));
    HIP_OK(hipMemcpyAsync(p->h_out2, p->d_out2, 2 * sizeof(double),
                          hipMemcpyDeviceToHost, 0));
    p->executed = 1;
  }
  HIP_OK(hipStreamSynchronize(0));

  for (m = 0; m < tr->NumberOfModels; m++) {
    int brIdx = (tr->numBranches > 1) ? m : 0;
    if (!S[m].executed) continue;
    _dlnLdlz[brIdx] += S[m].h_out2[0];
    _d2lnLdlz2[brIdx] += S[m].h_out2[1];
  }
}

/* topLevelMakenewz (makenewzGenericSpecial.c:1133): the nested
 * Newton-Raphson loop over (possibly per-partition, -M) branch lengths,
 * with the C2 all-reduce of {dlnL,d2lnL} per iteration (:1244) and the
 * exact step/clamp/convergence rules of :1260-1334. */
static void top_level_makenewz(tree *tr, double *z0, int _maxiter,
                               double *result)
{
  double z[NUM_BRANCHES], zprev[NUM_BRANCHES], zstep[NUM_BRANCHES];
  double dlnLdlz[NUM_BRANCHES], d2lnLdlz2[NUM_BRANCHES];
  int i, maxiter[NUM_BRANCHES], model;
  boolean firstIteration = TRUE;
  boolean outerConverged[NUM_BRANCHES];
  boolean loopConverged;

  for (i = 0; i < tr->numBranches; i++) {
    z[i] = z0[i];
    maxiter[i] = _maxiter;
    outerConverged[i] = FALSE;
    tr->curvatOK[i] = TRUE;
  }

  do {
    for (i = 0; i < tr->numBranches; i++) {
      if (outerConverged[i] == FALSE && tr->curvatOK[i] == TRUE) {
        tr->curvatOK[i] = FALSE;
        zprev[i] = z[i];
        zstep[i] = (1.0 - zmax) * z[i] + zmin;
      }
    }

    for (i = 0; i < tr->numBranches; i++) {
      if (outerConverged[i] == FALSE && tr->curvatOK[i] == FALSE) {
        double lz;
        if (z[i] < zmin)
          z[i] = zmin;
        else if (z[i] > zmax)
          z[i] = zmax;
        lz = log(z[i]);
        tr->coreLZ[i] = lz;
      }
    }

    if (tr->numBranches > 1) {
      assert(tr->numBranches == tr->NumberOfModels);
      for (model = 0; model < tr->NumberOfModels; model++) {
        if (tr->executeModel[model])
          tr->executeModel[model] = !tr->curvatOK[model];
      }
    } else {
      for (model = 0; model < tr->NumberOfModels; model++)
        tr->executeModel[model] = !tr->curvatOK[0];
    }

    storeExecuteMaskInTraversalDescriptor(tr);
    storeValuesInTraversalDescriptor(tr, &(tr->coreLZ[0]));

    if (firstIteration) {
      makenewzIterative(tr);
      firstIteration = FALSE;
    }
    execCore(tr, dlnLdlz, d2lnLdlz2);

    {
      double send[2 * NUM_BRANCHES], recv[2 * NUM_BRANCHES];
      memcpy(&send[0], dlnLdlz, sizeof(double) * tr->numBranches);
      memcpy(&send[tr->numBranches], d2lnLdlz2,
             sizeof(double) * tr->numBranches);
      MPI_Allreduce(send, recv, tr->numBranches * 2, MPI_DOUBLE, MPI_SUM,
                    MPI_COMM_WORLD);
      memcpy(dlnLdlz, &recv[0], sizeof(double) * tr->numBranches);
      memcpy(d2lnLdlz2, &recv[tr->numBranches],
             sizeof(double) * tr->numBranches);
    }

    for (i = 0; i < tr->numBranches; i++) {
      if (outerConverged[i] == FALSE && tr->curvatOK[i] == FALSE) {
        if ((d2lnLdlz2[i] >= 0.0) && (z[i] < zmax))
          zprev[i] = z[i] = 0.37 * z[i] + 0.63; /* bad curvature */
        else
          tr->curvatOK[i] = TRUE;
      }
    }

    for (i = 0; i < tr->numBranches; i++) {
      if (tr->curvatOK[i] == TRUE && outerConverged[i] == FALSE) {
        if (d2lnLdlz2[i] < 0.0) {
          double tantmp = -dlnLdlz[i] / d2lnLdlz2[i];
          if (tantmp < 100) {
            z[i] *= exp(tantmp);
            if (z[i] < zmin) z[i] = zmin;
            if (z[i] > 0.25 * zprev[i] + 0.75)
              z[i] = 0.25 * zprev[i] + 0.75;
          } else
            z[i] = 0.25 * zprev[i] + 0.75;
        }
        if (z[i] > zmax) z[i] = zmax;

        maxiter[i] = maxiter[i] - 1;
        if (fabs(z[i] - zprev[i]) > zstep[i]) {
          if (maxiter[i] < -20) {
            z[i] = z0[i];
            outerConverged[i] = TRUE;
          } else
            outerConverged[i] = FALSE;
        } else
          outerConverged[i] = TRUE;
      }
    }

    loopConverged = TRUE;
    for (i = 0; i < tr->numBranches; i++)
      loopConverged = loopConverged && outerConverged[i];
  } while (!loopConverged);

  for (model = 0; model < tr->NumberOfModels; model++)
    tr->executeModel[model] = TRUE;

  for (i = 0; i < tr->numBranches; i++) result[i] = z[i];
}

/* makenewzGeneric (makenewzGenericSpecial.c:1355). */
void makenewzGeneric(tree *tr, nodeptr p, nodeptr q, double *z0, int maxiter,
                     double *result, boolean mask)
{
  int i;

  tr->td[0].ti[0].pNumber = p->number;
  tr->td[0].ti[0].qNumber = q->number;

  for (i = 0; i < tr->numBranches; i++) {
    tr->td[0].ti[0].qz[i] = z0[i];
    if (mask) {
      tr->executeModel[i] = tr->partitionConverged[i] ? FALSE : TRUE;
    } else
      assert(tr->executeModel[i]);
  }

  tr->td[0].count = 1;
  if (!p->x)
    computeTraversalInfo(p, &(tr->td[0].ti[0]), &(tr->td[0].count),
                         tr->mxtips, tr->numBranches, TRUE);
  if (!q->x)
    computeTraversalInfo(q, &(tr->td[0].ti[0]), &(tr->td[0].count),
                         tr->mxtips, tr->numBranches, TRUE);

  top_level_makenewz(tr, z0, maxiter, result);

  for (i = 0; i < tr->numBranches; i++) tr->executeModel[i] = TRUE;
}

/* evaluatePartialGeneric (evaluatePartialGenericSpecial.c:259): the
 * CAT/PSR per-site rate probe (optimizeModel.c:1868-1892) — a single
 * site's lnL at an arbitrary rate ki over the CURRENT tree.  Host math,
 * as in the reference: a full traversal descriptor of the tree is built
 * locally (not via tr->td, which belongs to the optimizer's own calls)
 * and handed to the single-site recursion. */
double evaluatePartialGeneric(tree *tr, int i, double ki, int _model)
{
  static traversalInfo *ti = NULL;
  static examl_hip_trav_entry *pops = NULL;
  ShimPart *p;
  pInfo *pd;
  nodeptr start, back;
  int count = 0, n, e;

  shim_init(tr);
  p = &S[_model];
  pd = &tr->partitionData[_model];
  if (ti == NULL) {
    ti = (traversalInfo *)malloc(sizeof(traversalInfo) * tr->mxtips);
    pops = (examl_hip_trav_entry *)malloc(sizeof(examl_hip_trav_entry) *
                                          tr->mxtips);
  }
  start = tr->start; /* a tip (axml.c) */
  back = start->back;
  computeTraversalInfo(back, ti, &count, tr->mxtips, tr->numBranches,
                       FALSE);
  n = count;
  for (e = 0; e < n; e++) {
    examl_hip_trav_entry *o = &pops[e];
    o->tipCase = ti[e].tipCase;
    o->pNumber = ti[e].pNumber;
    o->qNumber = ti[e].qNumber;
    o->rNumber = ti[e].rNumber;
    o->x3Slot = ti[e].pNumber - tr->mxtips - 1;
    o->x1Slot = (ti[e].tipCase != INNER_INNER)
                    ? ti[e].qNumber
                    : ti[e].qNumber - tr->mxtips - 1;
    o->x2Slot = (ti[e].tipCase == TIP_TIP)
                    ? ti[e].rNumber
                    : ti[e].rNumber - tr->mxtips - 1;
    o->qz = ti[e].qz[0];
    o->rz = ti[e].rz[0];
  }
  if (p->states == 4)
    return examl_host_evaluate_partial_dna_cat(
        pops, n, start->number, back->number, start->z[0], i, ki,
        pd->wgt[i], pd->EIGN, pd->EI, pd->EV, pd->tipVector, p->h_tips,
        p->width, tr->mxtips);
  return examl_host_evaluate_partial_prot_cat(
      pops, n, start->number, back->number, start->z[0], i, ki,
      pd->wgt[i], pd->EIGN, pd->EI, pd->EV, pd->tipVector, p->h_tips,
      p->width, tr->mxtips);
}
