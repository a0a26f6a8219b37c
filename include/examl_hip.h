/* ============================================================================
 * examl_hip.h — C-ABI boundary of the MI355X-native ExaML likelihood core.
 *
 * This is the drop-in surface replacing ExaML's L0/L1 likelihood kernels
 * (SURVEY.md §8b).  Each entry point cites the reference function it
 * replaces (file:line into the upstream ExaML tree).  The host side of
 * ExaML (searchAlgo.c / optimizeModel.c, plain C) drives these through the
 * same call shapes its own dispatch layer uses; INTEGRATION.md shows the
 * binding a maintainer would add.
 *
 * Conventions:
 *   - "dev_" prefixed pointers are DEVICE (HIP) pointers; all others are
 *     host pointers.  No torch types anywhere.
 *   - `stream` is a hipStream_t passed as void* (0 = default stream).
 *   - All launches are stream-ordered and asynchronous; scalar results land
 *     in device buffers the caller reads back (or feeds to RCCL).
 *   - Return 0 on success, nonzero HIP error code otherwise;
 *     examl_hip_last_error_string() describes the last failure.
 *   - Memory layout is the reference's: CLV x[site*span + cat*states + state]
 *     fp64 with span = 4*states (GAMMA); P-matrices left/right
 *     [cat*states^2 + row*states + col]; tipVector[code*states + state];
 *     per-node scaler counts as unsigned int (axml.h:601 globalScaler).
 * ==========================================================================*/

#ifndef EXAML_HIP_H
#define EXAML_HIP_H

#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* tipCase values — reference examl/axml.h:302-304 */
#define EXAML_TIP_TIP 0
#define EXAML_TIP_INNER 1
#define EXAML_INNER_INNER 2

const char *examl_hip_version(void);
const char *examl_hip_last_error_string(void);

/* Kernel-time profiling for the bench's roofline leg: when enabled, every
 * newview launch of the traversal executor is bracketed with hipEvent pairs
 * on its launch stream; _get synchronizes outstanding pairs and returns the
 * accumulated {ms, launch count} per tipCase (index = EXAML_TIP_*). */
void examl_hip_profile_enable(int on);
void examl_hip_profile_reset(void);
void examl_hip_profile_get(double *ms_by_tc, long *cnt_by_tc);

/* hipGraph replay of repeated traversal shapes (on by default; bypassed
 * while profiling is enabled).  examl_hip_graphs_clear drops all cached
 * executable graphs (e.g. before freeing the device buffers they bind). */
void examl_hip_use_graphs(int on);
void examl_hip_graphs_clear(void);

/* Opt-in fused-multiply-add variant of the protein newview kernel (the
 * reference's own _FMA build class, avxLikelihood.c:17-19): ~1 ulp/op
 * difference vs the default bit-exact no-FMA path, roughly half the VALU
 * instructions.  Off by default; parity tests run with it off. */
void examl_hip_fast_math(int on);

/* ---------------------------------------------------------------------------
 * Host-side model math (runs once per model-parameter change; feeds the
 * kernels).  These replace the corresponding host functions in the
 * reference and keep bit-identical arithmetic.
 * ------------------------------------------------------------------------ */

/* replaces makeP (examl/newviewGenericSpecial.c:78); z1/z2 are
 * log-transformed branch lengths (caller clamps to zmin and takes log as
 * newviewGenericSpecial.c:982-983 does). */
void examl_host_make_p(double z1, double z2, const double *rates,
                       const double *EI, const double *EIGN, int numCats,
                       double *left, double *right, int states);

/* replaces calcDiagptable (examl/evaluateGenericSpecial.c:80); z is the raw
 * branch length (clamp+log inside). */
void examl_host_calc_diagptable(double z, int states, int numCats,
                                const double *rates, const double *EIGN,
                                double *diag);

/* d0/d1/d2 tables of coreGTRGAMMA (examl/makenewzGenericSpecial.c:2330-2346)
 * for states=4: out48 = {d0[16], d1[16], d2[16]}. */
void examl_host_core_dtables_dna(const double *EIGN, const double *gammaRates,
                                 double lz, double *out48);

/* replaces initReversibleGTR -> initGeneric (examl/models.c:3462/3234) for
 * DNA (states=4, 16 ambiguity codes): emits EIGN[4], EV[16], EI[16],
 * tipVector[64]. */
void examl_host_init_gtr_dna(const double *frequencies, const double *rates6,
                             double *EIGN, double *EV, double *EI,
                             double *tipVector);

/* replaces makeGammaCats (examl/models.c:3795), mean-rate (useMedian=FALSE)
 * discrete gamma. */
void examl_host_make_gamma_cats(double alpha, double *gammaRates, int K);

/* ---------------------------------------------------------------------------
 * L0 device kernels (one launch each).  DNA GTRGAMMA: states=4, span=16.
 * ------------------------------------------------------------------------ */

/* replaces newviewGTRGAMMA_AVX (examl/avxLikelihood.c:64, decl
 * examl/axml.h:1360): x3 = EV . ((P_L x1) o (P_R x2)) per (site, gamma cat),
 * with the reference's exact 2^-256 underflow rescale rule (all 16 span
 * entries below threshold; no scaling in TIP_TIP) and scaler counts
 * accumulated as sum of wgt[site] into *dev_scalerInc (device, caller
 * zeroes). x1/x2/x3/tipX1/tipX2/left/right/wgt/EV/tipVector are device
 * pointers. */
int examl_hip_newview_dna_gamma(int tipCase, const double *dev_x1,
                                const double *dev_x2, double *dev_x3,
                                const double *dev_EV,
                                const double *dev_tipVector,
                                const unsigned char *dev_tipX1,
                                const unsigned char *dev_tipX2, long n,
                                const double *dev_left,
                                const double *dev_right, const int *dev_wgt,
                                unsigned int *dev_scalerInc, void *stream);

/* replaces evaluateGTRGAMMA (examl/evaluateGenericSpecial.c:1879) plus the
 * scaler undo at :830: atomically accumulates
 *   sum_i wgt[i]*log(0.25*|term_i|)  +  (gs_p + gs_q) * log_minlik
 * into *dev_lnl (caller zeroes).  dev_tipX1 == NULL selects the
 * inner-inner body.  dev_gsP/dev_gsQ point at the two nodes' scaler counts
 * (device; pass NULL,NULL to skip the undo term). */
int examl_hip_evaluate_dna_gamma(const int *dev_wgt, const double *dev_x1,
                                 const double *dev_x2,
                                 const double *dev_tipVector,
                                 const unsigned char *dev_tipX1, long n,
                                 const double *dev_diag,
                                 const unsigned int *dev_gsP,
                                 const unsigned int *dev_gsQ,
                                 double log_minlik, double *dev_partials,
                                 double *dev_lnl, void *stream);

/* replaces sumGAMMA (examl/makenewzGenericSpecial.c:1798):
 * dev_sum[i,c,k] = x1'[i,c,k] * x2'[i,c,k] with tip expansion. */
int examl_hip_sum_dna_gamma(int tipCase, double *dev_sum,
                            const double *dev_x1, const double *dev_x2,
                            const double *dev_tipVector,
                            const unsigned char *dev_tipX1,
                            const unsigned char *dev_tipX2, long n,
                            void *stream);

/* replaces coreGTRGAMMA (examl/makenewzGenericSpecial.c:2309): accumulates
 * {dlnLdlz, d2lnLdlz2} into dev_out2[2] (caller zeroes).  dev_dtables holds
 * the 48 doubles from examl_host_core_dtables_dna, uploaded by the caller. */
int examl_hip_core_dna_gamma(long n, const double *dev_sum,
                             const double *dev_dtables, const int *dev_wgt,
                             double *dev_partials, double *dev_out2,
                             void *stream);

/* ---------------------------------------------------------------------------
 * L1 batched executors (the newviewIterative-shaped path: one host call per
 * traversal, kernels chained on `stream`).
 * ------------------------------------------------------------------------ */

/* One post-order CLV update op — the traversalInfo entry of
 * examl/axml.h:434-442 with CLV slot / tip-row bindings resolved by the
 * caller (the xVector/yVector indexing of newviewIterative,
 * examl/newviewGenericSpecial.c:1221-1261). */
typedef struct {
  int tipCase;           /* EXAML_TIP_* */
  int pNumber, qNumber, rNumber; /* node ids, index dev_scalers */
  int x1Slot, x2Slot, x3Slot;    /* CLV slots; for tip operands the slot is
                                    the tip row in dev_tips (x1Slot for q's
                                    tip, x2Slot for r when TIP_TIP) */
  double qz, rz;                 /* raw branch lengths */
} examl_hip_trav_entry;

/* replaces the per-partition body of newviewIterative
 * (examl/newviewGenericSpecial.c:917): computes all P-matrix pairs on the
 * host (makeP), uploads them in one copy, launches one newview kernel per
 * entry in post-order, then one finalize kernel that applies the recursive
 * scaler accumulation globalScaler[p] = gs[q] + gs[r] + inc
 * (newviewGenericSpecial.c:1503-1510; gs of tip nodes stays 0).
 * dev_pbuf: >= numOps*128 doubles; dev_inc: >= numOps uints (zeroed here). */
int examl_hip_newview_traversal_dna_gamma(
    const examl_hip_trav_entry *ops, int numOps, const double *EIGN,
    const double *EI, const double *gammaRates, const double *dev_EV,
    const double *dev_tipVector, double *dev_clv, long clvStrideDoubles,
    const unsigned char *dev_tips, long tipStrideBytes, const int *dev_wgt,
    long n, unsigned int *dev_scalers, unsigned int *dev_inc,
    double *dev_pbuf, void *stream);

/* replaces the per-partition body of evaluateIterative
 * (examl/evaluateGenericSpecial.c:403): calcDiagptable on the host for the
 * root branch z, upload, then the evaluate kernel including the scaler
 * undo.  rootTipCase: EXAML_TIP_INNER (tipSlot = tip row of the tip node)
 * or EXAML_INNER_INNER.  dev_diag_scratch: >= 16 doubles.  *dev_lnl is
 * accumulated (caller zeroes; feed to RCCL all-reduce afterwards —
 * replaces the MPI_Allreduce at evaluateGenericSpecial.c:969). */
int examl_hip_evaluate_root_dna_gamma(
    int rootTipCase, int pNumber, int qNumber, int x1Slot, int x2Slot,
    int tipSlot, double z, const double *EIGN, const double *gammaRates,
    const double *dev_tipVector, double *dev_clv, long clvStrideDoubles,
    const unsigned char *dev_tips, long tipStrideBytes, const int *dev_wgt,
    long n, const unsigned int *dev_scalers, double *dev_diag_scratch,
    double *dev_partials, double *dev_lnl, void *stream);

/* replaces the per-partition body of makenewzIterative's sum precompute
 * (examl/makenewzGenericSpecial.c:628,673-839) for the branch p--q. */
int examl_hip_sum_root_dna_gamma(int rootTipCase, int x1Slot, int x2Slot,
                                 int tipSlot, int tipSlot2,
                                 const double *dev_tipVector, double *dev_clv,
                                 long clvStrideDoubles,
                                 const unsigned char *dev_tips,
                                 long tipStrideBytes, double *dev_sum, long n,
                                 void *stream);

/* replaces the per-partition body of execCore
 * (examl/makenewzGenericSpecial.c:849): host dtables + upload + core
 * kernel; dev_dtab_scratch >= 48 doubles; dev_out2 accumulated (caller
 * zeroes; reduce across ranks afterwards — replaces the MPI_Allreduce at
 * makenewzGenericSpecial.c:1244). */
int examl_hip_core_root_dna_gamma(long n, const double *dev_sum,
                                  const double *EIGN,
                                  const double *gammaRates, double lz,
                                  const int *dev_wgt,
                                  double *dev_dtab_scratch,
                                  double *dev_partials, double *dev_out2,
                                  void *stream);

/* ---------------------------------------------------------------------------
 * DNA CAT (PSR, -m PSR) surface — span 4, per-site rate category cptr[i],
 * numCats <= 25 rate categories (maxCategories).  Each function replaces:
 *   newview  — newviewGTRCAT_AVX (examl/avxLikelihood.c:326)
 *   evaluate — evaluateGTRCAT (examl/evaluateGenericSpecial.c:1988; no 0.25)
 *   sum      — sumCAT (examl/makenewzGenericSpecial.c:1850)
 *   core     — coreGTRCAT (examl/makenewzGenericSpecial.c:2402)
 * P blocks are numCats*32 doubles (left|right); diag numCats*4; the core
 * dtable scratch is numCats*4 + 8 + numCats doubles.  The CAT traversal
 * executor and the optimizeRateCategories/evaluatePartialGeneric host loop
 * are scheduled for round 2.
 * ------------------------------------------------------------------------ */

void examl_host_core_dtables_dna_cat(const double *EIGN, const double *rptr,
                                     int numCats, double lz, double *out);

int examl_hip_newview_dna_cat(int tipCase, const double *dev_EV,
                              const int *dev_cptr, const double *dev_x1,
                              const double *dev_x2, double *dev_x3,
                              const double *dev_tipVector,
                              const unsigned char *dev_tipX1,
                              const unsigned char *dev_tipX2, long n,
                              const double *dev_P, int numCats,
                              const int *dev_wgt,
                              unsigned int *dev_scalerInc, void *stream);

int examl_hip_evaluate_dna_cat(const int *dev_cptr, const int *dev_wgt,
                               const double *dev_x1, const double *dev_x2,
                               const double *dev_tipVector,
                               const unsigned char *dev_tipX1, long n,
                               const double *dev_diag, int numCats,
                               const unsigned int *dev_gsP,
                               const unsigned int *dev_gsQ, double log_minlik,
                               double *dev_partials, double *dev_lnl,
                               void *stream);

int examl_hip_sum_dna_cat(int tipCase, double *dev_sum, const double *dev_x1,
                          const double *dev_x2,
                          const double *dev_tipVector,
                          const unsigned char *dev_tipX1,
                          const unsigned char *dev_tipX2, long n,
                          void *stream);

int examl_hip_core_root_dna_cat(long n, const double *dev_sum,
                                const double *EIGN, const double *rptr,
                                int numCats, double lz, const int *dev_wgt,
                                const int *dev_cptr, double *dev_dtab_scratch,
                                double *dev_partials, double *dev_out2,
                                void *stream);

/* CAT executors (the newviewIterative/evaluateIterative/makenewzIterative
 * CAT bodies; dev_pbuf >= numOps*numCats*32 doubles, dev_diag >= numCats*4).
 * The optimizeRateCategories/evaluatePartialGeneric host loop is round-2. */
int examl_hip_newview_traversal_dna_cat(
    const examl_hip_trav_entry *ops, int numOps, const double *EIGN,
    const double *EI, const double *perSiteRates, int numCats,
    const double *dev_EV, const double *dev_tipVector, const int *dev_cptr,
    double *dev_clv, long clvStrideDoubles, const unsigned char *dev_tips,
    long tipStrideBytes, const int *dev_wgt, long n,
    unsigned int *dev_scalers, unsigned int *dev_inc, double *dev_pbuf,
    void *stream);

int examl_hip_evaluate_root_dna_cat_x(
    int rootTipCase, int pNumber, int qNumber, int x1Slot, int x2Slot,
    int tipSlot, double z, const double *EIGN, const double *perSiteRates,
    int numCats, const double *dev_tipVector, const int *dev_cptr,
    double *dev_clv, long clvStrideDoubles, const unsigned char *dev_tips,
    long tipStrideBytes, const int *dev_wgt, long n,
    const unsigned int *dev_scalers, double *dev_diag_scratch,
    double *dev_partials, double *dev_lnl, void *stream);

int examl_hip_sum_root_dna_cat(int rootTipCase, int x1Slot, int x2Slot,
                               int tipSlot, int tipSlot2,
                               const double *dev_tipVector, double *dev_clv,
                               long clvStrideDoubles,
                               const unsigned char *dev_tips,
                               long tipStrideBytes, double *dev_sum, long n,
                               void *stream);

/* replaces evaluatePartialGeneric for DNA CAT
 * (examl/evaluatePartialGenericSpecial.c:259): weighted single-site lnL at
 * an arbitrary rate ki over the LAST full traversal (ops excludes the root
 * entry).  Pure host math, as in the reference. */
double examl_host_evaluate_partial_dna_cat(
    const void *ops, int numOps, int rootTipNumber, int rootQNumber,
    double root_z, long site, double ki, int w, const double *EIGN,
    const double *EI, const double *EV, const double *tipVector,
    const unsigned char *tips, long tipStrideBytes, int mxtips);

/* ---------------------------------------------------------------------------
 * Protein (20-state) GTRGAMMA surface — span 80, tip codes 1..22.  Each
 * function replaces the 20-state counterpart of the DNA one above:
 *   newview  — newviewGTRGAMMAPROT_AVX (examl/avxLikelihood.c:1312)
 *   evaluate — evaluateGTRGAMMAPROT (examl/evaluateGenericSpecial.c:1393)
 *   sum      — sumGAMMAPROT (examl/makenewzGenericSpecial.c:2083)
 *   core     — coreGTRGAMMAPROT (examl/makenewzGenericSpecial.c:2581)
 * P blocks are 3200 doubles (left|right), diag 80, dtables 240.
 * ------------------------------------------------------------------------ */

/* replaces initReversibleGTR for AA_DATA/GTR (examl/models.c:3495 ->
 * initGeneric with bitVectorAA); rates190 = upper-triangle exchangeabilities
 * (e.g. the LG model, examl_amd/data/lg_model.npz). */
void examl_host_init_gtr_aa(const double *frequencies, const double *rates190,
                            double *EIGN, double *EV, double *EI,
                            double *tipVector);

void examl_host_core_dtables_prot(const double *EIGN,
                                  const double *gammaRates, double lz,
                                  double *out240);

int examl_hip_newview_prot_gamma(int tipCase, const double *dev_x1,
                                 const double *dev_x2, double *dev_x3,
                                 const double *dev_EV,
                                 const double *dev_tipVector,
                                 const unsigned char *dev_tipX1,
                                 const unsigned char *dev_tipX2, long n,
                                 const double *dev_left,
                                 const double *dev_right, const int *dev_wgt,
                                 unsigned int *dev_scalerInc, void *stream);

int examl_hip_evaluate_prot_gamma(const int *dev_wgt, const double *dev_x1,
                                  const double *dev_x2,
                                  const double *dev_tipVector,
                                  const unsigned char *dev_tipX1, long n,
                                  const double *dev_diag,
                                  const unsigned int *dev_gsP,
                                  const unsigned int *dev_gsQ,
                                  double log_minlik, double *dev_partials,
                                  double *dev_lnl, void *stream);

int examl_hip_sum_prot_gamma(int tipCase, double *dev_sum,
                             const double *dev_x1, const double *dev_x2,
                             const double *dev_tipVector,
                             const unsigned char *dev_tipX1,
                             const unsigned char *dev_tipX2, long n,
                             void *stream);

int examl_hip_core_prot_gamma(long n, const double *dev_sum,
                              const double *dev_dtables, const int *dev_wgt,
                              double *dev_partials, double *dev_out2,
                              void *stream);

int examl_hip_newview_traversal_prot_gamma(
    const examl_hip_trav_entry *ops, int numOps, const double *EIGN,
    const double *EI, const double *gammaRates, const double *dev_EV,
    const double *dev_tipVector, double *dev_clv, long clvStrideDoubles,
    const unsigned char *dev_tips, long tipStrideBytes, const int *dev_wgt,
    long n, unsigned int *dev_scalers, unsigned int *dev_inc,
    double *dev_pbuf, void *stream);

int examl_hip_evaluate_root_prot_gamma(
    int rootTipCase, int pNumber, int qNumber, int x1Slot, int x2Slot,
    int tipSlot, double z, const double *EIGN, const double *gammaRates,
    const double *dev_tipVector, double *dev_clv, long clvStrideDoubles,
    const unsigned char *dev_tips, long tipStrideBytes, const int *dev_wgt,
    long n, const unsigned int *dev_scalers, double *dev_diag_scratch,
    double *dev_partials, double *dev_lnl, void *stream);

int examl_hip_sum_root_prot_gamma(int rootTipCase, int x1Slot, int x2Slot,
                                  int tipSlot, int tipSlot2,
                                  const double *dev_tipVector,
                                  double *dev_clv, long clvStrideDoubles,
                                  const unsigned char *dev_tips,
                                  long tipStrideBytes, double *dev_sum,
                                  long n, void *stream);

int examl_hip_core_root_prot_gamma(long n, const double *dev_sum,
                                   const double *EIGN,
                                   const double *gammaRates, double lz,
                                   const int *dev_wgt,
                                   double *dev_dtab_scratch,
                                   double *dev_partials, double *dev_out2,
                                   void *stream);


/* ---- LG4 (LG4M/LG4X): per-gamma-category matrices --------------------- */
/* Host model math (model_prep.cpp): EIGN4 stride 20 (scaled), EI4 stride
 * 400, tipVector4 stride 460, EV4 stride 400. */
void examl_host_make_gamma_cats_median(double alpha, double *gammaRates,
                                       int K);
void examl_host_make_p_lg4(double z1, double z2, const double *gammaRates,
                           const double *EI4, const double *EIGN4,
                           double *left, double *right);
void examl_host_calc_diag_lg4(double z, const double *gammaRates,
                              const double *EIGN4, double *diag /*80*/);
void examl_host_core_dtables_prot_lg4(const double *EIGN4,
                                      const double *gammaRates, double lz,
                                      double *dtab /*240*/);

/* L1 executors (replacing the LG4 branches of newviewIterative /
 * evaluateIterative / makenewzIterative+execCore). */
int examl_hip_newview_traversal_prot_lg4(
    const void *ops, int numOps, const double *EIGN4, const double *EI4,
    const double *gammaRates, const double *dev_EV4,
    const double *dev_tipVec4, double *dev_clv, long clvStride,
    const unsigned char *dev_tips, long tipStride, const int *dev_wgt,
    long n, unsigned int *dev_scalers, unsigned int *dev_inc,
    double *dev_pbuf, void *stream);
int examl_hip_evaluate_root_prot_lg4(
    int rootTipCase, int pNumber, int qNumber, int x1Slot, int x2Slot,
    int tipSlot, double z, const double *EIGN4, const double *gammaRates,
    const double *weights, const double *dev_tipVec4, double *dev_clv,
    long clvStride, const unsigned char *dev_tips, long tipStride,
    const int *dev_wgt, long n, const unsigned int *dev_scalers,
    double *dev_diag /*>=84*/, double *dev_partials, double *dev_lnl,
    void *stream);
int examl_hip_sum_root_prot_lg4(
    int rootTipCase, int x1Slot, int x2Slot, int tipSlot, int tipSlot2,
    const double *dev_tipVec4, double *dev_clv, long clvStride,
    const unsigned char *dev_tips, long tipStride, double *dev_sum, long n,
    void *stream);
int examl_hip_core_root_prot_lg4(
    long n, const double *dev_sum, const double *EIGN4,
    const double *gammaRates, const double *weights, double lz,
    const int *dev_wgt, double *dev_dtab /*>=244*/, double *dev_partials,
    double *dev_out2, void *stream);


/* ---- -S (saveMemory, SEV) DNA GTRGAMMA: compacted CLVs + gap columns -- */
int examl_hip_gap_and_prefix(const unsigned int *g1, const unsigned int *g2,
                             unsigned int *g3, int *prefix /*gvl+1*/,
                             int gvl, long n, void *stream);
int examl_hip_newview_dna_save(
    int tipCase, const double *x1, const double *x2, double *x3,
    const double *P /*left|right, 128*/, const double *EV,
    const double *tipVec, const unsigned char *tipX1,
    const unsigned char *tipX2, const int *wgt, long n,
    unsigned int *scalerInc, const unsigned int *g1, const unsigned int *g2,
    const unsigned int *g3, const int *pre1, const int *pre2,
    const int *pre3, const double *x1_gapcol, const double *x2_gapcol,
    double *x3_gapcol, int *scaleGap, void *stream);
int examl_hip_evaluate_dna_save(
    int tipCase, const double *x1, const double *x2, const double *tipVec,
    const unsigned char *tipX1, const int *wgt, const double *diag, long n,
    const unsigned int *g1, const unsigned int *g2, const int *pre1,
    const int *pre2, const double *x1_gapcol, const double *x2_gapcol,
    int pNumber, int qNumber, const unsigned int *dev_scalers,
    double *dev_partials, double *dev_lnl, void *stream);
int examl_hip_sum_dna_save(
    int tipCase, double *dev_sum, const double *x1, const double *x2,
    const double *tipVec, const unsigned char *tipX1,
    const unsigned char *tipX2, long n, const unsigned int *g1,
    const unsigned int *g2, const int *pre1, const int *pre2,
    const double *x1_gapcol, const double *x2_gapcol, void *stream);


/* ---- -S protein GTRGAMMA + CAT families --------------------------------
 * Same compaction design as the DNA GAMMA SAVE entries above; each
 * replaces the corresponding *_GAPPED_SAVE reference kernel:
 *   newview prot  — newviewGTRGAMMAPROT_AVX_GAPPED_SAVE (avxLikelihood.c:3125)
 *   evaluate prot — evaluateGTRGAMMAPROT_GAPPED_SAVE (evaluateGenericSpecial.c:1291)
 *   sum prot      — sumGAMMAPROT_GAPPED_SAVE (makenewzGenericSpecial.c:1896)
 *   CAT (states 4/20) — newviewGTRCAT_AVX_GAPPED_SAVE (avxLikelihood.c:2306),
 *   newviewGTRCATPROT_AVX_GAPPED_SAVE (:2607) + their evaluate/sum twins.
 * examl_host_make_p_save adds the saveMem rate-1.0 P pair at slot maxCats
 * (makeP's saveMem branch, newviewGenericSpecial.c:140-165); CAT P blocks
 * are [(maxCats+1)*S^2 left | (maxCats+1)*S^2 right]. */
void examl_host_make_p_save(double z1, double z2, const double *rptr,
                            const double *EI, const double *EIGN,
                            int numCats, double *left, double *right,
                            int maxCats, int states);
int examl_hip_newview_prot_save(
    int tipCase, const double *x1, const double *x2, double *x3,
    const double *P /*3200*/, const double *EV, const double *tipVec,
    const unsigned char *tipX1, const unsigned char *tipX2, const int *wgt,
    long n, unsigned int *scalerInc, const unsigned int *g1,
    const unsigned int *g2, const unsigned int *g3, const int *pre1,
    const int *pre2, const int *pre3, const double *x1_gapcol,
    const double *x2_gapcol, double *x3_gapcol, int *scaleGap, void *stream);
int examl_hip_evaluate_prot_save(
    int tipCase, const double *x1, const double *x2, const double *tipVec,
    const unsigned char *tipX1, const int *wgt, const double *diag, long n,
    const unsigned int *g1, const unsigned int *g2, const int *pre1,
    const int *pre2, const double *x1_gapcol, const double *x2_gapcol,
    int pNumber, int qNumber, const unsigned int *dev_scalers,
    double *dev_partials, double *dev_lnl, void *stream);
int examl_hip_sum_prot_save(
    int tipCase, double *dev_sum, const double *x1, const double *x2,
    const double *tipVec, const unsigned char *tipX1,
    const unsigned char *tipX2, long n, const unsigned int *g1,
    const unsigned int *g2, const int *pre1, const int *pre2,
    const double *x1_gapcol, const double *x2_gapcol, void *stream);
int examl_hip_newview_cat_save(
    int states, int tipCase, const double *EV, const int *cptr,
    const double *x1, const double *x2, double *x3, const double *tipVec,
    const unsigned char *tipX1, const unsigned char *tipX2, const int *wgt,
    long n, const double *P, int maxCats, unsigned int *scalerInc,
    const unsigned int *g1, const unsigned int *g2, const unsigned int *g3,
    const int *pre1, const int *pre2, const int *pre3,
    const double *x1_gapcol, const double *x2_gapcol, double *x3_gapcol,
    int *scaleGap, void *stream);
int examl_hip_evaluate_cat_save(
    int states, const int *cptr, const int *wgt, const double *x1,
    const double *x2, const double *tipVec, const unsigned char *tipX1,
    long n, const double *diag, const unsigned int *g1,
    const unsigned int *g2, const int *pre1, const int *pre2,
    const double *x1_gapcol, const double *x2_gapcol, int pNumber,
    int qNumber, const unsigned int *dev_scalers, double *dev_partials,
    double *dev_lnl, void *stream);
int examl_hip_sum_cat_save(
    int states, int tipCase, double *dev_sum, const double *x1,
    const double *x2, const double *tipVec, const unsigned char *tipX1,
    const unsigned char *tipX2, long n, const unsigned int *g1,
    const unsigned int *g2, const int *pre1, const int *pre2,
    const double *x1_gapcol, const double *x2_gapcol, void *stream);

/* ---- Protein CAT (-m PSR on AA partitions) ---------------------------- */
double examl_host_evaluate_partial_prot_cat(
    const void *ops, int numOps, int rootTipNumber, int rootQNumber,
    double root_z, long site, double ki, int w, const double *EIGN,
    const double *EI, const double *EV, const double *tipVector,
    const unsigned char *tips, long tipStride, int mxtips);
void examl_host_core_dtables_prot_cat(const double *EIGN, const double *rptr,
                                      int numCats, double lz,
                                      double *dtab /*cats*20+40+cats*/);
int examl_hip_newview_traversal_prot_cat(
    const examl_hip_trav_entry *ops, int numOps, const double *EIGN,
    const double *EI, const double *perSiteRates, int numCats,
    const double *dev_EV, const double *dev_tipVec, const int *dev_cptr,
    double *dev_clv, long clvStride, const unsigned char *dev_tips,
    long tipStride, const int *dev_wgt, long n, unsigned int *dev_scalers,
    unsigned int *dev_inc, double *dev_pbuf, void *stream);
int examl_hip_evaluate_root_prot_cat(
    int rootTipCase, int pNumber, int qNumber, int x1Slot, int x2Slot,
    int tipSlot, double z, const double *EIGN, const double *perSiteRates,
    int numCats, const double *dev_tipVec, const int *dev_cptr,
    double *dev_clv, long clvStride, const unsigned char *dev_tips,
    long tipStride, const int *dev_wgt, long n,
    const unsigned int *dev_scalers, double *dev_diag, double *dev_partials,
    double *dev_lnl, void *stream);
int examl_hip_sum_root_prot_cat(
    int rootTipCase, int x1Slot, int x2Slot, int tipSlot, int tipSlot2,
    const double *dev_tipVec, double *dev_clv, long clvStride,
    const unsigned char *dev_tips, long tipStride, double *dev_sum, long n,
    void *stream);
int examl_hip_core_root_prot_cat(
    long n, const double *dev_sum, const double *EIGN, const double *rptr,
    int numCats, double lz, const int *dev_wgt, const int *dev_cptr,
    double *dev_dtab, double *dev_partials, double *dev_out2, void *stream);

/* ---------------------------------------------------------------------------
 * Multi-partition fused executors: one launch per (traversal level x
 * tipCase) covering ALL partitions, replacing the per-(partition, op)
 * launch loop for partitioned data (newviewIterative's partition loop,
 * newviewGenericSpecial.c:1064; execCore's, makenewzGenericSpecial.c:885).
 * P matrices are computed ON DEVICE from per-partition EIGN/EI/rates and
 * the ops' branch lengths, so the fused path agrees with the
 * single-partition executors to <=1e-11 relative (device exp vs libm),
 * not bit-exact.  activeMask (host, numParts bytes, NULL = all) carries
 * the executeModel gating; masked partitions' CLVs/scalers/outputs are
 * untouched.  qzOv/rzOv (numOps*numParts, NULL = use ops[].qz/rz) carry
 * per-partition branch lengths under -M.
 * ------------------------------------------------------------------------ */

int examl_hip_multi_create(
    int states, int numParts, const long *widths, double *const *dev_clvs,
    const long *clvStrides, const unsigned char *const *dev_tips,
    const long *tipStrides, const int *const *dev_wgts,
    unsigned int *const *dev_scalers, const double *const *dev_EVs,
    const double *const *dev_tipVecs, int maxOps, void **out);
void examl_hip_multi_destroy(void *h);
int examl_hip_newview_traversal_multi(
    void *h, const examl_hip_trav_entry *ops, int numOps,
    const double *const *EIGNs, const double *const *EIs,
    const double *const *gammaRates, const unsigned char *activeMask,
    const double *qzOv, const double *rzOv, void *stream);
int examl_hip_evaluate_root_multi(
    void *h, int rootTipCase, int pNumber, int qNumber, int x1Slot,
    int x2Slot, int tipSlot, const double *zs, int zPerPart,
    const double *const *EIGNs, const double *const *gammaRates,
    const unsigned char *activeMask, double *dev_lnl /* numParts, caller
    zeroes */, void *stream);
int examl_hip_sum_root_multi(void *h, int rootTipCase, int x1Slot,
                             int x2Slot, int tipSlot, int tipSlot2,
                             const unsigned char *activeMask, void *stream);
int examl_hip_core_root_multi(
    void *h, const double *lzs, int lzPerPart, const double *const *EIGNs,
    const double *const *gammaRates, const unsigned char *activeMask,
    double *dev_out2 /* 2*numParts, caller zeroes */, void *stream);

#ifdef __cplusplus
}
#endif

#endif /* EXAML_HIP_H */
