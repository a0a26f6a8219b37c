/* oracle/ref_shim.c — TEST INFRASTRUCTURE ONLY (part of oracle/_ref).
 *
 * Link shim that lets the reference's kernel translation units
 * (/root/reference/examl/{avxLikelihood,newviewGenericSpecial,
 * evaluateGenericSpecial,makenewzGenericSpecial,models}.c, compiled in place
 * by oracle/Makefile) be loaded as a shared library so tests can call the
 * reference kernels directly for golden-vector generation.  No reference
 * SOURCE is copied: the global data tables come from including the
 * reference's own globalVariables.h in place, and every function that the
 * exercised kernels never reach is an abort() stub.
 */

#include <stdio.h>
#include <stdlib.h>
#include <string.h>

/* Pull in the reference's global data definitions (mask32, bitVectorIdentity,
 * bitVectorAA, ...) exactly as examl/axml.c does. */
#include "axml.h"
#include "globalVariables.h"

/* axml.c:142 (malloc_aligned) — faithful reimplementation */
void *malloc_aligned(size_t size) {
  void *ptr = NULL;
  if (posix_memalign(&ptr, BYTE_ALIGNMENT, size) != 0) {
    fprintf(stderr, "malloc_aligned failed\n");
    abort();
  }
  return ptr;
}

/* axml.c:301 — faithful reimplementation (CAT=1, GAMMA=4) */
size_t discreteRateCategories(int rateHetModel) {
  return (rateHetModel == CAT) ? 1 : 4;
}

/* axml.c getBitVector/getUndetermined — faithful for DNA/AA only */
const unsigned int *getBitVector(int dataType) {
  if (dataType == DNA_DATA) return bitVectorIdentity;
  if (dataType == AA_DATA) return bitVectorAA;
  abort();
}

int getUndetermined(int dataType) {
  if (dataType == DNA_DATA) return 15;
  if (dataType == AA_DATA) return 22;
  abort();
}

/* None of these are reachable from the kernel entry points the golden
 * generator calls (newviewGTRGAMMA_AVX, newviewGTRGAMMAPROT_AVX,
 * evaluateGTRGAMMA[PROT], sumGAMMA[PROT], coreGTRGAMMA[PROT], makeP,
 * calcDiagptable, initGeneric, makeGammaCats). */
void getxnode(nodeptr p) { (void)p; abort(); }
/* axml.c isTip — faithful (needed by evaluatePartialGTRCAT's assert) */
int isTip(int number, int maxTips) {
  return number > 0 && number <= maxTips;
}
void checkPerSiteRates(const tree *const tr) { (void)tr; abort(); }
void storeExecuteMaskInTraversalDescriptor(tree *tr) { (void)tr; abort(); }
void storeValuesInTraversalDescriptor(tree *tr, double *v) {
  (void)tr;
  (void)v;
  abort();
}
void scaleLG4X_EIGN(tree *tr, int model) { (void)tr; (void)model; abort(); }
unsigned int precomputed16_bitcount(unsigned int n, char *b) {
  (void)n;
  (void)b;
  abort();
}
