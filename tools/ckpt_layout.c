/* TEST INFRASTRUCTURE: prints the reference's checkpoint binary layout
   (struct offsets/sizes on this x86-64 gcc ABI) so the Python reader in
   examl_amd/checkpoint.py can parse reference checkpoints.  Includes the
   reference's axml.h in place; no source copied. */
#include <stdio.h>
#include <stddef.h>
#include "axml.h"

#define P(s, f) printf("%s.%s off=%zu size=%zu\n", #s, #f, offsetof(s, f), sizeof(((s*)0)->f))

int main(int argc, char *argv[]) {
  (void)argc; (void)argv;
  printf("sizeof(checkPointState)=%zu\n", sizeof(checkPointState));
  printf("sizeof(node)=%zu\n", sizeof(node));
  printf("sizeof(commandLine)=%zu\n", sizeof(commandLine));
  P(checkPointState, state);
  P(checkPointState, accumulatedTime);
  P(checkPointState, tr_likelihood);
  P(checkPointState, optimizeRateCategoryInvocations);
  P(checkPointState, catOpt);
  P(checkPointState, treeIteration);
  P(checkPointState, seed);
  P(checkPointState, quartetCounter);
  P(checkPointState, filePosition);
  P(checkPointState, quartetFileName);
  P(checkPointState, cmd);
  P(checkPointState, constraintTree);
  P(checkPointState, tr_NumberOfCategories);
  P(commandLine, useMedian);
  P(commandLine, perGeneBranchLengths);
  P(commandLine, likelihoodEpsilon);
  P(commandLine, categories);
  P(commandLine, mode);
  P(commandLine, rateHetModel);
  P(node, z);
  P(node, next);
  P(node, back);
  P(node, hash);
  P(node, number);
  P(node, x);
  return 0;
}
/* second entry point: pLengths rows (compiled with -DPLENGTHS via main2) */
