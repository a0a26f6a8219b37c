"""How the 12-taxon golden fixtures were generated (not run in CI).

Alignments: random-walk sequences (a root sequence mutated at 15% per
branch of a random join order) so the ML search has real signal:
  12.phy    12 x 400 DNA   (numpy default_rng(42))
  12aa.phy  12 x 400 AA    (numpy default_rng(77))
Starting tree 12.tree: examl_amd PhyloTree.random(12, seed=7), taxa
T01..T12, topology-only newick.

Binaries, via the reference's own parser (oracle/_ref/parse-examl):
  12.binary    parse-examl -s 12.phy -m DNA -n 12
  12m.binary   parse-examl -s 12.phy   -q dna2.part -m DNA  -n 12m
  12aa.binary  parse-examl -s 12aa.phy -q aa2.part  -m PROT -n 12aa
  12lg4.binary parse-examl -s 12aa.phy -q lg4.part  -m PROT -n 12lg4
with partition files
  dna2.part: DNA, p1 = 1-200 / DNA, p2 = 201-400
  aa2.part:  WAG, p1 = 1-200 / JTT, p2 = 201-400
  lg4.part:  LG4X, p1 = 1-200 / LG4M, p2 = 201-400

Goldens (reference examl-AVX on these inputs):
  -f E GAMMA               -3650.993621
  -f E PSR                 -3233.904617
  -f E GAMMA -M            -3634.341296
  -f E PSR -M              -3219.085042
  -f E GAMMA (WAG+JTT)     -7246.699416
  -f E GAMMA (LG4X+LG4M)   -7387.472983
  -f d -D                  -2741.473102  (+ 12.result.tree, RF trajectory
                            "fast cycle 0->1 rrf 0.555556")
"""
