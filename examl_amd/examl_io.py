"""Readers for ExaML's on-disk formats: the parse-examl binary alignment
(examl/byteFile.c) and topology-only Newick trees (examl/treeIO.c
treeReadLen semantics: missing branch lengths default to z = 0.9)."""

import struct

import numpy as np

from .tree import DEFAULTZ, PhyloTree

BYTEFILE_VERSION = 3022  # versionHeader/version.h:3
BYTEFILE_MAGIC = 6517718  # byteFile.c:134

DNA_DATA = 2  # axml.h dataType enum (DNA_DATA)
AA_DATA = 3


class BytePartition:
    pass


def read_byte_file(path):
    """Parse a .binary alignment (byteFile.c layout:
    [3-int header][weights][taxa][partitions][alignment bytes]).
    Returns (taxa_names, partitions); each partition carries its full
    site range (an un-sharded single-rank read, readMyData with one
    assignment covering lower..upper)."""
    with open(path, "rb") as f:
        data = f.read()
    off = 0

    def rd(fmt):
        nonlocal off
        vals = struct.unpack_from("<" + fmt, data, off)
        off += struct.calcsize("<" + fmt)
        return vals if len(vals) > 1 else vals[0]

    size_of_size_t = rd("i")
    assert size_of_size_t == 8, "byte file from a 32-bit parser"
    version = rd("i")
    assert version == BYTEFILE_VERSION, f"parser version {version}"
    magic = rd("i")
    assert magic == BYTEFILE_MAGIC

    num_tax = rd("i")
    num_pattern = rd("q")  # size_t
    num_partitions = rd("i")
    rd("d")  # gappyness

    weights = np.frombuffer(data, dtype=np.int32, count=num_pattern,
                            offset=off).copy()
    off += 4 * num_pattern

    taxa = []
    for _ in range(num_tax):
        ln = rd("i")
        taxa.append(data[off:off + ln].split(b"\0")[0].decode())
        off += ln

    parts = []
    for _ in range(num_partitions):
        p = BytePartition()
        p.states = rd("i")
        p.maxTipStates = rd("i")
        p.lower = rd("q")
        p.upper = rd("q")
        rd("q")  # width (unused, byteFile.c:228)
        p.dataType = rd("i")
        p.protModels = rd("i")
        p.protFreqs = rd("i")
        p.nonGTR = rd("i")
        p.optimizeBaseFrequencies = rd("i")
        ln = rd("I")
        p.name = data[off:off + ln].split(b"\0")[0].decode()
        off += ln
        p.frequencies = np.frombuffer(data, dtype=np.float64,
                                      count=p.states, offset=off).copy()
        off += 8 * p.states
        parts.append(p)

    # alignment: partition-major, per taxon rows of (upper-lower) bytes
    # (byteFile.c:318-345, new layout)
    aln_pos = off
    for p in parts:
        width = p.upper - p.lower
        p.width = width
        tips = np.zeros((num_tax + 1, width), dtype=np.uint8)
        base = aln_pos + p.lower * num_tax
        for j in range(1, num_tax + 1):
            o = base + (j - 1) * width
            tips[j] = np.frombuffer(data, dtype=np.uint8, count=width,
                                    offset=o)
        p.tips = tips
        p.wgt = weights[p.lower:p.upper].copy()
    return taxa, parts


def read_newick_trees(path, taxa_names):
    """All trees in the file (one per line — the -f E/-f e multi-tree
    input, getNumberOfTrees/optimizeTrees)."""
    with open(path) as f:
        lines = [ln.strip() for ln in f if ln.strip()]
    return [parse_newick_topology(ln, taxa_names) for ln in lines]


def read_newick_topology(path, taxa_names):
    with open(path) as f:
        s = f.read().strip().splitlines()[0]
    return parse_newick_topology(s, taxa_names)


def parse_newick_topology(s, taxa_names, read_bl=False):
    """Parse a (possibly multifurcating-root) Newick tree over the given
    taxa into a PhyloTree, branch lengths defaulting to z = 0.9 (defaultz,
    treeReadLen behavior for topology-only trees); with read_bl the
    branch lengths are read back as z = exp(-bl) (the inverse of
    getBranchLength, treeIO.c:176).  Ring order of inner nodes follows
    parse order (the reference's p->next chain)."""
    s = s.strip()
    if s.endswith(";"):
        s = s[:-1]
    name_to_id = {n: i + 1 for i, n in enumerate(taxa_names)}
    ntips = len(taxa_names)
    tree = PhyloTree(ntips)
    next_inner = [ntips + 1]

    pos = [0]

    def parse():
        """returns (node_id, z) of a subtree"""
        if s[pos[0]] == "(":
            pos[0] += 1
            children = []
            while True:
                children.append(parse())
                if s[pos[0]] == ",":
                    pos[0] += 1
                    continue
                assert s[pos[0]] == ")"
                pos[0] += 1
                break
            # optional label / branch length
            z = _read_label_bl()
            node = next_inner[0]
            next_inner[0] += 1
            for (c, cz) in children:
                tree.add_edge(node, c, cz)
            return node, z
        else:
            j = pos[0]
            while s[j] not in ",():;":
                j += 1
            name = s[pos[0]:j]
            pos[0] = j
            z = _read_label_bl()
            return name_to_id[name], z

    def _read_label_bl():
        z = DEFAULTZ
        # skip inner label
        j = pos[0]
        while j < len(s) and s[j] not in ",():;":
            j += 1
        pos[0] = j
        if j < len(s) and s[j] == ":":
            j += 1
            k = j
            while k < len(s) and s[k] not in ",();":
                k += 1
            # branch length present in file: ExaML's topology-only flow
            # ignores it (defaultz) unless read_bl is requested
            if read_bl:
                import math
                z = math.exp(-float(s[j:k]))
            pos[0] = k
        return z

    root_children = []
    assert s[0] == "("
    pos[0] += 1
    while True:
        root_children.append(parse())
        if s[pos[0]] == ",":
            pos[0] += 1
            continue
        assert s[pos[0]] == ")"
        pos[0] += 1
        break

    # unrooted trifurcation at the outermost node (standard phylip style);
    # a bifurcating root would create a degree-2 node — collapse it.
    node = next_inner[0]
    next_inner[0] += 1
    if len(root_children) == 3:
        for (c, cz) in root_children:
            tree.add_edge(node, c, cz)
    elif len(root_children) == 2:
        # collapse: connect the two children directly
        next_inner[0] -= 1
        (a, za), (b, zb) = root_children
        tree.add_edge(a, b, za)
    else:
        raise ValueError("unsupported root degree")
    assert next_inner[0] <= 2 * ntips - 1
    return tree


def to_newick_topology(tree, taxa_names):
    """Topology-only Tree2String (printBranchLengths=FALSE), as used for
    the -D convergence tree strings tr->tree0/tree1
    (searchAlgo.c:2178)."""

    def sub(p, parent):
        if tree.is_tip(p):
            return taxa_names[p - 1]
        if hasattr(tree, "ring_children"):
            kids = tree.ring_children(p, parent)
        else:
            kids = [w for w in tree.adj[p] if w != parent]
        return "(" + ",".join(sub(w, p) for w in kids) + ")"

    start = 1
    back = next(iter(tree.adj[start]))
    if hasattr(tree, "ring_children"):
        kids = tree.ring_children(back, start)
    else:
        kids = [w for w in tree.adj[back] if w != start]
    parts = [sub(start, back)] + [sub(w, back) for w in kids]
    return "(" + ",".join(parts) + ");"


def to_newick(tree, taxa_names, digits=20):
    """Write the tree with branch lengths as the reference's Tree2String
    does (treeIO.c:234, branch length = -log(z), SUMMARIZE_LH averages
    the per-partition -log z under -M), rooted at tr->start->back with
    the start tip as the first child."""
    import math

    def bl(a, b):
        zv = tree.get_zv(a, b)
        x = 0.0
        for z in zv:
            z = max(z, 1.0e-15)
            x += -math.log(z)
        return x / len(zv)

    def sub(p, parent):
        if tree.is_tip(p):
            return f"{taxa_names[p - 1]}:{bl(p, parent):.{digits}f}"
        if hasattr(tree, "ring_children"):
            kids = tree.ring_children(p, parent)
        else:
            kids = [w for w in tree.adj[p] if w != parent]
        inner = ",".join(sub(w, p) for w in kids)
        return f"({inner}):{bl(p, parent):.{digits}f}"

    start = 1
    back = next(iter(tree.adj[start]))
    if hasattr(tree, "ring_children"):
        kids = tree.ring_children(back, start)
    else:
        kids = [w for w in tree.adj[back] if w != start]
    parts = [sub(start, back)] + [sub(w, back) for w in kids]
    return "(" + ",".join(parts) + ");"
