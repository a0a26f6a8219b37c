"""The reference's binary checkpoint format (searchAlgo.c:1153
writeCheckpointInner / :1502 readCheckpoint): read AND write, so a run can
move between the reference and this framework mid-flight (-R restart).

The format is a raw x86-64 struct dump:

  checkPointState (1320 B)                      axml.h:682
  [constraintVector, only if constraintTree]
  tree0, tree1            (treeStringLength B each)  axml.c:583
  [CAT only] rateCategory (int32  x crunchedLength)
             patrat       (double x crunchedLength)
  per model:
      numberOfCategories  int32
      perSiteRates        double x maxCategories (25)
      EIGN,EV,EI          per-dataType lengths (DNA 4/16/16, AA 20/400/400)
      freqExponents, frequencies   (4 | 20)
      tipVector                    (64 | 460)
      substRates                   (6 | 190)
      weights, weightExponents     (4 + 4, LG4X)
      [LG4 blocks if protModels is LG4M/LG4X]
      alpha double, gammaRates double x 4
      protModels int32, autoProtModels int32
  [state == MOD_OPT] likelihoods (double x numberOfTrees),
                     treeStrings (treeStringLength x numberOfTrees)
  writeTree: start->number int32, nodeBaseAddress (8 B),
             node[mxtips + 3*(mxtips-1)] x 2080 B          axml.h:492

node layout (2080 B): z[256] doubles, next ptr, back ptr, hash u32,
number i32, x/xPars/xBips chars.  next/back are the WRITER's raw
addresses; readTree (searchAlgo.c:1311) relocates them against the stored
nodeBaseAddress.  Record order (axml.c setupTree:608): records
0..mxtips-1 are tips 1..mxtips; each inner node i occupies a block of 3
(ring b+2 -> b+1 -> b -> b+2, nodep[i] = b+2, x=1 on record b).

All offsets verified against the reference compiled in place
(tools/ckpt_layout.c)."""

import struct

import numpy as np

from .tree import PhyloTree

CKP_SIZE = 1320
NODE_SIZE = 2080
MAX_CATEGORIES = 25
NMLNGTH = 256

# pLengths rows (globalVariables.h): states -> (eign, ev, ei, freq, tipvec,
# subst)
_PLEN = {4: (4, 16, 16, 4, 64, 6), 20: (20, 400, 400, 20, 460, 190)}

REARR_SETTING = 1  # ckp.state (axml.h:655-659)
FAST_SPRS = 2
SLOW_SPRS = 3
MOD_OPT = 4
QUARTETS = 5

# rateHetModel values as stored by the reference (axml.h): CAT=0, GAMMA=1
RATE_HET_CAT = 0
RATE_HET_GAMMA = 1


def tree_string_length(mxtips):
    return mxtips * (NMLNGTH + 128) + 256 + mxtips * 2  # axml.c:583


def _node_count(mxtips):
    return mxtips + 3 * (mxtips - 1)


class Checkpoint:
    pass


def read_checkpoint(path, mxtips, states_per_model, rate_het="GAMMA",
                    crunched_length=None, num_trees=1, prot_models=None):
    """Parse a reference checkpoint into plain arrays.  states_per_model:
    4 or 20 per partition; crunched_length: total pattern count (CAT
    only); prot_models: the byte file's per-partition protModels ids —
    needed to know which partitions carry the LG4 per-category array
    block (writeCheckpointInner stores it between weightExponents and
    alpha, searchAlgo.c:1248-1262, and only the caller knows which
    partitions are LG4M/LG4X)."""
    d = open(path, "rb").read()
    ck = Checkpoint()
    ck.state = struct.unpack_from("<i", d, 0)[0]
    ck.optimize_rate_category_invocations = struct.unpack_from("<i", d, 48)[0]
    ck.accumulated_time = struct.unpack_from("<d", d, 56)[0]
    ck.cat_opt = struct.unpack_from("<i", d, 180)[0]
    ck.tree_iteration = struct.unpack_from("<i", d, 184)[0]
    # search-algorithm state (checkPointState, axml.h:679-720)
    (ck.rearrangements_max, ck.rearrangements_min, ck.thorough_iterations,
     ck.fast_iterations, ck.tree_vector_length, ck.mintrav, ck.maxtrav,
     ck.best_trav, ck.thorough) = struct.unpack_from("<9i", d, 12)
    (ck.start_lh, ck.lh, ck.previous_lh, ck.difference,
     ck.epsilon) = struct.unpack_from("<5d", d, 64)
    ck.impr, ck.cutoff = struct.unpack_from("<2i", d, 104)
    (ck.tr_start_lh, ck.tr_end_lh, ck.tr_likelihood, ck.tr_best_of_node,
     ck.tr_lh_cutoff, ck.tr_lh_avg,
     ck.tr_lh_dec) = struct.unpack_from("<7d", d, 112)
    (ck.tr_number_of_categories, ck.tr_it_count,
     ck.tr_do_cutoff) = struct.unpack_from("<3i", d, 168)
    # quartet state (-f q -I): seed at 192 (8-aligned after 184+4 pad),
    # flavor 200, quartetCounter 208, filePosition 216, fileName 224
    ck.seed = struct.unpack_from("<q", d, 192)[0]
    ck.flavor = struct.unpack_from("<i", d, 200)[0]
    ck.quartet_counter = struct.unpack_from("<Q", d, 208)[0]
    ck.file_position = struct.unpack_from("<q", d, 216)[0]
    constraint = struct.unpack_from("<i", d, 8)[0]
    c = 1248
    ck.likelihood_epsilon = struct.unpack_from("<d", d, c + 24)[0]
    ck.categories = struct.unpack_from("<i", d, c + 32)[0]
    ck.rate_het_model = struct.unpack_from("<i", d, c + 52)[0]
    ck.per_gene_bl = bool(struct.unpack_from("<i", d, c + 16)[0])

    off = CKP_SIZE
    assert not constraint, "constraint trees not supported"
    tsl = tree_string_length(mxtips)
    ck.tree0 = d[off:off + tsl]
    off += tsl
    ck.tree1 = d[off:off + tsl]
    off += tsl

    if rate_het == "CAT":
        assert crunched_length is not None
        ck.rate_category = np.frombuffer(d, np.int32, crunched_length, off)
        off += 4 * crunched_length
        ck.patrat = np.frombuffer(d, np.float64, crunched_length, off)
        off += 8 * crunched_length

    ck.models = []
    for mi, states in enumerate(states_per_model):
        eign, ev, ei, freq, tipvec, subst = _PLEN[states]
        m = {}

        def rd(n, kind="d"):
            nonlocal off
            if kind == "i":
                v = struct.unpack_from("<i", d, off)[0]
                off += 4
            else:
                v = np.frombuffer(d, np.float64, n, off).copy()
                off += 8 * n
            return v

        m["num_cats"] = rd(1, "i")
        m["per_site_rates"] = rd(MAX_CATEGORIES)
        m["EIGN"] = rd(eign)
        m["EV"] = rd(ev)
        m["EI"] = rd(ei)
        m["freqExponents"] = rd(freq)
        m["frequencies"] = rd(freq)
        m["tipVector"] = rd(tipvec)
        m["substRates"] = rd(subst)
        m["weights"] = rd(4)
        m["weightExponents"] = rd(4)
        if prot_models is not None and prot_models[mi] in (20, 21):
            # LG4M/LG4X: four per-category eigensystems
            # (writeCheckpointInner, searchAlgo.c:1248-1262)
            for key, n in (("rawEIGN_LG4", eign), ("EIGN_LG4", eign),
                           ("EV_LG4", ev), ("EI_LG4", ei),
                           ("frequencies_LG4", freq),
                           ("tipVector_LG4", tipvec),
                           ("substRates_LG4", subst)):
                m[key] = []
            for _k in range(4):
                m["rawEIGN_LG4"].append(rd(eign))
                m["EIGN_LG4"].append(rd(eign))
                m["EV_LG4"].append(rd(ev))
                m["EI_LG4"].append(rd(ei))
                m["frequencies_LG4"].append(rd(freq))
                m["tipVector_LG4"].append(rd(tipvec))
                m["substRates_LG4"].append(rd(subst))
        m["alpha"] = float(rd(1)[0])
        m["gammaRates"] = rd(4)
        m["protModels"] = rd(1, "i")
        m["autoProtModels"] = rd(1, "i")
        if m["protModels"] in (20, 21):
            assert prot_models is not None, \
                "pass prot_models to read LG4 checkpoints"
        ck.models.append(m)

    if ck.state == MOD_OPT:
        ck.likelihoods = np.frombuffer(d, np.float64, num_trees, off).copy()
        off += 8 * num_trees
        ck.tree_strings = d[off:off + tsl * num_trees]
        off += tsl * num_trees

    # readTree (searchAlgo.c:1311)
    ck.start_number = struct.unpack_from("<i", d, off)[0]
    off += 4
    base = struct.unpack_from("<Q", d, off)[0]
    ck._base = base
    off += 8
    x = _node_count(mxtips)
    recs = []
    for k in range(x):
        o = off + k * NODE_SIZE
        z = np.frombuffer(d, np.float64, 256, o)
        nxt, bck = struct.unpack_from("<QQ", d, o + 2048)
        number = struct.unpack_from("<i", d, o + 2068)[0]
        recs.append((z, nxt, bck, number))
    off += x * NODE_SIZE
    assert off == len(d), (off, len(d))

    def idx(ptr):
        if ptr == 0:
            return None
        r = (ptr - base) // NODE_SIZE
        assert 0 <= r < x and (ptr - base) % NODE_SIZE == 0
        return int(r)

    nb = len(states_per_model) if ck.per_gene_bl else 1
    tree = PhyloTree.__new__(PhyloTree)
    tree.ntips = mxtips
    tree.nnodes = 2 * mxtips - 1
    tree.adj = {i: {} for i in range(1, 2 * mxtips - 1)}
    for z, nxt, bck, number in recs:
        b = idx(bck)
        if b is None or not (1 <= number <= 2 * mxtips - 2):
            continue  # spare ring block (setupTree's inter = mxtips-1)
        nbr = recs[b][3]
        if nbr in tree.adj.get(number, {}):
            continue
        zv = np.array(z[:nb]) if nb > 1 else float(z[0])
        tree.adj[number][nbr] = zv
        tree.adj[nbr][number] = zv
    ck.tree = tree
    ck._recs = recs
    ck._nb = nb
    return ck


def spr_tree(ck, mxtips):
    """Reconstruct the EXACT ring structure (member identity, ring
    order, back pointers, shared branch vectors) from the checkpoint's
    node image — resumed SPR searches must traverse the very rings the
    reference serialized (writeTree/readTree, searchAlgo.c:1311)."""
    from .spr import RingNode, SprTree
    recs = ck._recs
    nb = ck._nb
    x = len(recs)
    st = SprTree.__new__(SprTree)
    st.ntips = mxtips
    st.nnodes = 2 * mxtips - 1
    members = [RingNode(recs[k][3]) for k in range(x)]
    # image layout (writeTree): tips 0..mxtips-1; inner node i's block
    # b = mxtips + 3*(i - mxtips - 1) holds [b]=p->next->next, [b+1]=
    # p->next, [b+2]=p with ring b+2 -> b+1 -> b -> b+2
    st.nodep = [None] * (2 * mxtips)
    st.ring = [None] * (2 * mxtips)
    for i in range(1, mxtips + 1):
        st.nodep[i] = st.ring[i] = members[i - 1]
    for i in range(mxtips + 1, 2 * mxtips - 1):
        b = mxtips + 3 * (i - mxtips - 1)
        members[b + 2].next = members[b + 1]
        members[b + 1].next = members[b]
        members[b].next = members[b + 2]
        st.nodep[i] = st.ring[i] = members[b + 2]
    first = {}  # per-edge shared branch vector under -M
    ck_base = ck._base
    nsz = NODE_SIZE

    def idx(ptr):
        return int((ptr - ck_base) // nsz) if ptr else None

    for k, (z, nxt, bck, number) in enumerate(recs):
        if not (1 <= number <= 2 * mxtips - 2):
            continue
        b = idx(bck)
        if b is None:
            continue
        m = members[k]
        mb = members[b]
        m.back = mb
        key = (min(k, b), max(k, b))
        if nb > 1:
            if key not in first:
                first[key] = np.array(z[:nb])
            m.z = first[key]
        else:
            m.z = float(z[0])
    st.start = ck.start_number
    return st


def write_checkpoint(path, tree, models, mxtips, *, state=MOD_OPT,
                     cat_opt=0, tree_iteration=0, invocations=1,
                     rate_het="GAMMA", per_gene_bl=False,
                     likelihood_epsilon=0.1, rate_category=None,
                     patrat=None, likelihoods=None, start_number=1,
                     accumulated_time=1.0, spr=None, use_median=False,
                     save_best_trees=0, save_memory=False,
                     search_convergence=False, categories=MAX_CATEGORIES,
                     initial_set=False, initial=10, tree0=None,
                     tree1=None):
    """Emit a checkpoint the reference's readCheckpoint accepts.  models:
    one dict per partition with the arrays of read_checkpoint's layout
    (build_model_entry converts our model objects).  spr: the
    search-state fields for FAST_SPRS/SLOW_SPRS checkpoints
    (checkPointState, axml.h:679-720)."""
    tsl = tree_string_length(mxtips)
    out = bytearray(CKP_SIZE)
    struct.pack_into("<i", out, 0, state)
    struct.pack_into("<i", out, 48, invocations)
    struct.pack_into("<d", out, 56, accumulated_time)
    struct.pack_into("<i", out, 180, cat_opt)
    struct.pack_into("<i", out, 184, tree_iteration)
    if spr is not None:
        struct.pack_into(
            "<9i", out, 12, spr.get("rearrangements_max", 0),
            spr.get("rearrangements_min", 0),
            spr.get("thorough_iterations", 0),
            spr.get("fast_iterations", 0),
            spr.get("tree_vector_length", 1), spr.get("mintrav", 1),
            spr.get("maxtrav", 5), spr.get("best_trav", 5),
            spr.get("thorough", 0))
        struct.pack_into(
            "<5d", out, 64, spr.get("start_lh", 0.0), spr.get("lh", 0.0),
            spr.get("previous_lh", 0.0), spr.get("difference", 10.0),
            spr.get("epsilon", 0.01))
        struct.pack_into("<2i", out, 104, int(spr.get("impr", 1)),
                         int(spr.get("cutoff", 1)))
        struct.pack_into(
            "<7d", out, 112, spr.get("tr_start_lh", 0.0),
            spr.get("tr_end_lh", 0.0), spr.get("tr_likelihood", 0.0),
            spr.get("tr_best_of_node", 0.0),
            spr.get("tr_lh_cutoff", 0.0), spr.get("tr_lh_avg", 0.0),
            spr.get("tr_lh_dec", 0.0))
        struct.pack_into("<3i", out, 168, MAX_CATEGORIES,
                         spr.get("tr_it_count", 0),
                         int(spr.get("tr_do_cutoff", 1)))
    # commandLine block (axml.h:660-679) — the reference's
    # checkCommandLineArguments (searchAlgo.c:1383) hard-fails on any
    # mismatch with the restart invocation, so every option the caller
    # actually ran with must land here.
    c = 1248
    struct.pack_into("<i", out, c + 0, 1 if use_median else 0)        # -a
    struct.pack_into("<i", out, c + 4, int(save_best_trees))          # -B
    struct.pack_into("<i", out, c + 8, 1 if save_memory else 0)       # -S
    struct.pack_into("<i", out, c + 12, 1 if search_convergence else 0)
    struct.pack_into("<i", out, c + 16, 1 if per_gene_bl else 0)      # -M
    struct.pack_into("<d", out, c + 24, likelihood_epsilon)           # -e
    struct.pack_into("<i", out, c + 32, int(categories))              # -c
    # adef->mode: BIG_RAPID_MODE for SPR-state checkpoints (axml.h:226)
    struct.pack_into("<i", out, c + 36,
                     1 if state in (REARR_SETTING, FAST_SPRS, SLOW_SPRS)
                     else 0)
    struct.pack_into("<i", out, c + 44, 1 if initial_set else 0)
    struct.pack_into("<i", out, c + 48, int(initial))                 # -i
    struct.pack_into("<i", out, c + 52,
                     RATE_HET_CAT if rate_het == "CAT" else RATE_HET_GAMMA)

    buf = bytearray(bytes(out))

    def tstr(t):
        """tree0/tree1 topology string padded to treeStringLength
        (searchAlgo.c:1207-1208); empty when the slot was never
        stored."""
        if t is None:
            return bytes(tsl)
        b = t.encode() if isinstance(t, str) else bytes(t)
        assert len(b) < tsl, "tree string exceeds treeStringLength"
        return b + bytes(tsl - len(b))

    buf += tstr(tree0)
    buf += tstr(tree1)

    if rate_het == "CAT":
        buf += np.ascontiguousarray(rate_category, np.int32).tobytes()
        buf += np.ascontiguousarray(patrat, np.float64).tobytes()

    for m in models:
        states = 4 if len(m["substRates"]) == 6 else 20
        buf += struct.pack("<i", m["num_cats"])
        psr = np.zeros(MAX_CATEGORIES)
        psr[:len(m["per_site_rates"])] = m["per_site_rates"]
        buf += psr.tobytes()
        for key in ("EIGN", "EV", "EI", "freqExponents", "frequencies",
                    "tipVector", "substRates"):
            a = np.ascontiguousarray(m[key], np.float64)
            buf += a.tobytes()
        buf += np.asarray(m.get("weights", np.full(4, 0.25))).tobytes()
        buf += np.asarray(m.get("weightExponents", np.zeros(4))).tobytes()
        if m.get("protModels", 0) in (20, 21):
            # LG4M/LG4X: the four per-category eigensystem blocks between
            # weightExponents and alpha (writeCheckpointInner,
            # searchAlgo.c:1244-1260); readCheckpoint gates this read on
            # the byte file's protModels, so it must be present
            for k in range(4):
                for key in ("rawEIGN_LG4", "EIGN_LG4", "EV_LG4", "EI_LG4",
                            "frequencies_LG4", "tipVector_LG4",
                            "substRates_LG4"):
                    buf += np.ascontiguousarray(m[key][k],
                                                np.float64).tobytes()
        buf += struct.pack("<d", m["alpha"])
        buf += np.ascontiguousarray(m["gammaRates"], np.float64).tobytes()
        buf += struct.pack("<ii", m.get("protModels", 0 if states == 4
                                        else m.get("protModels", 0)),
                           m.get("autoProtModels", 2))

    if state == MOD_OPT:
        lk = likelihoods if likelihoods is not None else [0.0]
        buf += np.asarray(lk, np.float64).tobytes()
        buf += bytes(tsl * len(lk))

    # writeTree image: tips at records 0..mxtips-1; inner node i at block
    # b = mxtips + 3*(i - mxtips - 1), ring b+2 -> b+1 -> b -> b+2,
    # nodep[i] = b+2 (setupTree, axml.c:608)
    x = _node_count(mxtips)
    base = NODE_SIZE  # arbitrary nonzero "address"

    def addr(r):
        return base + r * NODE_SIZE

    nb = len(models) if per_gene_bl else 1

    def zvec(a, b):
        zv = tree.get_zv(a, b)
        z = np.full(256, 0.9)
        z[:nb] = zv[:nb] if len(zv) >= nb else float(zv[0])
        return z

    # slot assignment: inner node's ring records in order [b+2, b+1, b]
    # take its (up to 3) neighbors in adjacency order
    rec_back = [-1] * x          # record -> record index of back (or -1)
    rec_z = [None] * x
    rec_number = [0] * x
    rec_next = [0] * x
    rec_x = [0] * x
    inner_slots = {}
    for i in range(1, mxtips + 1):
        rec_number[i - 1] = i
        rec_next[i - 1] = addr(i - 1)  # p->next = p for tips
    # setupTree allocates inter = mxtips-1 inner blocks (one spare ring
    # beyond the mxtips-2 used inner nodes) — number & ring the spare too
    for i in range(mxtips + 1, 2 * mxtips):
        b = mxtips + 3 * (i - mxtips - 1)
        for k in (b, b + 1, b + 2):
            rec_number[k] = i
            rec_back[k] = -1
        rec_next[b + 2] = addr(b + 1)
        rec_next[b + 1] = addr(b)
        rec_next[b] = addr(b + 2)
        rec_x[b] = 1
        inner_slots[i] = [b + 2, b + 1, b]
    used = {i: 0 for i in inner_slots}

    def take_slot(i):
        if i <= mxtips:
            return i - 1
        s = inner_slots[i][used[i]]
        used[i] += 1
        return s

    if hasattr(tree, "ring"):
        # SprTree: serialize the LIVE rings so a resumed reference run
        # traverses exactly the member structure this search had —
        # ring[i] maps to record b+2, its .next to b+1, .next.next to b
        # (the image's fixed next-pointer pattern above)
        memrec = {}
        for i in range(1, mxtips + 1):
            memrec[id(tree.ring[i])] = i - 1
        for i in range(mxtips + 1, 2 * mxtips - 1):
            b = mxtips + 3 * (i - mxtips - 1)
            m3 = tree.ring[i]
            memrec[id(m3)] = b + 2
            memrec[id(m3.next)] = b + 1
            memrec[id(m3.next.next)] = b

        def mzvec(mz):
            z = np.full(256, 0.9)
            if isinstance(mz, np.ndarray):
                z[:nb] = mz[:nb]
            else:
                z[0] = float(mz)
            return z

        for i in range(1, 2 * mxtips - 1):
            for m in tree.members(i):
                if m.back is None:
                    continue
                r = memrec[id(m)]
                rec_back[r] = memrec[id(m.back)]
                rec_z[r] = mzvec(m.z)
    else:
        for a, bn in tree.edges():
            ra, rb = take_slot(a), take_slot(bn)
            rec_back[ra] = rb
            rec_back[rb] = ra
            rec_z[ra] = rec_z[rb] = zvec(a, bn)

    buf += struct.pack("<i", start_number)
    buf += struct.pack("<Q", base)
    for r in range(x):
        z = rec_z[r] if rec_z[r] is not None else np.full(256, 0.9)
        rec = bytearray(NODE_SIZE)
        rec[0:2048] = z.tobytes()
        struct.pack_into("<Q", rec, 2048, rec_next[r])
        struct.pack_into("<Q", rec, 2056,
                         0 if rec_back[r] < 0 else addr(rec_back[r]))
        struct.pack_into("<i", rec, 2068, rec_number[r])
        rec[2072] = rec_x[r]
        buf += bytes(rec)
    open(path, "wb").write(bytes(buf))


def build_model_entry(model, num_cats=1, per_site_rates=(1.0,),
                      freq_exponents=None):
    """Convert one of our model objects into the per-model dict of the
    checkpoint layout."""
    states = model.states
    if states == 4:
        subst = model.rates6
        prot = 0
        auto = 2
    else:
        subst = model.rates190
        prot = getattr(model, "prot_model_id", 19)
        auto = getattr(model, "auto_prot_model", 2)
    return {
        "num_cats": num_cats,
        "per_site_rates": np.asarray(per_site_rates, float),
        "EIGN": model.EIGN[:states],
        "EV": model.EV,
        "EI": model.EI,
        "freqExponents": (freq_exponents if freq_exponents is not None
                          else np.zeros(states)),
        "frequencies": model.frequencies,
        "tipVector": model.tipVector,
        "substRates": subst,
        "alpha": model.alpha,
        "gammaRates": model.gammaRates,
        "protModels": prot,
        "autoProtModels": auto,
    }
