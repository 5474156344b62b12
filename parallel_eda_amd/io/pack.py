"""Greedy packing: LUT/FF primitives -> CLB clusters.

Re-implements the semantics of the reference's clusterer
(vpr/SRC/pack/cluster.c:do_clustering, pack.c:20 try_pack): seed a cluster
with the highest-gain unclustered primitive, then greedily absorb
connected primitives (attraction = shared signals) while the cluster has
room (N BLEs, clb_in input pins). Produces the block-level NetlistPy the
placer/router consume, where a block is one CLB (or an IO pad).
"""
from collections import defaultdict

import numpy as np

from ..arch.archdef import ArchDef, BLK_IO, BLK_CLB, BLK_RAM, BLK_DSP
from .blif import BlifModel, subckt_class
from .synth import NetlistPy


def pack_blif(model: BlifModel, arch: ArchDef, n_ble: int = 10):
    """Pack `model` into CLBs of n_ble BLEs with arch.clb_in input pins.

    Returns (NetlistPy, cluster_of_prim dict, block_names list).
    """
    prims = model.prims
    np_prims = len(prims)

    def prim_outputs(i):
        pp = prims[i]
        return pp.outputs if pp.kind == "subckt" else [pp.name]

    # hard blocks: RAM/DSP subckt instances are not clustered — each is
    # its own block on a matching column tile (reference: VPR packs
    # memory/mult molecules into their own block types, cluster.c)
    hard_prims = [i for i, pp in enumerate(prims)
                  if pp.kind == "subckt" and
                  subckt_class(pp.model) in ("ram", "dsp")]
    hard_set = set(hard_prims)

    # signal -> driving primitive index (or input pad)
    drv_of_sig = {}
    for s in model.inputs:
        drv_of_sig[s] = ("pad", s)
    for i in range(np_prims):
        for o in prim_outputs(i):
            drv_of_sig[o] = ("prim", i)

    # primitive adjacency (shared signals), for attraction gain
    sig_users = defaultdict(list)
    for i, p in enumerate(prims):
        for s in p.inputs:
            sig_users[s].append(i)

    cluster_of = [-1] * np_prims
    clusters = []

    def cluster_inputs(members, cand=None):
        """Distinct external input signals of members (+cand)."""
        mem = set(members)
        if cand is not None:
            mem.add(cand)
        produced = {o for i in mem for o in prim_outputs(i)}
        ins = set()
        for i in mem:
            for s in prims[i].inputs:
                if s not in produced:
                    ins.add(s)
            if prims[i].clock:
                pass  # clock nets are global, don't count against pins
        return ins

    model_outs = set(model.outputs)

    def cluster_outputs(members, cand=None):
        """Member outputs that leave the cluster (consumed by an outside
        primitive or a model output) — bounded by clb_out (reference:
        cluster_legality.c output-pin feasibility)."""
        mem = set(members)
        if cand is not None:
            mem.add(cand)
        outs = set()
        for i in mem:
            for o in prim_outputs(i):
                if o in model_outs or any(j not in mem
                                          for j in sig_users.get(o, ())):
                    outs.add(o)
        return outs

    def cluster_clock(members):
        """The single clock shared by the cluster's latches (None if
        purely combinational). VPR legality: one clock per cluster
        (cluster_legality.c clock feasibility)."""
        for i in members:
            if prims[i].clock:
                return prims[i].clock
        return None

    unclustered = set(range(np_prims)) - hard_set
    while unclustered:
        # seed: primitive with most inputs (hardest to place later)
        seed = max(unclustered, key=lambda i: (len(prims[i].inputs), -i))
        members = [seed]
        unclustered.discard(seed)
        ins = cluster_inputs(members)
        while len(members) < n_ble:
            # candidates: users/drivers of member signals
            gain = defaultdict(int)
            for i in members:
                p = prims[i]
                for s in p.inputs:
                    d = drv_of_sig.get(s)
                    if d and d[0] == "prim" and d[1] in unclustered:
                        gain[d[1]] += 1
                for o in prim_outputs(i):
                    for j in sig_users.get(o, ()):
                        if j in unclustered:
                            gain[j] += 1
            # best gain, legality-filtered: input pins, output pins, and
            # single-clock (reference: cluster_legality.c feasibility)
            def legal(cand):
                cclk = prims[cand].clock
                if cclk and cur_clk and cclk != cur_clk:
                    return False
                if len(cluster_inputs(members, cand)) > arch.clb_in:
                    return False
                return len(cluster_outputs(members, cand)) <= arch.clb_out

            best = None
            cur_clk = cluster_clock(members)
            for cand, gn in sorted(gain.items(), key=lambda kv: (-kv[1], kv[0])):
                if legal(cand):
                    best = cand
                    break
            if best is None:
                # unrelated fill (reference: do_clustering's
                # allow_unrelated_clustering — when no connected
                # candidate fits, pack the hardest-to-place remaining
                # primitive that is still legal, keeping cluster count
                # near capacity instead of leaving fragments)
                for cand in sorted(unclustered,
                                   key=lambda i: (-len(prims[i].inputs),
                                                  i))[:64]:
                    if legal(cand):
                        best = cand
                        break
            if best is None:
                break
            members.append(best)
            unclustered.discard(best)
        clusters.append(members)
        for i in members:
            cluster_of[i] = len(clusters) - 1

    # ---- block-level netlist ----
    # blocks: input pads, output pads, clusters, hard blocks (RAM/DSP)
    n_in = len(model.inputs)
    n_out = len(model.outputs)
    nb = n_in + n_out + len(clusters) + len(hard_prims)
    block_type = np.full(nb, BLK_CLB, dtype=np.int8)
    block_type[:n_in + n_out] = BLK_IO
    block_is_seq = np.zeros(nb, dtype=np.uint8)
    block_is_seq[:n_in + n_out] = 1
    names = ([f"ipad:{s}" for s in model.inputs] +
             [f"opad:{s}" for s in model.outputs] +
             [f"clb_{k}" for k in range(len(clusters))])
    clb0 = n_in + n_out
    hard0 = clb0 + len(clusters)
    blk_of_hard = {}
    for h, i in enumerate(hard_prims):
        pp = prims[i]
        cls = subckt_class(pp.model)
        blk = hard0 + h
        blk_of_hard[i] = blk
        block_type[blk] = BLK_RAM if cls == "ram" else BLK_DSP
        # RAM outputs are registered; DSP (multiply/adder) is comb
        block_is_seq[blk] = 1 if cls == "ram" else 0
        names.append(f"{pp.model}_{h}")
    # a cluster is sequential if it contains any latch; its clock domain
    # is the (majority) latch clock. Clock signals are GLOBAL nets: latch
    # clock pins are not data sinks, so clock-pad nets end up sinkless and
    # are dropped from routing (VPR routes clocks on dedicated networks).
    clock_names = []
    clock_id = {}
    block_clock = np.full(nb, -1, dtype=np.int32)
    def domain_of(cname):
        if cname not in clock_id:
            clock_id[cname] = len(clock_names)
            clock_names.append(cname)
        return clock_id[cname]

    for k, members in enumerate(clusters):
        latch_clocks = [prims[i].clock for i in members
                        if prims[i].kind == "latch" and prims[i].clock]
        if any(prims[i].kind == "latch" for i in members):
            block_is_seq[clb0 + k] = 1
        if latch_clocks:
            from collections import Counter
            cname = Counter(latch_clocks).most_common(1)[0][0]
            block_clock[clb0 + k] = domain_of(cname)
    for i in hard_prims:
        if prims[i].clock and block_is_seq[blk_of_hard[i]]:
            block_clock[blk_of_hard[i]] = domain_of(prims[i].clock)
    # IO pads: assign to domain 0 if any clock exists
    if clock_names:
        block_clock[:n_in + n_out] = 0

    in_pad_of = {s: i for i, s in enumerate(model.inputs)}
    out_pad_of = {s: n_in + i for i, s in enumerate(model.outputs)}

    def blk_of_prim(i):
        if i in blk_of_hard:
            return blk_of_hard[i]
        return clb0 + cluster_of[i]

    # nets: one per signal that crosses a block boundary (or feeds a pad)
    drivers, sink_lists = [], []
    for sig, d in drv_of_sig.items():
        if d[0] == "pad":
            src_blk = in_pad_of[sig]
        else:
            src_blk = blk_of_prim(d[1])
        sinks = set()
        for j in sig_users.get(sig, ()):
            sb = blk_of_prim(j)
            if sb != src_blk:
                sinks.add(sb)
        if sig in out_pad_of:
            sinks.add(out_pad_of[sig])
        sinks.discard(src_blk)
        if sinks:
            drivers.append(src_blk)
            sink_lists.append(sorted(sinks))

    sink_ptr = np.zeros(len(drivers) + 1, dtype=np.int64)
    for i, s in enumerate(sink_lists):
        sink_ptr[i + 1] = sink_ptr[i] + len(s)
    net_sinks = (np.concatenate([np.asarray(s, dtype=np.int32)
                                 for s in sink_lists])
                 if sink_lists else np.zeros(0, dtype=np.int32))
    # clustering can create cluster-level combinational cycles even from an
    # acyclic primitive graph (p1 in A -> p2 in B -> p3 in A). The timing
    # model needs an acyclic comb graph, so blocks stuck in a cycle are
    # marked sequential (they contain registered paths in any real mapping).
    _break_comb_cycles(block_is_seq, drivers, sink_ptr, net_sinks, nb)
    nl = NetlistPy(block_type, block_is_seq,
                   np.asarray(drivers, dtype=np.int32), sink_ptr, net_sinks,
                   names=names)
    nl.block_clock = block_clock
    nl.clock_names = clock_names
    # carry-chain macros (reference: place_macro.c follows the carry
    # links between chain primitives): adder/DSP hard blocks whose cout
    # actual feeds another's cin form a vertical macro, placed and moved
    # atomically by the placer.
    nl.macros = _detect_carry_chains(prims, hard_prims, blk_of_hard)
    return nl, cluster_of, names


def _detect_carry_chains(prims, hard_prims, blk_of_hard):
    cin_of = {}    # cin actual signal -> hard prim index
    cout_of = {}   # hard prim index -> cout actual signal
    for i in hard_prims:
        pp = prims[i]
        if subckt_class(pp.model) != "dsp":
            continue
        # blif.py stores actuals only; carry linkage = an output of one
        # instance used as an input of another DSP instance
        cout_of[i] = set(pp.outputs)
    users = {}
    for i in hard_prims:
        pp = prims[i]
        if subckt_class(pp.model) != "dsp":
            continue
        for a in pp.inputs:
            users.setdefault(a, []).append(i)
    nxt = {}
    prev = {}
    for i, outs in cout_of.items():
        for a in outs:
            for j in users.get(a, []):
                if j != i and i not in nxt and j not in prev:
                    nxt[i] = j
                    prev[j] = i
                    break
            if i in nxt:
                break
    macros = []
    for i in cout_of:
        if i in prev:
            continue  # not a chain head
        chain = [i]
        while chain[-1] in nxt:
            chain.append(nxt[chain[-1]])
        if len(chain) >= 2:
            macros.append([(blk_of_hard[c], 0, k)
                           for k, c in enumerate(chain)])
    return macros


def _break_comb_cycles(block_is_seq, drivers, sink_ptr, net_sinks, nb):
    """Flip one member of each combinational cycle to sequential so the
    netlist levelizes. This CHANGES timing semantics (the flipped block
    becomes a clocked endpoint), so every conversion is reported with a
    warning and the list of converted blocks is returned (reference VPR
    errors out on combinational loops during timing-graph build;
    path_delay.c levelization)."""
    converted = []
    while True:
        indeg = np.zeros(nb, dtype=np.int64)
        adj = defaultdict(list)
        for n, drv in enumerate(drivers):
            for s in net_sinks[sink_ptr[n]:sink_ptr[n + 1]]:
                if not block_is_seq[s]:
                    adj[drv].append(int(s))
                    indeg[s] += 1
        q = [b for b in range(nb)
             if block_is_seq[b] or indeg[b] == 0]
        seen = np.zeros(nb, dtype=bool)
        seen[q] = True
        head = 0
        q = list(q)
        while head < len(q):
            b = q[head]; head += 1
            for s in adj.get(b, ()):
                indeg[s] -= 1
                if indeg[s] == 0 and not seen[s]:
                    seen[s] = True
                    q.append(s)
        stuck = np.nonzero(~seen)[0]
        if len(stuck) == 0:
            if converted:
                import warnings
                warnings.warn(
                    f"netlist has combinational cycle(s): converted "
                    f"block(s) {converted} to sequential endpoints to "
                    f"break them — timing through these loops is NOT "
                    f"analyzed as combinational", stacklevel=3)
            return converted
        converted.append(int(stuck[0]))
        block_is_seq[stuck[0]] = 1  # break one cycle member, re-check
