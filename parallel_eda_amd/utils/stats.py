"""Per-run statistics files + routing stats.

Mirrors the reference's observability surface: the per-iteration
`iter_stats.txt` and end-of-run `final_stats.txt` written into a per-run
stats directory (reference: partitioning_multi_sink...cxx:5618-5650 stats
dir creation, :5926-5932 iter stats, :6347-6362 final stats), and the
wirelength/bends/channel-occupancy report of base/stats.c:27
routing_stats_new / :355 get_num_bends_and_length.
"""
import hashlib
import json
import os
import time

import numpy as np

from ..arch.archdef import RR_CHANX, RR_CHANY


class StatsWriter:
    def __init__(self, stats_dir, run_name="route"):
        self.dir = stats_dir
        os.makedirs(stats_dir, exist_ok=True)
        self.run_name = run_name
        self.iter_path = os.path.join(stats_dir, "iter_stats.txt")
        self.final_path = os.path.join(stats_dir, "final_stats.txt")
        self._t0 = time.perf_counter()
        self._rows = []
        with open(self.iter_path, "w") as f:
            f.write("# iter time_s overused rerouted_nets heap_pops "
                    "crit_path_ns\n")

    def iteration(self, it, overused, rerouted=-1, heap_pops=-1, cpd=0.0):
        t = time.perf_counter() - self._t0
        row = (it, t, overused, rerouted, heap_pops, cpd * 1e9)
        self._rows.append(row)
        with open(self.iter_path, "a") as f:
            f.write(f"{it} {t:.3f} {overused} {rerouted} {heap_pops} "
                    f"{cpd*1e9:.4f}\n")

    def net_costs(self, it, costs, rank=0):
        """Per-net measured route cost for one iteration (reference:
        net_route_time_iter_N_rank_R.txt, mpi_route...cxx:1171): one
        '<net> <cost>' line per net, nonzero entries only."""
        path = os.path.join(self.dir,
                            f"net_cost_iter_{it}_rank_{rank}.txt")
        c = list(costs)
        with open(path, "w") as f:
            for n, v in enumerate(c):
                if v:
                    f.write(f"{n} {v}\n")

    def final(self, success, wirelength, cpd, extra=None):
        total = time.perf_counter() - self._t0
        data = {
            "run": self.run_name,
            "success": bool(success),
            "iterations": len(self._rows),
            "total_route_time_s": round(total, 3),
            "wirelength": int(wirelength),
            "crit_path_ns": round(cpd * 1e9, 4),
        }
        if extra:
            data.update(extra)
        with open(self.final_path, "w") as f:
            for k, v in data.items():
                f.write(f"{k} {v}\n")
        with open(os.path.join(self.dir, "final_stats.json"), "w") as f:
            json.dump(data, f, indent=1)
        return data


def routing_serial_num(net_ids, tree_fn):
    """Cross-run routing fingerprint (reference: get_serial_num,
    route_common.c — a hash of every net's traceback, printed so two runs
    can be diffed without storing .route files)."""
    h = hashlib.sha256()
    for k in range(len(net_ids)):
        nodes, parents, sws, delays = tree_fn(k)
        h.update(np.ascontiguousarray(nodes, dtype=np.int32).tobytes())
        h.update(np.ascontiguousarray(parents, dtype=np.int32).tobytes())
    return h.hexdigest()[:16]


def mem_usage_mb():
    """Host RSS in MiB (reference: main.c:287 get_mem_usage via
    /proc/self/statm)."""
    try:
        with open("/proc/self/statm") as f:
            pages = int(f.read().split()[1])
        return pages * os.sysconf("SC_PAGE_SIZE") / (1 << 20)
    except Exception:
        return -1.0


def gpu_mem_usage_mb(device=None):
    try:
        import torch
        if not torch.cuda.is_available():
            return -1.0
        return torch.cuda.memory_allocated(device) / (1 << 20)
    except Exception:
        return -1.0


def routing_stats(g, arch, net_ids, tree_fn):
    """Wirelength / bends / segment & channel occupancy report.

    tree_fn: k -> (nodes, parents, sws, delays). Reference: stats.c:27
    routing_stats_new + get_num_bends_and_length:355."""
    ty = np.asarray(g.type)
    xl = np.asarray(g.xlow); xh = np.asarray(g.xhigh)
    yl = np.asarray(g.ylow); yh = np.asarray(g.yhigh)
    total_wl = 0
    total_bends = 0
    total_segments = 0
    max_net_wl = 0
    for k in range(len(net_ids)):
        nodes, parents, sws, delays = tree_fn(k)
        tyk = ty[nodes]
        chan = (tyk == RR_CHANX) | (tyk == RR_CHANY)
        wl = int((xh[nodes] - xl[nodes] + yh[nodes] - yl[nodes] + 1)[chan].sum())
        total_wl += wl
        total_segments += int(chan.sum())
        max_net_wl = max(max_net_wl, wl)
        # bends: parent chan type differs from child chan type
        for i in range(len(nodes)):
            p = parents[i]
            if p < 0:
                continue
            if chan[i] and (ty[nodes[p]] in (RR_CHANX, RR_CHANY)) \
                    and ty[nodes[p]] != tyk[i]:
                total_bends += 1
    # channel occupancy histogram
    return {
        "total_wirelength": total_wl,
        "total_segments": total_segments,
        "total_bends": total_bends,
        "avg_bends_per_net": total_bends / max(1, len(net_ids)),
        "max_net_wirelength": max_net_wl,
        "avg_wirelength_per_net": total_wl / max(1, len(net_ids)),
    }
