"""Router-iteration checkpoint / resume.

The reference writes per-iteration route state (write_routes:2105,
write_congestion_state:2247) but never reads it back; here the congestion
arrays + route trees ARE the complete router state, so a checkpoint can
resume PathFinder mid-run (SURVEY.md section 5.4's planned extension).
"""
import numpy as np


def save_router_state(path, router, pres_fac, iteration):
    """router: GpuRouter (device tensors are pulled to host)."""
    np.savez_compressed(
        path,
        occ=router.t_occ.cpu().numpy(),
        acc=router.t_acc.cpu().numpy(),
        tree_node=router.t_tree_node.cpu().numpy(),
        tree_parent=router.t_tree_parent.cpu().numpy(),
        tree_sw=router.t_tree_sw.cpu().numpy(),
        tree_delay=router.t_tree_delay.cpu().numpy(),
        tree_len=router.t_tree_len.cpu().numpy(),
        sink_delay=router.t_sink_delay.cpu().numpy(),
        tree_off=router.tree_off,
        bb=router.bb,
        bb_margin=router.bb_margin_per_net,
        pres_fac=np.float64(pres_fac),
        iteration=np.int64(iteration),
    )


def load_router_state(path, router):
    """Restore a checkpoint into a freshly-constructed GpuRouter over the
    same graph+nets. Returns (pres_fac, iteration)."""
    import torch
    d = np.load(path)
    if len(d["tree_off"]) != len(router.tree_off) or \
            not np.array_equal(d["tree_off"], router.tree_off):
        raise ValueError("checkpoint tree layout mismatch (different nets?)")
    dev = router.device
    router.t_occ.copy_(torch.from_numpy(d["occ"]).to(dev))
    router.t_acc.copy_(torch.from_numpy(d["acc"]).to(dev))
    router.t_tree_node.copy_(torch.from_numpy(d["tree_node"]).to(dev))
    router.t_tree_parent.copy_(torch.from_numpy(d["tree_parent"]).to(dev))
    router.t_tree_sw.copy_(torch.from_numpy(d["tree_sw"]).to(dev))
    router.t_tree_delay.copy_(torch.from_numpy(d["tree_delay"]).to(dev))
    router.t_tree_len.copy_(torch.from_numpy(d["tree_len"]).to(dev))
    router.t_sink_delay.copy_(torch.from_numpy(d["sink_delay"]).to(dev))
    router.bb = d["bb"]
    router.bb_margin_per_net = d["bb_margin"]
    router.t_bb.copy_(torch.from_numpy(d["bb"]).to(dev))
    router._bb_version += 1
    return float(d["pres_fac"]), int(d["iteration"])
