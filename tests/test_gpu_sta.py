"""GPU STA numerics vs CPU fp32 reference (same graph, random delays)."""
import numpy as np
import pytest

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_netlist, SynthSpec
from parallel_eda_amd.timing.sta import STA

pytestmark = pytest.mark.gpu


def test_gpu_sta_matches_cpu():
    from parallel_eda_amd.timing.gpu_sta import GpuSTA
    arch = get_arch("tseng")
    nl = synth_netlist(SynthSpec(n_clb=300, n_in=12, n_out=12, seed=21,
                                 max_fanin=16))
    cpu_sta = STA(nl, arch)
    gpu_sta = GpuSTA(nl, arch)
    rng = np.random.default_rng(3)
    for trial in range(3):
        d = (rng.random(nl.num_conns) * 2e-9).astype(np.float32)
        cpd_c, slack_c, crit_c = cpu_sta.analyze(d)
        cpd_g, slack_g, crit_g = gpu_sta.analyze(d)
        assert cpd_g == pytest.approx(cpd_c, rel=1e-5)
        assert np.allclose(slack_g, slack_c, rtol=1e-4, atol=1e-13)
        # GPU clamps crit to 0.99; apply same clamp to CPU reference
        crit_c2 = np.minimum(crit_c, 0.99)
        assert np.allclose(crit_g, crit_c2, rtol=1e-4, atol=1e-5)
