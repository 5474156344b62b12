"""Round-2 first GPU probe (call 1, ~22 min): the measurements that decide
this round's design.

  1. gpu test suite (regression incl. new incomplete-nets tracking)
  2. calendar vs ping-pong frontier A/B at LU32 (promote the winner)
  3. tseng GPU-vs-oracle quality with calendar on
  4. multi-domain GPU STA vs CPU oracle (validates the dormant kernels)
  5. RCCL hardware: ws=1 nccl init+allreduce; ws=2 on ONE GPU attempt
     (documents whether 2-rank RCCL testing is possible on a 1-GPU box)

Run: gpurun --timeout 1500 -- 'python tools/r2probe.py > gpurun_out/r2probe.log 2>&1'
"""
import os
import subprocess
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
os.chdir(ROOT)


def run(name, cmd, env=None, timeout=600):
    print(f"\n===== {name}: {cmd}", flush=True)
    e = dict(os.environ)
    e.update(env or {})
    t0 = time.time()
    try:
        r = subprocess.run(cmd, shell=True, env=e, timeout=timeout)
        rc = r.returncode
    except subprocess.TimeoutExpired:
        rc = -9
        print(f"===== {name} TIMEOUT", flush=True)
    print(f"===== {name} rc={rc} ({time.time()-t0:.0f}s)", flush=True)
    return rc


run("gpu-tests", "python -m pytest tests/ -q -m gpu", timeout=900)
run("lu32-pingpong", "python tools/gpu_sweep_one.py 1.2 3.0 5", timeout=420)
run("lu32-calendar", "python tools/gpu_sweep_one.py 1.2 3.0 5",
    env={"PNR_CALENDAR": "1"}, timeout=420)
run("tseng-quality-calendar",
    "python -m pytest tests/test_gpu_router.py -q",
    env={"PNR_CALENDAR": "1"}, timeout=300)
run("sta-domains-gpu", "python - <<'P'\n"
    "import sys; sys.path.insert(0, '.')\n"
    "import numpy as np\n"
    "from parallel_eda_amd.arch.archdef import get_arch\n"
    "from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch\n"
    "from parallel_eda_amd.timing.sta import STA\n"
    "from parallel_eda_amd.timing.gpu_sta import GpuSTA\n"
    "arch = get_arch('tseng')\n"
    "nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=3))\n"
    "rng = np.random.default_rng(3)\n"
    "bc = np.where(np.asarray(nl.block_is_seq) > 0,\n"
    "              rng.integers(0, 2, nl.num_blocks), -1).astype(np.int32)\n"
    "per = np.asarray([5e-9, 8e-9], dtype=np.float32)\n"
    "dly = rng.uniform(0.1e-9, 2e-9, nl.num_conns).astype(np.float32)\n"
    "wp_c, sl_c, cr_c = STA(nl, arch).analyze_domains(dly, bc, per)\n"
    "g = GpuSTA(nl, arch)\n"
    "wp_g, sl_g, cr_g = g.analyze_domains(dly, bc, per)\n"
    "print('slack match:', np.allclose(sl_c, sl_g, rtol=1e-4, atol=1e-12))\n"
    "print('crit match:', np.allclose(np.minimum(cr_c, 0.99), cr_g, rtol=1e-4))\n"
    "assert np.allclose(sl_c, sl_g, rtol=1e-4, atol=1e-12)\n"
    "P", timeout=300)
run("rccl-ws1", "python - <<'P'\n"
    "import os, torch, torch.distributed as dist\n"
    "os.environ.setdefault('MASTER_ADDR', '127.0.0.1')\n"
    "os.environ.setdefault('MASTER_PORT', '29531')\n"
    "dist.init_process_group('nccl', rank=0, world_size=1)\n"
    "t = torch.ones(1 << 20, dtype=torch.int32, device='cuda:0')\n"
    "dist.all_reduce(t)\n"
    "torch.cuda.synchronize()\n"
    "print('rccl ws=1 allreduce OK', int(t[0]))\n"
    "dist.destroy_process_group()\n"
    "P", timeout=300)
# ws=2 on one GPU: RCCL may refuse duplicate devices in one communicator —
# this documents whether 2-rank hardware tests are possible on a 1-GPU box
run("rccl-ws2-1gpu",
    "python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 "
    "--master-addr 127.0.0.1 --master-port 29532 tools/rccl_ws2_probe.py",
    timeout=300)
print("\nPROBE DONE", flush=True)
