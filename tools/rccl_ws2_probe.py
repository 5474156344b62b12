"""torchrun target: 2 ranks sharing one GPU, RCCL all-reduce.

Documents whether multi-rank RCCL collectives can be hardware-tested on a
1-GPU box (NCCL historically refuses duplicate devices in a communicator;
RCCL behavior measured here). Launched by tools/r2probe.py.
"""
import os

import torch
import torch.distributed as dist

rank = int(os.environ["RANK"])
dist.init_process_group("nccl")
torch.cuda.set_device(0)
t = torch.full((1024,), rank + 1, dtype=torch.int32, device="cuda:0")
dist.all_reduce(t)
torch.cuda.synchronize()
print(f"rank {rank}: sum={int(t[0])} (expect 3)", flush=True)
dist.destroy_process_group()
