"""Compare MIOpen find modes at the bench config; capture the tuned DB.

Writes per-mode steady step times; MIOPEN_USER_DB_PATH must point into
gpurun_out so the tuned find-db merges back for committing in-tree.
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
mode = sys.argv[1] if len(sys.argv) > 1 else "HYBRID"
os.environ["MIOPEN_FIND_MODE"] = mode
dtype = sys.argv[2] if len(sys.argv) > 2 else "bf16"

import torch
from mine_amd.config import default_config
from mine_amd.data import SyntheticMPIDataset, collate_src_tgt
from mine_amd.engine import SynthesisTask

cfg = default_config(**{
    "data.name": "realestate10k", "data.img_h": 256, "data.img_w": 384,
    "mpi.num_bins_coarse": 64, "data.per_gpu_batch_size": 4,
    "data.visible_point_count": 256, "lr.decay_steps": [4, 8],
    "training.amp_dtype": dtype,
})
ds = SyntheticMPIDataset(cfg, length=4)
items = collate_src_tgt([ds[i] for i in range(4)])
task = SynthesisTask(cfg, device="cuda:0")
t0 = time.time(); task.train_step(items); torch.cuda.synchronize()
print(f"[{mode} {dtype}] first step {time.time()-t0:.1f}s", flush=True)
t0 = time.time()
n = 4
for _ in range(n):
    task.train_step(items)
torch.cuda.synchronize()
dt = (time.time()-t0)/n
print(f"[{mode} {dtype}] steady {dt*1000:.0f} ms/step -> {4/dt:.1f} imgs/s/gpu", flush=True)
