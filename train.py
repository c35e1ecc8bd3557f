#!/usr/bin/env python3
"""Training entry point.

CLI contract matches the reference (ref train.py:18-25):
    python train.py --config_path configs/params_llff.yaml \
        --workspace /path/ws --version v1 --extra_config '{"key": val}'

Launch one process per GPU with torchrun / torch.distributed.run
(RANK/LOCAL_RANK/WORLD_SIZE from the environment; the legacy
--local_rank flag is also accepted). Each rank masks its own GPU via
CUDA_VISIBLE_DEVICES from `training.gpus` (ref train.py:58-60) so every
process sees its device as cuda:0, and collectives run over RCCL.
"""
from __future__ import annotations

import argparse
import os
import shutil
import sys


def parse_args():
    parser = argparse.ArgumentParser(description="mine_amd training")
    parser.add_argument("--config_path", default="configs/params_default.yaml", type=str)
    parser.add_argument("--workspace", type=str, required=True)
    parser.add_argument("--version", type=str, required=True)
    parser.add_argument("--extra_config", type=str, default="{}")
    parser.add_argument("--local_rank", default=None, type=int,
                        help="legacy torch.distributed.launch rank flag")
    return parser.parse_args()


def main():
    args = parse_args()
    local_rank = args.local_rank
    if local_rank is None:
        local_rank = int(os.environ.get("LOCAL_RANK", 0))

    from mine_amd.config import RuntimeState, load_config

    config = load_config(args.config_path, args.extra_config)

    # Per-rank GPU masking BEFORE importing anything that initializes HIP.
    gpus = config.get("training.gpus", [0])
    if not isinstance(gpus, list):
        gpus = [int(s) for s in str(gpus).split(",")]
    if os.environ.get("CUDA_VISIBLE_DEVICES") is None and len(gpus) > local_rank:
        os.environ["CUDA_VISIBLE_DEVICES"] = str(gpus[local_rank])

    import torch
    from torch.utils.data import DataLoader

    from mine_amd.data import get_dataset
    from mine_amd.engine import SynthesisTask
    from mine_amd.parallel import init_distributed
    from mine_amd.utils import setup_logger

    rank, local_rank, world_size = init_distributed()
    state = RuntimeState(global_rank=rank, local_rank=local_rank,
                         world_size=world_size)

    # Device selection: when training.gpus masked CUDA_VISIBLE_DEVICES
    # above, every rank sees its GPU as cuda:0; otherwise (e.g. default
    # `training.gpus: 0` under torchrun with all GPUs visible) each rank
    # takes device local_rank — without this, ranks 1+ would all land on
    # cuda:0.
    device = None
    if torch.cuda.is_available():
        dev_idx = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(dev_idx)
        device = f"cuda:{dev_idx}"

    workspace = os.path.join(args.workspace, args.version)
    state.local_workspace = workspace
    logger = None
    if rank == 0:
        os.makedirs(workspace, exist_ok=True)
        state.log_file = os.path.join(workspace, "training.log")
        logger = setup_logger("mine_amd", state.log_file)
        logger.info("Training config: {}".format(dict(config)))
        config.dump_yaml(os.path.join(workspace, "params.yaml"))
        from mine_amd.utils.summary import create_summary_writer
        state.tb_writer = create_summary_writer(workspace)
    state.logger = logger
    if world_size > 1:
        import torch.distributed as dist
        dist.barrier()

    train_dataset = get_dataset(config, logger, is_validation=False)
    val_dataset = get_dataset(config, logger, is_validation=True)

    if world_size > 1:
        sampler = torch.utils.data.distributed.DistributedSampler(train_dataset)
    else:
        sampler = None
    train_loader = DataLoader(train_dataset,
                              batch_size=config["data.per_gpu_batch_size"],
                              shuffle=(sampler is None), sampler=sampler,
                              drop_last=True,
                              num_workers=int(config.get("data.num_workers", 0)),
                              collate_fn=train_dataset.collate_fn,
                              pin_memory=torch.cuda.is_available())
    val_loader = DataLoader(val_dataset,
                            batch_size=config["data.per_gpu_batch_size"],
                            shuffle=False, drop_last=False, num_workers=0,
                            collate_fn=val_dataset.collate_fn)

    task = SynthesisTask(config, state=state, logger=logger, device=device)
    task.train(train_loader, val_loader)


if __name__ == "__main__":
    sys.exit(main())
