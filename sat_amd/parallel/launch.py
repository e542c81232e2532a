"""Distributed bring-up + per-rank data sharding.

Replaces the reference's Clusterone env plumbing (`clusterone_config.py`:
JOB_NAME/TASK_INDEX/PS_HOSTS/WORKER_HOSTS → tf.train.Server) with the
torchrun convention: RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR from the
environment, `nccl` (=RCCL over xGMI) backend on GPU, `gloo` on CPU.
"""

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend=None):
    """Initialize torch.distributed from torchrun env vars.

    Returns (rank, world_size, local_rank); no-op (0,1,0) outside torchrun.
    """
    if 'WORLD_SIZE' not in os.environ \
            or int(os.environ['WORLD_SIZE']) <= 1:
        return 0, 1, 0
    rank = int(os.environ['RANK'])
    world = int(os.environ['WORLD_SIZE'])
    local_rank = int(os.environ.get('LOCAL_RANK', rank))
    if backend is None:
        backend = 'nccl' if torch.cuda.is_available() else 'gloo'
    if backend == 'nccl':
        torch.cuda.set_device(local_rank)
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(minutes=10))
    return rank, world, local_rank


def shard_dataset(dataset, rank, world):
    """Restrict a DataSet to this rank's contiguous shard (per-rank data
    sharding replaces the reference's identical-data async PS free-for-all,
    SURVEY.md §5.8)."""
    if world <= 1:
        return dataset
    n = len(dataset.image_ids)
    per = (n + world - 1) // world
    lo, hi = rank * per, min(n, (rank + 1) * per)
    dataset.image_ids = dataset.image_ids[lo:hi]
    dataset.image_files = dataset.image_files[lo:hi]
    if dataset.word_idxs is not None:
        dataset.word_idxs = dataset.word_idxs[lo:hi]
    if dataset.masks is not None:
        dataset.masks = dataset.masks[lo:hi]
    dataset.setup()
    return dataset
