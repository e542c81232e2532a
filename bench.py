#!/usr/bin/env python3
"""Flagship training benchmark — the driver contract.

Measures training images/sec (the BASELINE.json metric; the reference
publishes no throughput, so vs_baseline is null) on the BASELINE Config #2
flagship: VGG16 encoder (frozen) + 512-unit soft-attention LSTM decoder,
batch 32 per GPU, bf16 compute, synthetic COCO-shaped data (random-init
weights; no network/datasets in this environment), teacher-forced T=20.

    python bench.py [--gpus N] [--steps K] [--warmup W] [--batch B]
                    [--cnn vgg16|resnet50] [--no-hip-graph]

For N>1 the driver launches this under torch.distributed.run with one rank
per GPU (RCCL over xGMI); per-GPU work is fixed (weak scaling).  Timing: W
untimed warmup steps, then exactly K steps bracketed by dist barrier +
torch.cuda.synchronize on both sides; elapsed is MAX over ranks; rank 0
prints one JSON line.

A training step = CNN forward + 20-step attention-LSTM forward + backward +
bucketed gradient all-reduce (N>1) + global-norm clip + Adam update — no
work is skipped or cached inside the timed region.
"""

import argparse
import json
import os
import time

import numpy as np
import torch


def make_batches(cfg, device, n_batches=4):
    """Pre-generate synthetic device-resident batches (input pipeline is
    outside the timed region; shapes/dtypes identical to the real loader)."""
    g = torch.Generator(device='cpu').manual_seed(cfg.seed)
    batches = []
    T = cfg.max_caption_length
    for _ in range(n_batches):
        images = torch.randn(cfg.batch_size, 3, 224, 224,
                             generator=g) * 50.0
        sentences = torch.randint(1, cfg.vocabulary_size,
                                  (cfg.batch_size, T), generator=g)
        lengths = torch.randint(8, T + 1, (cfg.batch_size,), generator=g)
        masks = (torch.arange(T).unsqueeze(0)
                 < lengths.unsqueeze(1)).float()
        batches.append((images.to(device), sentences.to(device),
                        masks.to(device)))
    return batches


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=30)
    p.add_argument('--warmup', type=int, default=10)
    p.add_argument('--batch', type=int, default=32)
    p.add_argument('--cnn', default='vgg16',
                   choices=['vgg16', 'resnet50'])
    p.add_argument('--train-cnn', action='store_true',
                   help='BASELINE config #4 shape: end-to-end CNN+RNN')
    p.add_argument('--no-hip-graph', action='store_true')
    args = p.parse_args()

    from config import Config
    from sat_amd.parallel.launch import init_distributed
    from sat_amd.models.base_model import BaseModel

    rank, world, local_rank = init_distributed()
    dist = torch.distributed if world > 1 else None

    cfg = Config()
    cfg.phase = 'train'
    cfg.train_cnn = args.train_cnn
    cfg.cnn = args.cnn
    cfg.synthetic_data = True
    cfg.batch_size = args.batch
    cfg.compute_dtype = 'bf16'
    cfg.use_hip_graph = not args.no_hip_graph
    torch.manual_seed(cfg.seed)

    have_gpu = torch.cuda.is_available()
    device = torch.device('cuda', local_rank if world > 1 else 0) \
        if have_gpu else torch.device('cpu')
    cfg.device = 'cuda' if have_gpu else 'cpu'

    model = BaseModel(cfg)
    batches = make_batches(cfg, device)

    def sync():
        if dist is not None:
            dist.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        model.train_step(*batches[i % len(batches)])

    sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        model.train_step(*batches[i % len(batches)])
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if have_gpu else 'cpu')
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    total_images = args.steps * cfg.batch_size * (world if world > 1 else 1)
    value = total_images / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": "training images/sec",
            "value": round(value, 2),
            "unit": "images/sec",
            "n_gpus": world if world > 1 else 1,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if have_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": "%s_%s+attention_lstm512"
                % (cfg.cnn,
                   'train_cnn' if cfg.train_cnn else 'frozen'),
                "global_batch": cfg.batch_size *
                (world if world > 1 else 1),
                "seq_len": cfg.max_caption_length,
                "vocab": cfg.vocabulary_size,
                "parallelism": "dp%d" % (world if world > 1 else 1),
            },
        }))

    if dist is not None and dist.is_initialized():
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
