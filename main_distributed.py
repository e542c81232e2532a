#!/usr/bin/env python3
"""Distributed (multi-GPU) entry point.

Replaces the reference's unfinished TF parameter-server launcher
(`main_distributed.py` + `clusterone_config.py`) with the MI355X-native
scheme: one process per GPU under torchrun, synchronous DP with RCCL
all-reduce over xGMI, per-rank dataset sharding, rank-0 chief for
checkpoints/summaries.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 main_distributed.py --phase=train [...]

Same flags as main.py.  eval/test phases run on rank 0 only (they are
single-GPU workloads in the reference too).
"""

import torch.distributed as dist

from config import Config
from main import build_parser
from sat_amd.data.dataset import (prepare_eval_data, prepare_test_data,
                                  prepare_train_data)
from sat_amd.models.base_model import BaseModel
from sat_amd.parallel.launch import init_distributed, shard_dataset


def main(argv=None):
    args = build_parser().parse_args(argv)
    rank, world, local_rank = init_distributed()

    config = Config()
    config.phase = args.phase
    config.train_cnn = args.train_cnn
    config.beam_size = args.beam_size
    config.device = args.device
    if args.synthetic:
        config.synthetic_data = True
    if args.compute_dtype:
        config.compute_dtype = args.compute_dtype

    if args.phase == 'train':
        data = prepare_train_data(config)
        data = shard_dataset(data, rank, world)
        model = BaseModel(config)
        if args.load:
            model.load(args.model_file)
        if args.load_cnn:
            model.load_cnn(args.cnn_model_file)
        model.train(data)
    elif rank == 0:
        if args.phase == 'eval':
            config.batch_size = 1
            coco, data, vocabulary = prepare_eval_data(config)
            model = BaseModel(config)
            model.load(args.model_file)
            model.eval(coco, data, vocabulary)
        else:
            data, vocabulary = prepare_test_data(config)
            model = BaseModel(config)
            model.load(args.model_file)
            model.test(data, vocabulary)

    if world > 1 and dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
