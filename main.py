#!/usr/bin/env python3
"""CLI entry point (parity with reference `main.py`).

    python main.py --phase=train
    python main.py --phase=eval --model_file='./models/xxxxxx.npy' [--beam_size=3]
    python main.py --phase=test --model_file='./models/xxxxxx.npy' [--beam_size=3]

Same seven flags as reference main.py:15-36 (+ MI355X extras: --device,
--synthetic, --compute_dtype).  Dispatches to train/eval/test on the
runtime (sat_amd.models.BaseModel), loading checkpoints / pretrained CNNs
exactly like reference main.py:49-72.
"""

import argparse

import torch

from config import Config
from sat_amd.data.dataset import (prepare_eval_data, prepare_test_data,
                                  prepare_train_data)
from sat_amd.models.base_model import BaseModel


def build_parser():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument('--phase', default='train',
                   choices=['train', 'eval', 'test'],
                   help='The phase can be train, eval or test')
    p.add_argument('--load', action='store_true',
                   help='Turn on to load a pretrained model from either '
                        'the latest checkpoint or a specified file')
    p.add_argument('--model_file', default=None,
                   help='If sepcified, load a pretrained model from this '
                        'file')
    p.add_argument('--load_cnn', action='store_true',
                   help='Turn on to load a pretrained CNN model')
    p.add_argument('--cnn_model_file', default='./vgg16_no_fc.npy',
                   help='The file containing a pretrained CNN model')
    p.add_argument('--train_cnn', action='store_true',
                   help='Turn on to train both CNN and RNN. Otherwise, '
                        'only RNN is trained')
    p.add_argument('--beam_size', type=int, default=3,
                   help='The size of beam search for caption generation')
    # MI355X-native extras
    p.add_argument('--device', default='auto',
                   choices=['auto', 'cuda', 'cpu'])
    p.add_argument('--synthetic', action='store_true',
                   help='Use synthetic COCO-shaped data (no files/network)')
    p.add_argument('--compute_dtype', default=None,
                   choices=['bf16', 'fp32'])
    return p


def main(argv=None):
    args = build_parser().parse_args(argv)
    config = Config()
    config.phase = args.phase
    config.train_cnn = args.train_cnn
    config.beam_size = args.beam_size
    config.device = args.device
    if args.synthetic:
        config.synthetic_data = True
    if args.compute_dtype:
        config.compute_dtype = args.compute_dtype

    torch.manual_seed(config.seed)

    if args.phase == 'train':
        data = prepare_train_data(config)
        model = BaseModel(config)
        if args.load:
            model.load(args.model_file)
        if args.load_cnn:
            model.load_cnn(args.cnn_model_file)
        model.train(data)

    elif args.phase == 'eval':
        config.batch_size = 1  # reference main.py:59 forces eval batch 1
        coco, data, vocabulary = prepare_eval_data(config)
        model = BaseModel(config)
        model.load(args.model_file)
        model.eval(coco, data, vocabulary)

    else:
        data, vocabulary = prepare_test_data(config)
        model = BaseModel(config)
        model.load(args.model_file)
        model.test(data, vocabulary)


if __name__ == '__main__':
    main()
