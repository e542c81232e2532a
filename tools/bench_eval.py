#!/usr/bin/env python3
"""Eval (beam search) throughput: captions/sec at beam=3, batched.

    python tools/bench_eval.py [--batch 64] [--batches 5] [--beam 3]
                               [--host-beam]

Measures the device-resident beam path (or the host-heap reference path
with --host-beam) on the flagship eval model shape: VGG16 frozen,
LSTM-512, vocab 5000, synthetic images.
"""
import argparse
import sys
import time

sys.path.insert(0, '.')

import torch  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--batch', type=int, default=64)
    p.add_argument('--batches', type=int, default=5)
    p.add_argument('--beam', type=int, default=3)
    p.add_argument('--host-beam', action='store_true')
    args = p.parse_args()

    from config import Config
    from sat_amd.models.base_model import BaseModel
    from sat_amd.data.vocabulary import Vocabulary

    cfg = Config()
    cfg.phase = 'eval'
    cfg.train_cnn = False
    cfg.synthetic_data = True
    cfg.beam_size = args.beam
    cfg.use_device_beam = not args.host_beam
    torch.manual_seed(cfg.seed)

    vocab = Vocabulary(cfg.vocabulary_size)
    vocab.build(['a man rides a horse down the street .',
                 'a dog sits on the beach sand .'])
    m = BaseModel(cfg)
    files = ['synthetic://%d' % i for i in range(args.batch)]

    m.beam_search(files, vocab)  # warmup
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.batches):
        m.beam_search(files, vocab)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    n = args.batch * args.batches
    print('%s beam=%d batch=%d: %.1f captions/sec (%.1f ms/batch)'
          % ('host' if args.host_beam else 'device', args.beam,
             args.batch, n / dt, dt / args.batches * 1000))


if __name__ == '__main__':
    main()
