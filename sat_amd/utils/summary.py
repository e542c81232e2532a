"""Training observability (parity with reference TensorBoard summaries,
`base_model.py:46-47,57-63` + `model.py:515-542`).

Scalars are written twice: to a native TensorBoard event file (hand-
encoded Event protobuf + TFRecord framing, sat_amd/utils/tb_events.py —
a stock TensorBoard opens it) and to a JSONL sidecar (one
{'step', tag: value} record per step, grep/pandas-friendly).  The scalar
set matches what the reference logs (cross_entropy/attention/reg/total
loss, accuracy, attention-map stats, per-variable mean/std/max/min).
"""

import json
import os
import time

import torch

from .tb_events import TBEventWriter


class SummaryWriter(object):
    def __init__(self, log_dir):
        os.makedirs(log_dir, exist_ok=True)
        self.path = os.path.join(log_dir, 'events.jsonl')
        self._f = open(self.path, 'a')
        self._tb = TBEventWriter(log_dir)

    def add_scalar(self, tag, value, step):
        self._f.write(json.dumps(
            {'step': int(step), 'tag': tag, 'value': float(value),
             'wall_time': time.time()}) + '\n')
        self._tb.add_scalar(tag, value, step)

    def add_scalars(self, scalars, step):
        rec = {'step': int(step), 'wall_time': time.time()}
        rec.update({k: float(v) for k, v in scalars.items()})
        self._f.write(json.dumps(rec) + '\n')
        self._tb.add_scalars(scalars, step)

    def variable_summary(self, name, tensor, step):
        """mean/stddev/max/min + histogram per variable (reference
        model.py:534-542 logs exactly this set)."""
        t = tensor.detach().float()
        self.add_scalars({
            '%s/mean' % name: t.mean().item(),
            '%s/stddev' % name: t.std().item() if t.numel() > 1 else 0.0,
            '%s/max' % name: t.max().item(),
            '%s/min' % name: t.min().item(),
        }, step)
        # histogram on a bounded sample (full tensors are ~20 MB)
        flat = t.reshape(-1)
        if flat.numel() > 4096:
            idx = torch.linspace(0, flat.numel() - 1, 4096,
                                 dtype=torch.int64)
            flat = flat.cpu()[idx]
        self._tb.add_histogram('%s/hist' % name,
                               flat.cpu().tolist(), step)

    def flush(self):
        self._f.flush()
        self._tb.flush()

    def close(self):
        self._f.close()
        self._tb.close()
