"""Training observability (parity with reference TensorBoard summaries,
`base_model.py:46-47,57-63` + `model.py:515-542`).

The tensorboard package is not in this image, so scalars go to a JSONL event
file (one {'step', tag: value} record per step) plus optional per-variable
mean/std/max/min statistics — the same scalar set the reference logs
(cross_entropy/attention/reg/total loss, accuracy, attention-map stats).
A JSONL file is trivially convertible to TB events offline.
"""

import json
import os
import time


class SummaryWriter(object):
    def __init__(self, log_dir):
        os.makedirs(log_dir, exist_ok=True)
        self.path = os.path.join(log_dir, 'events.jsonl')
        self._f = open(self.path, 'a')

    def add_scalar(self, tag, value, step):
        self._f.write(json.dumps(
            {'step': int(step), 'tag': tag, 'value': float(value),
             'wall_time': time.time()}) + '\n')

    def add_scalars(self, scalars, step):
        rec = {'step': int(step), 'wall_time': time.time()}
        rec.update({k: float(v) for k, v in scalars.items()})
        self._f.write(json.dumps(rec) + '\n')

    def variable_summary(self, name, tensor, step):
        """mean/stddev/max/min per variable (reference model.py:534-542)."""
        t = tensor.detach().float()
        self.add_scalars({
            '%s/mean' % name: t.mean().item(),
            '%s/stddev' % name: t.std().item() if t.numel() > 1 else 0.0,
            '%s/max' % name: t.max().item(),
            '%s/min' % name: t.min().item(),
        }, step)

    def flush(self):
        self._f.flush()

    def close(self):
        self._f.close()
