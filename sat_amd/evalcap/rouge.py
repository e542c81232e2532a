"""ROUGE-L for caption evaluation (clean-room, pure Python).

Interface parity with reference `utils/coco/pycocoevalcap/rouge/rouge.py`:
`compute_score(gts, res)` -> (mean score, per-image scores).  Semantics:
longest-common-subsequence precision/recall per reference, F-score with
beta = 1.2 (rouge.py:72), max over references per image, mean over images
(rouge.py:45-99).
"""

import numpy as np


def _lcs_len(a, b):
    """Length of the longest common subsequence of token lists a, b."""
    if not a or not b:
        return 0
    prev = [0] * (len(b) + 1)
    for x in a:
        cur = [0] * (len(b) + 1)
        for j, y in enumerate(b, 1):
            cur[j] = prev[j - 1] + 1 if x == y else max(prev[j], cur[j - 1])
        prev = cur
    return prev[-1]


class Rouge(object):
    def __init__(self):
        self.beta = 1.2

    def method(self):
        return "Rouge"

    def calc_score(self, candidate, refs):
        assert len(candidate) == 1 and len(refs) >= 1
        hyp = candidate[0].split()
        prec, rec = [], []
        for ref in refs:
            r = ref.split()
            lcs = _lcs_len(hyp, r)
            prec.append(lcs / len(hyp) if hyp else 0.0)
            rec.append(lcs / len(r) if r else 0.0)
        p, r = max(prec), max(rec)
        if p != 0 and r != 0:
            return ((1 + self.beta ** 2) * p * r) / (r + self.beta ** 2 * p)
        return 0.0

    def compute_score(self, gts, res):
        scores = []
        for iid in gts.keys():
            scores.append(self.calc_score(res[iid], gts[iid]))
        return float(np.mean(scores)), scores
