"""CIDEr for caption evaluation (clean-room, pure Python).

Interface parity with reference `utils/coco/pycocoevalcap/cider/cider.py`:
`compute_score(gts, res)` -> (mean score, per-image array).  Semantics follow
cider_scorer.py:47-181: TF-IDF-weighted 1..4-gram vectors (document frequency
over the reference corpus, log(N/df) IDF clipped at 0), cosine similarity
hypothesis-vs-each-reference with count clipping, a length-difference gaussian
penalty (sigma = 6), average over references, mean over n, x10.
"""

import math
from collections import Counter, defaultdict

import numpy as np


def _ngram_counts(tokens, n_max=4):
    out = []
    for n in range(1, n_max + 1):
        out.append(Counter(tuple(tokens[i:i + n])
                           for i in range(len(tokens) - n + 1)))
    return out


class Cider(object):
    def __init__(self, n=4, sigma=6.0):
        self.n = n
        self.sigma = sigma

    def method(self):
        return "CIDEr"

    def compute_score(self, gts, res):
        img_ids = list(gts.keys())
        crefs = {i: [_ngram_counts(r.split(), self.n) for r in gts[i]]
                 for i in img_ids}
        ctests = {i: _ngram_counts(res[i][0].split(), self.n) for i in img_ids}

        # document frequency over reference sets
        df = defaultdict(float)
        for i in img_ids:
            seen = set()
            for ref in crefs[i]:
                for n in range(self.n):
                    seen.update(ref[n].keys())
            for ng in seen:
                df[ng] += 1.0
        log_n_imgs = math.log(max(len(img_ids), 1))

        def vec(counts):
            """TF-IDF vector + per-n norms + token length."""
            v = [defaultdict(float) for _ in range(self.n)]
            norm = [0.0] * self.n
            length = 0
            for n in range(self.n):
                for ng, c in counts[n].items():
                    idf = log_n_imgs - math.log(max(df[ng], 1.0))
                    idf = max(idf, 0.0)
                    v[n][ng] = c * idf
                    norm[n] += v[n][ng] ** 2
                    if n == 0:
                        length += c
            return v, [math.sqrt(x) for x in norm], length

        def sim(vh, nh, lh, vr, nr, lr):
            delta = float(lh - lr)
            val = np.zeros(self.n)
            for n in range(self.n):
                s = 0.0
                for ng, w in vh[n].items():
                    s += min(w, vr[n].get(ng, 0.0)) * vr[n].get(ng, 0.0)
                if nh[n] != 0 and nr[n] != 0:
                    s /= (nh[n] * nr[n])
                val[n] = s * math.exp(-(delta ** 2) / (2 * self.sigma ** 2))
            return val

        scores = []
        for i in img_ids:
            vh, nh, lh = vec(ctests[i])
            score = np.zeros(self.n)
            for ref in crefs[i]:
                vr, nr, lr = vec(ref)
                score += sim(vh, nh, lh, vr, nr, lr)
            score_avg = np.mean(score / max(len(crefs[i]), 1)) * 10.0
            scores.append(score_avg)
        scores = np.array(scores)
        return float(np.mean(scores)), scores
