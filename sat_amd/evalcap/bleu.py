"""Corpus BLEU-1..4 for caption evaluation (clean-room, pure Python).

Interface parity with the reference metric module
(`utils/coco/pycocoevalcap/bleu/bleu.py:21` — `compute_score(gts, res)` ->
(4 corpus scores, 4 per-image score lists)).  Semantics follow the classic
Papineni BLEU as the reference's BleuScorer implements it
(`bleu_scorer.py:199-264`): clipped modified n-gram precision accumulated over
the corpus, brevity penalty from the *closest* reference length per image,
and the small/tiny epsilon smoothing so empty counts don't zero the corpus.
"""

import math
from collections import Counter


def _ngrams(tokens, n):
    return Counter(tuple(tokens[i:i + n]) for i in range(len(tokens) - n + 1))


class Bleu(object):
    def __init__(self, n=4):
        self.n = n

    def method(self):
        return "Bleu"

    def compute_score(self, gts, res):
        small = 1e-9
        tiny = 1e-15
        n = self.n

        total_correct = [0.0] * n
        total_guess = [0.0] * n
        total_hyp_len = 0
        total_ref_len = 0
        per_image = [[] for _ in range(n)]

        img_ids = list(gts.keys())
        for iid in img_ids:
            assert iid in res and len(res[iid]) >= 1
            hyp = res[iid][0].split()
            refs = [r.split() for r in gts[iid]]
            assert len(refs) >= 1

            hyp_len = len(hyp)
            # closest reference length (ties -> shorter, as in bleu_scorer)
            ref_len = min((abs(len(r) - hyp_len), len(r)) for r in refs)[1]
            total_hyp_len += hyp_len
            total_ref_len += ref_len

            correct = [0.0] * n
            guess = [0.0] * n
            for k in range(1, n + 1):
                hyp_ng = _ngrams(hyp, k)
                max_ref = Counter()
                for r in refs:
                    for ng, c in _ngrams(r, k).items():
                        if c > max_ref[ng]:
                            max_ref[ng] = c
                correct[k - 1] = float(sum(min(c, max_ref[ng])
                                           for ng, c in hyp_ng.items()))
                guess[k - 1] = float(max(0, hyp_len - k + 1))
                total_correct[k - 1] += correct[k - 1]
                total_guess[k - 1] += guess[k - 1]

            # per-sentence score (with its own brevity penalty)
            bp_s = 1.0
            if hyp_len < ref_len:
                bp_s = math.exp(1.0 - ref_len / (hyp_len + tiny)) \
                    if hyp_len > 0 else 0.0
            logp = 0.0
            for k in range(n):
                logp += math.log((correct[k] + tiny) / (guess[k] + small))
                per_image[k].append(math.exp(logp / (k + 1)) * bp_s)

        bp = 1.0
        if total_hyp_len < total_ref_len:
            bp = math.exp(1.0 - total_ref_len / (total_hyp_len + tiny)) \
                if total_hyp_len > 0 else 0.0
        scores = []
        logp = 0.0
        for k in range(n):
            logp += math.log((total_correct[k] + tiny)
                             / (total_guess[k] + small))
            scores.append(math.exp(logp / (k + 1)) * bp)
        return scores, per_image
