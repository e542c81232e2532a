"""METEOR for caption evaluation (pure Python, no Java).

The reference shells out to the METEOR-1.5 Java jar through a persistent
subprocess pipe (`utils/coco/pycocoevalcap/meteor/meteor.py:15-58`); the jar
is a missing git-LFS blob in this environment.  This module re-implements the
METEOR scoring *formula* in Python with the exact-match and simple-stem
matcher stages (the paraphrase-table stage needs the missing
`paraphrase-en.gz` data file and is omitted — documented approximation).

Parameters are METEOR 1.5's English defaults: alpha=0.85, beta=0.2,
gamma=0.6 (fragmentation penalty on chunk count).  Interface matches the
reference: `compute_score(gts, res)` -> (mean score, per-image scores).
"""

import numpy as np

_ALPHA = 0.85
_BETA = 0.2
_GAMMA = 0.6


def _stem(w):
    """Tiny Porter-ish suffix stripper — approximates METEOR's stem module."""
    for suf in ('ing', 'edly', 'ed', 'es', 's', 'ly'):
        if w.endswith(suf) and len(w) - len(suf) >= 3:
            return w[: len(w) - len(suf)]
    return w


def _align(hyp, ref):
    """Greedy left-to-right alignment: exact first, then stem matches.

    Returns (num_matches, num_chunks) where chunks are maximal runs of
    matches that are contiguous and order-preserving in both strings.
    """
    m = len(hyp)
    used_ref = [False] * len(ref)
    match_of = [None] * m  # hyp position -> ref position

    # stage 1: exact
    for i, w in enumerate(hyp):
        for j, r in enumerate(ref):
            if not used_ref[j] and match_of[i] is None and w == r:
                match_of[i] = j
                used_ref[j] = True
                break
    # stage 2: stem
    hs = [_stem(w) for w in hyp]
    rs = [_stem(w) for w in ref]
    for i in range(m):
        if match_of[i] is not None:
            continue
        for j in range(len(ref)):
            if not used_ref[j] and hs[i] == rs[j]:
                match_of[i] = j
                used_ref[j] = True
                break

    pairs = [(i, j) for i, j in enumerate(match_of) if j is not None]
    matches = len(pairs)
    if matches == 0:
        return 0, 0
    chunks = 1
    for (i0, j0), (i1, j1) in zip(pairs, pairs[1:]):
        if not (i1 == i0 + 1 and j1 == j0 + 1):
            chunks += 1
    return matches, chunks


def _score_pair(hyp_toks, ref_toks):
    if not hyp_toks or not ref_toks:
        return 0.0
    m, ch = _align(hyp_toks, ref_toks)
    if m == 0:
        return 0.0
    p = m / len(hyp_toks)
    r = m / len(ref_toks)
    fmean = (p * r) / (_ALPHA * p + (1 - _ALPHA) * r)
    frag = ch / m
    penalty = _GAMMA * (frag ** _BETA)
    return fmean * (1.0 - penalty)


class Meteor(object):
    def method(self):
        return "METEOR"

    def compute_score(self, gts, res):
        scores = []
        for iid in gts.keys():
            hyp = res[iid][0].split()
            best = max(_score_pair(hyp, ref.split()) for ref in gts[iid])
            scores.append(best)
        return float(np.mean(scores)), scores
