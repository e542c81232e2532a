"""METEOR for caption evaluation (pure Python, no Java).

The reference shells out to the METEOR-1.5 Java jar through a persistent
subprocess pipe (`utils/coco/pycocoevalcap/meteor/meteor.py:15-58`); the
jar is a missing git-LFS blob in this environment.  This module
re-implements METEOR 1.5's scoring pipeline:

  * candidate matches from the EXACT (weight 1.0) and STEM (weight 0.6)
    matcher stages;
  * alignment resolution as METEOR does it — a BEAM SEARCH over match
    permutations selecting the alignment with (1) the most matched
    words, then (2) the fewest chunks, then (3) the highest stage
    weight — not a greedy left-to-right pass;
  * weighted P/R fmean (alpha=0.85) and the fragmentation penalty
    gamma * (chunks/matches)^beta with METEOR 1.5's English defaults
    (beta=0.2, gamma=0.6);
  * per-segment best-reference selection.

Documented approximations vs METEOR 1.5 proper (both need data files
that are missing LFS blobs in the reference checkout too):
  * no SYNONYM stage (needs WordNet) and no PARAPHRASE stage (needs
    paraphrase-en.gz);
  * no function-word discounting (delta) — without a function-word list
    every token is a content word, which is the delta-neutral case.
Scores are therefore NOT comparable to published METEOR numbers; eval
output labels them "METEOR (approx)".

Interface matches the reference: `compute_score(gts, res)` ->
(mean score, per-image scores).
"""

import numpy as np

_ALPHA = 0.85
_BETA = 0.2
_GAMMA = 0.6
_W_EXACT = 1.0
_W_STEM = 0.6
_BEAM = 128


def _stem(w):
    """Tiny Porter-ish suffix stripper — approximates METEOR's stemmer."""
    for suf in ('ing', 'edly', 'ed', 'es', 's', 'ly'):
        if w.endswith(suf) and len(w) - len(suf) >= 3:
            return w[: len(w) - len(suf)]
    return w


def _candidates(hyp, ref):
    """Per hyp position: list of (ref position, stage weight) options,
    exact matches preferred over stem matches for the same pair."""
    rs = [_stem(r) for r in ref]
    out = []
    for w in hyp:
        ws = _stem(w)
        opts = []
        for j, r in enumerate(ref):
            if w == r:
                opts.append((j, _W_EXACT))
            elif ws == rs[j]:
                opts.append((j, _W_STEM))
        out.append(opts)
    return out


def _align(hyp, ref):
    """METEOR-style alignment via beam search.

    State per partial alignment (hyp prefix processed):
      (used-ref bitmask, matches, chunks, weight, last matched (i, j)).
    Returns (matches, chunks, total stage weight).
    """
    cands = _candidates(hyp, ref)
    # beam entries: key = used mask; value tuple
    # (matches, -chunks, weight) is the comparison order METEOR uses
    beam = [(0, 0, 0, 0.0, None)]  # mask, matches, chunks, weight, last
    for i, opts in enumerate(cands):
        nxt = []
        for mask, m, ch, wsum, last in beam:
            # skip this hyp word
            nxt.append((mask, m, ch, wsum, last))
            for j, w in opts:
                bit = 1 << j
                if mask & bit:
                    continue
                contig = (last is not None
                          and last[0] == i - 1 and last[1] == j - 1)
                nxt.append((mask | bit, m + 1,
                            ch + (0 if contig else 1),
                            wsum + w, (i, j)))
        # prune: best state per (comparable) rank
        nxt.sort(key=lambda s: (-s[1], s[2], -s[3]))
        seen = set()
        beam = []
        for s in nxt:
            if s[0] in seen:
                continue
            seen.add(s[0])
            beam.append(s)
            if len(beam) >= _BEAM:
                break
    best = max(beam, key=lambda s: (s[1], -s[2], s[3]))
    return best[1], best[2], best[3]


def _score_pair(hyp_toks, ref_toks):
    if not hyp_toks or not ref_toks:
        return 0.0
    m, ch, wsum = _align(hyp_toks, ref_toks)
    if m == 0:
        return 0.0
    p = wsum / len(hyp_toks)
    r = wsum / len(ref_toks)
    fmean = (p * r) / (_ALPHA * p + (1 - _ALPHA) * r)
    frag = ch / m
    penalty = _GAMMA * (frag ** _BETA)
    return fmean * (1.0 - penalty)


class Meteor(object):
    def method(self):
        return "METEOR (approx)"

    def compute_score(self, gts, res):
        scores = []
        for iid in gts.keys():
            hyp = res[iid][0].split()
            best = max(_score_pair(hyp, ref.split()) for ref in gts[iid])
            scores.append(best)
        return float(np.mean(scores)), scores
