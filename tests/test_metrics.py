import math

import numpy as np

from sat_amd.evalcap.bleu import Bleu
from sat_amd.evalcap.cider import Cider
from sat_amd.evalcap.meteor import Meteor
from sat_amd.evalcap.rouge import Rouge


GTS = {1: ['a man riding a horse'], 2: ['a dog on the beach']}
PERFECT = {1: ['a man riding a horse'], 2: ['a dog on the beach']}
WRONG = {1: ['blue elephants fly'], 2: ['purple trains sing loud']}


def test_bleu_perfect_is_one():
    scores, per_img = Bleu(4).compute_score(GTS, PERFECT)
    for s in scores:
        assert abs(s - 1.0) < 1e-6
    assert len(per_img[0]) == 2


def test_bleu_wrong_is_near_zero():
    scores, _ = Bleu(4).compute_score(GTS, WRONG)
    assert scores[0] < 0.1


def test_bleu_brevity_penalty():
    gts = {1: ['a b c d e f g h']}
    res = {1: ['a b c d']}
    scores, _ = Bleu(4).compute_score(gts, res)
    # unigram precision 1.0 but hyp len 4 vs ref 8 -> bp = e^(1-2)
    assert abs(scores[0] - math.exp(-1.0)) < 1e-6


def test_bleu_clipping():
    gts = {1: ['the cat']}
    res = {1: ['the the the the']}
    scores, _ = Bleu(1).compute_score(gts, res)
    # clipped count 1 of 4 guesses; no bp (hyp longer)
    assert abs(scores[0] - 0.25) < 1e-6


def test_rouge_perfect_is_one():
    score, per = Rouge().compute_score(GTS, PERFECT)
    assert abs(score - 1.0) < 1e-6


def test_rouge_orders():
    s_good, _ = Rouge().compute_score(GTS, {1: ['a man riding'],
                                            2: ['a dog beach']})
    s_bad, _ = Rouge().compute_score(GTS, WRONG)
    assert s_good > s_bad


def test_cider_perfect_beats_wrong():
    s_good, arr = Cider().compute_score(GTS, PERFECT)
    s_bad, _ = Cider().compute_score(GTS, WRONG)
    assert s_good > s_bad
    assert len(arr) == 2


def test_meteor_perfect_beats_wrong():
    s_good, _ = Meteor().compute_score(GTS, PERFECT)
    s_bad, _ = Meteor().compute_score(GTS, WRONG)
    assert s_good > s_bad >= 0.0
    assert s_good <= 1.0


def test_meteor_stem_matching():
    s, _ = Meteor().compute_score({1: ['a man rides a horse']},
                                  {1: ['a man riding a horse']})
    assert s > 0.5  # 'riding' should stem-match 'rides'


def test_full_eval_driver(tmp_path):
    import json
    from sat_amd.data.coco import COCO
    from sat_amd.data.synthetic import make_coco
    from sat_amd.evalcap.eval import COCOEvalCap

    gt = make_coco(3, 2, seed=3)
    results = [{'image_id': i,
                'caption': gt.imgToAnns[i][0]['caption']}
               for i in gt.imgs.keys()]
    rf = tmp_path / 'res.json'
    rf.write_text(json.dumps(results))
    res = gt.loadRes(str(rf))
    scorer = COCOEvalCap(gt, res, None)
    scorer.evaluate()
    assert scorer.eval['Bleu_1'] > 0.9  # echoing a GT caption scores high
    assert set(scorer.eval) == {'Bleu_1', 'Bleu_2', 'Bleu_3', 'Bleu_4',
                                'METEOR', 'ROUGE_L', 'CIDEr'}


def test_metrics_tolerate_empty_candidate():
    gts = {1: ['a man riding a horse']}
    res = {1: ['']}
    scores, _ = Bleu(4).compute_score(gts, res)
    assert all(0.0 <= s < 1e-6 for s in scores)
    s, _ = Rouge().compute_score(gts, res)
    assert s == 0.0
    s, _ = Meteor().compute_score(gts, res)
    assert s == 0.0
    s, _ = Cider().compute_score(gts, res)
    assert s == 0.0


def test_bleu_single_word():
    scores, _ = Bleu(4).compute_score({1: ['dog']}, {1: ['dog']})
    assert scores[0] > 0.9  # unigram perfect; higher n-grams degenerate


def test_meteor_beam_alignment_hand_computed():
    """Hand-checkable METEOR alignments: the beam aligner must find the
    chunk-minimizing permutation a greedy left-to-right matcher misses."""
    from sat_amd.evalcap.meteor import _align, _score_pair

    # identical sentences: all matched, one chunk
    m, ch, w = _align(['a', 'b', 'c'], ['a', 'b', 'c'])
    assert (m, ch, w) == (3, 1, 3.0)

    # duplicate word: a greedy left-to-right matcher binds hyp 'the' to
    # ref[0], splitting the alignment into 3 chunks.  The optimal
    # permutation (dog->1, and->2, the->3, cat->4) is contiguous in
    # both sentences: ONE chunk.
    hyp = ['dog', 'and', 'the', 'cat']
    ref = ['the', 'dog', 'and', 'the', 'cat']
    m, ch, w = _align(hyp, ref)
    assert m == 4 and w == 4.0
    assert ch == 1, 'expected chunk-minimizing alignment, got %d' % ch

    # stem-stage match carries weight 0.6
    m, ch, w = _align(['walked'], ['walks'])
    assert m == 1 and abs(w - 0.6) < 1e-9

    # no matches
    assert _score_pair(['x'], ['y']) == 0.0

    # perfect 2-word match: fmean=1, frag = 1 chunk / 2 matches = 0.5,
    # penalty = gamma * 0.5^beta
    s = _score_pair(['a', 'b'], ['a', 'b'])
    assert abs(s - (1.0 - 0.6 * 0.5 ** 0.2)) < 1e-9
