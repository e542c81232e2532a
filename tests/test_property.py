"""Property-based tests (hypothesis) for the pure-Python data layer —
invariants that hold for ANY input, not just the fixtures."""

import string

from hypothesis import given, settings
from hypothesis import strategies as st

from sat_amd.data.vocabulary import Vocabulary
from sat_amd.evalcap.bleu import Bleu
from sat_amd.evalcap.meteor import _align, _score_pair
from sat_amd.evalcap.rouge import Rouge
from sat_amd.utils.topn import CaptionData, TopN

words = st.text(alphabet=string.ascii_lowercase, min_size=1, max_size=6)
sentences = st.lists(words, min_size=1, max_size=12).map(' '.join)


@settings(max_examples=60, deadline=None)
@given(st.lists(sentences, min_size=1, max_size=8),
       st.integers(min_value=2, max_value=50))
def test_vocabulary_roundtrip_properties(corpus, size):
    v = Vocabulary(size)
    v.build([s + ' .' for s in corpus])
    # invariant: index 0 reserved, size bounded, indices consistent
    assert v.words[0] == '<start>'
    assert v.size <= size
    assert all(v.word2idx[w] == i for i, w in enumerate(v.words))
    # process/get round-trip: any processed sentence detokenizes to a
    # string ending in '.' containing only vocabulary words
    for s in corpus:
        idxs = v.process_sentence(s + ' .')
        out = v.get_sentence(idxs)
        assert out.endswith('.')
        for tok in out[:-1].split():
            assert tok in v.word2idx or tok in string.punctuation


@settings(max_examples=60, deadline=None)
@given(st.lists(st.tuples(st.floats(min_value=0, max_value=1,
                                    allow_nan=False),
                          st.integers(0, 1000)),
                min_size=1, max_size=30),
       st.integers(min_value=1, max_value=5))
def test_topn_keeps_the_n_best(items, n):
    t = TopN(n)
    for score, tag in items:
        t.push(CaptionData([tag], None, None, score))
    got = t.extract(sort=True)
    assert len(got) == min(n, len(items))
    scores = [c.score for c in got]
    assert scores == sorted(scores, reverse=True)
    # they are the global top-n scores
    best = sorted((s for s, _ in items), reverse=True)[:n]
    assert [round(s, 9) for s in scores] == [round(s, 9) for s in best]


@settings(max_examples=40, deadline=None)
@given(sentences)
def test_metrics_identity_bounds(s):
    gts = {1: [s]}
    res = {1: [s]}
    bleu, _ = Bleu(4).compute_score(gts, res)
    assert all(0.0 <= b <= 1.0 + 1e-9 for b in bleu)
    assert bleu[0] > 0.99  # identical unigram match
    rouge, _ = Rouge().compute_score(gts, res)
    assert abs(rouge - 1.0) < 1e-6


@settings(max_examples=60, deadline=None)
@given(st.lists(words, min_size=1, max_size=8),
       st.lists(words, min_size=1, max_size=8))
def test_meteor_alignment_invariants(hyp, ref):
    m, ch, w = _align(hyp, ref)
    assert 0 <= m <= min(len(hyp), len(ref))
    assert (m == 0 and ch == 0) or 1 <= ch <= m
    assert 0.0 <= w <= m + 1e-9
    sc = _score_pair(hyp, ref)
    assert 0.0 <= sc <= 1.0
    # symmetry of the match COUNT under exact-only corpora is not
    # guaranteed (stems differ), but score of identical sentences must
    # dominate any other hypothesis of the same length
    assert _score_pair(ref, ref) >= sc - 1e-9
