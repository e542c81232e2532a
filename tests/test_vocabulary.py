import numpy as np

from sat_amd.data.vocabulary import Vocabulary


def _build():
    v = Vocabulary(10)
    v.build(['a man riding a horse.', 'a dog.', 'a man walking.'])
    return v


def test_start_token_is_index_zero():
    v = _build()
    assert v.words[0] == '<start>'
    assert v.word2idx['<start>'] == 0


def test_frequency_ranked():
    v = _build()
    # 'a' (4x) must be the most frequent word after <start>
    assert v.words[1] == 'a'


def test_log_frequencies_max_zero():
    v = _build()
    assert np.isclose(v.word_frequencies.max(), 0.0)


def test_process_sentence_roundtrip():
    v = _build()
    idxs = v.process_sentence('a man riding a horse.')
    assert [v.words[i] for i in idxs] == \
        ['a', 'man', 'riding', 'a', 'horse', '.']


def test_get_sentence_truncates_at_period():
    v = _build()
    idxs = v.process_sentence('a man riding.') + \
        v.process_sentence('a dog.')
    s = v.get_sentence(idxs)
    assert s == 'a man riding.'


def test_get_sentence_appends_period():
    v = _build()
    idxs = v.process_sentence('a man')
    assert v.get_sentence(idxs) == 'a man.'


def test_save_load_roundtrip(tmp_path):
    v = _build()
    f = str(tmp_path / 'vocab.csv')
    v.save(f)
    v2 = Vocabulary(10, f)
    assert list(v2.words) == list(v.words)
    assert v2.word2idx == v.word2idx
