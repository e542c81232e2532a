from sat_amd.data.tokenizer import PTBTokenizer, word_tokenize


def test_word_tokenize_basic():
    assert word_tokenize("A man riding a horse.") == \
        ['a', 'man', 'riding', 'a', 'horse', '.']


def test_word_tokenize_contractions():
    assert word_tokenize("don't stop") == ['do', "n't", 'stop']
    assert word_tokenize("the dog's ball") == ['the', 'dog', "'s", 'ball']


def test_word_tokenize_punct_split():
    assert word_tokenize("red, white and blue!") == \
        ['red', ',', 'white', 'and', 'blue', '!']


def test_ptb_tokenizer_strips_punct():
    tok = PTBTokenizer()
    out = tok.tokenize({1: [{'caption': 'A man, riding a horse.'}]})
    assert out == {1: ['a man riding a horse']}


def test_ptb_tokenizer_multiple_captions():
    tok = PTBTokenizer()
    out = tok.tokenize({1: ['a dog.', 'a cat!'], 2: ['a bus.']})
    assert out[1] == ['a dog', 'a cat'] and out[2] == ['a bus']


def test_word_tokenize_numbers_and_unicode():
    assert word_tokenize("2 dogs and 3.5 cats") == \
        ['2', 'dogs', 'and', '3.5', 'cats']
    # non-ASCII letters are dropped rather than crashing
    out = word_tokenize("café naïve dog")
    assert 'dog' in out


def test_word_tokenize_empty():
    assert word_tokenize("") == []
    assert word_tokenize("   ") == []
