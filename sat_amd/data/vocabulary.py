"""Frequency-ranked vocabulary with CSV persistence.

Behavioral parity with reference `utils/vocabulary.py`:
  * index 0 is reserved for '<start>' (vocabulary.py:27-28);
  * remaining slots are the top (size-1) corpus words by frequency
    (vocabulary.py:31-39), size shrinking to the corpus if smaller (:25-26);
  * `word_frequencies` is the log of the normalized frequency, shifted so
    max == 0 (vocabulary.py:41-44);
  * `get_sentence` truncates at the first '.' (appending one if absent) and
    re-attaches punctuation/apostrophes without a leading space
    (vocabulary.py:53-63);
  * persisted as a CSV with word,index,frequency columns (vocabulary.py:65-70).

Differences: tokenization is the in-repo PTB-style tokenizer (no NLTK), and
`process_sentence` skips out-of-vocabulary words instead of raising KeyError —
the reference only ever calls it on pre-filtered corpora where OOV words have
been removed (dataset.py:92), so on those inputs behavior is identical.
"""

import os
import string

import numpy as np
import pandas as pd

from .tokenizer import word_tokenize


class Vocabulary(object):
    def __init__(self, size, save_file=None):
        self.words = []
        self.word2idx = {}
        self.word_frequencies = []
        self.size = size
        if save_file is not None:
            self.load(save_file)

    def build(self, sentences):
        """Build the vocabulary and compute the frequency of each word."""
        counts = {}
        for sentence in sentences:
            for w in word_tokenize(sentence.lower()):
                counts[w] = counts.get(w, 0) + 1.0

        if self.size - 1 > len(counts):
            self.size = len(counts) + 1

        self.words = ['<start>']
        self.word2idx = {'<start>': 0}
        freqs = [1.0]

        ranked = sorted(counts.items(), key=lambda kv: kv[1], reverse=True)
        for word, freq in ranked[: self.size - 1]:
            self.word2idx[word] = len(self.words)
            self.words.append(word)
            freqs.append(freq)

        f = np.array(freqs, dtype=np.float64)
        f /= f.sum()
        f = np.log(f)
        f -= f.max()
        self.word_frequencies = f
        self.size = len(self.words)

    def process_sentence(self, sentence):
        """Tokenize a sentence and map each in-vocabulary token to its index."""
        return [self.word2idx[w]
                for w in word_tokenize(sentence.lower())
                if w in self.word2idx]

    def get_sentence(self, idxs):
        """Translate a vector of indices back into a sentence string.

        Semantics (parity contract with reference vocabulary.py:53-63,
        verified by the reference-CSV round-trip test): the caption stops
        at the first '.' token inclusive — one is supplied if the indices
        never produce it — and tokens are joined with a single space,
        except that punctuation and apostrophe-led clitics ("'s", "n't")
        attach directly to the preceding token.
        """
        pieces = []
        terminated = False
        for i in idxs:
            w = self.words[int(i)]
            attaches = w.startswith("'") or w in string.punctuation
            if pieces and not attaches:
                pieces.append(' ')
            pieces.append(w)
            if w == '.':
                terminated = True
                break
        if not terminated:
            pieces.append('.')
        return ''.join(pieces)

    def save(self, save_file):
        """Save the vocabulary as a CSV (word, index, frequency)."""
        pd.DataFrame({
            'word': list(self.words),
            'index': list(range(len(self.words))),
            'frequency': list(self.word_frequencies),
        }).to_csv(save_file)

    def load(self, save_file):
        """Load the vocabulary from a CSV produced by `save`."""
        assert os.path.exists(save_file), save_file
        data = pd.read_csv(save_file)
        self.words = [str(w) for w in data['word'].values]
        n = min(self.size, len(self.words))
        self.words = self.words[:n]
        self.word2idx = {self.words[i]: i for i in range(n)}
        self.word_frequencies = np.asarray(data['frequency'].values[:n])
        self.size = n
