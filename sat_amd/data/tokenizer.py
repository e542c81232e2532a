"""Pure-Python tokenization.

Replaces two external dependencies of the reference:
  * NLTK `word_tokenize` used for vocabulary building / caption filtering
    (reference `utils/vocabulary.py:21`, `utils/coco/coco.py:328`);
  * the Stanford CoreNLP PTBTokenizer Java subprocess used at metric time
    (reference `utils/coco/pycocoevalcap/tokenizer/ptbtokenizer.py:27-66`).

Both are approximated by a single regex tokenizer with PTB-ish behavior:
lower-casing, splitting standard punctuation off words, keeping contraction
tails ("n't", "'s") attached the way PTB does.
"""

import re

# Contractions PTB splits: don't -> do n't, it's -> it 's, etc.
_CONTRACTION = re.compile(r"(?i)\b(\w+)(n't|'s|'m|'re|'ve|'ll|'d)\b")

_TOKEN = re.compile(
    r"n't|'s|'m|'re|'ve|'ll|'d"   # contraction tails (after splitting)
    r"|[a-zA-Z]+"                  # words
    r"|[0-9]+(?:\.[0-9]+)?"        # numbers
    r"|[.,!?;:\"'()\[\]{}<>@#$%^&*\-+=/\\|~`]"  # punctuation, one char each
)

# Punctuation the PTBTokenizer wrapper strips from metric-time tokens
# (reference ptbtokenizer.py:21-22).
PUNCTUATIONS = [
    "''", "'", "``", "`", "-LRB-", "-RRB-", "-LCB-", "-RCB-",
    ".", "?", "!", ",", ":", "-", "--", "...", ";",
]


def word_tokenize(sentence):
    """PTB-style tokenization of one sentence -> list of lower-case tokens."""
    s = _CONTRACTION.sub(r"\1 \2", sentence.lower())
    return _TOKEN.findall(s)


class PTBTokenizer(object):
    """Metric-time tokenizer with the reference wrapper's interface.

    `tokenize` maps {image_id: [{'caption': str}, ...]} to
    {image_id: [token-joined str, ...]} with PUNCTUATIONS removed, matching
    reference ptbtokenizer.py:27-66 (minus the Java subprocess).
    """

    def tokenize(self, captions_for_image):
        out = {}
        for k, caps in captions_for_image.items():
            out[k] = []
            for cap in caps:
                text = cap['caption'] if isinstance(cap, dict) else cap
                toks = [t for t in word_tokenize(text) if t not in PUNCTUATIONS]
                out[k].append(' '.join(toks))
        return out
