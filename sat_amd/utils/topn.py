"""Beam-search bookkeeping structures.

Behavioral parity with reference `utils/misc.py:38-88`:
  * `CaptionData` — one hypothesis: (sentence word-idx list, LSTM memory,
    LSTM output, cumulative score); ordered by score (misc.py:54-60);
  * `TopN` — heapq-backed bounded best-N container with push / extract(sort) /
    reset (misc.py:62-88).
"""

import heapq


class CaptionData(object):
    __slots__ = ('sentence', 'memory', 'output', 'score')

    def __init__(self, sentence, memory, output, score):
        self.sentence = sentence
        self.memory = memory
        self.output = output
        self.score = score

    def __lt__(self, other):
        return self.score < other.score

    def __eq__(self, other):
        return self.score == other.score


class TopN(object):
    def __init__(self, n):
        self._n = n
        self._data = []

    def size(self):
        assert self._data is not None
        return len(self._data)

    def push(self, x):
        assert self._data is not None
        if len(self._data) < self._n:
            heapq.heappush(self._data, x)
        else:
            heapq.heappushpop(self._data, x)

    def extract(self, sort=False):
        """Return contents, invalidating the container (reference misc.py:78)."""
        assert self._data is not None
        data = self._data
        self._data = None
        if sort:
            data.sort(reverse=True)
        return data

    def reset(self):
        self._data = []
