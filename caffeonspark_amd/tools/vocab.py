"""Vocabulary build/save/load + sentence embedding (reference:
tools/Vocab.scala + Conversions.scala sentence->word-id path for the COCO
caption pipeline)."""

from __future__ import annotations

import json
import re
from collections import Counter
from typing import Dict, Iterable, List

UNK = "<unk>"
EOS = "<eos>"  # id 0, marks end of sentence (caffe LRCN convention)

_WORD_RE = re.compile(r"[\w']+")


def tokenize(sentence: str) -> List[str]:
    return _WORD_RE.findall(sentence.lower())


class Vocab:
    def __init__(self, words: List[str]):
        self.words = words  # index = id
        self.index: Dict[str, int] = {w: i for i, w in enumerate(words)}

    @classmethod
    def build(cls, sentences: Iterable[str], size: int) -> "Vocab":
        counts = Counter()
        for s in sentences:
            counts.update(tokenize(s))
        words = [EOS, UNK] + [w for w, _ in counts.most_common(size - 2)]
        return cls(words)

    def embed(self, sentence: str, length: int) -> List[int]:
        """sentence -> fixed-length id list terminated by EOS (0),
        padded with -1 (the loss ignore_label)."""
        ids = [self.index.get(w, self.index[UNK])
               for w in tokenize(sentence)][:length - 1]
        ids.append(self.index[EOS])
        ids += [-1] * (length - len(ids))
        return ids

    def save(self, path: str) -> None:
        with open(path, "w") as fh:
            json.dump(self.words, fh)

    @classmethod
    def load(cls, path: str) -> "Vocab":
        with open(path) as fh:
            return cls(json.load(fh))

    def __len__(self):
        return len(self.words)
