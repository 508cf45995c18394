"""Element-wise mean of a float-array column (reference: VectorMean.scala,
the UDAF behind distributed `test` aggregation)."""

from __future__ import annotations

from typing import Iterable, List


def vector_mean(column: Iterable) -> List[float]:
    total: List[float] = []
    n = 0
    for v in column:
        if v is None:
            continue
        if isinstance(v, (int, float)):
            v = [float(v)]
        v = list(v)
        if not total:
            total = [0.0] * len(v)
        for i, x in enumerate(v):
            total[i] += float(x)
        n += 1
    return [t / max(1, n) for t in total]
