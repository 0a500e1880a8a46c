"""Dataset helpers (reference stdlib/ml/datasets — fetches public datasets;
offline image: generators for synthetic equivalents)."""
from __future__ import annotations


def synthetic_classification(n: int = 1000, seed: int = 0):
    """Synthetic 2-class points [(x, y, label)] for classifier examples."""
    import random

    rng = random.Random(seed)
    rows = []
    for _ in range(n):
        label = rng.randint(0, 1)
        cx = 2.0 * label
        rows.append((rng.gauss(cx, 1.0), rng.gauss(cx, 1.0), label))
    return rows
