"""ml helpers (reference stdlib/ml/utils.py)."""
from __future__ import annotations


def classifier_accuracy(predictions, labels):
    """Fraction of matching (prediction, label) pairs."""
    pairs = list(zip(predictions, labels))
    if not pairs:
        return 0.0
    return sum(1 for p, l in pairs if p == l) / len(pairs)
