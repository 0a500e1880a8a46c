"""ML utilities (reference stdlib/ml)."""
from pathway_amd.stdlib.ml import classifiers, index, smart_table_ops
from pathway_amd.stdlib.ml import datasets, hmm, utils
from pathway_amd.stdlib.ml.index import KNNIndex

__all__ = ["classifiers", "index", "smart_table_ops", "KNNIndex"]
