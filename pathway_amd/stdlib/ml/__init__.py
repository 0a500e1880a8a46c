"""ML utilities (reference stdlib/ml): classifiers, index, smart_table_ops."""
from pathway_amd.stdlib.ml import index

__all__ = ["index"]
