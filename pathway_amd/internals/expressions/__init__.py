from pathway_amd.internals.expressions.date_time import DateTimeNamespace
from pathway_amd.internals.expressions.numerical import NumericalNamespace
from pathway_amd.internals.expressions.string import BinaryNamespace, StringNamespace

__all__ = ["DateTimeNamespace", "NumericalNamespace", "StringNamespace", "BinaryNamespace"]
