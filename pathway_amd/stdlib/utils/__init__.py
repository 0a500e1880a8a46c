from pathway_amd.stdlib.utils import bucketing, col, filtering
from pathway_amd.stdlib.utils.async_transformer import AsyncTransformer
from pathway_amd.stdlib.utils.pandas_transformer import pandas_transformer

__all__ = ["bucketing", "col", "filtering", "AsyncTransformer", "pandas_transformer"]
