from pathway_amd.stdlib.utils import col, filtering
from pathway_amd.stdlib.utils.async_transformer import AsyncTransformer
from pathway_amd.stdlib.utils.pandas_transformer import pandas_transformer

__all__ = ["col", "filtering", "AsyncTransformer", "pandas_transformer"]
