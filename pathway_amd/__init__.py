"""pathway_amd — an MI355X-native incremental streaming dataflow framework.

A from-scratch implementation of the pathwaycom/pathway feature set
 with a GPU-columnar engine: delta batches on PyTorch-ROCm
tensors, hand-written HIP/CDNA4 kernels for the hot operators, RCCL over
xGMI for the worker exchange.  Import as ``import pathway_amd as pw``.
"""

from __future__ import annotations

import pathway_amd.reducers as reducers
import pathway_amd.universes as universes
from pathway_amd import debug, udfs
from pathway_amd.internals import (
    ERROR,
    PENDING,
    UDF,
    ColumnExpression,
    ColumnReference,
    DateTimeNaive,
    DateTimeUtc,
    Duration,
    G,
    GroupedJoinResult,
    GroupedTable,
    Joinable,
    JoinMode,
    JoinResult,
    Json,
    LiveTable,
    MonitoringLevel,
    Pointer,
    PyObjectWrapper,
    Schema,
    SchemaProperties,
    Table,
    TableLike,
    TableSlice,
    Universe,
    __version__,
    apply,
    apply_async,
    apply_full_async,
    apply_with_type,
    assert_table_has_schema,
    cast,
    coalesce,
    column_definition,
    declare_type,
    enable_interactive_mode,
    fill_error,
    global_error_log,
    groupby,
    if_else,
    iterate,
    iterate_universe,
    join,
    join_inner,
    join_left,
    join_outer,
    join_right,
    left,
    load_yaml,
    local_error_log,
    make_tuple,
    require,
    right,
    run,
    run_all,
    schema_builder,
    schema_from_csv,
    schema_from_dict,
    schema_from_types,
    set_license_key,
    set_monitoring_config,
    sql,
    table_transformer,
    this,
    udf,
    unwrap,
    wrap_py_object,
)
from pathway_amd.internals import dtype as _dt
from pathway_amd.internals.custom_reducers import BaseCustomAccumulator


class Type:
    """PathwayType surface (reference api.PathwayType)."""

    ANY = _dt.ANY
    STRING = _dt.STR
    INT = _dt.INT
    BOOL = _dt.BOOL
    FLOAT = _dt.FLOAT
    POINTER = _dt.POINTER
    DATE_TIME_NAIVE = _dt.DATE_TIME_NAIVE
    DATE_TIME_UTC = _dt.DATE_TIME_UTC
    DURATION = _dt.DURATION
    ARRAY = _dt.Array()
    JSON = _dt.JSON
    BYTES = _dt.BYTES
    PY_OBJECT_WRAPPER = _dt.PY_OBJECT_WRAPPER
    FUTURE = _dt.FUTURE


class PersistenceMode:
    REALTIME_REPLAY = "realtime_replay"
    SPEEDRUN_REPLAY = "speedrun_replay"
    BATCH = "batch"
    PERSISTING = "persisting"
    SELECTIVE_PERSISTING = "selective_persisting"
    UDF_CACHING = "udf_caching"
    OPERATOR_PERSISTING = "operator_persisting"


import pathway_amd.io as io  # noqa: E402
import pathway_amd.persistence as persistence  # noqa: E402
from pathway_amd import demo  # noqa: E402
from pathway_amd.stdlib import (  # noqa: E402
    graphs,
    indexing,
    ml,
    ordered,
    stateful,
    statistical,
    temporal,
    utils,
    viz,
)
from pathway_amd.internals.exported import (  # noqa: E402
    ExportedTable,
    export_table,
    import_table,
)
from pathway_amd.stdlib.utils.async_transformer import AsyncTransformer  # noqa: E402
from pathway_amd.stdlib.utils.pandas_transformer import pandas_transformer  # noqa: E402

from pathway_amd.internals.row_transformer import (  # noqa: E402
    ClassArg,
    input_attribute,
    input_method,
    method,
    output_attribute,
    transformer,
)

# reference-surface aliases
from pathway_amd.stdlib.temporal._asof_join import AsofJoinResult  # noqa: E402
from pathway_amd.stdlib.temporal._interval_join import IntervalJoinResult  # noqa: E402
from pathway_amd.stdlib.temporal._window_join import WindowJoinResult  # noqa: E402

OuterJoinResult = JoinResult
asynchronous = udfs  # reference-deprecated alias for pw.udfs
window = temporal  # reference alias module for window constructors

# temporal joins attached like the reference does
Table.windowby = temporal.windowby
Table.asof_join = temporal.asof_join
Table.inactivity_detection = temporal.inactivity_detection
Table.add_update_timestamp_utc = temporal.add_update_timestamp_utc
Table.asof_join_left = temporal.asof_join_left
Table.asof_join_right = temporal.asof_join_right
Table.asof_join_outer = temporal.asof_join_outer
Table.asof_now_join = temporal.asof_now_join
Table.asof_now_join_inner = temporal.asof_now_join_inner
Table.asof_now_join_left = temporal.asof_now_join_left
Table.window_join = temporal.window_join
Table.window_join_inner = temporal.window_join_inner
Table.window_join_left = temporal.window_join_left
Table.window_join_right = temporal.window_join_right
Table.window_join_outer = temporal.window_join_outer
Table.interval_join = temporal.interval_join
Table.interval_join_inner = temporal.interval_join_inner
Table.interval_join_left = temporal.interval_join_left
Table.interval_join_right = temporal.interval_join_right
Table.interval_join_outer = temporal.interval_join_outer

__all__ = [
    "udfs",
    "graphs",
    "utils",
    "debug",
    "demo",
    "indexing",
    "ml",
    "io",
    "persistence",
    "apply",
    "udf",
    "UDF",
    "apply_async",
    "apply_with_type",
    "apply_full_async",
    "declare_type",
    "cast",
    "GroupedTable",
    "iterate",
    "iterate_universe",
    "ExportedTable",
    "export_table",
    "import_table",
    "JoinResult",
    "JoinMode",
    "GroupedJoinResult",
    "reducers",
    "schema_from_types",
    "schema_from_dict",
    "schema_from_csv",
    "schema_builder",
    "Table",
    "TableLike",
    "TableSlice",
    "ColumnReference",
    "ColumnExpression",
    "Schema",
    "SchemaProperties",
    "Pointer",
    "PyObjectWrapper",
    "wrap_py_object",
    "MonitoringLevel",
    "Universe",
    "this",
    "left",
    "right",
    "Joinable",
    "coalesce",
    "require",
    "sql",
    "run",
    "run_all",
    "if_else",
    "make_tuple",
    "Type",
    "PersistenceMode",
    "__version__",
    "universes",
    "temporal",
    "statistical",
    "stateful",
    "ordered",
    "viz",
    "column_definition",
    "unwrap",
    "fill_error",
    "assert_table_has_schema",
    "DateTimeNaive",
    "DateTimeUtc",
    "Duration",
    "Json",
    "table_transformer",
    "BaseCustomAccumulator",
    "AsyncTransformer",
    "pandas_transformer",
    "join",
    "join_inner",
    "join_left",
    "join_right",
    "join_outer",
    "groupby",
    "enable_interactive_mode",
    "LiveTable",
    "set_license_key",
    "set_monitoring_config",
    "global_error_log",
    "local_error_log",
    "load_yaml",
    "ERROR",
    "PENDING",
    "AsofJoinResult",
    "IntervalJoinResult",
    "WindowJoinResult",
    "OuterJoinResult",
    "asynchronous",
    "window",
    "transformer",
    "ClassArg",
    "input_attribute",
    "input_method",
    "output_attribute",
    "method",
]
