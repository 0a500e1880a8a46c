"""Internal API aggregation (mirrors reference pathway.internals exports)."""

from __future__ import annotations

__version__ = "0.1.0"

from pathway_amd.internals import dtype
from pathway_amd.internals.api import (
    ERROR,
    PENDING,
    BasePointer,
    Pointer,
    PyObjectWrapper,
    wrap_py_object,
)
from pathway_amd.internals.common import (
    UDF,
    apply,
    apply_async,
    apply_full_async,
    apply_with_type,
    cast,
    coalesce,
    declare_type,
    fill_error,
    if_else,
    make_tuple,
    require,
    table_transformer,
    udf,
    unwrap,
)
from pathway_amd.internals.config import (
    set_license_key,
    set_monitoring_config,
)
from pathway_amd.internals.datetime_types import DateTimeNaive, DateTimeUtc, Duration
from pathway_amd.internals.expression import (
    ColumnExpression,
    ColumnReference,
    ReducerExpression,
)
from pathway_amd.internals.groupbys import GroupedTable
from pathway_amd.internals.joins import Joinable, JoinMode, JoinResult
from pathway_amd.internals.json import Json
from pathway_amd.internals.rungraph import G, run, run_all
from pathway_amd.internals.schema import (
    ColumnDefinition,
    Schema,
    SchemaProperties,
    assert_table_has_schema,
    column_definition,
    schema_builder,
    schema_from_csv,
    schema_from_dict,
    schema_from_types,
)
from pathway_amd.internals.table import Table, TableLike
from pathway_amd.internals.thisclass import left, right, this
from pathway_amd.internals.universe import Universe


class MonitoringLevel:
    NONE = "none"
    IN_OUT = "in_out"
    ALL = "all"
    AUTO = "auto"
    AUTO_ALL = "auto_all"


class TableSlice:
    def __init__(self, mapping, table):
        self._mapping = mapping
        self._table = table

    def __getitem__(self, name):
        return self._mapping[name]

    def keys(self):
        return list(self._mapping.keys())


GroupedJoinResult = JoinResult
LiveTable = Table


def iterate(func, iteration_limit: int | None = None, **kwargs):
    """Fixpoint iteration (reference pw.iterate, dataflow.rs:5060).

    Round-1 implementation: batch-mode fixpoint — static inputs only.
    """
    from pathway_amd.internals.iterate import run_iterate

    return run_iterate(func, iteration_limit, **kwargs)


def iterate_universe(func, **kwargs):
    return iterate(func, **kwargs)


def join(left_table, right_table, *on, **kwargs):
    return left_table.join(right_table, *on, **kwargs)


def join_inner(left_table, right_table, *on, **kwargs):
    return left_table.join_inner(right_table, *on, **kwargs)


def join_left(left_table, right_table, *on, **kwargs):
    return left_table.join_left(right_table, *on, **kwargs)


def join_right(left_table, right_table, *on, **kwargs):
    return left_table.join_right(right_table, *on, **kwargs)


def join_outer(left_table, right_table, *on, **kwargs):
    return left_table.join_outer(right_table, *on, **kwargs)


def groupby(table, *args, **kwargs):
    return table.groupby(*args, **kwargs)


def global_error_log():
    from pathway_amd.internals.errors import global_error_log as _gel

    return _gel()


def local_error_log():
    return global_error_log()


def load_yaml(stream):
    from pathway_amd.internals.yaml_loader import load_yaml as _ly

    return _ly(stream)


def enable_interactive_mode():
    pass


def sql(query: str, **tables):
    from pathway_amd.internals.sql import sql as _sql

    return _sql(query, **tables)
