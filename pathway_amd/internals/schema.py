"""Schema system (reference internals/schema.py, 1042 LoC — behavior parity).

Schemas are classes: ``class S(pw.Schema): x: int``; columns carry dtype,
primary-key flag, and default values via ``column_definition``.
"""

from __future__ import annotations

import dataclasses
from typing import Any, Iterable, Mapping

from pathway_amd.internals import dtype as dt

_NO_DEFAULT = object()


@dataclasses.dataclass(frozen=True)
class ColumnDefinition:
    primary_key: bool = False
    default_value: Any = _NO_DEFAULT
    dtype: Any = None
    name: str | None = None
    append_only: bool | None = None

    @property
    def has_default_value(self) -> bool:
        return self.default_value is not _NO_DEFAULT


def column_definition(
    *,
    primary_key: bool = False,
    default_value: Any = _NO_DEFAULT,
    dtype: Any = None,
    name: str | None = None,
    append_only: bool | None = None,
) -> Any:
    return ColumnDefinition(primary_key, default_value, dtype, name, append_only)


@dataclasses.dataclass(frozen=True)
class ColumnSchema:
    name: str
    dtype: dt.DType
    primary_key: bool = False
    default_value: Any = _NO_DEFAULT
    append_only: bool = False

    @property
    def has_default_value(self) -> bool:
        return self.default_value is not _NO_DEFAULT


@dataclasses.dataclass(frozen=True)
class SchemaProperties:
    append_only: bool = False


class SchemaMetaclass(type):
    __columns__: dict[str, ColumnSchema]
    __properties__: SchemaProperties

    def __init__(cls, name, bases, namespace, append_only: bool | None = None):
        super().__init__(name, bases, namespace)
        columns: dict[str, ColumnSchema] = {}
        for base in reversed(bases):
            if hasattr(base, "__columns__"):
                columns.update(base.__columns__)
        annotations = namespace.get("__annotations__", {})
        for field, hint in annotations.items():
            if field.startswith("__"):
                continue
            definition = namespace.get(field, None)
            if isinstance(definition, ColumnDefinition):
                col_dtype = dt.wrap(definition.dtype) if definition.dtype is not None else dt.wrap(hint)
                columns[definition.name or field] = ColumnSchema(
                    name=definition.name or field,
                    dtype=col_dtype,
                    primary_key=definition.primary_key,
                    default_value=definition.default_value,
                    append_only=bool(definition.append_only)
                    if definition.append_only is not None
                    else bool(append_only),
                )
            else:
                columns[field] = ColumnSchema(
                    name=field, dtype=dt.wrap(hint), append_only=bool(append_only)
                )
        cls.__columns__ = columns
        cls.__properties__ = SchemaProperties(append_only=bool(append_only))

    def columns(cls) -> Mapping[str, ColumnSchema]:
        return dict(cls.__columns__)

    def column_names(cls) -> list[str]:
        return list(cls.__columns__.keys())

    def keys(cls) -> list[str]:
        return cls.column_names()

    def typehints(cls) -> dict[str, Any]:
        return {n: c.dtype.typehint for n, c in cls.__columns__.items()}

    def dtypes(cls) -> dict[str, dt.DType]:
        return {n: c.dtype for n, c in cls.__columns__.items()}

    def primary_key_columns(cls) -> list[str] | None:
        pkeys = [n for n, c in cls.__columns__.items() if c.primary_key]
        return pkeys or None

    def default_values(cls) -> dict[str, Any]:
        return {
            n: c.default_value for n, c in cls.__columns__.items() if c.has_default_value
        }

    def __or__(cls, other: "SchemaMetaclass") -> "SchemaMetaclass":
        columns = dict(cls.__columns__)
        for n, c in other.__columns__.items():
            if n in columns and columns[n].dtype != c.dtype:
                raise ValueError(f"schema union conflict on column {n}")
            columns[n] = c
        return schema_from_columns(columns)

    def __getitem__(cls, name: str) -> ColumnSchema:
        return cls.__columns__[name]

    def __eq__(cls, other: object) -> bool:
        if not isinstance(other, SchemaMetaclass):
            return NotImplemented
        return cls.__columns__ == other.__columns__

    def __hash__(cls) -> int:
        return hash(tuple(cls.__columns__.items()))

    def __repr__(cls) -> str:
        cols = ", ".join(f"{n}: {c.dtype!r}" for n, c in cls.__columns__.items())
        return f"<pw.Schema {{{cols}}}>"

    def with_types(cls, **kwargs: Any) -> "SchemaMetaclass":
        columns = dict(cls.__columns__)
        for n, hint in kwargs.items():
            if n not in columns:
                raise ValueError(f"column {n} not present in schema")
            columns[n] = dataclasses.replace(columns[n], dtype=dt.wrap(hint))
        return schema_from_columns(columns)

    def without(cls, *names: str) -> "SchemaMetaclass":
        columns = {n: c for n, c in cls.__columns__.items() if n not in names}
        return schema_from_columns(columns)

    def update_properties(cls, **kwargs: Any) -> "SchemaMetaclass":
        return schema_from_columns(dict(cls.__columns__), SchemaProperties(**kwargs))


class Schema(metaclass=SchemaMetaclass):
    """Base class for user-declared schemas."""


def schema_from_columns(
    columns: Mapping[str, ColumnSchema], properties: SchemaProperties | None = None
) -> SchemaMetaclass:
    cls = SchemaMetaclass("Schema", (Schema,), {"__annotations__": {}})
    cls.__columns__ = dict(columns)
    cls.__properties__ = properties or SchemaProperties()
    return cls


def schema_from_types(_name: str | None = None, **kwargs: Any) -> SchemaMetaclass:
    columns = {n: ColumnSchema(name=n, dtype=dt.wrap(t)) for n, t in kwargs.items()}
    return schema_from_columns(columns)


def schema_from_dict(
    types: Mapping[str, Any],
    id_from: Iterable[str] | None = None,
    default_values: Mapping[str, Any] | None = None,
    name: str | None = None,
) -> SchemaMetaclass:
    id_from = set(id_from or ())
    default_values = default_values or {}
    columns = {}
    for n, t in types.items():
        columns[n] = ColumnSchema(
            name=n,
            dtype=dt.wrap(t),
            primary_key=n in id_from,
            default_value=default_values.get(n, _NO_DEFAULT),
        )
    return schema_from_columns(columns)


def schema_from_csv(
    path: str,
    *,
    name: str | None = None,
    properties: SchemaProperties | None = None,
    delimiter: str = ",",
    comment_character: str | None = None,
    escape: str | None = None,
    quote: str = '"',
    enable_double_quote_escapes: bool = True,
    num_parsed_rows: int | None = None,
) -> SchemaMetaclass:
    """Infer a schema from a CSV file header + sampled rows."""
    import csv as _csv

    with open(path, newline="") as f:
        reader = _csv.reader(f, delimiter=delimiter, quotechar=quote)
        rows = []
        header = None
        for i, row in enumerate(reader):
            if comment_character and row and row[0].startswith(comment_character):
                continue
            if header is None:
                header = row
                continue
            rows.append(row)
            if num_parsed_rows is not None and len(rows) >= num_parsed_rows:
                break
    if header is None:
        raise ValueError(f"empty csv file {path}")

    def infer(values: list[str]) -> dt.DType:
        kinds = set()
        for v in values:
            try:
                int(v)
                kinds.add("int")
                continue
            except ValueError:
                pass
            try:
                float(v)
                kinds.add("float")
                continue
            except ValueError:
                pass
            if v in ("true", "false", "True", "False"):
                kinds.add("bool")
            else:
                kinds.add("str")
        if kinds <= {"int"}:
            return dt.INT
        if kinds <= {"int", "float"}:
            return dt.FLOAT
        if kinds <= {"bool"}:
            return dt.BOOL
        return dt.STR

    columns = {}
    for j, col in enumerate(header):
        values = [r[j] for r in rows if j < len(r)]
        columns[col] = ColumnSchema(name=col, dtype=infer(values) if values else dt.STR)
    return schema_from_columns(columns, properties)


def schema_builder(
    columns: Mapping[str, ColumnDefinition],
    *,
    name: str | None = None,
    properties: SchemaProperties | None = None,
) -> SchemaMetaclass:
    out = {}
    for n, d in columns.items():
        out[n] = ColumnSchema(
            name=d.name or n,
            dtype=dt.wrap(d.dtype) if d.dtype is not None else dt.ANY,
            primary_key=d.primary_key,
            default_value=d.default_value,
        )
    return schema_from_columns(out, properties)


def assert_table_has_schema(
    table: Any,
    schema: SchemaMetaclass,
    *,
    allow_superset: bool = True,
    ignore_primary_keys: bool = True,
) -> None:
    table_dtypes = table.schema.dtypes()
    for n, c in schema.__columns__.items():
        if n not in table_dtypes:
            raise AssertionError(f"table is missing column {n}")
        if table_dtypes[n] != c.dtype and c.dtype != dt.ANY:
            raise AssertionError(
                f"column {n} has dtype {table_dtypes[n]!r}, expected {c.dtype!r}"
            )
    if not allow_superset:
        extra = set(table_dtypes) - set(schema.__columns__)
        if extra:
            raise AssertionError(f"table has extra columns: {sorted(extra)}")
