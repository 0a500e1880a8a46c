"""pw.io — connectors (reference python/pathway/io, 46 modules).

Working offline: fs, csv, jsonlines, plaintext, python (ConnectorSubject),
subscribe, sqlite, http (+rest_connector), null, debezium-envelope parsing.
Service-backed connectors (kafka/nats/postgres/...) expose the reference
API and activate when their client library is installed.
"""
from pathway_amd.io import (
    airbyte,
    azure,
    bigquery,
    clickhouse,
    csv,
    debezium,
    deltalake,
    dynamodb,
    elasticsearch,
    fs,
    gdrive,
    http,
    iceberg,
    jsonlines,
    kafka,
    kinesis,
    logstash,
    minio,
    mongodb,
    mqtt,
    mssql,
    mysql,
    nats,
    null,
    plaintext,
    postgres,
    pubsub,
    pulsar,
    python,
    questdb,
    rabbitmq,
    redpanda,
    s3,
    sqlite,
    chroma,
    duckdb,
    leann,
    milvus,
    pinecone,
    pyfilesystem,
    qdrant,
    slack,
    weaviate,
)
from pathway_amd.io._subscribe import (
    OnChangeCallback,
    OnChangeCallbackAsync,
    OnFinishCallback,
    subscribe,
)
from pathway_amd.io._utils import (
    ENGINE_TIME,
    CsvParserSettings,
    DurationLike,
    TLSSettings,
)
from pathway_amd.io.synchronization import (
    SynchronizedColumn,
    register_input_synchronization_group,
)

__all__ = [
    "airbyte", "azure", "bigquery", "clickhouse", "csv", "debezium",
    "deltalake", "dynamodb", "elasticsearch", "fs", "gdrive", "http",
    "iceberg", "jsonlines", "kafka", "kinesis", "logstash", "minio",
    "mongodb", "mqtt", "mssql", "mysql", "nats", "null", "plaintext",
    "postgres", "pubsub", "pulsar", "python", "questdb", "rabbitmq",
    "redpanda", "s3", "sqlite", "chroma", "duckdb", "leann", "milvus", "pinecone", "pyfilesystem", "qdrant", "slack", "weaviate", "subscribe", "register_input_synchronization_group",
    "SynchronizedColumn", "ENGINE_TIME", "CsvParserSettings",
    "DurationLike", "TLSSettings", "OnChangeCallback",
    "OnChangeCallbackAsync", "OnFinishCallback",
]
