"""Data formats for connectors (reference src/connectors/data_format/).

Pure-python binary codecs — the reference links avro/bson crates; this
build implements the wire formats directly so every connector is testable
offline:

  * avro      — Avro binary encoding + Object Container Files + the
                Confluent schema-registry wire framing (avro.rs)
  * bson      — BSON documents, used by the MongoDB wire client (bson.rs)
  * debezium  — Debezium CDC envelope -> insert/delete events (debezium.rs)
  * registry  — Confluent schema-registry REST client (avro.rs:SchemaRegistry)
"""

from pathway_amd.io.formats import avro, bson, debezium, registry  # noqa: F401
