"""pw.io — connectors (reference python/pathway/io, 46 modules).

Round-1 set: csv, jsonlines, fs, plaintext, python (ConnectorSubject),
subscribe, null.  Streaming connector runtime lands in the streaming phase;
message-queue and DB connectors are stubbed with the reference API surface.
"""
from pathway_amd.io import csv, fs, jsonlines, plaintext, python
from pathway_amd.io._subscribe import subscribe
from pathway_amd.io import null

__all__ = ["csv", "fs", "jsonlines", "plaintext", "python", "subscribe", "null"]
