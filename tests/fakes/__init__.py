"""In-process fake services for offline connector tests.

Each fake speaks the REAL wire protocol of the service it stands in for,
so the connector code under test is the production code path (client,
framing, offsets) — only the remote endpoint is simulated.
"""
