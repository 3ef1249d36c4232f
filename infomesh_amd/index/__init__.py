"""Index layer: LocalStore (SQLite FTS5 ground truth), GPU hybrid shards,
ranking, link graph, snapshot format (reference parity: infomesh/index/)."""
