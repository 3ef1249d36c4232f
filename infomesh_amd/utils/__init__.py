"""Cross-cutting utilities: observability, resource governor, preflight,
plugins, diagnostics, benchmarks, SLO tracking (reference layer 10)."""
