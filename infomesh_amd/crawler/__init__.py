"""Crawler layer: fetch -> parse -> dedup -> index pipeline (CPU-side),
streaming into the GPU index via the services layer.
Reference parity: infomesh/crawler/ (SURVEY.md §2.3)."""
