"""Local HTTP admin API (reference parity: infomesh/api/)."""
