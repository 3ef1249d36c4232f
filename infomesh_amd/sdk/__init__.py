"""Python SDK (reference parity: infomesh/sdk/client.py)."""
from .client import InfoMeshClient, AsyncInfoMeshClient  # noqa: F401
