"""Framework integrations (reference parity: infomesh/integrations/ —
LangChain/LlamaIndex/Haystack adapters). The heavy frameworks are
optional: adapters duck-type their interfaces and work standalone."""
from .adapters import (InfoMeshRetriever, InfoMeshReader,  # noqa: F401
                       InfoMeshDocumentStore)
