"""Shared fixtures (reference test-strategy parity: SURVEY.md §4 —
isolation via :memory:/tmp-path SQLite; gpu marker for MI355X-only tests)."""
from __future__ import annotations

import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X GPU (run with -m gpu on a GPU box)")


@pytest.fixture
def tmp_data_dir(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_DATA_DIR", str(tmp_path))
    return tmp_path


@pytest.fixture
def store():
    from infomesh_amd.index.local_store import LocalStore
    s = LocalStore(":memory:")
    yield s
    s.close()


@pytest.fixture
def seeded_store(store):
    """A store with a small deterministic corpus."""
    from infomesh_amd.index.local_store import Document
    docs = [
        ("https://docs.python.org/3/tutorial/", "Python Tutorial",
         "The Python tutorial covers functions, classes and modules. "
         "Learn how to write python code with examples.", "en"),
        ("https://docs.python.org/3/library/asyncio.html", "asyncio — Asynchronous I/O",
         "asyncio is a library to write concurrent code using the async await "
         "syntax. It provides event loops, tasks and coroutines.", "en"),
        ("https://pytorch.org/docs/", "PyTorch documentation",
         "PyTorch is a machine learning framework with GPU tensors and "
         "automatic differentiation. Train neural networks fast.", "en"),
        ("https://en.wikipedia.org/wiki/Okapi_BM25", "Okapi BM25",
         "BM25 is a ranking function used by search engines to estimate the "
         "relevance of documents to a given search query.", "en"),
        ("https://rocm.docs.amd.com/", "ROCm documentation",
         "ROCm is AMD's open software platform for GPU computing, including "
         "HIP kernels, rocBLAS and RCCL collective communication.", "en"),
        ("https://example.de/seite", "Deutsche Seite",
         "Dies ist eine deutsche Seite über Suchmaschinen und Indexierung.", "de"),
    ]
    for url, title, text, lang in docs:
        store.add_document(Document(url=url, title=title, text=text, language=lang))
    return store
