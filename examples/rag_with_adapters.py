#!/usr/bin/env python3
"""Framework-adapter example: the LangChain-style retriever feeding a
RAG answer, fully in-process (no network, no framework install).

    python examples/rag_with_adapters.py
"""
from __future__ import annotations

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from infomesh_amd.config import Config
from infomesh_amd.index.local_store import Document
from infomesh_amd.integrations.adapters import InfoMeshRetriever
from infomesh_amd.search.rag import format_rag_output
from infomesh_amd.services import AppContext


def main() -> None:
    ctx = AppContext.create(config=Config(), with_worker=False,
                            with_engine=False, in_memory=True)
    try:
        pages = [
            ("https://rocm.docs/lds", "Local Data Share",
             "The LDS is a 160 KB per-CU scratchpad; tiles staged "
             "through it feed the MFMA matrix cores."),
            ("https://rocm.docs/xgmi", "xGMI links",
             "xGMI provides point-to-point GPU links; RCCL runs ring "
             "collectives across the seven links of each MI355X."),
        ]
        for url, title, text in pages:
            ctx.index_document(Document(url=url, title=title,
                                        text=text * 4))

        retriever = InfoMeshRetriever(ctx=ctx, k=2)
        docs = retriever.get_relevant_documents("lds mfma tiles")
        print("retrieved:", [d.metadata["url"] for d in docs])

        rag = format_rag_output(
            "lds mfma tiles",
            [{"url": d.metadata["url"], "title": "", "snippet":
              d.page_content, "score": 1.0} for d in docs],
            answer_mode=True)
        print("chunks:", len(rag.chunks), "answer:",
              (rag.answer or "")[:60])
    finally:
        ctx.close()


if __name__ == "__main__":
    main()
