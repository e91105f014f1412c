"""RAGService — index/query/delete over the vector store with embeddings
from the provider layer (parity with the reference RAG interface,
api/pkg/rag/rag.go:11, and its semantic+keyword merged query,
rag_kodit.go:150-209).
"""
from __future__ import annotations

import logging
import re
from typing import List, Optional

from helix_amd.rag.chunker import chunk_text
from helix_amd.rag.vectorstore import VectorStore

log = logging.getLogger("helix_amd.rag")


class RAGService:
    def __init__(self, cfg, store, providers):
        self.cfg = cfg
        self.store = store
        self.providers = providers
        self.vs = VectorStore(store)

    async def _embed(self, texts: List[str]) -> List[List[float]]:
        client = self.providers.get_client(self.cfg.rag.embeddings_provider)
        resp = await client.embeddings({
            "model": self.cfg.rag.embeddings_model, "input": texts})
        data = sorted(resp["data"], key=lambda d: d["index"])
        return [d["embedding"] for d in data]

    async def index(self, knowledge_id: str, documents: List[dict]):
        """documents: [{text, metadata?}] -> chunk, embed, store."""
        chunks: List[dict] = []
        for doc in documents:
            chunks.extend(chunk_text(doc.get("text", ""),
                                     self.cfg.rag.chunk_size,
                                     self.cfg.rag.chunk_overlap,
                                     doc.get("metadata", {})))
        if not chunks:
            return 0
        B = 64
        for i in range(0, len(chunks), B):
            batch = chunks[i:i + B]
            vecs = await self._embed([c["text"] for c in batch])
            self.vs.add(knowledge_id, batch, vecs)
        return len(chunks)

    async def query(self, knowledge_id: str, text: str,
                    k: Optional[int] = None) -> List[dict]:
        k = k or self.cfg.rag.results_count
        vec = (await self._embed([text]))[0]
        sem = self.vs.query(knowledge_id, vec, k * 2,
                            self.cfg.rag.distance_threshold)
        # keyword boost (merged ranking, reference rag_kodit.go:150-209)
        terms = set(re.findall(r"\w+", text.lower()))
        for r in sem:
            hits = sum(1 for t in set(re.findall(r"\w+", r["text"].lower()))
                       if t in terms)
            r["score"] += 0.01 * hits
        sem.sort(key=lambda r: -r["score"])
        return sem[:k]

    def delete(self, knowledge_id: str):
        self.vs.delete(knowledge_id)

    def chunk_count(self, knowledge_id: str) -> int:
        return self.vs.count(knowledge_id)
