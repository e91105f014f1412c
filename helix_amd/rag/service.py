"""RAGService — index/query/delete over the vector store with embeddings
from the provider layer (parity with the reference RAG interface,
api/pkg/rag/rag.go:11, and its semantic+keyword merged query,
rag_kodit.go:150-209).

Round 2: versioned index namespaces. Each (re)index writes chunks into a
fresh `{kid}@v{n}` namespace and atomically swaps an alias row when
complete, so queries keep serving the previous version during a long
reindex (the reference's knowledge versioning does the same swap,
api/pkg/controller/knowledge versioning).
"""
from __future__ import annotations

import logging
import re
from typing import List, Optional

from helix_amd.rag.chunker import chunk_any, chunk_text, detect_language
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

    # -- versioned namespaces ----------------------------------------------
    def _active_ns(self, knowledge_id: str) -> str:
        row = self.store.get("rag_alias", knowledge_id)
        return row["ns"] if row else knowledge_id

    def _chunks_of(self, documents: List[dict]) -> List[dict]:
        chunks: List[dict] = []
        for doc in documents:
            md = doc.get("metadata", {})
            path = md.get("path", "") or md.get("source", "")
            if path and detect_language(path):
                chunks.extend(chunk_any(doc.get("text", ""), path,
                                        metadata=md))
            else:
                chunks.extend(chunk_text(doc.get("text", ""),
                                         self.cfg.rag.chunk_size,
                                         self.cfg.rag.chunk_overlap, md))
        return chunks

    async def index(self, knowledge_id: str, documents: List[dict],
                    progress=None) -> int:
        """documents: [{text, metadata?}] -> chunk, embed, store, swap."""
        return await self.index_chunks(knowledge_id,
                                       self._chunks_of(documents),
                                       progress=progress)

    async def index_chunks(self, knowledge_id: str,
                           chunks: List[dict],
                           progress=None) -> int:
        alias = self.store.get("rag_alias", knowledge_id) or \
            {"id": knowledge_id, "ns": knowledge_id, "version": 0}
        old_ns = alias["ns"]
        ver = int(alias.get("version", 0)) + 1
        ns = f"{knowledge_id}@v{ver}"
        B = 64
        for i in range(0, len(chunks), B):
            batch = chunks[i:i + B]
            vecs = await self._embed([c["text"] for c in batch])
            self.vs.add(ns, batch, vecs)
            if progress is not None:
                progress(min(i + B, len(chunks)), len(chunks))
        # atomic swap: queries resolve through the alias row
        self.store.put("rag_alias", knowledge_id,
                       {"id": knowledge_id, "ns": ns, "version": ver})
        if old_ns != ns:
            self.vs.delete(old_ns)
        return len(chunks)

    async def query(self, knowledge_id: str, text: str,
                    k: Optional[int] = None) -> List[dict]:
        k = k or self.cfg.rag.results_count
        vec = (await self._embed([text]))[0]
        sem = self.vs.query(self._active_ns(knowledge_id), vec, k * 2,
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
        self.vs.delete(self._active_ns(knowledge_id))
        self.vs.delete(knowledge_id)
        self.store.delete("rag_alias", knowledge_id)

    def chunk_count(self, knowledge_id: str) -> int:
        return self.vs.count(self._active_ns(knowledge_id))
