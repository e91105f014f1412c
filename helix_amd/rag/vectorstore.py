"""Vector store — replaces the reference's VectorChord/pgvector backend
(SURVEY.md §2.1 Kodit integration) with an in-process index: vectors
persisted in the document store, brute-force cosine top-k via numpy
(exact, fine up to ~1M chunks; the GPU bge encoder is the expensive part).
"""
from __future__ import annotations

import json
from typing import Dict, List, Optional

import numpy as np


class VectorStore:
    def __init__(self, store):
        self.store = store                      # helix_amd.store.Store
        self._cache: Dict[str, tuple] = {}      # knowledge_id -> (ids, mat, docs)

    def _load(self, knowledge_id: str):
        if knowledge_id in self._cache:
            return self._cache[knowledge_id]
        rows = self.store.list("rag_chunks", parent=knowledge_id,
                               limit=1000000, desc=False)
        ids = [r["id"] for r in rows]
        docs = rows
        mat = (np.array([r["vector"] for r in rows], dtype=np.float32)
               if rows else np.zeros((0, 1), dtype=np.float32))
        if mat.size:
            norms = np.linalg.norm(mat, axis=1, keepdims=True)
            mat = mat / np.maximum(norms, 1e-9)
        self._cache[knowledge_id] = (ids, mat, docs)
        return self._cache[knowledge_id]

    def add(self, knowledge_id: str, chunks: List[dict],
            vectors: List[List[float]]):
        for i, (c, v) in enumerate(zip(chunks, vectors)):
            cid = f"{knowledge_id}-{self.store.count('rag_chunks')}-{i}"
            self.store.put("rag_chunks", cid,
                           {"id": cid, "text": c["text"],
                            "metadata": c.get("metadata", {}), "vector": v},
                           parent=knowledge_id)
        self._cache.pop(knowledge_id, None)

    def query(self, knowledge_id: str, vector: List[float],
              k: int = 4, threshold: float = 0.0) -> List[dict]:
        ids, mat, docs = self._load(knowledge_id)
        if not ids:
            return []
        q = np.array(vector, dtype=np.float32)
        q = q / max(np.linalg.norm(q), 1e-9)
        sims = mat @ q
        order = np.argsort(-sims)[:k]
        out = []
        for i in order:
            if sims[i] < threshold:
                continue
            d = docs[int(i)]
            out.append({"id": d["id"], "text": d["text"],
                        "metadata": d.get("metadata", {}),
                        "score": float(sims[int(i)])})
        return out

    def delete(self, knowledge_id: str):
        for r in self.store.list("rag_chunks", parent=knowledge_id,
                                 limit=1000000):
            self.store.delete("rag_chunks", r["id"])
        self._cache.pop(knowledge_id, None)

    def count(self, knowledge_id: str) -> int:
        ids, _, _ = self._load(knowledge_id)
        return len(ids)
