"""Code intelligence (parity with the reference's in-process kodit
library, SURVEY.md §2.1 Kodit integration): index git repositories into
the vector store using the native bge embeddings (the same
route-through-/v1/embeddings seam, kodit_init.go:203), query by
semantic+keyword merge, expose as an agent-visible search."""
from __future__ import annotations

import logging
import os
from typing import List, Optional

from helix_amd.rag.chunker import chunk_code

log = logging.getLogger("helix_amd.code_intel")

CODE_EXT = {".py", ".go", ".rs", ".c", ".cc", ".cpp", ".h", ".hpp", ".hip",
            ".cu", ".js", ".ts", ".tsx", ".java", ".rb", ".sh", ".md",
            ".yaml", ".yml", ".toml", ".json"}


class CodeIntelService:
    def __init__(self, rag, git):
        self.rag = rag
        self.git = git

    async def index_repo(self, repo_id: str, ref: str = "HEAD") -> int:
        """RegisterDirectory-equivalent (reference rag_kodit.go:62)."""
        kid = f"code:{repo_id}"
        self.rag.delete(kid)
        docs = []
        for path in self.git.ls_tree(repo_id, ref):
            if os.path.splitext(path)[1].lower() not in CODE_EXT:
                continue
            try:
                content = self.git.read_file(repo_id, path, ref)
            except Exception:
                continue
            docs.extend(chunk_code(content, path))
        if not docs:
            return 0
        # already chunked: index verbatim through the vector store
        B = 64
        total = 0
        for i in range(0, len(docs), B):
            batch = docs[i:i + B]
            vecs = await self.rag._embed([d["text"] for d in batch])
            self.rag.vs.add(kid, batch, vecs)
            total += len(batch)
        return total

    async def query(self, repo_id: str, query: str, k: int = 6) -> List[dict]:
        return await self.rag.query(f"code:{repo_id}", query, k)
