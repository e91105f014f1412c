"""Text chunking (parity with the reference knowledge splitter,
controller/knowledge splitter.go): paragraph-aware sliding window with
overlap, character-budgeted."""
from __future__ import annotations

from typing import List


def chunk_text(text: str, chunk_size: int = 512, overlap: int = 64,
               metadata: dict | None = None) -> List[dict]:
    metadata = metadata or {}
    if not text.strip():
        return []
    # split on paragraph boundaries first, then pack into windows
    paras = [p.strip() for p in text.split("\n\n") if p.strip()]
    pieces: List[str] = []
    for p in paras:
        if len(p) <= chunk_size:
            pieces.append(p)
        else:
            step = max(1, chunk_size - overlap)
            for i in range(0, len(p), step):
                pieces.append(p[i:i + chunk_size])
                if i + chunk_size >= len(p):
                    break
    chunks: List[dict] = []
    cur = ""
    for piece in pieces:
        if cur and len(cur) + len(piece) + 2 > chunk_size:
            chunks.append(cur)
            tail = cur[-overlap:] if overlap else ""
            cur = (tail + "\n" + piece) if tail else piece
        else:
            cur = (cur + "\n\n" + piece) if cur else piece
    if cur:
        chunks.append(cur)
    return [{"text": c, "metadata": dict(metadata, chunk=i)}
            for i, c in enumerate(chunks)]
