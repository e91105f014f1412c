"""Chunking (parity with the reference knowledge splitter,
controller/knowledge splitter.go, and kodit's code-aware splitting,
rag_kodit.go:150): paragraph-aware sliding window with overlap for
prose; declaration-boundary splitting for code in the top languages."""
from __future__ import annotations

import re
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
            continue
        # long paragraph: pack whole words into windows, overlap on a
        # word boundary — a blind p[i:i+size] slice can split a word
        # longer than the overlap across two chunks, losing it from
        # retrieval (caught by the chunker coverage property test)
        cur = ""
        for w in p.split():
            while len(w) > chunk_size:          # single oversized token
                if cur:
                    pieces.append(cur)
                    cur = ""
                pieces.append(w[:chunk_size])
                w = w[max(1, chunk_size - overlap):]
            if cur and len(cur) + 1 + len(w) > chunk_size:
                pieces.append(cur)
                tail = cur[-overlap:] if overlap else ""
                sp = tail.find(" ")
                tail = tail[sp + 1:] if sp >= 0 else ""
                cur = (tail + " " + w) if tail else w
            else:
                cur = (cur + " " + w) if cur else w
        if cur:
            pieces.append(cur)
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


# ---------------------------------------------------------------------------
# Code-aware chunking: split on top-level declaration boundaries so a
# retrieved chunk is a whole function/class where possible (reference
# kodit indexes code units, rag_kodit.go:150; the round-1 fallback was a
# blind 60-line window).

_DECL_RES = {
    "python": re.compile(r"^(def |class |async def |@)", re.M),
    "go": re.compile(r"^(func |type \w+ (struct|interface)|var \(|const \()", re.M),
    "js": re.compile(r"^(export\s+)?(async\s+)?(function\b|class\b|const \w+\s*=|interface\b|type \w+\s*=)", re.M),
    "c": re.compile(r"^[A-Za-z_][\w\s\*:<>,&]*\([^;]*$|^(class|struct|namespace|template)\b", re.M),
    "rust": re.compile(r"^(pub\s+)?(fn |struct |enum |impl |trait |mod |macro_rules!)", re.M),
    "java": re.compile(r"^\s{0,4}(public|private|protected|static|final|abstract|class|interface|enum)\b", re.M),
    "ruby": re.compile(r"^(\s*)(def |class |module )", re.M),
    "shell": re.compile(r"^(\w+\s*\(\)\s*\{|function \w+)", re.M),
}

EXT_LANG = {
    ".py": "python", ".pyi": "python",
    ".go": "go",
    ".js": "js", ".jsx": "js", ".ts": "js", ".tsx": "js", ".mjs": "js",
    ".c": "c", ".h": "c", ".cc": "c", ".cpp": "c", ".cxx": "c",
    ".hpp": "c", ".hip": "c", ".cu": "c", ".cuh": "c",
    ".rs": "rust",
    ".java": "java", ".kt": "java", ".scala": "java",
    ".rb": "ruby",
    ".sh": "shell", ".bash": "shell",
}


def detect_language(path: str) -> str:
    m = re.search(r"\.[A-Za-z0-9]+$", path or "")
    return EXT_LANG.get(m.group(0).lower() if m else "", "")


def _decl_starts(lines: List[str], lang: str) -> List[int]:
    """Line indices where a top-level declaration begins."""
    rx = _DECL_RES.get(lang)
    if rx is None:
        return []
    starts = []
    for i, ln in enumerate(lines):
        if not ln or ln[0] in " \t":
            # top-level only, except java/ruby which allow indentation
            if lang not in ("java", "ruby"):
                continue
        if rx.match(ln):
            # pull leading decorators/comments/attributes into the unit
            j = i
            while j > 0 and re.match(
                    r"^\s*(@|#|//|/\*|\*|#\[)", lines[j - 1] or "#"):
                if not lines[j - 1].strip():
                    break
                j -= 1
            starts.append(j)
    # dedupe while keeping order
    seen, out = set(), []
    for s in starts:
        if s not in seen:
            seen.add(s)
            out.append(s)
    return sorted(out)


def chunk_code(text: str, path: str, max_lines: int = 80,
               overlap: int = 8, metadata: dict | None = None) -> List[dict]:
    """Split code at declaration boundaries, packing whole units into
    <=max_lines windows; units longer than max_lines fall back to a
    sliding line window. Unknown languages use the line window. Each
    chunk is prefixed with a path:line header for retrieval grounding."""
    metadata = metadata or {}
    lines = text.splitlines()
    if not lines:
        return []
    lang = detect_language(path)
    starts = _decl_starts(lines, lang)
    bounds: List[tuple] = []              # (start, end) line windows
    if len(starts) >= 2:
        if starts[0] != 0:
            starts = [0] + starts
        for a, b in zip(starts, starts[1:] + [len(lines)]):
            bounds.append((a, b))
    else:
        bounds = [(0, len(lines))]

    windows: List[tuple] = []
    cur_a = cur_b = None
    for a, b in bounds:
        if b - a > max_lines:
            if cur_a is not None:
                windows.append((cur_a, cur_b))
                cur_a = None
            step = max(1, max_lines - overlap)
            for i in range(a, b, step):
                windows.append((i, min(i + max_lines, b)))
                if i + max_lines >= b:
                    break
        elif cur_a is None:
            cur_a, cur_b = a, b
        elif cur_b - cur_a + (b - a) <= max_lines:
            cur_b = b
        else:
            windows.append((cur_a, cur_b))
            cur_a, cur_b = a, b
    if cur_a is not None:
        windows.append((cur_a, cur_b))

    out = []
    for a, b in windows:
        body = "\n".join(lines[a:b]).strip("\n")
        if not body.strip():
            continue
        out.append({"text": f"// {path}:{a + 1}\n{body}",
                    "metadata": dict(metadata, path=path, start_line=a + 1,
                                     language=lang or "text")})
    return out


def chunk_any(text: str, path: str = "", chunk_size: int = 512,
              overlap: int = 64, metadata: dict | None = None) -> List[dict]:
    """Dispatch: code files get declaration-aware chunking, everything
    else the paragraph window."""
    if detect_language(path):
        return chunk_code(text, path, metadata=metadata)
    return chunk_text(text, chunk_size, overlap,
                      dict(metadata or {}, **({"path": path} if path else {})))
