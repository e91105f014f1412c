"""Text extraction (parity with the reference's extraction service,
api/pkg/extract/extract.go — Tika + readability in the crawler,
api/pkg/controller/knowledge/browser crawler readability pass). Offline
implementation: a readability-style HTML main-content extractor built on
html.parser plus per-format dispatch (html, markdown, code, plain).

The reference shells out to Apache Tika for binary formats and runs a
Chrome pool for JS rendering; neither exists offline, so HTML extraction
is a block-scoring pass (text density vs link density, boilerplate tag
pruning) which covers the crawler's readability behavior, and binary
formats degrade cleanly to an error the reconciler surfaces on the
knowledge row.
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field
from html.parser import HTMLParser
from typing import List, Optional

# Tags whose subtree is never content.
_DROP = {"script", "style", "noscript", "svg", "iframe",
         "nav", "footer", "aside", "form", "button", "select"}
# Block-level tags that delimit candidate text blocks.
_BLOCK = {"p", "div", "article", "section", "main", "li", "td", "th",
          "blockquote", "pre", "h1", "h2", "h3", "h4", "h5", "h6",
          "figcaption", "summary", "dd", "dt"}
_HEADING = {"h1", "h2", "h3", "h4", "h5", "h6"}

_BOILER_RE = re.compile(
    r"comment|sidebar|share|social|advert|promo|cookie|banner|menu|"
    r"breadcrumb|related|popup|modal", re.I)


@dataclass
class _Block:
    tag: str
    text: str = ""
    link_chars: int = 0
    boiler: bool = False
    depth: int = 0

    def score(self) -> float:
        n = len(self.text.strip())
        if n == 0:
            return 0.0
        link_density = self.link_chars / max(1, n)
        s = n * (1.0 - link_density)
        if self.tag in _HEADING:
            s *= 1.5
        if self.boiler:
            s *= 0.1
        return s


@dataclass
class _Parse:
    title: str = ""
    blocks: List[_Block] = field(default_factory=list)
    links: List[str] = field(default_factory=list)


class _Extractor(HTMLParser):
    def __init__(self):
        super().__init__(convert_charrefs=True)
        self.out = _Parse()
        self._drop_depth = 0
        self._in_title = False
        self._in_link = 0
        self._boiler_depth = 0
        self._stack: List[_Block] = []

    def handle_starttag(self, tag, attrs):
        ad = dict(attrs)
        if tag == "a":
            # links are collected even inside dropped/nav subtrees: the
            # crawler follows site navigation, readability only drops
            # its text
            href = ad.get("href")
            if href:
                self.out.links.append(href)
        if tag in _DROP:
            self._drop_depth += 1
            return
        if self._drop_depth:
            return
        if tag == "title":
            self._in_title = True
        if tag == "a":
            self._in_link += 1
        idcls = f"{ad.get('id', '')} {ad.get('class', '')}"
        boiler = bool(_BOILER_RE.search(idcls)) or \
            ad.get("role") in ("navigation", "banner", "contentinfo")
        if boiler:
            self._boiler_depth += 1
        if tag in _BLOCK:
            self._stack.append(_Block(
                tag=tag, boiler=boiler or self._boiler_depth > (1 if boiler else 0),
                depth=len(self._stack)))
        elif tag == "br" and self._stack:
            self._stack[-1].text += "\n"

    def handle_endtag(self, tag):
        if tag in _DROP:
            self._drop_depth = max(0, self._drop_depth - 1)
            return
        if self._drop_depth:
            return
        if tag == "title":
            self._in_title = False
        if tag == "a":
            self._in_link = max(0, self._in_link - 1)
        if tag in _BLOCK and self._stack:
            blk = self._stack.pop()
            txt = blk.text.strip()
            if txt:
                self.out.blocks.append(blk)
                # bubble nothing: nested text was consumed by this block

    def handle_data(self, data):
        if self._drop_depth:
            return
        if self._in_title:
            self.out.title += data
            return
        if self._stack:
            self._stack[-1].text += data
            if self._in_link:
                self._stack[-1].link_chars += len(data)


def extract_html(html: str, min_block_score: float = 10.0) -> dict:
    """Readability-style extraction: returns {title, text, links}.

    Blocks are scored by text length discounted by link density, halved
    for boilerplate-classed containers; the kept set is every block
    scoring >= min_block_score OR >= 20% of the best block's score, in
    document order — this keeps article bodies with their headings and
    drops nav/footer/share rows.
    """
    p = _Extractor()
    try:
        p.feed(html)
        p.close()
    except Exception:
        pass
    blocks = p.out.blocks
    if not blocks:
        text = re.sub(r"<[^>]+>", " ", html)
        text = re.sub(r"\s+", " ", text).strip()
        return {"title": p.out.title.strip(), "text": text,
                "links": p.out.links}
    best = max(b.score() for b in blocks)
    kept = [b for b in blocks
            if b.score() >= min_block_score or
            (best > 0 and b.score() >= 0.2 * best)]
    if not kept:
        kept = sorted(blocks, key=lambda b: -b.score())[:3]
    parts = []
    for b in kept:
        t = re.sub(r"[ \t]+", " ", b.text).strip()
        t = re.sub(r"\n{2,}", "\n", t)
        if b.tag in _HEADING:
            t = f"\n# {t}\n"
        parts.append(t)
    text = "\n".join(parts).strip()
    text = re.sub(r"\n{3,}", "\n\n", text)
    return {"title": p.out.title.strip(), "text": text, "links": p.out.links}


_MD_CODE = re.compile(r"```.*?```", re.S)
_MD_LINK = re.compile(r"\[([^\]]*)\]\([^)]*\)")
_MD_IMG = re.compile(r"!\[([^\]]*)\]\([^)]*\)")

TEXT_EXT = {".txt", ".text", ".rst", ".adoc", ".csv", ".tsv", ".log",
            ".json", ".yaml", ".yml", ".toml", ".ini", ".cfg", ".xml"}
HTML_EXT = {".html", ".htm", ".xhtml"}
MD_EXT = {".md", ".markdown", ".mdx"}
BINARY_EXT = {".pdf", ".doc", ".docx", ".ppt", ".pptx", ".xls", ".xlsx",
              ".zip", ".gz", ".png", ".jpg", ".jpeg", ".gif", ".so",
              ".bin", ".exe", ".tar"}


def extract_text(content: str, path: str = "",
                 content_type: str = "") -> str:
    """Dispatch extraction by path extension / content type. Binary
    formats raise ValueError (the reference sends these to Tika, which
    is unavailable offline — the reconciler records the error)."""
    ext = ""
    if path:
        m = re.search(r"\.[A-Za-z0-9]+$", path)
        ext = m.group(0).lower() if m else ""
    if ext in BINARY_EXT:
        raise ValueError(
            f"binary format {ext} needs the Tika extraction service "
            "(unavailable offline)")
    ct = (content_type or "").lower()
    if ext in HTML_EXT or "text/html" in ct or \
            (not ext and content.lstrip()[:200].lower().startswith(
                ("<!doctype html", "<html"))):
        r = extract_html(content)
        title = f"# {r['title']}\n\n" if r["title"] else ""
        return title + r["text"]
    if ext in MD_EXT or "markdown" in ct:
        text = _MD_IMG.sub(r"\1", content)
        text = _MD_LINK.sub(r"\1", text)
        return text
    return content
