"""Knowledge reconciler — per-knowledge state machine (parity with the
reference api/pkg/controller/knowledge: states preparing -> pending ->
indexing -> ready/error, knowledge.go:133-140 sources, versioning,
refresh). Sources: inline text, filestore files, web crawl (needs
network; errors cleanly in air-gapped deployments).
"""
from __future__ import annotations

import asyncio
import logging
import os
import time
from typing import List, Optional

from helix_amd.server.types import new_id

log = logging.getLogger("helix_amd.knowledge")

STATES = ("preparing", "pending", "indexing", "ready", "error")


class KnowledgeReconciler:
    def __init__(self, cfg, store, rag, filestore_path: str = ""):
        self.cfg = cfg
        self.store = store
        self.rag = rag
        self.filestore_path = filestore_path or cfg.filestore.path
        self._task: Optional[asyncio.Task] = None

    # -- CRUD ---------------------------------------------------------------
    def create(self, owner: str, name: str, source: dict,
               app_id: str = "") -> dict:
        kid = new_id("kno")
        doc = {"id": kid, "name": name, "owner": owner, "app_id": app_id,
               "source": source, "state": "preparing", "message": "",
               "version": 0, "chunks": 0, "created": time.time()}
        self.store.put("knowledge", kid, doc, owner=owner, parent=app_id)
        return doc

    def get(self, kid: str) -> Optional[dict]:
        return self.store.get("knowledge", kid)

    def list(self, owner: str) -> List[dict]:
        return self.store.list("knowledge", owner=owner)

    def delete(self, kid: str) -> bool:
        if self.rag is not None:
            self.rag.delete(kid)
        return self.store.delete("knowledge", kid)

    def request_refresh(self, kid: str):
        doc = self.get(kid)
        if doc:
            doc["state"] = "preparing"
            self.store.put("knowledge", kid, doc, owner=doc["owner"],
                           parent=doc.get("app_id", ""))

    # -- reconcile loop ------------------------------------------------------
    async def reconcile_once(self) -> int:
        """Advance every knowledge row one state; returns #processed."""
        n = 0
        for doc in self.store.list("knowledge", limit=10000):
            if doc.get("state") == "preparing":
                doc["state"] = "pending"
                self._save(doc)
                n += 1
            elif doc.get("state") == "pending":
                doc["state"] = "indexing"
                self._save(doc)
                try:
                    count = await self._index(doc)
                    doc["state"] = "ready"
                    doc["chunks"] = count
                    doc["version"] = doc.get("version", 0) + 1
                    doc["message"] = f"indexed {count} chunks"
                    vid = new_id("kver")
                    self.store.put("knowledge_versions", vid,
                                   {"id": vid, "knowledge_id": doc["id"],
                                    "version": doc["version"],
                                    "chunks": count, "ts": time.time()},
                                   parent=doc["id"])
                except Exception as e:
                    log.exception("indexing failed for %s", doc["id"])
                    doc["state"] = "error"
                    doc["message"] = str(e)
                self._save(doc)
                n += 1
        return n

    def _save(self, doc):
        self.store.put("knowledge", doc["id"], doc, owner=doc["owner"],
                       parent=doc.get("app_id", ""))

    async def _index(self, doc) -> int:
        if self.rag is None:
            raise RuntimeError("RAG service unavailable")
        src = doc.get("source", {})
        documents = []
        if "text" in src:
            content = src["text"]
            if isinstance(content, dict):
                content = content.get("content", "")
            documents.append({"text": content,
                              "metadata": {"source": "text"}})
        elif "filestore" in src:
            # owner-scoped + containment-checked (same rule as
            # FileStore._resolve): a knowledge source must not read
            # outside the owner's filestore namespace
            owner_root = os.path.abspath(os.path.join(
                self.filestore_path, "users", doc.get("owner", "")))
            rel = str(src["filestore"].get("path", "")).lstrip("/")
            base = os.path.abspath(os.path.join(owner_root, rel))
            if base != owner_root and \
                    not base.startswith(owner_root + os.sep):
                raise PermissionError("path escapes filestore root")
            if os.path.isdir(base):
                for root, _, files in os.walk(base):
                    for f in files:
                        p = os.path.join(root, f)
                        try:
                            with open(p, "r", errors="ignore") as fh:
                                documents.append({
                                    "text": fh.read(),
                                    "metadata": {"source": p}})
                        except OSError:
                            continue
            elif os.path.isfile(base):
                with open(base, "r", errors="ignore") as fh:
                    documents.append({"text": fh.read(),
                                      "metadata": {"source": base}})
            else:
                raise FileNotFoundError(base)
        elif "web" in src:
            urls = src["web"].get("urls", [])
            documents = await self._crawl(urls)
        else:
            raise ValueError(f"unsupported knowledge source: {list(src)}")
        self.rag.delete(doc["id"])  # reindex from scratch (versioned)
        return await self.rag.index(doc["id"], documents)

    async def _crawl(self, urls: List[str]) -> List[dict]:
        import httpx
        docs = []
        async with httpx.AsyncClient(timeout=20) as http:
            for u in urls:
                r = await http.get(u, follow_redirects=True)
                text = r.text
                # crude readability: strip tags
                import re
                text = re.sub(r"<script.*?</script>", " ", text, flags=re.S)
                text = re.sub(r"<style.*?</style>", " ", text, flags=re.S)
                text = re.sub(r"<[^>]+>", " ", text)
                text = re.sub(r"\s+", " ", text)
                docs.append({"text": text, "metadata": {"source": u}})
        return docs

    async def run(self, interval: float = 5.0):
        while True:
            try:
                await self.reconcile_once()
            except Exception:
                log.exception("reconcile loop error")
            await asyncio.sleep(interval)
