"""Knowledge reconciler — per-knowledge state machine (parity with the
reference api/pkg/controller/knowledge: states preparing -> pending ->
indexing -> ready/error, knowledge.go:133-140 sources, versioning,
refresh schedules). Sources: inline text, filestore files, web crawl
(link-following with readability extraction; needs network — errors
cleanly in air-gapped deployments), and S3/GCS object stores via a
pluggable client (offline deployments use the local-directory client,
mirroring the reference's S3/GCS source config without cloud SDKs).

Round 2: indexing runs as per-knowledge
async jobs with a concurrency cap, so one large document set no longer
blocks state transitions for other rows; the RAG layer swaps versions
atomically, so queries serve the previous index during reindex; cron
refresh schedules re-queue ready rows (reference knowledge cron).
"""
from __future__ import annotations

import asyncio
import fnmatch
import logging
import os
import time
from typing import Dict, List, Optional
from urllib.parse import urljoin, urlparse

from helix_amd.server.extract import extract_html, extract_text
from helix_amd.server.triggers import CronSchedule
from helix_amd.server.types import new_id

log = logging.getLogger("helix_amd.knowledge")

STATES = ("preparing", "pending", "indexing", "ready", "error")


class LocalObjectStore:
    """Offline stand-in for the reference's S3/GCS knowledge sources
    (knowledge.go source config): buckets are directories under a
    configured root. Cloud deployments would swap in a real client with
    the same two methods."""

    def __init__(self, root: str):
        self.root = root

    def _bucket(self, bucket: str) -> str:
        base = os.path.abspath(self.root)
        p = os.path.abspath(os.path.join(base, bucket))
        if p != base and not p.startswith(base + os.sep):
            raise PermissionError("bucket escapes object-store root")
        return p

    def list(self, bucket: str, prefix: str = "") -> List[str]:
        root = self._bucket(bucket)
        if not os.path.isdir(root):
            raise FileNotFoundError(f"bucket not found: {bucket}")
        keys = []
        for r, _, files in os.walk(root):
            for f in files:
                key = os.path.relpath(os.path.join(r, f), root)
                if not prefix or key.startswith(prefix):
                    keys.append(key)
        return sorted(keys)

    def get(self, bucket: str, key: str) -> bytes:
        root = self._bucket(bucket)
        p = os.path.abspath(os.path.join(root, key))
        if not p.startswith(root + os.sep):
            raise PermissionError("key escapes bucket")
        with open(p, "rb") as fh:
            return fh.read()


class KnowledgeReconciler:
    def __init__(self, cfg, store, rag, filestore_path: str = "",
                 object_store=None, max_concurrent_indexing: int = 3):
        self.cfg = cfg
        self.store = store
        self.rag = rag
        self.filestore_path = filestore_path or cfg.filestore.path
        self.object_store = object_store
        self.max_concurrent = max_concurrent_indexing
        self._jobs: Dict[str, asyncio.Task] = {}
        self._task: Optional[asyncio.Task] = None

    def _get_object_store(self, kind: str):
        if self.object_store is not None:
            return self.object_store
        root = getattr(self.cfg.rag, "object_store_path", "") or \
            os.environ.get("HELIX_OBJECT_STORE_PATH", "")
        if root:
            return LocalObjectStore(root)
        raise RuntimeError(
            f"{kind} source needs an object-store client; set "
            "HELIX_OBJECT_STORE_PATH for the local client or inject one")

    # -- CRUD ---------------------------------------------------------------
    def create(self, owner: str, name: str, source: dict,
               app_id: str = "", refresh_schedule: str = "") -> dict:
        kid = new_id("kno")
        if refresh_schedule:
            CronSchedule(refresh_schedule)      # validate early
        doc = {"id": kid, "name": name, "owner": owner, "app_id": app_id,
               "source": source, "state": "preparing", "message": "",
               "version": 0, "chunks": 0, "created": time.time(),
               "refresh_schedule": refresh_schedule, "last_indexed": 0.0}
        self.store.put("knowledge", kid, doc, owner=owner, parent=app_id)
        return doc

    def get(self, kid: str) -> Optional[dict]:
        return self.store.get("knowledge", kid)

    def list(self, owner: str) -> List[dict]:
        return self.store.list("knowledge", owner=owner)

    def delete(self, kid: str) -> bool:
        job = self._jobs.pop(kid, None)
        if job:
            job.cancel()
        if self.rag is not None:
            self.rag.delete(kid)
        return self.store.delete("knowledge", kid)

    def request_refresh(self, kid: str):
        doc = self.get(kid)
        if doc:
            doc["state"] = "preparing"
            self._save(doc)

    # -- reconcile loop ------------------------------------------------------
    async def reconcile_once(self, wait: bool = True) -> int:
        """Advance every knowledge row one state. Indexing runs as a
        background task per row (<= max_concurrent at once); wait=True
        (tests / synchronous callers) gathers the jobs spawned by this
        pass, wait=False (the serve loop) returns immediately so a slow
        index cannot stall other rows' transitions."""
        n = 0
        spawned: List[asyncio.Task] = []
        now = time.time()
        for doc in self.store.list("knowledge", limit=10000):
            kid = doc["id"]
            state = doc.get("state")
            if state == "preparing":
                doc["state"] = "pending"
                self._save(doc)
                n += 1
            elif state == "pending":
                if kid in self._jobs or \
                        len(self._jobs) >= self.max_concurrent:
                    continue
                doc["state"] = "indexing"
                self._save(doc)
                t = asyncio.get_event_loop().create_task(
                    self._run_index(doc))
                self._jobs[kid] = t
                spawned.append(t)
                n += 1
            elif state == "indexing" and kid not in self._jobs:
                # crashed/restarted mid-index: re-queue (startup recovery)
                doc["state"] = "pending"
                self._save(doc)
                n += 1
            elif state == "ready" and doc.get("refresh_schedule"):
                try:
                    sched = CronSchedule(doc["refresh_schedule"])
                except ValueError:
                    continue
                last = float(doc.get("last_indexed") or 0)
                if now - last >= 90 and sched.matches(time.localtime(now)):
                    doc["state"] = "preparing"
                    self._save(doc)
                    n += 1
        if wait and spawned:
            await asyncio.gather(*spawned, return_exceptions=True)
        return n

    async def _run_index(self, doc):
        kid = doc["id"]
        try:
            count = await self._index(doc)
            doc["state"] = "ready"
            doc["chunks"] = count
            doc["version"] = doc.get("version", 0) + 1
            doc["last_indexed"] = time.time()
            doc["message"] = f"indexed {count} chunks"
            if doc.pop("_skipped", None):
                doc["message"] += "; " + doc.pop("_skip_note", "")
            vid = new_id("kver")
            self.store.put("knowledge_versions", vid,
                           {"id": vid, "knowledge_id": kid,
                            "version": doc["version"],
                            "chunks": count, "ts": time.time()},
                           parent=kid)
        except asyncio.CancelledError:
            raise
        except Exception as e:
            log.exception("indexing failed for %s", kid)
            doc["state"] = "error"
            doc["message"] = str(e)
        finally:
            self._jobs.pop(kid, None)
        self._save(doc)

    def _save(self, doc):
        self.store.put("knowledge", doc["id"], doc, owner=doc["owner"],
                       parent=doc.get("app_id", ""))

    # -- sources -------------------------------------------------------------
    async def _index(self, doc) -> int:
        if self.rag is None:
            raise RuntimeError("RAG service unavailable")
        src = doc.get("source", {})
        documents = []
        skipped: List[str] = []
        if "text" in src:
            content = src["text"]
            if isinstance(content, dict):
                content = content.get("content", "")
            documents.append({"text": content,
                              "metadata": {"source": "text"}})
        elif "filestore" in src:
            documents = self._read_filestore(doc, src, skipped)
        elif "web" in src:
            documents = await self._crawl(src["web"])
        elif "s3" in src or "gcs" in src:
            kind = "s3" if "s3" in src else "gcs"
            cfg = src[kind]
            client = self._get_object_store(kind)
            bucket = cfg.get("bucket", "")
            prefix = cfg.get("prefix", cfg.get("path", "")) or ""
            glob = cfg.get("glob", "")
            for key in client.list(bucket, prefix):
                if glob and not fnmatch.fnmatch(key, glob):
                    continue
                raw = client.get(bucket, key)
                try:
                    text = extract_text(
                        raw.decode("utf-8", errors="replace"), path=key)
                except ValueError as e:
                    skipped.append(f"{key}: {e}")
                    continue
                documents.append({
                    "text": text,
                    "metadata": {"source": f"{kind}://{bucket}/{key}",
                                 "path": key}})
        else:
            raise ValueError(f"unsupported knowledge source: {list(src)}")
        # versioned swap happens inside the RAG layer: the previous
        # index keeps serving queries until the new one is complete;
        # batch progress lands on the row (reference progress_percent)
        def _progress(done: int, total: int):
            doc["progress_percent"] = int(100 * done / max(1, total))
            self._save(doc)

        count = await self.rag.index(doc["id"], documents,
                                     progress=_progress)
        doc["progress_percent"] = 100
        if skipped:
            doc["_skipped"] = True
            doc["_skip_note"] = f"skipped {len(skipped)}: " + \
                "; ".join(skipped[:5])
        return count

    def _read_filestore(self, doc, src, skipped: List[str]) -> List[dict]:
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
        documents = []

        def _read_one(p: str):
            try:
                with open(p, "r", errors="ignore") as fh:
                    raw = fh.read()
            except OSError:
                return
            try:
                text = extract_text(raw, path=p)
            except ValueError as e:
                skipped.append(f"{os.path.basename(p)}: {e}")
                return
            documents.append({"text": text,
                              "metadata": {"source": p, "path": p}})

        if os.path.isdir(base):
            for root, _, files in os.walk(base):
                for f in files:
                    _read_one(os.path.join(root, f))
        elif os.path.isfile(base):
            _read_one(base)
        else:
            raise FileNotFoundError(base)
        return documents

    async def _crawl(self, web: dict) -> List[dict]:
        """Breadth-first crawl with readability extraction (reference
        crawler: Chrome pool + readability; this is the no-JS httpx
        equivalent). Follows same-host links up to max_pages."""
        import httpx
        urls = list(web.get("urls", []))
        max_pages = int(web.get("max_pages", 5) or 5)
        max_depth = int(web.get("max_depth", 1) or 1)
        seen = set()
        docs = []
        queue = [(u, 0) for u in urls]
        async with httpx.AsyncClient(timeout=20) as http:
            while queue and len(docs) < max_pages:
                u, depth = queue.pop(0)
                if u in seen:
                    continue
                seen.add(u)
                r = await http.get(u, follow_redirects=True)
                page = extract_html(r.text)
                title = f"# {page['title']}\n\n" if page["title"] else ""
                docs.append({"text": title + page["text"],
                             "metadata": {"source": u,
                                          "title": page["title"]}})
                if depth < max_depth:
                    host = urlparse(u).netloc
                    for href in page["links"]:
                        nxt = urljoin(u, href.split("#")[0])
                        if urlparse(nxt).netloc == host and \
                                nxt not in seen and \
                                nxt.startswith(("http://", "https://")):
                            queue.append((nxt, depth + 1))
        return docs

    async def run(self, interval: float = 5.0):
        while True:
            try:
                await self.reconcile_once(wait=False)
            except Exception:
                log.exception("reconcile loop error")
            await asyncio.sleep(interval)
