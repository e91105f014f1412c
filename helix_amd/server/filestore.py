"""Filestore — local FS backend with per-owner namespaces (parity with
api/pkg/filestore local driver: session/app file trees, avatars).
Uploads are raw-body PUTs (no multipart dependency in this image)."""
from __future__ import annotations

import os
import shutil
import time
from typing import List


class FileStore:
    def __init__(self, root: str):
        self.root = os.path.abspath(root)
        os.makedirs(self.root, exist_ok=True)

    def _resolve(self, owner: str, path: str) -> str:
        base = os.path.abspath(os.path.join(self.root, "users", owner))
        full = os.path.abspath(os.path.join(base, path.lstrip("/")))
        # note the os.sep suffix: plain startswith would accept a
        # sibling dir whose name has this owner's as a prefix
        if full != base and not full.startswith(base + os.sep):
            raise PermissionError("path escapes filestore root")
        return full

    def write(self, owner: str, path: str, data: bytes) -> dict:
        full = self._resolve(owner, path)
        os.makedirs(os.path.dirname(full), exist_ok=True)
        with open(full, "wb") as f:
            f.write(data)
        return self.stat(owner, path)

    def read(self, owner: str, path: str) -> bytes:
        with open(self._resolve(owner, path), "rb") as f:
            return f.read()

    def stat(self, owner: str, path: str) -> dict:
        full = self._resolve(owner, path)
        st = os.stat(full)
        return {"path": path, "size": st.st_size, "modified": st.st_mtime,
                "is_dir": os.path.isdir(full)}

    def list(self, owner: str, path: str = "") -> List[dict]:
        full = self._resolve(owner, path)
        if not os.path.exists(full):
            return []
        out = []
        for name in sorted(os.listdir(full)):
            p = os.path.join(path, name) if path else name
            out.append(self.stat(owner, p))
        return out

    def delete(self, owner: str, path: str) -> bool:
        full = self._resolve(owner, path)
        if os.path.isdir(full):
            shutil.rmtree(full)
            return True
        if os.path.exists(full):
            os.unlink(full)
            return True
        return False

    def user_root(self, owner: str) -> str:
        base = os.path.join(self.root, "users", owner)
        os.makedirs(base, exist_ok=True)
        return base
