"""helix.yaml loading (parity with api/pkg/apps NewLocalApp +
config.ProcessYAMLConfig: plain form and CRD form, file refs inlined)."""
from __future__ import annotations

import os
from typing import Any, Dict

import yaml

from helix_amd.server.types import AppHelixConfig


def parse_helix_yaml(text: str, base_dir: str = ".") -> AppHelixConfig:
    doc = yaml.safe_load(text) or {}
    return parse_helix_config(doc, base_dir)


def parse_helix_config(doc: Dict[str, Any],
                       base_dir: str = ".") -> AppHelixConfig:
    # CRD form: apiVersion/kind/metadata/spec (reference AgentHelixConfigCRD)
    if "spec" in doc and ("apiVersion" in doc or "kind" in doc):
        spec = doc.get("spec") or {}
        meta = doc.get("metadata") or {}
        if "name" not in spec and meta.get("name"):
            spec = {**spec, "name": meta["name"]}
        doc = spec
    doc = _inline_file_refs(doc, base_dir)
    return AppHelixConfig.model_validate(doc)


def _inline_file_refs(node, base_dir: str):
    """Replace {"from_file": path} / "file://path" strings with file
    contents (reference process_files.go)."""
    if isinstance(node, dict):
        if set(node.keys()) == {"from_file"}:
            return _read(node["from_file"], base_dir)
        return {k: _inline_file_refs(v, base_dir) for k, v in node.items()}
    if isinstance(node, list):
        return [_inline_file_refs(v, base_dir) for v in node]
    if isinstance(node, str) and node.startswith("file://"):
        return _read(node[len("file://"):], base_dir)
    return node


def _read(path: str, base_dir: str) -> str:
    full = path if os.path.isabs(path) else os.path.join(base_dir, path)
    with open(full, "r") as f:
        return f.read()


def load_app_file(path: str) -> AppHelixConfig:
    with open(path) as f:
        return parse_helix_yaml(f.read(), os.path.dirname(
            os.path.abspath(path)))
