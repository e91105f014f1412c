"""GPU detection with rocm-smi / torch — parity with the reference's
gpudetect (gpudetect.go:125-145 rocm-smi CSV path) plus the gfx950/CDNA4
entry the reference's arch table lacked (SURVEY.md §2.2 gpuarch:
"MI355X = gfx950/CDNA4 is NOT in this table").
"""
from __future__ import annotations

import csv
import io
import subprocess
from typing import List

from helix_amd.server.types import GPUStatus

# gfx -> canonical arch (reference canonical.go:29-35 + the new CDNA4 row)
AMD_GFX_ARCH = {
    "gfx950": "cdna4",
    "gfx942": "cdna3",
    "gfx90a": "cdna2",
    "gfx908": "cdna1",
    "gfx1100": "rdna3",
    "gfx1030": "rdna2",
    "gfx906": "vega",
    "gfx900": "vega",
}


def detect() -> List[GPUStatus]:
    gpus = _detect_torch()
    if gpus:
        return gpus
    return _detect_rocm_smi()


def _detect_torch() -> List[GPUStatus]:
    try:
        import torch
        if not torch.cuda.is_available():
            return []
        out = []
        for i in range(torch.cuda.device_count()):
            props = torch.cuda.get_device_properties(i)
            free, total = torch.cuda.mem_get_info(i)
            arch = getattr(props, "gcnArchName", "").split(":")[0]
            out.append(GPUStatus(
                index=i, vendor="amd",
                arch=AMD_GFX_ARCH.get(arch, arch or "unknown"),
                name=props.name, total_memory=total, free_memory=free,
                used_memory=total - free))
        return out
    except Exception:
        return []


def _detect_rocm_smi() -> List[GPUStatus]:
    try:
        res = subprocess.run(
            ["rocm-smi", "--showproductname", "--showmeminfo", "vram",
             "--csv"],
            capture_output=True, text=True, timeout=20)
        if res.returncode != 0:
            return []
        out = []
        rdr = csv.DictReader(io.StringIO(res.stdout))
        for i, row in enumerate(rdr):
            total = int(row.get("VRAM Total Memory (B)", 0) or 0)
            used = int(row.get("VRAM Total Used Memory (B)", 0) or 0)
            name = row.get("Card Series", row.get("Card Model", "AMD GPU"))
            out.append(GPUStatus(index=i, vendor="amd", arch="cdna4",
                                 name=name or "AMD GPU", total_memory=total,
                                 used_memory=used,
                                 free_memory=total - used))
        return out
    except Exception:
        return []
