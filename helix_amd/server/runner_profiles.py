"""Runner profiles + GPU compatibility (parity with the reference's
api/pkg/runner/profile: RunnerProfile{Models, GPURequirement} and the
AND-composed Compatibility check, compatibility.go:50,138 — re-based on
model manifests instead of compose YAML, with the CDNA4/gfx950 arch rows
the reference lacked)."""
from __future__ import annotations

from typing import List, Optional

from pydantic import BaseModel, Field

from helix_amd.server.types import GPUStatus, new_id


class ProfileGPURequirement(BaseModel):
    """AND-composed requirement (reference runner_profile.go:32-39)."""
    count: int = 1
    vendor: str = ""                  # '' = any
    architectures: List[str] = []     # e.g. ["cdna4", "cdna3"]
    min_vram_bytes: int = 0
    model_match: str = ""             # substring of GPU name


class RunnerProfile(BaseModel):
    id: str = Field(default_factory=lambda: new_id("prof"))
    name: str = ""
    description: str = ""
    models: List[dict] = []           # ModelSpec dicts (name, preset, ...)
    gpu_requirement: ProfileGPURequirement = ProfileGPURequirement()


def gpu_matches(req: ProfileGPURequirement, gpu: GPUStatus) -> bool:
    if req.vendor and gpu.vendor != req.vendor:
        return False
    if req.architectures and gpu.arch not in req.architectures:
        return False
    if req.min_vram_bytes and gpu.total_memory < req.min_vram_bytes:
        return False
    if req.model_match and req.model_match.lower() not in gpu.name.lower():
        return False
    return True


def compatibility(req: ProfileGPURequirement,
                  gpus: List[GPUStatus]) -> tuple[bool, str]:
    """All conditions AND-composed; needs `count` matching GPUs
    (reference compatibility.go:50)."""
    matching = [g for g in gpus if gpu_matches(req, g)]
    if len(matching) < req.count:
        return False, (f"requires {req.count} matching GPU(s), "
                       f"found {len(matching)} of {len(gpus)}")
    return True, ""


def filter_compatible(profiles: List[RunnerProfile],
                      gpus: List[GPUStatus]) -> List[RunnerProfile]:
    return [p for p in profiles if compatibility(p.gpu_requirement, gpus)[0]]


class ProfileService:
    """CRUD + runner assignment (reference profile/store.go + the
    assign-profile flow)."""

    def __init__(self, store):
        self.store = store

    def create(self, profile: RunnerProfile) -> RunnerProfile:
        self.store.put("runner_profiles", profile.id, profile.model_dump())
        return profile

    def get(self, pid: str) -> Optional[RunnerProfile]:
        doc = self.store.get("runner_profiles", pid)
        return RunnerProfile.model_validate(doc) if doc else None

    def list(self) -> List[RunnerProfile]:
        return [RunnerProfile.model_validate(d)
                for d in self.store.list("runner_profiles", limit=1000)]

    def delete(self, pid: str) -> bool:
        return self.store.delete("runner_profiles", pid)

    def assign(self, runner_id: str, profile_id: str,
               gpus: List[GPUStatus]) -> tuple[bool, str]:
        prof = self.get(profile_id)
        if prof is None:
            return False, "profile not found"
        ok, why = compatibility(prof.gpu_requirement, gpus)
        if not ok:
            return False, why
        self.store.put("runner_assignments", runner_id,
                       {"id": runner_id, "profile_id": profile_id})
        return True, ""

    def assignment(self, runner_id: str) -> Optional[dict]:
        doc = self.store.get("runner_assignments", runner_id)
        if doc is None:
            return None
        prof = self.get(doc["profile_id"])
        return prof.model_dump() if prof else None

    def clear_assignment(self, runner_id: str) -> bool:
        return self.store.delete("runner_assignments", runner_id)
