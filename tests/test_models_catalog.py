"""ModelCatalog (reference api/pkg/model: model_info.json +
DynamicModelInfoProvider): static catalog, DB overrides winning,
merged listing, context/cost helpers.
"""
import pytest

from helix_amd.server.models_catalog import (STATIC_MODEL_INFO,
                                             ModelCatalog)
from helix_amd.store import Store


@pytest.fixture()
def cat():
    return ModelCatalog(Store(":memory:"))


def test_static_lookup_and_miss(cat):
    info = cat.get("llama3-8b")
    assert info["context_length"] == 8192
    assert info["family"] == "llama"
    assert cat.get("no-such-model") is None
    # every serving family is priced + kinded
    for mid in ("llama3-70b", "bge-base", "flux-lite"):
        assert STATIC_MODEL_INFO[mid]["kind"]


def test_override_wins_and_merges(cat):
    cat.set_override("llama3-8b", {"context_length": 16384})
    info = cat.get("llama3-8b")
    assert info["context_length"] == 16384
    # non-overridden static fields survive the merge
    assert info["family"] == "llama"
    # a dynamic-only model appears in get and list
    cat.set_override("custom-ft", {"context_length": 4096,
                                   "kind": "chat"})
    assert cat.get("custom-ft")["context_length"] == 4096
    listed = {m["id"]: m for m in cat.list()}
    assert "custom-ft" in listed
    assert listed["llama3-8b"]["context_length"] == 16384


def test_context_length_default(cat):
    assert cat.context_length("llama3.1-8b") == 32768
    assert cat.context_length("unknown", default=2048) == 2048


def test_cost_usd(cat):
    # llama3-8b: 0.05/M prompt + 0.10/M completion
    cost = cat.cost_usd("llama3-8b", 1_000_000, 500_000)
    assert cost == pytest.approx(0.05 + 0.05)
    assert cat.cost_usd("unknown", 1000, 1000) == 0.0


def test_get_returns_copies(cat):
    """Mutating a returned dict must not poison the static catalog."""
    info = cat.get("llama3-8b")
    info["context_length"] = 1
    assert cat.get("llama3-8b")["context_length"] == 8192
