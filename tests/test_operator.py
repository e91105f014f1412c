"""K8s operator reconcile loop (deploy/operator.py) against a fake
API server: CR -> children created, spec changes -> children updated,
CR deletion -> children garbage-collected.
"""
import importlib.util
import os

import pytest

_spec = importlib.util.spec_from_file_location(
    "helix_operator",
    os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "deploy", "operator.py"))
operator = importlib.util.module_from_spec(_spec)
_spec.loader.exec_module(operator)


class FakeK8s:
    """Stores objects by path; speaks the subset the operator uses."""

    def __init__(self):
        self.objects = {}
        self.crs = []

    def request(self, method, path, body=None):
        if method == "GET" and path.endswith(
                f"/{operator.PLURAL}"):
            return 200, {"items": self.crs}
        if method == "GET" and "labelSelector" in path:
            kind = path.split("?")[0].rsplit("/", 1)[-1]
            items = [o for p, o in self.objects.items()
                     if f"/{kind}/" in p + "/"
                     and p.split("/")[-2] == kind]
            return 200, {"items": items}
        if method == "GET":
            if path in self.objects:
                return 200, self.objects[path]
            return 404, {}
        if method == "POST":
            name = body["metadata"]["name"]
            self.objects[f"{path}/{name}"] = body
            return 201, body
        if method == "PUT":
            self.objects[path] = body
            return 200, body
        if method == "DELETE":
            self.objects.pop(path, None)
            return 200, {}
        return 400, {}


def _cr(name, **spec):
    return {"metadata": {"name": name}, "spec": spec}


def test_reconcile_creates_updates_and_gcs():
    api = FakeK8s()
    rec = operator.Reconciler(api, "helix")
    api.crs = [_cr("prod", image="helix-amd:v1", replicas=2,
                   runnerToken="rt", adminApiKey="ak")]
    assert rec.reconcile_once() == 1
    dep = api.objects[
        "/apis/apps/v1/namespaces/helix/deployments/prod-cp"]
    assert dep["spec"]["replicas"] == 2
    assert dep["spec"]["template"]["spec"]["containers"][0][
        "image"] == "helix-amd:v1"
    ds = api.objects[
        "/apis/apps/v1/namespaces/helix/daemonsets/prod-runner"]
    assert ds["spec"]["template"]["spec"]["containers"][0][
        "resources"]["limits"]["amd.com/gpu"] == 1
    assert "/api/v1/namespaces/helix/services/prod-api" in api.objects
    sec = api.objects["/api/v1/namespaces/helix/secrets/prod-auth"]
    assert sec["stringData"]["runner-token"] == "rt"

    # spec update flows through
    api.crs = [_cr("prod", image="helix-amd:v2", replicas=3,
                   runnerToken="rt", adminApiKey="ak")]
    rec.reconcile_once()
    dep = api.objects[
        "/apis/apps/v1/namespaces/helix/deployments/prod-cp"]
    assert dep["spec"]["replicas"] == 3
    assert dep["spec"]["template"]["spec"]["containers"][0][
        "image"] == "helix-amd:v2"

    # runner disabled removes nothing automatically but new CR set
    # without the instance GCs all its children
    api.crs = []
    rec.reconcile_once()
    assert not any("prod" in p for p in api.objects), api.objects.keys()


def test_runner_disabled_and_node_selector():
    api = FakeK8s()
    rec = operator.Reconciler(api, "ns1")
    api.crs = [_cr("edge", runner={"enabled": False})]
    rec.reconcile_once()
    assert not any("daemonsets" in p for p in api.objects)
    api.crs = [_cr("edge", runner={
        "enabled": True, "gpusPerPod": 8,
        "nodeSelector": {"pool": "mi355x"}})]
    rec.reconcile_once()
    ds = api.objects[
        "/apis/apps/v1/namespaces/ns1/daemonsets/edge-runner"]
    assert ds["spec"]["template"]["spec"]["nodeSelector"] == {
        "pool": "mi355x"}
    assert ds["spec"]["template"]["spec"]["containers"][0][
        "resources"]["limits"]["amd.com/gpu"] == 8
