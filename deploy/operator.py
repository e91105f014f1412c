#!/usr/bin/env python3
"""helix-amd Kubernetes operator (the reference ships Helm charts; an
operator closes the lifecycle loop: declare a HelixDeployment custom
resource, get the control-plane Deployment + runner DaemonSet +
Service + Secret reconciled continuously).

Runs with only the standard library + an injectable API client, so it
is fully testable offline and needs no kubernetes pip package in the
cluster image either (the in-cluster REST API + service-account token
is enough).

    python deploy/operator.py --namespace helix

CRD (apply once):
  apiVersion: apiextensions.k8s.io/v1
  kind: CustomResourceDefinition
  metadata: {name: helixdeployments.helix.amd}
  spec:
    group: helix.amd
    names: {kind: HelixDeployment, plural: helixdeployments,
            singular: helixdeployment, shortNames: [hxd]}
    scope: Namespaced
    versions:
      - name: v1
        served: true
        storage: true
        schema:
          openAPIV3Schema:
            type: object
            x-kubernetes-preserve-unknown-fields: true
"""
from __future__ import annotations

import argparse
import json
import logging
import os
import ssl
import time
import urllib.request
from typing import Dict, List, Optional

log = logging.getLogger("helix-operator")

GROUP = "helix.amd"
VERSION = "v1"
PLURAL = "helixdeployments"


class InClusterClient:
    """Minimal K8s REST client using the pod service account."""

    def __init__(self):
        host = os.environ["KUBERNETES_SERVICE_HOST"]
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        self.base = f"https://{host}:{port}"
        sa = "/var/run/secrets/kubernetes.io/serviceaccount"
        self.token = open(f"{sa}/token").read().strip()
        self.ctx = ssl.create_default_context(cafile=f"{sa}/ca.crt")

    def request(self, method: str, path: str,
                body: Optional[dict] = None) -> tuple:
        req = urllib.request.Request(
            self.base + path, method=method,
            data=json.dumps(body).encode() if body is not None else None,
            headers={"Authorization": f"Bearer {self.token}",
                     "Content-Type": "application/json",
                     "Accept": "application/json"})
        try:
            with urllib.request.urlopen(req, context=self.ctx) as r:
                return r.status, json.loads(r.read() or b"{}")
        except urllib.error.HTTPError as e:
            return e.code, json.loads(e.read() or b"{}")


# -- desired-state builders (mirrors deploy/helm/helix-amd) ------------------

def _labels(name: str, role: str) -> dict:
    return {"app.kubernetes.io/name": "helix-amd",
            "app.kubernetes.io/instance": name,
            "helix.amd/role": role}


def build_children(name: str, ns: str, spec: dict) -> List[dict]:
    image = spec.get("image", "helix-amd:latest")
    port = int(spec.get("port", 8080))
    admin_key = spec.get("adminApiKey", "")
    runner_token = spec.get("runnerToken", "")
    runner = spec.get("runner", {})
    objs = [{
        "apiVersion": "v1", "kind": "Secret",
        "metadata": {"name": f"{name}-auth", "namespace": ns,
                     "labels": _labels(name, "auth")},
        "stringData": {"admin-api-key": admin_key,
                       "runner-token": runner_token},
    }, {
        "apiVersion": "apps/v1", "kind": "Deployment",
        "metadata": {"name": f"{name}-cp", "namespace": ns,
                     "labels": _labels(name, "control-plane")},
        "spec": {
            "replicas": int(spec.get("replicas", 1)),
            "selector": {"matchLabels": _labels(name, "control-plane")},
            "template": {
                "metadata": {"labels": _labels(name, "control-plane")},
                "spec": {"containers": [{
                    "name": "control-plane", "image": image,
                    "command": ["python", "-m", "helix_amd.cli",
                                "serve"],
                    "ports": [{"containerPort": port}],
                    "env": [
                        {"name": "SERVER_PORT", "value": str(port)},
                        {"name": "HELIX_ADMIN_API_KEY",
                         "valueFrom": {"secretKeyRef": {
                             "name": f"{name}-auth",
                             "key": "admin-api-key"}}},
                        {"name": "HELIX_RUNNER_TOKEN",
                         "valueFrom": {"secretKeyRef": {
                             "name": f"{name}-auth",
                             "key": "runner-token"}}},
                    ],
                }]},
            },
        },
    }, {
        "apiVersion": "v1", "kind": "Service",
        "metadata": {"name": f"{name}-api", "namespace": ns,
                     "labels": _labels(name, "api")},
        "spec": {"selector": _labels(name, "control-plane"),
                 "ports": [{"port": port, "targetPort": port}]},
    }]
    if runner.get("enabled", True):
        objs.append({
            "apiVersion": "apps/v1", "kind": "DaemonSet",
            "metadata": {"name": f"{name}-runner", "namespace": ns,
                         "labels": _labels(name, "runner")},
            "spec": {
                "selector": {"matchLabels": _labels(name, "runner")},
                "template": {
                    "metadata": {"labels": _labels(name, "runner")},
                    "spec": {
                        "nodeSelector": runner.get("nodeSelector", {}),
                        "containers": [{
                            "name": "runner", "image": image,
                            "command": [
                                "python", "-m", "helix_amd.cli",
                                "runner", "--api-url",
                                f"http://{name}-api:{port}",
                                "--runner-id", "$(NODE_NAME)",
                                "--tunnel"],
                            "env": [
                                {"name": "NODE_NAME", "valueFrom": {
                                    "fieldRef": {
                                        "fieldPath": "spec.nodeName"}}},
                                {"name": "HELIX_RUNNER_TOKEN",
                                 "valueFrom": {"secretKeyRef": {
                                     "name": f"{name}-auth",
                                     "key": "runner-token"}}},
                                {"name": "HSA_ENABLE_IPC_MODE_LEGACY",
                                 "value": "0"},
                            ],
                            "resources": {"limits": {
                                "amd.com/gpu":
                                    int(runner.get("gpusPerPod", 1))}},
                        }],
                    },
                },
            },
        })
    return objs


def _api_path(obj: dict, ns: str, name: str = "") -> str:
    kind = obj["kind"].lower() + "s"
    core = obj["apiVersion"] == "v1"
    base = f"/api/v1/namespaces/{ns}/{kind}" if core else \
        f"/apis/{obj['apiVersion']}/namespaces/{ns}/{kind}"
    return f"{base}/{name}" if name else base


class Reconciler:
    def __init__(self, client, namespace: str):
        self.client = client
        self.ns = namespace

    def list_crs(self) -> List[dict]:
        status, body = self.client.request(
            "GET", f"/apis/{GROUP}/{VERSION}/namespaces/{self.ns}/"
                   f"{PLURAL}")
        if status != 200:
            log.warning("list CRs: HTTP %s", status)
            return []
        return body.get("items", [])

    def apply(self, obj: dict):
        name = obj["metadata"]["name"]
        path = _api_path(obj, self.ns, name)
        status, current = self.client.request("GET", path)
        if status == 404:
            st, _ = self.client.request(
                "POST", _api_path(obj, self.ns), obj)
            log.info("create %s/%s: HTTP %s", obj["kind"], name, st)
        else:
            # preserve resourceVersion for replace semantics
            obj = dict(obj)
            obj["metadata"] = dict(obj["metadata"],
                                   resourceVersion=current.get(
                                       "metadata", {}).get(
                                       "resourceVersion", ""))
            st, _ = self.client.request("PUT", path, obj)
            log.info("update %s/%s: HTTP %s", obj["kind"], name, st)

    def delete_stale(self, cr_names: List[str]):
        """Remove children whose owning CR is gone (label-selected)."""
        for api, kind in (("/apis/apps/v1", "deployments"),
                          ("/apis/apps/v1", "daemonsets"),
                          ("/api/v1", "services"),
                          ("/api/v1", "secrets")):
            status, body = self.client.request(
                "GET", f"{api}/namespaces/{self.ns}/{kind}"
                       "?labelSelector=app.kubernetes.io/name%3D"
                       "helix-amd")
            if status != 200:
                continue
            for item in body.get("items", []):
                inst = item["metadata"].get("labels", {}).get(
                    "app.kubernetes.io/instance", "")
                if inst and inst not in cr_names:
                    name = item["metadata"]["name"]
                    self.client.request(
                        "DELETE",
                        f"{api}/namespaces/{self.ns}/{kind}/{name}")
                    log.info("gc %s/%s (CR %s gone)", kind, name, inst)

    def reconcile_once(self) -> int:
        crs = self.list_crs()
        for cr in crs:
            name = cr["metadata"]["name"]
            for obj in build_children(name, self.ns,
                                      cr.get("spec", {})):
                self.apply(obj)
        self.delete_stale([c["metadata"]["name"] for c in crs])
        return len(crs)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--namespace",
                    default=os.environ.get("POD_NAMESPACE", "default"))
    ap.add_argument("--interval", type=float, default=15.0)
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)
    rec = Reconciler(InClusterClient(), args.namespace)
    log.info("helix-amd operator watching %s/%s in %s",
             GROUP, PLURAL, args.namespace)
    while True:
        try:
            rec.reconcile_once()
        except Exception:
            log.exception("reconcile error")
        time.sleep(args.interval)


if __name__ == "__main__":
    main()
