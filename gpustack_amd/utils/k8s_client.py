"""Minimal Kubernetes API client (first-party, httpx — no kubernetes lib).

The reference delegates GPU-instance pods to the gpustack-operator Go
binary (gpustack/gpu_instances/); here the operator's
pod lifecycle is driven in-process by the server's GPUInstanceController
through this client. Covers exactly what that controller needs: create /
get / delete Pods and Services in a namespace, with in-cluster
service-account auth or explicit endpoint+token configuration.
"""
from __future__ import annotations

import os

import httpx

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class KubeError(RuntimeError):
    def __init__(self, status: int, message: str):
        super().__init__(f"kubernetes API {status}: {message}")
        self.status = status


class KubeClient:
    """`api_server`/`token` explicit, or in-cluster defaults
    (KUBERNETES_SERVICE_HOST + mounted service-account token)."""

    def __init__(self, api_server: str | None = None, token: str | None = None,
                 namespace: str | None = None, verify: bool | str = True,
                 transport: httpx.BaseTransport | None = None):
        if api_server is None:
            host = os.environ.get("KUBERNETES_SERVICE_HOST")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            if not host:
                raise KubeError(0, "no api_server and not running in-cluster")
            api_server = f"https://{host}:{port}"
            ca = os.path.join(SA_DIR, "ca.crt")
            if verify is True and os.path.exists(ca):
                verify = ca
        if token is None:
            tok_path = os.path.join(SA_DIR, "token")
            if os.path.exists(tok_path):
                with open(tok_path) as f:
                    token = f.read().strip()
        self.namespace = namespace or self._default_namespace()
        headers = {"Accept": "application/json"}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._c = httpx.Client(base_url=api_server, headers=headers,
                               verify=verify if transport is None else True,
                               transport=transport, timeout=15.0)

    @staticmethod
    def _default_namespace() -> str:
        ns_path = os.path.join(SA_DIR, "namespace")
        if os.path.exists(ns_path):
            with open(ns_path) as f:
                return f.read().strip()
        return "default"

    def close(self) -> None:
        self._c.close()

    # -- raw ----------------------------------------------------------------
    def _req(self, method: str, path: str, body: dict | None = None,
             ok_missing: bool = False) -> dict | None:
        r = self._c.request(method, path, json=body)
        if r.status_code == 404 and ok_missing:
            return None
        if r.status_code >= 300:
            raise KubeError(r.status_code, r.text[:300])
        return r.json() if r.content else {}

    # -- pods / services -----------------------------------------------------
    def create_pod(self, manifest: dict) -> dict:
        ns = manifest.get("metadata", {}).get("namespace", self.namespace)
        return self._req("POST", f"/api/v1/namespaces/{ns}/pods", manifest)

    def get_pod(self, name: str, namespace: str | None = None) -> dict | None:
        ns = namespace or self.namespace
        return self._req("GET", f"/api/v1/namespaces/{ns}/pods/{name}",
                         ok_missing=True)

    def delete_pod(self, name: str, namespace: str | None = None) -> None:
        ns = namespace or self.namespace
        self._req("DELETE", f"/api/v1/namespaces/{ns}/pods/{name}",
                  ok_missing=True)

    def create_service(self, manifest: dict) -> dict:
        ns = manifest.get("metadata", {}).get("namespace", self.namespace)
        return self._req("POST", f"/api/v1/namespaces/{ns}/services", manifest)

    def get_service(self, name: str, namespace: str | None = None) -> dict | None:
        ns = namespace or self.namespace
        return self._req("GET", f"/api/v1/namespaces/{ns}/services/{name}",
                         ok_missing=True)

    def delete_service(self, name: str, namespace: str | None = None) -> None:
        ns = namespace or self.namespace
        self._req("DELETE", f"/api/v1/namespaces/{ns}/services/{name}",
                  ok_missing=True)
