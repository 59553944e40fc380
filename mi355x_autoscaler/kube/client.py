"""Minimal Kubernetes REST client (replaces pykube — unavailable offline).

Implements exactly the API surface the reference used through pykube
[R: autoscaler/kube.py, autoscaler/cluster.py — reconstructed]:

  * list all pods / nodes / PDBs (full relist per cycle)
  * PATCH a node's ``spec.unschedulable`` (cordon / uncordon)
  * DELETE a pod (drain) and a node object
  * watch streams (for the delta-ingest path; optional)

Auth: bearer-token service account (in-cluster) or a kubeconfig with
token/client-cert entries; the fake in-process API server
(kube/fake_api.py) serves the same endpoints over localhost.
"""

from __future__ import annotations

import json
import logging
import os
from typing import Dict, Iterator, List, Optional

import requests

logger = logging.getLogger(__name__)

_SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class KubeApiError(RuntimeError):
    def __init__(self, status: int, message: str):
        super().__init__(f"kube api {status}: {message}")
        self.status = status


class KubeApiClient:
    def __init__(
        self,
        base_url: str,
        token: Optional[str] = None,
        verify=True,
        session: Optional[requests.Session] = None,
        timeout: float = 30.0,
    ):
        self.base_url = base_url.rstrip("/")
        self.timeout = timeout
        self.session = session or requests.Session()
        self.session.verify = verify
        if token:
            self.session.headers["Authorization"] = f"Bearer {token}"

    # -- constructors ----------------------------------------------------
    @classmethod
    def in_cluster(cls) -> "KubeApiClient":
        host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        with open(os.path.join(_SA_DIR, "token")) as f:
            token = f.read().strip()
        ca = os.path.join(_SA_DIR, "ca.crt")
        return cls(f"https://{host}:{port}", token=token, verify=ca if os.path.exists(ca) else True)

    @classmethod
    def from_kubeconfig(cls, path: str) -> "KubeApiClient":
        import yaml

        with open(path) as f:
            cfg = yaml.safe_load(f)
        ctx_name = cfg.get("current-context")
        ctx = next(c["context"] for c in cfg["contexts"] if c["name"] == ctx_name)
        cluster = next(c["cluster"] for c in cfg["clusters"] if c["name"] == ctx["cluster"])
        user = next(u["user"] for u in cfg["users"] if u["name"] == ctx["user"])
        token = user.get("token")
        verify = cluster.get("certificate-authority", not cluster.get("insecure-skip-tls-verify", False))
        return cls(cluster["server"], token=token, verify=verify)

    # -- plumbing --------------------------------------------------------
    def _req(self, method: str, path: str, **kw):
        url = f"{self.base_url}{path}"
        resp = self.session.request(method, url, timeout=self.timeout, **kw)
        if resp.status_code >= 400:
            raise KubeApiError(resp.status_code, resp.text[:500])
        return resp

    def _list_all(self, path: str) -> List[Dict]:
        """Follow continue-tokens so large lists (1M-pod traces) page."""
        items: List[Dict] = []
        cont = None
        while True:
            params = {"limit": 500000}
            if cont:
                params["continue"] = cont
            data = self._req("GET", path, params=params).json()
            items.extend(data.get("items", []))
            cont = (data.get("metadata") or {}).get("continue")
            if not cont:
                return items

    # -- reads -----------------------------------------------------------
    def list_pods(self) -> List[Dict]:
        return self._list_all("/api/v1/pods")

    def list_pods_raw(self) -> bytes:
        """Raw PodList body for the native snapshot codec (single page;
        the codec parses it without building Python dicts)."""
        return self._req("GET", "/api/v1/pods").content

    def list_nodes(self) -> List[Dict]:
        return self._list_all("/api/v1/nodes")

    def list_pdbs(self) -> List[Dict]:
        try:
            return self._list_all("/apis/policy/v1/poddisruptionbudgets")
        except KubeApiError as e:
            if e.status == 404:
                return []
            raise

    def watch(self, path: str, resource_version: Optional[str] = None) -> Iterator[Dict]:
        """Yield watch events (``{"type": ..., "object": ...}``)."""
        for line in self.watch_raw(path, resource_version):
            yield json.loads(line)

    def watch_raw(self, path: str, resource_version: Optional[str] = None) -> Iterator[bytes]:
        """Raw JSONL watch lines — fed to the C++ batch event decoder
        without building Python dicts."""
        params = {"watch": "true"}
        if resource_version:
            params["resourceVersion"] = resource_version
        resp = self.session.get(
            f"{self.base_url}{path}", params=params, stream=True, timeout=None
        )
        # 1 MiB read chunks: a 10k-pod burst (~5 MB of JSONL) is a few
        # socket reads, not thousands of 512 B ones
        for line in resp.iter_lines(chunk_size=1 << 20):
            if line:
                yield line

    # -- mutations (cordon / drain / delete) ------------------------------
    def set_unschedulable(self, node_name: str, value: bool) -> None:
        """[R: autoscaler/kube.py · KubeNode.cordon — reconstructed]"""
        self._req(
            "PATCH",
            f"/api/v1/nodes/{node_name}",
            data=json.dumps({"spec": {"unschedulable": value}}),
            headers={"Content-Type": "application/strategic-merge-patch+json"},
        )

    def delete_pod(self, namespace: str, name: str, grace_seconds: int = 30) -> None:
        self._req(
            "DELETE",
            f"/api/v1/namespaces/{namespace}/pods/{name}",
            params={"gracePeriodSeconds": grace_seconds},
        )

    def delete_node(self, name: str) -> None:
        self._req("DELETE", f"/api/v1/nodes/{name}")
