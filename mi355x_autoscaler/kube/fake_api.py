"""In-process fake Kubernetes API server (fastapi).

Benchmark config #1 ("in-process fake k8s API on CPU") and the test
harness both use this: it serves the list/patch/delete surface of
kube/client.py over localhost, backed by in-memory pod/node stores, and
supports fault injection (delayed / failed responses) for the failure-
detection tests.

The reference's tests mocked pykube instead [R: test/test_cluster.py —
reconstructed]; a live fake server exercises the real HTTP client path.
"""

from __future__ import annotations

import copy
import threading
import time
from typing import Dict, List, Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse


class FakeKubeState:
    """Thread-safe in-memory object store with a watch event log."""

    def __init__(self):
        self.lock = threading.RLock()
        # watchers block on this instead of sleep-polling: every emit
        # wakes the streams, so watch latency is scheduler-bound (the
        # reaction-time benchmark measures this path)
        self.cond = threading.Condition(self.lock)
        self.pods: Dict[str, Dict] = {}   # "<ns>/<name>" -> obj
        self.nodes: Dict[str, Dict] = {}
        self.pdbs: Dict[str, Dict] = {}
        self.fault_delay_s: float = 0.0
        self.fail_next: int = 0           # fail the next N requests with 500
        self.drop_watch_after: int = -1   # fault injection: cut streams
        self.deleted_pods: List[str] = []
        self.deleted_nodes: List[str] = []
        self.patched_nodes: List[str] = []
        self.resource_version: int = 0
        self.pod_events: List[Dict] = []  # [{rv, type, object}]
        self.node_events: List[Dict] = []

    def load(self, pods: List[Dict], nodes: List[Dict], pdbs: Optional[List[Dict]] = None):
        with self.lock:
            self.pods = {
                f"{p['metadata'].get('namespace', 'default')}/{p['metadata']['name']}": copy.deepcopy(p)
                for p in pods
            }
            self.nodes = {n["metadata"]["name"]: copy.deepcopy(n) for n in nodes}
            self.pdbs = {d["metadata"]["name"]: copy.deepcopy(d) for d in (pdbs or [])}
            self.pod_events = []
            self.node_events = []
            self.resource_version += 1
            self.cond.notify_all()

    # -- watch-visible mutations ---------------------------------------
    def _emit(self, etype: str, obj: Dict):
        self.resource_version += 1
        obj = copy.deepcopy(obj)
        obj.setdefault("metadata", {})["resourceVersion"] = str(self.resource_version)
        self.pod_events.append(
            {"rv": self.resource_version, "type": etype, "object": obj}
        )
        self.cond.notify_all()

    def _emit_node(self, etype: str, obj: Dict):
        self.resource_version += 1
        obj = copy.deepcopy(obj)
        obj.setdefault("metadata", {})["resourceVersion"] = str(self.resource_version)
        self.node_events.append(
            {"rv": self.resource_version, "type": etype, "object": obj}
        )
        self.cond.notify_all()

    def add_node(self, node: Dict):
        with self.lock:
            self.nodes[node["metadata"]["name"]] = copy.deepcopy(node)
            self._emit_node("ADDED", node)

    def add_pod(self, pod: Dict):
        with self.lock:
            key = f"{pod['metadata'].get('namespace', 'default')}/{pod['metadata']['name']}"
            self.pods[key] = copy.deepcopy(pod)
            self._emit("ADDED", pod)

    def modify_pod(self, pod: Dict):
        with self.lock:
            key = f"{pod['metadata'].get('namespace', 'default')}/{pod['metadata']['name']}"
            self.pods[key] = copy.deepcopy(pod)
            self._emit("MODIFIED", pod)

    def remove_pod(self, namespace: str, name: str):
        with self.lock:
            key = f"{namespace}/{name}"
            pod = self.pods.pop(key, None)
            if pod is not None:
                self._emit("DELETED", pod)


def _event_stream(state: FakeKubeState, log: str, start_rv: int):
    """JSONL watch stream over an event log; wakes on state.cond
    (no poll-sleep on the reaction path), honors drop_watch_after."""
    import json as _json

    sent = 0
    last_rv = start_rv
    while True:
        batch = []
        with state.lock:
            for e in getattr(state, log):
                if e["rv"] > last_rv:
                    batch.append(e)
            if not batch:
                state.cond.wait(timeout=0.5)
        for e in batch:
            if state.drop_watch_after >= 0 and sent >= state.drop_watch_after:
                return  # injected disconnect
            last_rv = e["rv"]
            sent += 1
            yield _json.dumps({"type": e["type"], "object": e["object"]}) + "\n"


def build_app(state: FakeKubeState) -> FastAPI:
    app = FastAPI()

    @app.middleware("http")
    async def faults(request: Request, call_next):
        if state.fault_delay_s > 0:
            time.sleep(state.fault_delay_s)
        with state.lock:
            if state.fail_next > 0:
                state.fail_next -= 1
                return JSONResponse({"message": "injected fault"}, status_code=500)
        return await call_next(request)

    @app.get("/api/v1/pods")
    def list_pods(watch: bool = False, resourceVersion: str = "0"):
        if watch:
            from starlette.responses import StreamingResponse

            start_rv = int(resourceVersion or 0)

            return StreamingResponse(_event_stream(state, "pod_events", start_rv),
                                     media_type="application/json")
        with state.lock:
            return {
                "kind": "PodList",
                "items": list(state.pods.values()),
                "metadata": {"resourceVersion": str(state.resource_version)},
            }

    @app.get("/api/v1/nodes")
    def list_nodes(watch: bool = False, resourceVersion: str = "0"):
        if watch:
            from starlette.responses import StreamingResponse

            return StreamingResponse(
                _event_stream(state, "node_events", int(resourceVersion or 0)),
                media_type="application/json")
        with state.lock:
            return {"kind": "NodeList", "items": list(state.nodes.values()),
                    "metadata": {"resourceVersion": str(state.resource_version)}}

    @app.get("/apis/policy/v1/poddisruptionbudgets")
    def list_pdbs():
        with state.lock:
            return {"kind": "PodDisruptionBudgetList", "items": list(state.pdbs.values()), "metadata": {}}

    @app.patch("/api/v1/nodes/{name}")
    async def patch_node(name: str, request: Request):
        body = await request.json()
        with state.lock:
            node = state.nodes.get(name)
            if node is None:
                return JSONResponse({"message": "not found"}, status_code=404)
            unsched = (body.get("spec") or {}).get("unschedulable")
            if unsched is not None:
                node.setdefault("spec", {})["unschedulable"] = bool(unsched)
            state.patched_nodes.append(name)
            state._emit_node("MODIFIED", node)
            return node

    @app.delete("/api/v1/namespaces/{ns}/pods/{name}")
    def delete_pod(ns: str, name: str):
        with state.lock:
            key = f"{ns}/{name}"
            pod = state.pods.pop(key, None)
            if pod is not None:
                state._emit("DELETED", pod)
            state.deleted_pods.append(key)
            return {"status": "Success"}

    @app.delete("/api/v1/nodes/{name}")
    def delete_node(name: str):
        with state.lock:
            node = state.nodes.pop(name, None)
            if node is not None:
                state._emit_node("DELETED", node)
            state.deleted_nodes.append(name)
            return {"status": "Success"}

    return app


class FakeKubeServer:
    """Run the fake API in a background uvicorn thread on a free port."""

    def __init__(self, state: Optional[FakeKubeState] = None, host: str = "127.0.0.1"):
        import socket

        import uvicorn

        self.state = state or FakeKubeState()
        self.app = build_app(self.state)
        sock = socket.socket()
        sock.bind((host, 0))
        self.host, self.port = sock.getsockname()
        sock.close()
        config = uvicorn.Config(
            self.app, host=host, port=self.port, log_level="error", lifespan="off"
        )
        self.server = uvicorn.Server(config)
        self.thread = threading.Thread(target=self.server.run, daemon=True)

    @property
    def base_url(self) -> str:
        return f"http://{self.host}:{self.port}"

    def start(self, timeout: float = 10.0) -> "FakeKubeServer":
        self.thread.start()
        deadline = time.time() + timeout
        while not self.server.started:
            if time.time() > deadline:
                raise RuntimeError("fake kube api failed to start")
            time.sleep(0.01)
        return self

    def stop(self):
        self.server.should_exit = True
        self.thread.join(timeout=5)

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
