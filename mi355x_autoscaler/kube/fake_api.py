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
        self.pdb_events: List[Dict] = []

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
            self.pdb_events = []
            self.resource_version += 1
            self.cond.notify_all()

    # -- watch-visible mutations ---------------------------------------
    def _emit(self, etype: str, obj: Dict):
        import json as _json

        self.resource_version += 1
        obj = copy.deepcopy(obj)
        obj.setdefault("metadata", {})["resourceVersion"] = str(self.resource_version)
        self.pod_events.append(
            {"rv": self.resource_version, "type": etype, "object": obj,
             "line": _json.dumps({"type": etype, "object": obj}) + "\n"}
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

    def _emit_pdb(self, etype: str, obj: Dict):
        self.resource_version += 1
        obj = copy.deepcopy(obj)
        obj.setdefault("metadata", {})["resourceVersion"] = str(self.resource_version)
        self.pdb_events.append(
            {"rv": self.resource_version, "type": etype, "object": obj}
        )
        self.cond.notify_all()

    def add_pdb(self, pdb: Dict):
        with self.lock:
            self.pdbs[pdb["metadata"]["name"]] = copy.deepcopy(pdb)
            self._emit_pdb("ADDED", pdb)

    def modify_pdb(self, pdb: Dict):
        with self.lock:
            self.pdbs[pdb["metadata"]["name"]] = copy.deepcopy(pdb)
            self._emit_pdb("MODIFIED", pdb)

    def remove_pdb(self, name: str):
        with self.lock:
            pdb = self.pdbs.pop(name, None)
            if pdb is not None:
                self._emit_pdb("DELETED", pdb)

    def add_node(self, node: Dict):
        with self.lock:
            self.nodes[node["metadata"]["name"]] = copy.deepcopy(node)
            self._emit_node("ADDED", node)

    def bulk_add_pods(self, pods: List[Dict], copy_objs: bool = True,
                      lines: Optional[List[str]] = None):
        """Burst injection under ONE lock/notify (the reaction-time
        benchmark's POST); copy_objs=False trusts the caller not to
        mutate the objects afterwards. `lines` are optional
        pre-serialized watch lines ('{"type":"ADDED","object":...}\n')
        — the real apiserver receives serialized bodies on the wire,
        so a POSTing client can hand them over instead of paying a
        Python re-serialization here (those lines then lack the bumped
        resourceVersion, which watchers don't consume)."""
        import copy as _copy
        import json as _json

        with self.lock:
            for i, pod in enumerate(pods):
                obj = _copy.deepcopy(pod) if copy_objs else pod
                key = (f"{obj['metadata'].get('namespace', 'default')}/"
                       f"{obj['metadata']['name']}")
                self.pods[key] = obj
                self.resource_version += 1
                ev = {"rv": self.resource_version, "type": "ADDED",
                      "object": obj}
                if lines is not None:
                    ev["line"] = lines[i]
                else:
                    obj.setdefault("metadata", {})["resourceVersion"] = str(
                        self.resource_version)
                    ev["line"] = _json.dumps(
                        {"type": "ADDED", "object": obj}) + "\n"
                self.pod_events.append(ev)
            self.cond.notify_all()

    def bulk_remove_pods(self, keys: List[str]):
        """Delete many pods under one lock ('<ns>/<name>' keys)."""
        with self.lock:
            import json as _json

            for key in keys:
                pod = self.pods.pop(key, None)
                if pod is not None:
                    self.resource_version += 1
                    self.pod_events.append(
                        {"rv": self.resource_version, "type": "DELETED",
                         "object": pod,
                         "line": _json.dumps({"type": "DELETED",
                                              "object": pod}) + "\n"})
            self.cond.notify_all()

    def remove_node(self, name: str):
        with self.lock:
            node = self.nodes.pop(name, None)
            if node is not None:
                self._emit_node("DELETED", node)

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
    """JSONL watch stream over an event log; wakes on state.cond (no
    poll-sleep on the reaction path), scans the log positionally
    (O(new events), not O(log)), serializes outside the lock, and
    yields one CHUNK per batch so a 10k-pod burst is a handful of
    writes, not 10k. Honors drop_watch_after fault injection."""
    import json as _json

    sent = 0
    last_rv = start_rv
    pos = 0
    while True:
        with state.lock:
            log_list = getattr(state, log)
            if pos > len(log_list):
                pos = 0  # log was reset by load()
            batch = [e for e in log_list[pos:] if e["rv"] > last_rv]
            pos = len(log_list)
            if not batch:
                state.cond.wait(timeout=0.5)
                continue
        if state.drop_watch_after >= 0:
            room = state.drop_watch_after - sent
            if room <= 0:
                return  # injected disconnect
            batch = batch[: room]
        last_rv = batch[-1]["rv"]
        sent += len(batch)
        yield "".join(
            e.get("line") or
            _json.dumps({"type": e["type"], "object": e["object"]}) + "\n"
            for e in batch)
        if state.drop_watch_after >= 0 and sent >= state.drop_watch_after:
            return


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
    def list_pdbs(watch: bool = False, resourceVersion: str = "0"):
        if watch:
            from starlette.responses import StreamingResponse

            return StreamingResponse(
                _event_stream(state, "pdb_events", int(resourceVersion or 0)),
                media_type="application/json")
        with state.lock:
            return {"kind": "PodDisruptionBudgetList",
                    "items": list(state.pdbs.values()),
                    "metadata": {"resourceVersion": str(state.resource_version)}}

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
