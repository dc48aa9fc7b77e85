"""Autotune HTTP service + client.

Re-design of the reference's Flask control plane
(bagua/service/autotune_service.py:78-435) on the stdlib
ThreadingHTTPServer (no Flask in this image); same endpoint paths and
JSON bodies:

    POST /api/v1/register_tensors
    POST /api/v1/report_metrics
    POST /api/v1/ask_hyperparameters
    POST /api/v1/report_tensor_execution_order
    GET  /api/v1/health_check

The server runs in a daemon thread on rank 0 (the reference forked a
process; a thread suffices because the stdlib server holds no GIL-bound
work between requests).
"""

import json
import logging
import os
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Dict

from .. import env
from ..defines import BaguaHyperparameter, TensorDeclaration
from .autotune_task_manager import (
    AutotuneTaskManager,
    split_bucket_by_bucket_size,
)

logger = logging.getLogger(__name__)


class AutotuneServiceState:
    def __init__(self, world_size: int, autotune_level: int = 1,
                 max_samples: int = 60, sampling_confidence_time_s: float = 5.0,
                 warmup_time_s: float = 30.0,
                 default_bucket_size: int = 32 * 1024 * 1024,
                 nnodes: int = 1):
        self.world_size = world_size
        self.autotune_level = autotune_level
        self.max_samples = max_samples
        self.sampling_confidence_time_s = sampling_confidence_time_s
        self.warmup_time_s = warmup_time_s
        self.default_bucket_size = default_bucket_size
        self.nnodes = nnodes

        self.lock = threading.Lock()
        self.managers: Dict[str, AutotuneTaskManager] = {}
        # per-model tuning state
        self.current_hp: Dict[str, BaguaHyperparameter] = {}
        self.sample_count: Dict[str, int] = {}
        self.completed: Dict[str, bool] = {}
        self.last_proposal_time: Dict[str, float] = {}
        self.start_time = time.time()
        # metric reports: model -> {rank: (iter, speed)}
        self.metrics: Dict[str, Dict[int, tuple]] = {}
        # train_iter at which the current hp was handed out: a speed
        # sample only scores the current hp if it was MEASURED under it
        # (reference gated per-iteration with all ranks synced,
        # autotune_service.py:78-272; mixed-iteration samples would score
        # a hyperparameter with stale speeds)
        self.hp_installed_iter: Dict[str, int] = {}

    def manager(self, model_name: str) -> AutotuneTaskManager:
        if model_name not in self.managers:
            self.managers[model_name] = AutotuneTaskManager(
                model_name, env.is_output_autotune_log(),
                search_hierarchical=self.nnodes > 1)
        return self.managers[model_name]


class _Handler(BaseHTTPRequestHandler):
    state: AutotuneServiceState = None

    def log_message(self, fmt, *args):  # quiet
        logger.debug("autotune http: " + fmt, *args)

    def _json(self, code: int, payload: Dict):
        body = json.dumps(payload).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _body(self) -> Dict:
        length = int(self.headers.get("Content-Length", 0))
        if length == 0:
            return {}
        return json.loads(self.rfile.read(length))

    def do_GET(self):
        if self.path == "/api/v1/health_check":
            self._json(200, {"status": "ok"})
        else:
            self._json(404, {"error": "not found"})

    def do_POST(self):
        st = self.state
        try:
            req = self._body()
            if self.path == "/api/v1/register_tensors":
                model = req["model_name"]
                decls = [TensorDeclaration(**d) for d in req["tensor_list"]]
                with st.lock:
                    mgr = st.manager(model)
                    mgr.set_tensor_list(decls)
                    if model not in st.current_hp:
                        hp = BaguaHyperparameter(
                            bucket_size=st.default_bucket_size,
                            buckets=split_bucket_by_bucket_size(
                                decls, st.default_bucket_size))
                        st.current_hp[model] = hp
                        st.sample_count[model] = 0
                        st.completed[model] = st.autotune_level < 1
                self._json(200, {
                    "recommended_hyperparameters":
                        st.current_hp[model].dict()})

            elif self.path == "/api/v1/report_metrics":
                model = req["model_name"]
                rank = int(req["rank"])
                train_iter = int(req["train_iter"])
                speed = float(req["speed"])
                with st.lock:
                    st.metrics.setdefault(model, {})[rank] = (train_iter,
                                                              speed)
                self._json(200, {"status": "ok"})

            elif self.path == "/api/v1/ask_hyperparameters":
                model = req["model_name"]
                train_iter = int(req["train_iter"])
                with st.lock:
                    rsp = self._ask(st, model, train_iter)
                self._json(200, rsp)

            elif self.path == "/api/v1/report_tensor_execution_order":
                with st.lock:
                    for span in req.get("spans", []):
                        model = span.get("model_name", "")
                        for mgr_name, mgr in st.managers.items():
                            if not model or mgr_name == model:
                                mgr.report_span(span["tensor_name"],
                                                int(span["start_time"]))
                self._json(200, {"status": "ok"})
            else:
                self._json(404, {"error": "not found"})
        except Exception as e:  # noqa: BLE001 — report to client
            logger.exception("autotune handler error")
            self._json(500, {"error": str(e)})

    @staticmethod
    def _ask(st: AutotuneServiceState, model: str, train_iter: int) -> Dict:
        mgr = st.manager(model)
        hp = st.current_hp.get(model)
        if hp is None:
            hp = BaguaHyperparameter(bucket_size=st.default_bucket_size)
            st.current_hp[model] = hp
        if st.completed.get(model, False):
            best = mgr.best_hyperparameters() or hp
            return {"recommended_hyperparameters": best.dict(),
                    "is_autotune_completed": True}
        # warmup gate (reference: autotune_service.py:227-248)
        if time.time() - st.start_time < st.warmup_time_s:
            return {"recommended_hyperparameters": hp.dict(),
                    "is_autotune_completed": False}
        # wait until every rank reported a speed sample MEASURED UNDER the
        # current hp (report train_iter newer than the hp's install
        # iteration) — the all-ranks-synced-per-iteration gate
        reports = st.metrics.get(model, {})
        installed = st.hp_installed_iter.get(model, 0)
        fresh = {r: v for r, v in reports.items() if v[0] > installed}
        if len(fresh) < st.world_size:
            return {"recommended_hyperparameters": hp.dict(),
                    "is_autotune_completed": False}
        # confidence gate: at most one proposal per confidence window
        now = time.time()
        last = st.last_proposal_time.get(model, 0.0)
        if now - last < st.sampling_confidence_time_s:
            return {"recommended_hyperparameters": hp.dict(),
                    "is_autotune_completed": False}

        score = sum(v[1] for v in fresh.values()) / len(fresh)
        mgr.record(train_iter, hp, score)
        st.sample_count[model] = st.sample_count.get(model, 0) + 1
        st.metrics[model] = {}
        st.last_proposal_time[model] = now
        st.hp_installed_iter[model] = train_iter

        if st.sample_count[model] >= st.max_samples:
            st.completed[model] = True
            best = mgr.best_hyperparameters() or hp
            st.current_hp[model] = best
            logger.info("autotune completed for %s: bucket=%d hier=%s",
                        model, best.bucket_size,
                        best.is_hierarchical_reduce)
            return {"recommended_hyperparameters": best.dict(),
                    "is_autotune_completed": True}

        new_hp = mgr.tell_and_ask(hp, score)
        st.current_hp[model] = new_hp
        return {"recommended_hyperparameters": new_hp.dict(),
                "is_autotune_completed": False}


class AutotuneServer:
    def __init__(self, port: int, world_size: int):
        state = AutotuneServiceState(
            world_size,
            autotune_level=env.get_autotune_level(),
            max_samples=env.get_autotune_max_samples(),
            sampling_confidence_time_s=(
                env.get_autotune_sampling_confidence_time_s()),
            warmup_time_s=env.get_autotune_warmup_time_s(),
            default_bucket_size=env.get_default_bucket_size(),
            # LOCAL_WORLD_SIZE absent => launched outside torchrun;
            # assume a single node (searching hierarchical there is noise)
            nnodes=(max(1, world_size // max(1, env.get_local_size()))
                    if "LOCAL_WORLD_SIZE" in os.environ else 1),
        )
        handler = type("BoundHandler", (_Handler,), {"state": state})
        self.state = state
        self.httpd = ThreadingHTTPServer(("127.0.0.1", port), handler)
        self.port = self.httpd.server_address[1]
        self.thread = threading.Thread(target=self.httpd.serve_forever,
                                       daemon=True)
        self.thread.start()

    def shutdown(self):
        self.httpd.shutdown()
        self.thread.join(timeout=10)


def start_autotune_server(port: int, world_size: int) -> AutotuneServer:
    return AutotuneServer(port, world_size)


class AutotuneClient:
    """Retrying HTTP client (reference: autotune_service.py:306-435)."""

    def __init__(self, host: str, port: int, proxies=None):
        import requests

        self.base = "http://%s:%d" % (host, port)
        self.session = requests.Session()
        self.session.trust_env = False
        self.timeout = 30

    def _post(self, path: str, payload: Dict, retries: int = 3) -> Dict:
        last = None
        for _ in range(retries):
            try:
                r = self.session.post(self.base + path, json=payload,
                                      timeout=self.timeout)
                if r.status_code == 200:
                    return r.json()
                last = RuntimeError("HTTP %d: %s" % (r.status_code, r.text))
            except Exception as e:  # noqa: BLE001
                last = e
                time.sleep(0.5)
        raise RuntimeError("autotune request %s failed: %s" % (path, last))

    def health_check(self) -> bool:
        try:
            r = self.session.get(self.base + "/api/v1/health_check",
                                 timeout=5)
            return r.status_code == 200
        except Exception:  # noqa: BLE001
            return False

    def register_tensors(self, model_name: str, tensor_list) -> Dict:
        return self._post("/api/v1/register_tensors", {
            "model_name": model_name, "tensor_list": tensor_list})

    def report_metrics(self, model_name: str, rank: int, train_iter: int,
                       hyperparameters: Dict, speed: float) -> Dict:
        return self._post("/api/v1/report_metrics", {
            "model_name": model_name, "rank": rank,
            "train_iter": train_iter, "hyperparameters": hyperparameters,
            "speed": speed})

    def ask_hyperparameters(self, model_name: str, rank: int,
                            train_iter: int) -> Dict:
        return self._post("/api/v1/ask_hyperparameters", {
            "model_name": model_name, "rank": rank,
            "train_iter": train_iter})

    def report_tensor_execution_order(self, spans) -> Dict:
        return self._post("/api/v1/report_tensor_execution_order",
                          {"spans": spans})
