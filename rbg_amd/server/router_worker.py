"""Router role — HTTP front-end + prefill/decode dispatcher.

The in-repo realization of the reference's router (SMG) role
(reference examples/inference/pd-disagg-standalone.yaml:55-66): terminates
an OpenAI-style HTTP API, picks a prefill and a decode instance from the
discovery topology, drives the migration handshake and streams the decode
side's tokens back.  In colocated mode it simply load-balances generate
calls round-robin across worker instances.

No external web framework: stdlib ThreadingHTTPServer keeps the worker
process dependency-free and fork-fast.
"""
from __future__ import annotations

import itertools
import json
import logging
import os
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Any, Dict, List

from ..runtime.worker import WorkerContext, register_runner
from .rpc import RpcClient

log = logging.getLogger(__name__)


def simple_tokenize(text: str, vocab_size: int = 128000) -> List[int]:
    """Deterministic byte-pair-free stand-in tokenizer (no network for real
    tokenizer files; weights are random-init anyway)."""
    return [(b * 7 + i) % (vocab_size - 10) + 1
            for i, b in enumerate(text.encode())]


class _InstanceWatcher:
    """One poller per engine instance: batches every outstanding request
    into a single poll_many RPC per tick, so N concurrent requests cost one
    round trip instead of N."""

    def __init__(self, client: RpcClient, tick: float = 0.01):
        self.client = client
        self.tick = tick
        self._lock = threading.Lock()
        self._waiting: Dict[int, threading.Event] = {}
        self._results: Dict[int, Dict[str, Any]] = {}
        self._thread: threading.Thread = None
        self._stopped = False

    def stop(self) -> None:
        """Retire the watcher (its instance left the topology): fail any
        still-waiting requests so they re-dispatch, stop the poll loop,
        release the socket."""
        self._stopped = True
        self._fail_all("instance left the topology")
        try:
            self.client.close()
        except Exception:  # noqa: BLE001
            pass

    FAIL_WINDOW_S = 10.0    # continuous RPC failure -> declare seqs lost

    def wait_for(self, seq_id: int, timeout: float) -> Dict[str, Any]:
        ev = threading.Event()
        with self._lock:
            self._waiting[seq_id] = ev
            if self._thread is None or not self._thread.is_alive():
                self._thread = threading.Thread(target=self._loop, daemon=True)
                self._thread.start()
        if not ev.wait(timeout):
            with self._lock:
                self._waiting.pop(seq_id, None)
            raise TimeoutError(f"seq {seq_id} did not finish in {timeout}s")
        with self._lock:
            res = self._results.pop(seq_id)
        if res.get("lost"):
            # instance died / was recreated: its sequences are gone — the
            # router re-dispatches the whole request (InstanceLost)
            raise InstanceLost(res.get("reason", "instance lost"))
        return res

    def _fail_all(self, reason: str) -> None:
        with self._lock:
            for sid, ev in list(self._waiting.items()):
                self._results[sid] = {"lost": True, "reason": reason}
                ev.set()
            self._waiting.clear()

    def _loop(self):
        fail_since = None
        while not self._stopped:
            with self._lock:
                ids = list(self._waiting.keys())
            if not ids:
                time.sleep(self.tick)
                continue
            try:
                results = self.client.call("poll_many", seq_ids=ids)
                fail_since = None
            except Exception as e:  # noqa: BLE001
                msg = repr(e)
                if "unknown seq" in msg:
                    # the engine restarted (gang recreate): pre-restart
                    # sequences are gone — fail them NOW so the router
                    # retries elsewhere instead of waiting out the timeout
                    self._fail_all("engine restarted (unknown seq)")
                    continue
                # connection-level failure: the instance is down or
                # recreating; give it FAIL_WINDOW_S before declaring loss
                now = time.monotonic()
                if fail_since is None:
                    fail_since = now
                elif now - fail_since > self.FAIL_WINDOW_S:
                    self._fail_all(f"instance unreachable > "
                                   f"{self.FAIL_WINDOW_S}s: {msg}")
                    fail_since = None
                time.sleep(0.2)
                continue
            with self._lock:
                for sid, res in zip(ids, results):
                    if res.get("finished") and sid in self._waiting:
                        self._results[sid] = res
                        self._waiting.pop(sid).set()
            time.sleep(self.tick)


class InstanceLost(RuntimeError):
    """A dispatched instance died before finishing the sequence; the
    request is safe to re-dispatch (full re-prefill) elsewhere."""


class Router:
    def __init__(self, ctx: WorkerContext):
        self.ctx = ctx
        args = ctx.args
        self.mode = args.get("dispatch", "colocated")   # colocated | pd
        self.vocab_size = int(args.get("vocab_size", 128000))
        self.worker_roles = args.get("worker_roles", ["worker"])
        self.prefill_roles = args.get("prefill_roles", ["prefill"])
        self.decode_roles = args.get("decode_roles", ["decode"])
        self._clients: Dict[str, RpcClient] = {}
        self._watchers: Dict[str, _InstanceWatcher] = {}
        self._rr = itertools.count()
        self._lock = threading.Lock()
        # stale client/watcher GC: a recreated instance publishes new ports,
        # so (name, ports) keys accumulate across gang recreates — retire
        # entries absent from the topology for STALE_GRACE_S (grace covers
        # transient unready flaps during rolling updates)
        self._stale_since: Dict[Any, float] = {}
        self._last_gc = 0.0

    STALE_GRACE_S = 30.0

    def _gc_stale(self) -> None:
        now = time.monotonic()
        if now - self._last_gc < 5.0:
            return
        self._last_gc = now
        live = set()
        topo = self.ctx.load_topology().get("group", {})
        for role in topo.get("roles", []):
            for inst in role.get("instances", []):
                if inst.get("ports"):
                    live.add((inst["name"], tuple(inst["ports"])))
        with self._lock:
            for key in set(self._clients) | set(self._watchers):
                if key in live:
                    self._stale_since.pop(key, None)
                    continue
                t0 = self._stale_since.setdefault(key, now)
                if now - t0 < self.STALE_GRACE_S:
                    continue
                self._stale_since.pop(key, None)
                c = self._clients.pop(key, None)
                if c is not None:
                    try:
                        c.close()
                    except Exception:  # noqa: BLE001
                        pass
                w = self._watchers.pop(key, None)
                if w is not None:
                    w.stop()

    # -- topology ------------------------------------------------------------

    def _instances(self, roles: List[str]) -> List[Dict[str, Any]]:
        topo = self.ctx.load_topology().get("group", {})
        out = []
        for role in topo.get("roles", []):
            if role.get("name") in roles:
                for inst in role.get("instances", []):
                    if inst.get("ready") and inst.get("ports"):
                        out.append(inst)
        return out

    def _client(self, inst: Dict[str, Any]) -> RpcClient:
        # keyed by (name, ports): a recreated instance may publish a new
        # RPC port, and a stale cached socket must not shadow it
        key = (inst["name"], tuple(inst.get("ports", ())))
        with self._lock:
            c = self._clients.get(key)
            if c is None:
                c = RpcClient(inst.get("address", "127.0.0.1"),
                              int(inst["ports"][0]))
                self._clients[key] = c
            return c

    def _watcher(self, inst: Dict[str, Any]) -> _InstanceWatcher:
        key = (inst["name"], tuple(inst.get("ports", ())))
        with self._lock:
            w = self._watchers.get(key)
            if w is None:
                # dedicated polling connection (the control socket stays
                # free for submits)
                w = _InstanceWatcher(RpcClient(
                    inst.get("address", "127.0.0.1"), int(inst["ports"][0])))
                self._watchers[key] = w
            return w

    def _pick_role(self, roles: List[str], timeout: float = 15.0
                   ) -> Dict[str, Any]:
        """Round-robin over ready instances; brief retry covers the window
        between role readiness and the next discovery publish."""
        deadline = time.monotonic() + timeout
        while True:
            insts = self._instances(roles)
            if insts:
                return insts[next(self._rr) % len(insts)]
            if time.monotonic() >= deadline:
                raise RuntimeError(f"no ready instances for roles {roles}")
            time.sleep(0.1)

    # -- request paths -------------------------------------------------------

    def generate(self, tokens: List[int], max_new_tokens: int,
                 temperature: float = 0.0) -> Dict[str, Any]:
        """Linked-failover continuity: if the dispatched instance dies
        mid-request (InstanceLost from the watcher, or a submit-time
        connection error), the whole request re-dispatches — round-robin
        naturally lands on a surviving replica while the gang recreates
        the dead one."""
        t0 = time.monotonic()
        self._gc_stale()
        window = float(self.ctx.args.get("failover_window_s", 45.0))
        deadline = t0 + window
        last: Exception = None
        attempt = 0
        while True:
            try:
                if self.mode == "pd":
                    return self._generate_pd(tokens, max_new_tokens,
                                             temperature, t0)
                return self._generate_colocated(tokens, max_new_tokens,
                                                temperature, t0)
            except (InstanceLost, ConnectionError, OSError,
                    RuntimeError) as e:
                # retryable: instance death, connection refusal during a
                # gang recreate, or a transiently-empty ready set.  The
                # gang recreates in seconds; requests WAIT (bounded by
                # failover_window_s) instead of failing the client.
                retryable = isinstance(
                    e, (InstanceLost, ConnectionError, OSError)) or \
                    "no ready instances" in str(e) or \
                    "PeerDead" in str(e)
                if not retryable or time.monotonic() >= deadline:
                    raise
                last = e
                attempt += 1
                if attempt <= 3 or attempt % 20 == 0:
                    log.warning("request re-dispatch (attempt %d): %r",
                                attempt, e)
                time.sleep(0.25)

    def _generate_colocated(self, tokens, max_new_tokens, temperature,
                            t0) -> Dict[str, Any]:
        inst = self._pick_role(self.worker_roles)
        sid = self._client(inst).call("submit", tokens=tokens,
                                      max_new_tokens=max_new_tokens,
                                      temperature=temperature)
        res = self._watcher(inst).wait_for(sid, timeout=900.0)
        res["wall_s"] = time.monotonic() - t0
        res["instance"] = inst["name"]
        return res

    def _generate_pd(self, tokens: List[int], max_new_tokens: int,
                     temperature: float, t0: float) -> Dict[str, Any]:
        prefill = self._pick_role(self.prefill_roles)
        decode = self._pick_role(self.decode_roles)
        pres = self._client(prefill).call(
            "prefill", tokens=tokens, max_new_tokens=max_new_tokens,
            decode_instance=decode["name"], temperature=temperature)
        ttft = time.monotonic() - t0
        dclient = self._client(decode)
        seq_id = pres.get("decode_seq_id")
        if seq_id is None:
            # TP decode groups return a ticket that resolves once the
            # lockstep import lands
            deadline = time.monotonic() + 120
            while seq_id is None and time.monotonic() < deadline:
                seq_id = dclient.call("resolve_ticket",
                                      ticket=pres["ticket"])["seq_id"]
                if seq_id is None:
                    time.sleep(0.01)
            if seq_id is None:
                raise RuntimeError("KV import ticket never resolved")
        res = self._watcher(decode).wait_for(seq_id, timeout=900.0)
        res["ttft_s"] = ttft
        res["wall_s"] = time.monotonic() - t0
        res["prefill_instance"] = prefill["name"]
        res["decode_instance"] = decode["name"]
        return res

    def stats(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {"mode": self.mode, "instances": {}}
        for role_list in (self.worker_roles, self.prefill_roles,
                          self.decode_roles):
            for inst in self._instances(role_list):
                try:
                    out["instances"][inst["name"]] = \
                        self._client(inst).call("stats")
                except Exception as e:  # noqa: BLE001
                    out["instances"][inst["name"]] = {"error": repr(e)}
        return out


class _Handler(BaseHTTPRequestHandler):
    router: Router = None   # set by serve()

    def log_message(self, *a):  # quiet
        pass

    def _reply(self, code: int, obj: Any) -> None:
        data = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def do_GET(self):
        if self.path in ("/health", "/healthz"):
            self._reply(200, {"status": "ok"})
        elif self.path == "/metrics":
            self._reply(200, self.router.stats())
        else:
            self._reply(404, {"error": "not found"})

    def do_POST(self):
        try:
            n = int(self.headers.get("Content-Length", 0))
            body = json.loads(self.rfile.read(n) or b"{}")
            if self.path in ("/generate", "/v1/completions"):
                tokens = body.get("prompt_tokens")
                if tokens is None:
                    tokens = simple_tokenize(str(body.get("prompt", "")),
                                             self.router.vocab_size)
                res = self.router.generate(
                    tokens,
                    int(body.get("max_tokens", body.get("max_new_tokens", 64))),
                    float(body.get("temperature", 0.0)))
                if self.path == "/v1/completions":
                    res = {
                        "id": f"cmpl-{int(time.time()*1000)}",
                        "object": "text_completion",
                        "model": "rbg-mi355x",
                        "choices": [{"index": 0,
                                     "tokens": res["tokens"],
                                     "finish_reason": "length"}],
                        "usage": {"prompt_tokens": len(tokens),
                                  "completion_tokens": len(res["tokens"])},
                        "timing": {"ttft_s": res.get("ttft_s"),
                                   "wall_s": res.get("wall_s")},
                    }
                self._reply(200, res)
            else:
                self._reply(404, {"error": "not found"})
        except Exception as e:  # noqa: BLE001
            self._reply(500, {"error": repr(e)})


@register_runner("router")
def router_runner(ctx: WorkerContext) -> None:
    router = Router(ctx)
    port = 0
    raw = os.environ.get("PORT_HTTP", "")
    if raw:
        port = int(raw.split(",")[0])
    handler = type("BoundHandler", (_Handler,), {"router": router})
    httpd = ThreadingHTTPServer(("127.0.0.1", port), handler)
    port = httpd.server_address[1]
    thread = threading.Thread(target=httpd.serve_forever, daemon=True)
    thread.start()
    ctx.set_ready(http_port=port, mode=router.mode)
    try:
        while not ctx.should_stop():
            ctx.wait(0.2)
    finally:
        httpd.shutdown()
