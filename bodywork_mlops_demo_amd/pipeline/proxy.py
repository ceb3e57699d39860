"""Single-URL service front — the ClusterIP load balancer equivalent.

The reference's consumer sees ONE cluster-DNS URL fronting 2 replicas
(``stage_4_test_model_scoring_service.py:28``, ``bodywork.yaml:41-42``);
kubernetes' ClusterIP service round-robins connections across pods.  This
proxy restores that abstraction boundary for the in-process runner: the
declared ``service.port`` is the stable front, replicas bind behind it,
and clients never need the replica set.

Implementation: a threaded HTTP reverse proxy (one OS thread per in-flight
request, matching the replicas' serialize-within-replica model) with
round-robin backend selection and dead-backend failover — a request that
cannot reach its chosen replica is retried on the next one, which is what
keeps the load-test green while the watchdog respawns an injected failure.
Each proxy thread keeps one persistent connection per backend, so the
steady-state cost is one extra local TCP hop.  Clients that want the raw
shared-nothing GPU fan-out still get the replica URL set via
``BODYWORK_AMD_SERVICE_URLS`` and drive replicas directly.
"""
from __future__ import annotations

import http.client
import itertools
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)

# hop-by-hop headers a proxy must not forward (RFC 9110 §7.6.1)
_HOP_BY_HOP = {
    "connection", "keep-alive", "proxy-authenticate", "proxy-authorization",
    "te", "trailers", "transfer-encoding", "upgrade", "host",
}


class FrontProxy:
    """Round-robin HTTP front over a set of local replica ports."""

    def __init__(self, backend_ports: list[int], host: str = "127.0.0.1",
                 port: int = 0):
        self.host = host
        self.backend_ports = list(backend_ports)
        self._rr = itertools.count()
        self._local = threading.local()
        proxy = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, fmt, *args):  # quiet; we have our own log
                pass

            def _forward(self):
                n = len(proxy.backend_ports)
                body = None
                length = int(self.headers.get("Content-Length") or 0)
                if length:
                    body = self.rfile.read(length)
                start = next(proxy._rr)
                last_err: Exception | None = None
                for attempt in range(n):
                    bport = proxy.backend_ports[(start + attempt) % n]
                    try:
                        resp = proxy._backend_request(
                            bport, self.command, self.path,
                            body, self.headers)
                    except (ConnectionError, OSError,
                            http.client.HTTPException) as e:
                        last_err = e
                        proxy._drop_conn(bport)
                        continue
                    data = resp.read()
                    self.send_response(resp.status)
                    for k, v in resp.getheaders():
                        if k.lower() not in _HOP_BY_HOP:
                            self.send_header(k, v)
                    self.send_header("Content-Length", str(len(data)))
                    self.end_headers()
                    self.wfile.write(data)
                    return
                log.error(f"front proxy: all {n} replicas unreachable "
                          f"({last_err})")
                msg = b'{"error": "no healthy replica"}'
                self.send_response(502)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(msg)))
                self.end_headers()
                self.wfile.write(msg)

            do_GET = _forward
            do_POST = _forward

        self._server = ThreadingHTTPServer((host, port), Handler)
        self._server.daemon_threads = True
        self.port = self._server.server_address[1]
        self._thread: threading.Thread | None = None

    # one persistent connection per (proxy thread, backend port)
    def _backend_request(self, bport: int, method: str, path: str,
                         body, headers) -> http.client.HTTPResponse:
        conns = getattr(self._local, "conns", None)
        if conns is None:
            conns = self._local.conns = {}
        conn = conns.get(bport)
        fwd = {k: v for k, v in headers.items()
               if k.lower() not in _HOP_BY_HOP}
        for _ in range(2):  # one retry on a stale kept-alive connection
            if conn is None:
                conn = conns[bport] = http.client.HTTPConnection(
                    self.host, bport, timeout=300)
            try:
                conn.request(method, path, body=body, headers=fwd)
                return conn.getresponse()
            except (ConnectionError, http.client.HTTPException, OSError):
                conn.close()
                conn = conns[bport] = None
                if _ == 1:
                    raise
        raise ConnectionError("unreachable")  # pragma: no cover

    def _drop_conn(self, bport: int) -> None:
        conns = getattr(self._local, "conns", None)
        if conns and conns.get(bport) is not None:
            conns[bport].close()
            conns[bport] = None

    def start(self) -> "FrontProxy":
        self._thread = threading.Thread(
            target=self._server.serve_forever, daemon=True,
            name=f"front-proxy:{self.port}")
        self._thread.start()
        log.info(f"front proxy on :{self.port} -> replicas "
                 f"{self.backend_ports}")
        return self

    def stop(self) -> None:
        self._server.shutdown()
        self._server.server_close()
        if self._thread is not None:
            self._thread.join(timeout=10)
