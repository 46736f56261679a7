"""Minimal threaded JSON-over-HTTP server + retrying client.

The reference uses Flask for its control-plane APIs (coordinator, league);
this image has no Flask, so the control plane runs on a stdlib
ThreadingHTTPServer with a route table — same wire format (JSON bodies,
POST routes), zero extra dependencies, trivially portable.
"""
import json
import socket
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import requests


def pick_unused_port():
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(('127.0.0.1', 0))
    port = s.getsockname()[1]
    s.close()
    return port


def get_ip():
    try:
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.connect(('8.8.8.8', 80))
        ip = s.getsockname()[0]
        s.close()
        return ip
    except OSError:
        return '127.0.0.1'


class JsonHttpServer:
    """routes: {'/path': callable(body_dict) -> response_dict}."""

    def __init__(self, routes, host='0.0.0.0', port=None):
        self.routes = dict(routes)
        self.host = host
        self.port = port or pick_unused_port()
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, fmt, *args):
                pass

            def _respond(self, code, obj):
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header('Content-Type', 'application/json')
                self.send_header('Content-Length', str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_POST(self):
                length = int(self.headers.get('Content-Length', 0))
                try:
                    body = json.loads(self.rfile.read(length) or b'{}')
                except json.JSONDecodeError:
                    self._respond(400, {'error': 'bad json'})
                    return
                fn = outer.routes.get(self.path)
                if fn is None:
                    self._respond(404, {'error': f'no route {self.path}'})
                    return
                try:
                    self._respond(200, fn(body))
                except Exception as e:  # noqa: BLE001 - control plane must survive
                    self._respond(500, {'error': repr(e)})

            do_GET = do_POST

        self._server = ThreadingHTTPServer((host, self.port), Handler)
        self._thread = None

    def start(self, daemon=True):
        self._thread = threading.Thread(target=self._server.serve_forever, daemon=daemon)
        self._thread.start()
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()


def post_json(url, body=None, retries=3, timeout=10, backoff=0.5):
    last = None
    for i in range(retries):
        try:
            r = requests.post(url, json=body or {}, timeout=timeout)
            if r.status_code == 200:
                return r.json()
            last = RuntimeError(f'{url} -> {r.status_code}: {r.text[:200]}')
        except requests.RequestException as e:
            last = e
        time.sleep(backoff * (i + 1))
    raise last
