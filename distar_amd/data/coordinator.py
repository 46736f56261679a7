"""Metadata broker for the Adapter transport.

Functional parity with the reference's
`ctools/worker/coordinator/coordinator.py:20-256`: register/request datum
metadata per token (FIFO), queue introspection, dead-producer eviction after
repeated failures; the reference's Flask app is a stdlib JSON HTTP server
here (utils/http.py).  `Worker` mirrors the per-token broker the reference
spawns to scale hot tokens.
"""
import threading
from collections import defaultdict, deque

from ..utils.http import JsonHttpServer, pick_unused_port


class Coordinator:
    def __init__(self, cfg=None, host='0.0.0.0', port=None, maxlen=4096):
        comm = (cfg or {}).get('communication', {}) if cfg is not None else {}
        self.port = port or comm.get('coordinator_port', None) or pick_unused_port()
        self._queues = defaultdict(lambda: deque(maxlen=maxlen))
        self._strikes = defaultdict(int)
        self._lock = threading.Lock()
        self._server = JsonHttpServer({
            '/register_datum': self._register_datum,
            '/request_datum': self._request_datum,
            '/queue_length': self._queue_length,
            '/remove_server': self._remove_server,
            '/start_worker': self._start_worker,
        }, host=host, port=self.port)
        self._workers = {}

    # --------------------------------------------------------------- routes
    def _register_datum(self, body):
        with self._lock:
            self._queues[body['token']].append(
                {'ip': body['ip'], 'port': body['port'], 'id': body.get('id', 0)})
        return {'ok': True}

    def _request_datum(self, body):
        # LIFO: hand out the NEWEST datum.  Producers evict their OLDEST
        # payloads when a bounded queue overflows, so oldest-first metadata
        # lets a lagging consumer chase only already-evicted ids forever
        # (head-of-line livelock); newest-first always resolves, and for RL
        # trajectories fresher data also means lower staleness.
        with self._lock:
            q = self._queues.get(body['token'])
            if q:
                return q.pop()
        return {'ip': None}

    def _queue_length(self, body):
        with self._lock:
            return {'length': len(self._queues.get(body['token'], ()))}

    def _remove_server(self, body):
        """Evict all pending metadata from a dead producer after 5 strikes
        (reference coordinator.py:114-128)."""
        key = (body['ip'], body.get('port'))
        with self._lock:
            self._strikes[key] += 1
            if self._strikes[key] >= body.get('threshold', 5):
                for token, q in self._queues.items():
                    self._queues[token] = deque(
                        (m for m in q if m['ip'] != body['ip']), maxlen=q.maxlen)
                self._strikes[key] = 0
                return {'removed': True}
        return {'removed': False}

    def _start_worker(self, body):
        """Spawn a per-token Worker broker to offload a hot token
        (reference coordinator.py:156-165); returns its address."""
        token = body.get('token', 'default')
        with self._lock:
            if token not in self._workers:
                worker = Worker(host='0.0.0.0')
                worker.run()
                self._workers[token] = worker
            return {'ip': '127.0.0.1', 'port': self._workers[token].port}

    # ------------------------------------------------------------------ api
    def run(self, daemon=True):
        self._server.start(daemon=daemon)
        return self

    def close(self):
        self._server.stop()


class Worker(Coordinator):
    """Per-token broker process the coordinator can delegate hot tokens to
    (reference coordinator.py:20-59) — identical API on its own port."""
