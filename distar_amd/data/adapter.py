"""Brokered point-to-point trajectory/model transport.

Functional parity with the reference's
`ctools/worker/coordinator/adapter.py:27-412` (push/pull/length/full over a
coordinator-brokered direct TCP path, per-token bounded buffers, retrying
HTTP, per-datum failure tolerance) with one structural improvement: instead
of the reference's one-listening-socket-per-datum scheme, each producer runs
ONE persistent data server; payloads are addressed by id (registered with
the coordinator as {ip, port, id}).  No socket churn, no
close-a-listening-socket races, and a consumer finding an evicted payload
gets an explicit GONE reply instead of a connection error.

MI355X note: payloads land in host memory on the learner side and are fed to
pinned staging buffers (rl_dataloader) — the transport itself is
hardware-agnostic TCP, exactly as the reference intends.
"""
import socket
import struct
import threading
import time
from collections import OrderedDict

from ..utils.http import get_ip, post_json
from ..utils.serialize import dumps, loads

_LEN = struct.Struct('!Q')
_GONE = (1 << 64) - 1


def _recv_all(conn, n):
    chunks = []
    got = 0
    while got < n:
        b = conn.recv(min(1 << 20, n - got))
        if not b:
            raise ConnectionError('socket closed mid-payload')
        chunks.append(b)
        got += len(b)
    return b''.join(chunks)


class _DataServer:
    """One persistent listener serving id-addressed payloads."""

    def __init__(self, maxlen_per_token):
        self._payloads = OrderedDict()               # id -> payload bytes
        self._token_ids = {}                         # token -> list of ids
        self._maxlen = maxlen_per_token
        self._next_id = 0
        self._lock = threading.Lock()
        self._sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._sock.bind(('0.0.0.0', 0))
        self._sock.listen(64)
        self.port = self._sock.getsockname()[1]
        threading.Thread(target=self._accept_loop, daemon=True).start()

    def add(self, token, payload):
        with self._lock:
            datum_id = self._next_id
            self._next_id += 1
            self._payloads[datum_id] = payload
            ids = self._token_ids.setdefault(token, [])
            ids.append(datum_id)
            while len(ids) > self._maxlen:
                old = ids.pop(0)
                self._payloads.pop(old, None)
        return datum_id

    def mark_taken(self, datum_id, token):
        with self._lock:
            self._payloads.pop(datum_id, None)
            ids = self._token_ids.get(token)
            if ids and datum_id in ids:
                ids.remove(datum_id)

    def _accept_loop(self):
        while True:
            try:
                conn, _ = self._sock.accept()
            except OSError:
                return
            threading.Thread(target=self._serve, args=(conn,), daemon=True).start()

    def _serve(self, conn):
        try:
            with conn:
                conn.settimeout(60)
                (datum_id,) = _LEN.unpack(_recv_all(conn, _LEN.size))
                with self._lock:
                    payload = self._payloads.get(datum_id)
                if payload is None:
                    conn.sendall(_LEN.pack(_GONE))
                else:
                    conn.sendall(_LEN.pack(len(payload)) + payload)
        except (OSError, ConnectionError):
            pass


class Adapter:
    def __init__(self, cfg=None, coordinator_ip=None, coordinator_port=None,
                 maxlen=32):
        comm = (cfg or {}).get('communication', {}) if cfg is not None else {}
        self._coord_ip = coordinator_ip or comm.get('coordinator_ip', '127.0.0.1')
        self._coord_port = coordinator_port or comm.get('coordinator_port', 0)
        self._ip = get_ip()
        self._maxlen = maxlen
        self._server = None
        self._server_lock = threading.Lock()

    @property
    def _coord_url(self):
        return f'http://{self._coord_ip}:{self._coord_port}'

    def _ensure_server(self):
        with self._server_lock:
            if self._server is None:
                self._server = _DataServer(self._maxlen)
        return self._server

    # ------------------------------------------------------------ producer
    def push(self, data, token, fs_type='nppickle', compress=True):
        payload = dumps(data, fs_type=fs_type, compress=compress)
        srv = self._ensure_server()
        datum_id = srv.add(token, payload)
        post_json(self._coord_url + '/register_datum',
                  {'token': token, 'ip': self._ip, 'port': srv.port,
                   'id': datum_id})
        return True

    def length(self, token):
        return post_json(self._coord_url + '/queue_length',
                         {'token': token}).get('length', 0)

    def full(self, token):
        return self.length(token) >= self._maxlen

    # ------------------------------------------------------------ consumer
    def pull(self, token, fs_type='nppickle', sleep_time=0.5, size=1,
             worker_num=1, timeout=None):
        """Block until ``size`` payloads for ``token`` arrive; returns a list."""
        out = []
        out_lock = threading.Lock()
        deadline = time.time() + timeout if timeout else None

        def fetch_loop():
            while True:
                with out_lock:
                    if len(out) >= size:
                        return
                if deadline and time.time() > deadline:
                    return
                try:
                    meta = post_json(self._coord_url + '/request_datum',
                                     {'token': token})
                except Exception:
                    time.sleep(sleep_time)
                    continue
                if not meta.get('ip'):
                    time.sleep(sleep_time)
                    continue
                try:
                    data = self._fetch(meta['ip'], meta['port'], meta['id'], fs_type)
                except (OSError, ConnectionError):
                    continue
                if data is None:         # payload evicted at the producer
                    continue
                with out_lock:
                    out.append(data)

        threads = [threading.Thread(target=fetch_loop, daemon=True)
                   for _ in range(max(worker_num, 1))]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        return out[:size]

    @staticmethod
    def _fetch(ip, port, datum_id, fs_type, timeout=60):
        with socket.create_connection((ip, port), timeout=timeout) as conn:
            conn.sendall(_LEN.pack(datum_id))
            (n,) = _LEN.unpack(_recv_all(conn, _LEN.size))
            if n == _GONE:
                return None
            payload = _recv_all(conn, n)
        return loads(payload, fs_type=fs_type)
