"""Minimal RFC6455 websocket client (and the server-side handshake helper
the conformance tests use) over stdlib sockets.

The offline image ships no `websocket-client`; SC2's API is a plain
binary-frame websocket at ws://host:port/sc2api, which this ~150-line
client covers completely: handshake, masked client frames, fragmentation,
ping/pong, close.  Replaces the reference stack's dependency on the
`websocket` package (`distar/pysc2/lib/remote_controller.py:147-175`).
"""
import base64
import hashlib
import os
import socket
import struct

_GUID = '258EAFA5-E914-47DA-95CA-C5AB0DC85B11'

OP_CONT, OP_TEXT, OP_BINARY, OP_CLOSE, OP_PING, OP_PONG = 0, 1, 2, 8, 9, 10


class WebSocketError(ConnectionError):
    pass


def _accept_key(key: str) -> str:
    return base64.b64encode(
        hashlib.sha1((key + _GUID).encode()).digest()).decode()


class WebSocket:
    """Blocking websocket over an already-connected or fresh socket."""

    def __init__(self, sock: socket.socket):
        self._sock = sock
        self._buf = b''
        self.masking = True      # client frames are masked; servers set False

    # ------------------------------------------------------------ connect
    @classmethod
    def connect(cls, host: str, port: int, resource: str = '/sc2api',
                timeout: float = 120.0):
        sock = socket.create_connection((host, port), timeout=timeout)
        sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        key = base64.b64encode(os.urandom(16)).decode()
        handshake = (
            f'GET {resource} HTTP/1.1\r\n'
            f'Host: {host}:{port}\r\n'
            'Upgrade: websocket\r\n'
            'Connection: Upgrade\r\n'
            f'Sec-WebSocket-Key: {key}\r\n'
            'Sec-WebSocket-Version: 13\r\n\r\n')
        sock.sendall(handshake.encode())
        ws = cls(sock)
        status, headers = ws._read_http_head()
        if status != 101:
            sock.close()
            raise WebSocketError(f'websocket handshake rejected: {status}')
        if headers.get('sec-websocket-accept') != _accept_key(key):
            sock.close()
            raise WebSocketError('websocket handshake: bad accept key')
        return ws

    def _read_http_head(self):
        data = b''
        while b'\r\n\r\n' not in data:
            chunk = self._sock.recv(4096)
            if not chunk:
                raise WebSocketError('connection closed during handshake')
            data += chunk
        head, _, rest = data.partition(b'\r\n\r\n')
        self._buf = rest
        lines = head.decode('latin1').split('\r\n')
        status = int(lines[0].split(' ')[1])
        headers = {}
        for line in lines[1:]:
            if ':' in line:
                k, v = line.split(':', 1)
                headers[k.strip().lower()] = v.strip()
        return status, headers

    # ------------------------------------------------------------- frames
    def _read_exact(self, n: int) -> bytes:
        while len(self._buf) < n:
            chunk = self._sock.recv(max(4096, n - len(self._buf)))
            if not chunk:
                raise WebSocketError('connection closed mid-frame')
            self._buf += chunk
        out, self._buf = self._buf[:n], self._buf[n:]
        return out

    def _read_frame(self):
        b0, b1 = self._read_exact(2)
        fin = bool(b0 & 0x80)
        opcode = b0 & 0x0F
        masked = bool(b1 & 0x80)
        length = b1 & 0x7F
        if length == 126:
            (length,) = struct.unpack('>H', self._read_exact(2))
        elif length == 127:
            (length,) = struct.unpack('>Q', self._read_exact(8))
        mask = self._read_exact(4) if masked else None
        payload = self._read_exact(length)
        if mask:
            payload = bytes(c ^ mask[i % 4] for i, c in enumerate(payload))
        return fin, opcode, payload

    def _send_frame(self, opcode: int, payload: bytes, mask: bool = True):
        b0 = 0x80 | opcode
        header = bytes([b0])
        length = len(payload)
        mask_bit = 0x80 if mask else 0
        if length < 126:
            header += bytes([mask_bit | length])
        elif length < (1 << 16):
            header += bytes([mask_bit | 126]) + struct.pack('>H', length)
        else:
            header += bytes([mask_bit | 127]) + struct.pack('>Q', length)
        if mask:
            mkey = os.urandom(4)
            header += mkey
            payload = bytes(c ^ mkey[i % 4] for i, c in enumerate(payload))
        self._sock.sendall(header + payload)

    # ---------------------------------------------------------------- api
    def send_binary(self, data: bytes):
        self._send_frame(OP_BINARY, data, mask=self.masking)

    def recv(self) -> bytes:
        """Next complete binary/text message (reassembling fragments,
        answering pings)."""
        message = b''
        while True:
            fin, opcode, payload = self._read_frame()
            if opcode == OP_PING:
                self._send_frame(OP_PONG, payload, mask=self.masking)
                continue
            if opcode == OP_PONG:
                continue
            if opcode == OP_CLOSE:
                self._send_frame(OP_CLOSE, b'')
                raise WebSocketError('websocket closed by peer')
            message += payload
            if fin:
                return message

    def settimeout(self, timeout):
        self._sock.settimeout(timeout)

    def close(self):
        try:
            self._send_frame(OP_CLOSE, b'')
        except OSError:
            pass
        try:
            self._sock.close()
        except OSError:
            pass


def server_handshake(conn: socket.socket) -> 'WebSocket':
    """Accept one websocket client on `conn` (tests / fake SC2 servers)."""
    ws = WebSocket(conn)
    ws.masking = False
    data = b''
    while b'\r\n\r\n' not in data:
        chunk = conn.recv(4096)
        if not chunk:
            raise WebSocketError('client closed during handshake')
        data += chunk
    head, _, rest = data.partition(b'\r\n\r\n')
    ws._buf = rest
    key = None
    for line in head.decode('latin1').split('\r\n'):
        if line.lower().startswith('sec-websocket-key:'):
            key = line.split(':', 1)[1].strip()
    if key is None:
        raise WebSocketError('no Sec-WebSocket-Key in client handshake')
    conn.sendall((
        'HTTP/1.1 101 Switching Protocols\r\n'
        'Upgrade: websocket\r\n'
        'Connection: Upgrade\r\n'
        f'Sec-WebSocket-Accept: {_accept_key(key)}\r\n\r\n').encode())
    return ws
