"""Reserve unused TCP ports for SC2 game/base port sets.

The reference reserves distinct game/base ports via pysc2's portspicker
before create/join (`distar/envs/env.py:211-274`,
`pysc2/lib/portspicker.py`); reusing the websocket listen ports (already
bound by the SC2 processes) makes multiplayer join fail.  This picker
bind-tests candidate ports and keeps a process-local reservation set so
concurrent envs in one process can't hand out the same port twice.
"""
import random
import socket
import threading

_LOCK = threading.Lock()
_RESERVED = set()


def _can_bind(port: int) -> bool:
    try:
        with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
            s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            s.bind(('127.0.0.1', port))
        return True
    except OSError:
        return False


def pick_unused_ports(num_ports: int, retries: int = 200):
    """Return `num_ports` distinct currently-free ports (reserved
    process-locally until returned via return_ports)."""
    assert num_ports > 0
    rng = random.Random()
    picked = []
    with _LOCK:
        for _ in range(retries):
            if len(picked) == num_ports:
                break
            port = rng.randint(10000, 60000)
            if port in _RESERVED or not _can_bind(port):
                continue
            _RESERVED.add(port)
            picked.append(port)
        else:
            for p in picked:
                _RESERVED.discard(p)
            raise RuntimeError(f'could not reserve {num_ports} free ports')
    return picked


def return_ports(ports):
    with _LOCK:
        for p in ports:
            _RESERVED.discard(p)
