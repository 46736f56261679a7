"""Thread/process lock helpers (reference `ctools/utils/lock_helper.py:7-59`)."""
import multiprocessing
import threading
from enum import Enum, unique


@unique
class LockContextType(Enum):
    THREAD_LOCK = 1
    PROCESS_LOCK = 2


_LOCK_TYPES = {
    LockContextType.THREAD_LOCK: threading.Lock,
    LockContextType.PROCESS_LOCK: multiprocessing.Lock,
}


class LockContext:
    def __init__(self, type_=LockContextType.THREAD_LOCK):
        self.lock = _LOCK_TYPES[type_]()

    def __enter__(self):
        self.lock.acquire()
        return self

    def __exit__(self, *args):
        self.lock.release()

    def acquire(self):
        self.lock.acquire()

    def release(self):
        self.lock.release()
