"""Writer-priority readers/writer lock — parity with reference sparkflow/RWLock.py.

Used by the parameter server's ``acquire_lock=True`` ("async-with-lock")
mode: weight reads (pulls) take the read side, optimizer applies take the
write side; writers have priority so a stream of pulls cannot starve
updates (reference RWLock.py:10-66 semantics).
"""
import threading


class RWLock:
    """Many readers XOR one writer; waiting writers block new readers."""

    def __init__(self):
        self._cond = threading.Condition()
        self._readers = 0          # active readers
        self._writer = False       # active writer
        self._writers_waiting = 0

    def acquire_read(self):
        with self._cond:
            while self._writer or self._writers_waiting:
                self._cond.wait()
            self._readers += 1

    def acquire_write(self):
        with self._cond:
            self._writers_waiting += 1
            try:
                while self._writer or self._readers:
                    self._cond.wait()
            finally:
                self._writers_waiting -= 1
            self._writer = True

    def release(self):
        with self._cond:
            if self._writer:
                self._writer = False
            else:
                if self._readers <= 0:
                    raise RuntimeError("release() without a held lock")
                self._readers -= 1
            self._cond.notify_all()

    # context-manager helpers -------------------------------------------------
    class _Guard:
        def __init__(self, lock, write):
            self._lock, self._write = lock, write

        def __enter__(self):
            if self._write:
                self._lock.acquire_write()
            else:
                self._lock.acquire_read()
            return self

        def __exit__(self, *exc):
            self._lock.release()
            return False

    def reading(self):
        return RWLock._Guard(self, write=False)

    def writing(self):
        return RWLock._Guard(self, write=True)
