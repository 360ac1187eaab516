import threading
import time

from sparkflow_amd.parallel.rwlock import RWLock


def test_multiple_readers():
    lock = RWLock()
    lock.acquire_read()
    lock.acquire_read()
    lock.release()
    lock.release()


def test_writer_excludes_readers():
    lock = RWLock()
    state = {"in_read": False}
    lock.acquire_write()

    def reader():
        lock.acquire_read()
        state["in_read"] = True
        lock.release()

    t = threading.Thread(target=reader)
    t.start()
    time.sleep(0.1)
    assert not state["in_read"]
    lock.release()
    t.join(timeout=5)
    assert state["in_read"]


def test_writer_priority_blocks_new_readers():
    lock = RWLock()
    lock.acquire_read()
    events = []

    def writer():
        lock.acquire_write()
        events.append("w")
        lock.release()

    def late_reader():
        lock.acquire_read()
        events.append("r")
        lock.release()

    tw = threading.Thread(target=writer)
    tw.start()
    time.sleep(0.1)  # writer is now waiting
    tr = threading.Thread(target=late_reader)
    tr.start()
    time.sleep(0.1)
    assert events == []  # late reader must queue behind the waiting writer
    lock.release()
    tw.join(timeout=5)
    tr.join(timeout=5)
    assert events[0] == "w"


def test_concurrent_stress():
    lock = RWLock()
    counter = {"v": 0}

    def writer():
        for _ in range(50):
            with lock.writing():
                v = counter["v"]
                time.sleep(0)
                counter["v"] = v + 1

    def reader():
        for _ in range(50):
            with lock.reading():
                _ = counter["v"]

    threads = [threading.Thread(target=writer) for _ in range(4)] + [
        threading.Thread(target=reader) for _ in range(4)
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
    assert counter["v"] == 200
