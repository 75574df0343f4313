"""Capture-exclusivity RW lock + scheduler failure recovery (the r02
mixed-soak crash class: a hipGraph capture invalidated by concurrent
co-served launches aborted the server)."""

import threading
import time

import pytest

torch = pytest.importorskip("torch")


def test_rwlock_readers_share_writers_exclude():
    from client_amd.server.models import _RWLock

    lock = _RWLock()
    state = {"readers": 0, "max_readers": 0, "writer_active": 0}
    mu = threading.Lock()
    stop = threading.Event()

    def reader():
        while not stop.is_set():
            lock.acquire_read()
            with mu:
                assert state["writer_active"] == 0
                state["readers"] += 1
                state["max_readers"] = max(state["max_readers"],
                                           state["readers"])
            time.sleep(0.001)
            with mu:
                state["readers"] -= 1
            lock.release_read()

    def writer():
        for _ in range(5):
            lock.acquire_write()
            with mu:
                assert state["readers"] == 0
                state["writer_active"] += 1
            time.sleep(0.002)
            with mu:
                state["writer_active"] -= 1
            lock.release_write()
            time.sleep(0.001)

    readers = [threading.Thread(target=reader) for _ in range(4)]
    w = threading.Thread(target=writer)
    for t in readers:
        t.start()
    w.start()
    w.join(timeout=30)
    stop.set()
    for t in readers:
        t.join(timeout=10)
    assert not w.is_alive()
    # readers really did overlap each other at some point
    assert state["max_readers"] >= 2


def test_rwlock_writer_priority():
    """A pending writer blocks NEW readers (no starvation under a
    steady read stream)."""
    from client_amd.server.models import _RWLock

    lock = _RWLock()
    lock.acquire_read()
    got_write = threading.Event()

    def writer():
        lock.acquire_write()
        got_write.set()
        lock.release_write()

    w = threading.Thread(target=writer)
    w.start()
    time.sleep(0.05)  # writer now pending
    late_reader_in = threading.Event()

    def late_reader():
        lock.acquire_read()
        late_reader_in.set()
        lock.release_read()

    r = threading.Thread(target=late_reader)
    r.start()
    time.sleep(0.05)
    # the late reader must NOT get in ahead of the pending writer
    assert not late_reader_in.is_set()
    lock.release_read()
    assert got_write.wait(timeout=10)
    assert late_reader_in.wait(timeout=10)
    w.join(10)
    r.join(10)


def test_scheduler_survives_step_exception():
    """An unexpected exception inside a scheduler step must END the
    in-flight streams and leave the scheduler serving new requests —
    not kill the worker thread (which hung every stream in the r02
    soak)."""
    from client_amd.models.llama import LlamaModel, llama_tiny_config
    from client_amd.server.decode_scheduler import DecodeScheduler

    torch.manual_seed(0)
    cfg = llama_tiny_config()
    m = LlamaModel(cfg).eval()
    sched = DecodeScheduler(m, max_batch=2, device="cpu",
                            dtype=torch.float32)
    try:
        boom = {"armed": True}
        orig = sched._decode_step

        def exploding_step():
            if boom["armed"]:
                boom["armed"] = False
                raise RuntimeError("injected fault")
            return orig()

        sched._decode_step = exploding_step
        q1 = sched.submit([1, 2, 3], 4)
        # the faulted stream terminates with END (no tokens required)
        seen_end = False
        for _ in range(10):
            tok = q1.get(timeout=30)
            if tok is sched.END:
                seen_end = True
                break
        assert seen_end
        # and the scheduler still serves new requests afterwards
        q2 = sched.submit([4, 5, 6], 3)
        toks = []
        while True:
            tok = q2.get(timeout=30)
            if tok is sched.END:
                break
            toks.append(tok)
        assert len(toks) == 3
    finally:
        sched.shutdown()
