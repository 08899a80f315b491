"""Native C++ TCP store (csrc/rcclx.cpp) — the rendezvous layer the
reference inherits from PyTorch 0.x THD (tuto.md:404-419), tested
directly on CPU: wire protocol, blocking GET, atomic ADD, and
many-client concurrency.  Skipped when the native extension is not
built (hipcc cross-compiles it on CPU boxes, so it normally is)."""

import threading

import pytest

from dist_tuto_pth_amd.dist import _free_port
from dist_tuto_pth_amd.utils.native import load_native, native_available

if not native_available("_rcclx"):
    pytest.skip("_rcclx.so not built (run python build.py)",
                allow_module_level=True)


def _mk_store_pair():
    rx = load_native("_rcclx")
    port = _free_port()
    server = rx.TcpStore("127.0.0.1", port, 0, 2, True, 30_000)
    client = rx.TcpStore("127.0.0.1", port, 1, 2, False, 30_000)
    return server, client


def test_set_get_roundtrip():
    s, c = _mk_store_pair()
    s.set("k1", b"hello")
    assert c.get("k1") == b"hello"
    c.set("k2", b"\x00\xff" * 64)   # binary-safe (ncclUniqueId is raw)
    assert s.get("k2") == b"\x00\xff" * 64


def test_get_blocks_until_set():
    s, c = _mk_store_pair()
    out = {}

    def getter():
        out["v"] = c.get("late-key")

    t = threading.Thread(target=getter)
    t.start()
    import time
    time.sleep(0.2)
    assert "v" not in out          # still blocked
    s.set("late-key", b"now")
    t.join(10)
    assert out.get("v") == b"now"


def test_get_timeout_raises():
    rx = load_native("_rcclx")
    port = _free_port()
    s = rx.TcpStore("127.0.0.1", port, 0, 1, True, 300)  # 300 ms
    with pytest.raises(RuntimeError, match="timed out"):
        s.get("never-set")


def test_add_is_atomic_across_clients():
    rx = load_native("_rcclx")
    port = _free_port()
    server = rx.TcpStore("127.0.0.1", port, 0, 9, True, 30_000)
    clients = [rx.TcpStore("127.0.0.1", port, i + 1, 9, False, 30_000)
               for i in range(8)]

    def bump(st, n):
        for _ in range(n):
            st.add("ctr", 1)

    threads = [threading.Thread(target=bump, args=(st, 50))
               for st in clients]
    for t in threads:
        t.start()
    for t in threads:
        t.join(30)
    assert server.add("ctr", 0) == 8 * 50
    del clients, server


def test_many_keys_concurrent_set_get():
    rx = load_native("_rcclx")
    port = _free_port()
    server = rx.TcpStore("127.0.0.1", port, 0, 5, True, 30_000)
    clients = [rx.TcpStore("127.0.0.1", port, i + 1, 5, False, 30_000)
               for i in range(4)]
    errs = []

    def worker(idx, st):
        try:
            for j in range(40):
                st.set(f"w{idx}:{j}", bytes([idx]) * (j + 1))
            for j in range(40):
                v = st.get(f"w{(idx + 1) % 4}:{j}")
                assert v == bytes([(idx + 1) % 4]) * (j + 1)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(i, st))
               for i, st in enumerate(clients)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(30)
    assert not errs, errs
    del clients, server
