import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


def pytest_collection_modifyitems(config, items):
    try:
        import parsec_amd as pm
        has_gpu = pm.hip_device_count() > 0
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU visible")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="module")
def ctx():
    import parsec_amd as pm
    c = pm.Context(nworkers=2, rank=0, world=1)
    yield c
    del c


_port_locks = []  # flock fds held for the session lifetime
_port_cache = {}  # salt -> base: repeat calls must return the same base


def port_base(salt=0, span=64):
    """A base with `span` consecutive bindable ports. Two safeguards:
    ports stay BELOW the ephemeral range (32768+, where outgoing sockets
    land), and the chosen window is reserved via a session-lifetime
    flock so concurrent test sessions on one host never pick overlapping
    windows (probing alone races: both sessions can see the same window
    free before either binds)."""
    import fcntl
    import socket
    if salt in _port_cache:
        return _port_cache[salt]
    WINDOW = 12000  # 20000..31999
    start = 20000 + ((os.getpid() * 131 + salt * 977) % WINDOW)
    for attempt in range(200):
        base = 20000 + (start - 20000 + attempt * (span + 1)) % WINDOW
        try:
            lf = open(f"/tmp/.pa_test_ports_{base}.lock", "w")
            fcntl.flock(lf, fcntl.LOCK_EX | fcntl.LOCK_NB)
        except OSError:
            continue
        ok = True
        socks = []
        try:
            for off in range(span):
                sk = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
                sk.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
                try:
                    sk.bind(("127.0.0.1", base + off))
                except OSError:
                    ok = False
                    sk.close()
                    break
                socks.append(sk)
        finally:
            for sk in socks:
                sk.close()
        if ok:
            _port_locks.append(lf)  # released automatically at exit
            _port_cache[salt] = base
            return base
        lf.close()  # releases the flock; window is occupied anyway
    raise RuntimeError("no free port range found")
