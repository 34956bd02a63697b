"""Remote pdb session (reference: pdb_websocket + deep_breakpoint + `kt
debug`): a worker hits breakpoint(), a client attaches over TCP, inspects
a local variable and continues — driven end-to-end over the socket."""
import os
import socket
import subprocess
import sys
import time

import pytest

pytestmark = pytest.mark.flaky_retry

WORKER = """
import os, sys
os.environ["KT_DEBUG_PORT"] = "{port}"
sys.path.insert(0, {repo!r})
from kubetorch_amd.serving.pdb_ws import deep_breakpoint
secret_value = 40 + 2
deep_breakpoint()
print("RESUMED", secret_value + 1, flush=True)
"""


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.mark.timeout(120)
def test_breakpoint_attach_inspect_continue(tmp_path):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    port = _free_port()
    script = tmp_path / "worker.py"
    script.write_text(WORKER.format(port=port, repo=repo))
    proc = subprocess.Popen([sys.executable, str(script)],
                            stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                            text=True)
    try:
        # connect exactly ONCE, after the worker says it is listening —
        # a timed-out probe connection could otherwise be the one the
        # single-client accept() takes
        banner = proc.stderr.readline()
        assert "waiting for" in banner, banner
        conn = socket.create_connection(("127.0.0.1", port), 10)

        def read_until(needle, timeout=20):
            buf = ""
            end = time.time() + timeout
            conn.settimeout(1.0)
            while time.time() < end:
                try:
                    ch = conn.recv(4096).decode(errors="replace")
                except socket.timeout:
                    continue
                if not ch:
                    break
                buf += ch
                if needle in buf:
                    return buf
            raise AssertionError(f"never saw {needle!r} in {buf!r}")

        read_until("(Pdb)")
        conn.sendall(b"p secret_value\n")
        out = read_until("(Pdb)")
        assert "42" in out
        conn.sendall(b"c\n")
        stdout, _ = proc.communicate(timeout=30)
        assert "RESUMED 43" in stdout
    finally:
        if proc.poll() is None:
            proc.kill()
