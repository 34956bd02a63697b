"""Remote debugging: breakpoint() in user code opens a pdb session on a TCP
port inside the worker pod; `kt debug <service>` attaches interactively.
(Reference parity: serving/pdb_websocket.py + deep_breakpoint — socket
transport instead of WebSocket, same UX.)"""
import os
import pdb
import socket
import sys

DEBUG_PORT = int(os.environ.get("KT_DEBUG_PORT", "4444"))


class _SockIO:
    def __init__(self, conn):
        self.conn = conn
        self._file = conn.makefile("rw", buffering=1)

    def readline(self):
        return self._file.readline()

    def write(self, s):
        self._file.write(s)
        return len(s)

    def flush(self):
        self._file.flush()


_last_conn = [None]  # previous session's socket, closed on the next hit


def deep_breakpoint(*_, **__):
    """PYTHONBREAKPOINT target: wait for one debugger client, run pdb over
    the connection, resume on 'continue'. set_trace is the LAST statement
    (no finally, server socket closed beforehand): the first trace event
    must land on the CALLER's next line, not inside this function's
    cleanup code."""
    if _last_conn[0] is not None:
        try:
            _last_conn[0].close()
        except OSError:
            pass
        _last_conn[0] = None
    srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("0.0.0.0", DEBUG_PORT))
    srv.listen(1)
    sys.stderr.write(
        f"[kt-debug] breakpoint hit; waiting for `kt debug` on :{DEBUG_PORT}\n")
    conn, addr = srv.accept()
    _last_conn[0] = conn
    io = _SockIO(conn)
    io.write(f"[kt-debug] attached from {addr}\n")
    dbg = pdb.Pdb(stdin=io, stdout=io)
    srv.close()
    dbg.set_trace(sys._getframe(1))


def attach(host, port=DEBUG_PORT):
    """Client side: bridge the local terminal to the remote pdb session."""
    import threading

    conn = socket.create_connection((host, port))
    stop = False

    def pump_in():
        try:
            for line in sys.stdin:
                conn.sendall(line.encode())
                if stop:
                    return
        except (OSError, KeyboardInterrupt):
            pass

    t = threading.Thread(target=pump_in, daemon=True)
    t.start()
    try:
        while True:
            data = conn.recv(4096)
            if not data:
                break
            sys.stdout.write(data.decode(errors="replace"))
            sys.stdout.flush()
    finally:
        stop = True
        conn.close()
