"""Worker→driver log transport (N8 in SURVEY.md §2.2).

Implements the documented log plumbing of the reference runtime
(reference runner_base.py:62-72, horovod/__init__.py:20-25): workers send
framed messages over a local TCP socket to the driver, which prints
``log_to_driver`` messages to stdout and routes rank stdout/stderr
according to ``driver_log_verbosity``.  Rank 0's return value travels
over the same channel, cloudpickle-encoded (reference README.md:92-93).

Frame format (little-endian):
    1 byte  kind   (b'L' log_to_driver, b'R' return value)
    4 bytes rank   (int32)
    8 bytes length (int64)
    <length> bytes payload
"""

import os
import socket
import struct
import sys
import threading

_HDR = struct.Struct("<ciq")

# Environment variable carrying the driver log-server address "host:port".
LOG_ADDR_ENV = "SPARKDL_LOG_ADDR"
RANK_ENV = "RANK"


class LogServer:
    """Driver-side log sink.

    Accepts connections from workers, prints b'L' frames to stdout, and
    captures the b'R' (return value) frame from rank 0.
    """

    def __init__(self):
        self._sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._sock.bind(("127.0.0.1", 0))
        self._sock.listen(64)
        self.addr = "127.0.0.1:%d" % self._sock.getsockname()[1]
        self.return_value_bytes = None
        self._threads = []
        self._accept_thread = None
        self._closed = threading.Event()

    def start(self):
        self._accept_thread = threading.Thread(
            target=self._accept_loop, name="sparkdl-logserver", daemon=True)
        self._accept_thread.start()
        return self

    def _accept_loop(self):
        while not self._closed.is_set():
            try:
                conn, _ = self._sock.accept()
            except OSError:
                return
            t = threading.Thread(
                target=self._serve, args=(conn,), daemon=True)
            t.start()
            self._threads.append(t)

    def _recv_exact(self, conn, n):
        buf = b""
        while len(buf) < n:
            chunk = conn.recv(n - len(buf))
            if not chunk:
                return None
            buf += chunk
        return buf

    def _serve(self, conn):
        try:
            while True:
                hdr = self._recv_exact(conn, _HDR.size)
                if hdr is None:
                    return
                kind, rank, length = _HDR.unpack(hdr)
                payload = self._recv_exact(conn, length) if length else b""
                if payload is None and length:
                    return
                if kind == b"L":
                    # log_to_driver contract: driver prints to stdout.
                    sys.stdout.write(payload.decode("utf-8", "replace") + "\n")
                    sys.stdout.flush()
                elif kind == b"R":
                    self.return_value_bytes = payload
        finally:
            conn.close()

    def close(self):
        self._closed.set()
        try:
            self._sock.close()
        except OSError:
            pass
        for t in self._threads:
            t.join(timeout=2.0)


# ---------------------------------------------------------------------------
# Worker side
# ---------------------------------------------------------------------------

_client_lock = threading.Lock()
_client_sock = None


def _get_client():
    global _client_sock
    addr = os.environ.get(LOG_ADDR_ENV)
    if addr is None:
        return None
    if _client_sock is None:
        host, port = addr.rsplit(":", 1)
        s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        s.connect((host, int(port)))
        _client_sock = s
    return _client_sock


def _send_frame(kind, payload):
    rank = int(os.environ.get(RANK_ENV, "0"))
    with _client_lock:
        sock = _get_client()
        if sock is None:
            return False
        sock.sendall(_HDR.pack(kind, rank, len(payload)) + payload)
    return True


def forward_to_driver(message):
    """Send a log_to_driver message; print locally when not inside a run."""
    if not _send_frame(b"L", message.encode("utf-8")):
        sys.stdout.write(message + "\n")
        sys.stdout.flush()


def send_return_value(pickled_bytes):
    """Rank 0 ships its cloudpickled return value to the driver."""
    _send_frame(b"R", pickled_bytes)


def reset_client():
    """Drop the cached connection (used by tests / forked children)."""
    global _client_sock
    with _client_lock:
        if _client_sock is not None:
            try:
                _client_sock.close()
            except OSError:
                pass
        _client_sock = None
