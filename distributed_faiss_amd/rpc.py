# TCP transport for multi-NODE deployments.
#
# Restores the reference's many-node mode (distributed_faiss/rpc.py:
# pickle-over-TCP with a transparent method proxy, rpc.py:137-138, and
# server-side traceback-string error propagation, server.py:215-238 /
# rpc.py:126-131). Intra-node the in-process registry replaces this
# (BASELINE.json north_star retires rpc for that path); this module exists
# so a reference deployment spanning nodes still works.
#
# Framing difference from the reference (documented): 8-byte big-endian
# length prefix + pickle(protocol 4) payload instead of the FileSock
# block stream (rpc.py:42-93) — same payloads, simpler framing. The
# security posture matches the reference: pickle over a trusted network.

import pickle
import socket
import struct
import threading
import traceback

DEFAULT_PORT = 12032  # reference rpc.py:22


class ServerException(Exception):
    """Re-raised client-side with the server's traceback text
    (reference rpc.py:126-131)."""


def _send_msg(sock, obj):
    payload = pickle.dumps(obj, protocol=4)
    sock.sendall(struct.pack(">Q", len(payload)) + payload)


def _recv_exact(sock, n):
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(min(n - len(buf), 1 << 20))
        if not chunk:
            raise EOFError("connection closed")
        buf += chunk
    return buf


def _recv_msg(sock):
    (n,) = struct.unpack(">Q", _recv_exact(sock, 8))
    return pickle.loads(_recv_exact(sock, n))


class TcpServer:
    """Serve an IndexServer's methods over TCP (thread per connection,
    like reference server.py:95-135)."""

    def __init__(self, index_server, port=DEFAULT_PORT, host=""):
        self.index_server = index_server
        self.sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.sock.bind((host, port))
        self.sock.listen(16)
        self.port = self.sock.getsockname()[1]
        self._stop = threading.Event()

    def serve_forever(self):
        while not self._stop.is_set():
            try:
                conn, _addr = self.sock.accept()
            except OSError:
                break
            t = threading.Thread(target=self._conn_loop, args=(conn,),
                                 daemon=True)
            t.start()

    def start_background(self):
        t = threading.Thread(target=self.serve_forever, daemon=True)
        t.start()
        return t

    def stop(self):
        self._stop.set()
        try:
            self.sock.close()
        except OSError:
            pass

    def _conn_loop(self, conn):
        try:
            while True:
                try:
                    fname, args = _recv_msg(conn)
                except (EOFError, ConnectionError):
                    return
                st, ret = None, None
                try:
                    f = getattr(self.index_server, fname)
                except AttributeError:
                    st = "unknown method " + fname  # reference server.py:226-228
                if st is None:
                    try:
                        ret = f(*args)
                    except Exception:
                        # traceback-string propagation, reference server.py:229-233
                        st = traceback.format_exc()
                try:
                    _send_msg(conn, (st, ret))
                except (ConnectionError, BrokenPipeError):
                    return
        finally:
            try:
                conn.close()
            except OSError:
                pass


class TcpClient:
    """Transparent method proxy to a remote IndexServer (reference
    rpc.Client, rpc.py:95-138: every attribute is a remote call)."""

    def __init__(self, host, port):
        self.sock = socket.create_connection((host, port))
        self._lock = threading.Lock()

    def _call(self, fname, *args):
        with self._lock:
            _send_msg(self.sock, (fname, args))
            st, ret = _recv_msg(self.sock)
        if st is not None:
            raise ServerException(st)
        return ret

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return lambda *args: self._call(name, *args)

    def close(self):
        try:
            self.sock.close()
        except OSError:
            pass
