"""TCP port-forwarding for ``kray session`` (reference analog:
kubectl-plugin/pkg/cmd/session — kubectl port-forward to the head service).

The kubectl plugin shells out to ``kubectl port-forward``; here the CLI
talks to clusters whose head endpoints are directly routable (in-cluster,
or the operator's kube-API facade host), so a plain user-space TCP splice
does the same job with no kubectl dependency: one listener per local port,
each accepted connection is piped byte-for-byte to ``target_host:remote``.
"""
from __future__ import annotations

import socket
import threading
from typing import Callable, List, Optional, Sequence, Tuple


def _splice(src: socket.socket, dst: socket.socket) -> None:
    try:
        while True:
            data = src.recv(65536)
            if not data:
                break
            dst.sendall(data)
    except OSError:
        pass
    finally:
        for s in (src, dst):
            try:
                s.shutdown(socket.SHUT_RDWR)
            except OSError:
                pass


class PortForwarder:
    """Forward local ports to ``target_host`` until :meth:`stop`.

    ``mappings`` is a sequence of ``(local_port, remote_port)``;
    ``local_port`` 0 picks a free port (useful in tests — read the bound
    ports back from :attr:`local_ports` after :meth:`start`).
    """

    def __init__(self, target_host: str,
                 mappings: Sequence[Tuple[int, int]],
                 bind_host: str = "127.0.0.1",
                 on_connect: Optional[Callable[[int, int], None]] = None):
        self.target_host = target_host
        self.mappings = list(mappings)
        self.bind_host = bind_host
        self.on_connect = on_connect
        self.local_ports: List[int] = []
        self._listeners: List[socket.socket] = []
        self._threads: List[threading.Thread] = []
        self._stopped = threading.Event()

    def start(self) -> "PortForwarder":
        for local, remote in self.mappings:
            listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            listener.bind((self.bind_host, local))
            listener.listen(16)
            self.local_ports.append(listener.getsockname()[1])
            self._listeners.append(listener)
            t = threading.Thread(target=self._accept_loop,
                                 args=(listener, remote),
                                 name=f"kray-forward-{remote}", daemon=True)
            t.start()
            self._threads.append(t)
        return self

    def _accept_loop(self, listener: socket.socket, remote: int) -> None:
        while not self._stopped.is_set():
            try:
                conn, _ = listener.accept()
            except OSError:
                return  # listener closed by stop()
            try:
                upstream = socket.create_connection(
                    (self.target_host, remote), timeout=10)
            except OSError:
                conn.close()
                continue
            if self.on_connect is not None:
                self.on_connect(listener.getsockname()[1], remote)
            threading.Thread(target=_splice, args=(conn, upstream),
                             daemon=True).start()
            threading.Thread(target=_splice, args=(upstream, conn),
                             daemon=True).start()

    def stop(self) -> None:
        self._stopped.set()
        for listener in self._listeners:
            try:
                listener.close()
            except OSError:
                pass
        for t in self._threads:
            t.join(timeout=2)

    def wait(self) -> None:
        """Block until interrupted (Ctrl-C in the CLI)."""
        try:
            while not self._stopped.wait(3600):
                pass
        except KeyboardInterrupt:
            self.stop()
