"""Batch queue: the rank-partitioned, epoch-windowed rendezvous between the
shuffle producer and the trainers.

MI355X-native re-design of the reference's Ray ``BatchQueue`` client +
``_QueueActor`` (reference: ray_shuffling_data_loader/batch_queue.py:24-509).
The queue state machine lives in C++ (``_rsdl_cpp.BatchQueueCore``, one mutex +
condition variables — the moral equivalent of the reference's single asyncio
event loop). Two transports:

  * **in-process** (the hot path): the per-GPU trainer process produces and
    consumes its own epoch partitions, so queue items are plain Python
    objects / torch tensors handed over with zero serialization.
  * **named** (parity with the reference's named actor,
    batch_queue.py:358-380): ``name=...`` + ``connect=False`` starts a
    Unix-domain-socket server thread around the core; ``connect=True``
    dials it with exponential-backoff retries.

Exceptions ``Empty`` / ``Full`` mirror the reference's
(batch_queue.py:13-18).
"""

import asyncio
import logging
import os
import pickle
import socket
import socketserver
import struct
import tempfile
import threading
import time
from typing import Any, Iterable, Optional

from ray_shuffling_data_loader_amd._rsdl_cpp import (  # noqa: F401
    BatchQueueCore,
    Closed,
    Empty,
    Full,
)

logger = logging.getLogger(__name__)

_HDR = struct.Struct("<Q")


def _queue_socket_path(name: str) -> str:
    """Socket path inside a per-uid mode-0700 directory: the server handler
    unpickles requests, so the rendezvous must not be dialable by other
    local users (a world-writable /tmp socket would hand them arbitrary
    code execution in the trainer process)."""
    base = os.environ.get("RSDL_QUEUE_DIR", tempfile.gettempdir())
    sockdir = os.path.join(base, f"rsdl_queue_u{os.getuid()}")
    os.makedirs(sockdir, mode=0o700, exist_ok=True)
    st = os.stat(sockdir)
    if st.st_uid != os.getuid():
        raise RuntimeError(
            f"queue socket directory {sockdir} is owned by uid {st.st_uid}, "
            "not us; refusing to rendezvous there (set RSDL_QUEUE_DIR)"
        )
    if st.st_mode & 0o077:
        os.chmod(sockdir, 0o700)
    return os.path.join(sockdir, f"rsdl_queue_{name}.sock")


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("queue server connection closed")
        buf += chunk
    return buf


# RSDL_SHM_QUEUE (default ON): serialize messages with
# torch.multiprocessing's ForkingPickler so tensor PAYLOADS cross the
# socket as shared-memory / device-IPC handles instead of bytes (the
# socket carries only metadata) — the analog of the reference's plasma
# zero-copy get (reference dataset.py:136-139). CPU storages move to shm
# files (sharing strategy "file_system" — the "file_descriptor" default
# cannot pass fds between UNRELATED processes, which is exactly what the
# named-queue rendezvous connects); CUDA storages travel as dmabuf IPC
# handles (HSA_ENABLE_IPC_MODE_LEGACY=0). Set RSDL_SHM_QUEUE=0 for plain
# pickle. The collective mode (the production path) never serializes
# batches at all.
_FP = None


def _shm_pickler():
    """Lazy setup of the ForkingPickler path (checked per call so tests can
    toggle RSDL_SHM_QUEUE without re-importing the module)."""
    global _FP
    if os.environ.get("RSDL_SHM_QUEUE", "1") != "1":
        return None
    if _FP is None:
        import torch.multiprocessing as _tmp  # registers torch reducers

        _tmp.set_sharing_strategy("file_system")
        from multiprocessing.reduction import ForkingPickler

        _FP = ForkingPickler
    return _FP


def _send_msg(sock: socket.socket, obj: Any) -> None:
    fp = _shm_pickler()
    if fp is not None:
        import io

        buf = io.BytesIO()
        fp(buf, pickle.HIGHEST_PROTOCOL).dump(obj)
        payload = buf.getvalue()
    else:
        payload = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    sock.sendall(_HDR.pack(len(payload)) + payload)


def _recv_msg(sock: socket.socket) -> Any:
    (n,) = _HDR.unpack(_recv_exact(sock, _HDR.size))
    return pickle.loads(_recv_exact(sock, n))


_EXC_BY_NAME = {
    "Empty": Empty,
    "Full": Full,
    "Closed": Closed,
    "ValueError": ValueError,
    "IndexError": IndexError,
    "RuntimeError": RuntimeError,
}


class _QueueRequestHandler(socketserver.BaseRequestHandler):
    """One thread per client connection; loops over length-prefixed pickled
    (method, args, kwargs) requests against the shared core."""

    def handle(self):
        core = self.server.core  # type: ignore[attr-defined]
        while True:
            try:
                method, args, kwargs = _recv_msg(self.request)
            except (ConnectionError, EOFError, OSError):
                return
            try:
                result = getattr(core, method)(*args, **kwargs)
                reply = ("ok", result)
            except Exception as e:  # noqa: BLE001 - forwarded to client
                reply = ("err", type(e).__name__, str(e))
            try:
                _send_msg(self.request, reply)
            except (ConnectionError, OSError):
                return


class _ThreadedQueueServer(socketserver.ThreadingUnixStreamServer):
    daemon_threads = True
    allow_reuse_address = True


def connect_queue_actor(name: str, num_retries: int = 5):
    """Connect to a named queue server with exponential-backoff retries
    (parity with reference batch_queue.py:358-380)."""
    path = _queue_socket_path(name)
    delay = 0.1
    last_exc: Optional[Exception] = None
    for _ in range(max(1, num_retries)):
        try:
            sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            sock.connect(path)
            return sock
        except OSError as e:
            last_exc = e
            logger.info(
                "Couldn't connect to queue server %s, retrying in %.2fs",
                name,
                delay,
            )
            time.sleep(delay)
            delay *= 2
    raise ValueError(
        f"Unable to connect to queue server {name} after "
        f"{num_retries} retries. Last error: {last_exc!s}"
    )


class BatchQueue:
    """First-in, first-out per-(epoch, rank) batch queue.

    Same surface as the reference client (batch_queue.py:24-355): sync and
    async put/get, batched variants, nowait variants, qsize/empty/full,
    epoch-window ``new_epoch``, ``producer_done``, ``task_done``,
    ``wait_until_all_epochs_done`` and ``shutdown``.
    """

    def __init__(
        self,
        num_epochs: int,
        num_trainers: int,
        max_concurrent_epochs: int,
        maxsize: int = 0,
        name: Optional[str] = None,
        connect: bool = False,
        connect_retries: int = 5,
    ) -> None:
        self._name = name
        self._server = None
        self._server_thread = None
        self._sock = None
        self._sock_lock = threading.Lock()
        self._closed = False
        if connect:
            assert name is not None, "connect=True requires a queue name"
            self.core = None
            self._sock = connect_queue_actor(name, connect_retries)
        else:
            self.core = BatchQueueCore(
                max_concurrent_epochs, num_epochs, num_trainers, maxsize
            )
            if name is not None:
                path = _queue_socket_path(name)
                if os.path.exists(path):
                    os.unlink(path)
                self._server = _ThreadedQueueServer(
                    path, _QueueRequestHandler
                )
                os.chmod(path, 0o600)  # server unpickles: owner-only
                self._server.core = self.core
                self._server_thread = threading.Thread(
                    target=self._server.serve_forever,
                    name=f"rsdl-queue-server-{name}",
                    daemon=True,
                )
                self._server_thread.start()

    @property
    def actor(self):
        """Reference-API alias (batch_queue.py:63-65 exposes `.actor`): the
        backing queue core (local mode) or client socket (connected mode);
        None after shutdown."""
        return self.core if self.core is not None else self._sock

    # ----- transport ---------------------------------------------------------

    def _call(self, method: str, *args, **kwargs):
        if self._closed:
            raise RuntimeError("BatchQueue has been shut down")
        if self.core is not None:
            return getattr(self.core, method)(*args, **kwargs)
        with self._sock_lock:
            _send_msg(self._sock, (method, args, kwargs))
            reply = _recv_msg(self._sock)
        if reply[0] == "ok":
            return reply[1]
        exc_cls = _EXC_BY_NAME.get(reply[1], RuntimeError)
        raise exc_cls(reply[2])

    # A blocking remote call must not serialize-starve other threads using
    # the same client socket; the hot path is in-process so one socket with a
    # lock is acceptable for the parity/test transport.

    # ----- lifecycle ---------------------------------------------------------

    def ready(self) -> None:
        """Wait until the queue core/server is ready."""
        self._call("size")

    def shutdown(self, force: bool = False, grace_period_s: int = 5) -> None:
        """Tear down the queue (server + core). Parity with reference
        batch_queue.py:333-355: like the reference's killed actor, any
        consumer/producer blocked inside a queue op wakes up and raises
        (``Closed``, a RuntimeError subclass) instead of hanging forever."""
        del force, grace_period_s  # no child process to kill; kept for parity
        if self.core is not None:
            # Wake every thread blocked in get/put/new_epoch/join waits.
            self.core.close()
        if self._server is not None:
            self._server.shutdown()
            self._server.server_close()
            try:
                os.unlink(self._server.server_address)
            except OSError:
                pass
            self._server = None
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
            self._sock = None
        self.core = None
        self._closed = True

    # ----- epoch window ------------------------------------------------------

    def new_epoch(self, epoch: int) -> None:
        """Block until the queue has window capacity for ``epoch``
        (max_concurrent_epochs backpressure; reference batch_queue.py:73-82,
        395-418)."""
        self._call("new_epoch", epoch)

    def producer_done(self, rank: int, epoch: int) -> None:
        self._call("producer_done", rank, epoch)

    def task_done(self, rank: int, epoch: int, num_items: int = 1) -> None:
        self._call("task_done", rank, epoch, num_items)

    def wait_until_all_epochs_done(self) -> None:
        self._call("wait_until_all_epochs_done")

    # ----- size --------------------------------------------------------------

    def __len__(self) -> int:
        return self._call("size")

    def size(self, rank: int, epoch: int) -> int:
        return self._call("qsize", rank, epoch)

    def qsize(self, rank: int, epoch: int) -> int:
        return self.size(rank, epoch)

    def empty(self, rank: int, epoch: int) -> bool:
        return self._call("empty", rank, epoch)

    def full(self, rank: int, epoch: int) -> bool:
        return self._call("full", rank, epoch)

    # ----- put ---------------------------------------------------------------

    @staticmethod
    def _norm_timeout(timeout: Optional[float]) -> float:
        if timeout is None:
            return -1.0
        if timeout < 0:
            raise ValueError("'timeout' must be a non-negative number")
        return float(timeout)

    def put(
        self,
        rank: int,
        epoch: int,
        item: Any,
        block: bool = True,
        timeout: Optional[float] = None,
    ) -> None:
        t = self._norm_timeout(timeout)
        self._call("put", rank, epoch, item, block, t)

    def put_nowait(self, rank: int, epoch: int, item: Any) -> None:
        self.put(rank, epoch, item, block=False)

    def put_batch(
        self,
        rank: int,
        epoch: int,
        items: Iterable,
        block: bool = True,
        timeout: Optional[float] = None,
    ) -> None:
        t = self._norm_timeout(timeout)
        self._call("put_batch", rank, epoch, list(items), block, t)

    def put_nowait_batch(self, rank: int, epoch: int, items: Iterable) -> None:
        self._call("put_nowait_batch", rank, epoch, list(items))

    async def put_async(
        self,
        rank: int,
        epoch: int,
        item: Any,
        block: bool = True,
        timeout: Optional[float] = None,
    ) -> None:
        t = self._norm_timeout(timeout)
        await asyncio.to_thread(self._call, "put", rank, epoch, item, block, t)

    # ----- get ---------------------------------------------------------------

    def get(
        self,
        rank: int,
        epoch: int,
        block: bool = True,
        timeout: Optional[float] = None,
    ) -> Any:
        t = self._norm_timeout(timeout)
        return self._call("get", rank, epoch, block, t)

    def get_nowait(self, rank: int, epoch: int) -> Any:
        return self.get(rank, epoch, block=False)

    def get_batch(self, rank: int, epoch: int) -> list:
        """Block for >= 1 item, then greedily drain the sub-queue
        (reference batch_queue.py:286-287, 468-475)."""
        return self._call("get_batch", rank, epoch)

    def get_nowait_batch(
        self, rank: int, epoch: int, num_items: Optional[int] = None
    ) -> list:
        n = -1 if num_items is None else num_items
        return self._call("get_nowait_batch", rank, epoch, n)

    async def get_async(
        self,
        rank: int,
        epoch: int,
        block: bool = True,
        timeout: Optional[float] = None,
    ) -> Any:
        t = self._norm_timeout(timeout)
        return await asyncio.to_thread(self._call, "get", rank, epoch, block, t)
