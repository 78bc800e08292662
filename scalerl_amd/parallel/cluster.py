"""Multi-node control plane: TCP transport for off-node CPU actors.

Capability parity with the reference's ``scalerl/hpc/`` layer
(worker.py / connection.py / parameter_server.py — HandyRL heritage), which
ships broken (imports nonexistent modules, SURVEY.md "Broken-as-shipped").
This implementation works and is covered by localhost tests:

- :class:`FrameConnection` — length-prefixed binary frames over TCP (the
  reference's PickledConnection, connection.py:26-84, with tensors encoded
  as raw buffers instead of pickle for the bulk payloads);
- :class:`ParameterServer` — versioned weight blob with push/pull
  (parameter_server.py:4-33 semantics);
- :class:`WorkerServer` / :class:`RemoteWorkerCluster` — entry handshake
  (worker ids + config ship to remote nodes) and gather loops that batch
  episode uploads and weight requests (worker.py:153-341 call structure).

Scope note: single-node xGMI is the benchmark scope; bulk
intra-node traffic goes over RCCL/shared memory.  This plane exists for
cross-node CPU actor farms: weights flow node→actors, rollout payloads
actors→node, both as raw tensor frames.
"""

from __future__ import annotations

import hashlib
import hmac
import json
import os
import secrets as _secrets
import socket
import struct
import threading
from typing import Any, Callable, Dict, List, Optional, Tuple

import torch

_HDR = struct.Struct("!Q")
# untrusted-input bounds: a peer can never make us allocate more than this
_MAX_META = 16 << 20      # 16 MiB of JSON metadata
_MAX_TENSOR = 4 << 30     # 4 GiB per tensor frame
# dtype whitelist — `getattr(torch, name)` on attacker bytes is not ok
_DTYPES = {
    "float32": torch.float32, "float64": torch.float64,
    "float16": torch.float16, "bfloat16": torch.bfloat16,
    "int64": torch.int64, "int32": torch.int32, "int16": torch.int16,
    "int8": torch.int8, "uint8": torch.uint8, "bool": torch.bool,
}


def cluster_secret() -> bytes:
    """Shared secret for the HMAC handshake (SCALERL_CLUSTER_SECRET).
    Every node of a multi-node run must export the same value."""
    return os.environ.get("SCALERL_CLUSTER_SECRET", "").encode()


class FrameConnection:
    """Length-prefixed frames; payloads are (header dict, tensor list).

    Wire safety (this is the only code that parses bytes off the network):
    metadata frames are JSON — never pickle, so a malicious peer can at
    worst send bad values, not code; tensor frames are raw buffers decoded
    against a dtype whitelist with size caps.  Connections are authenticated
    with an HMAC-SHA256 challenge/response over a shared secret before any
    frame is parsed (reference's PickledConnection, connection.py:26-84,
    had neither property).
    """

    def __init__(self, sock: socket.socket):
        self.sock = sock
        self.lock = threading.Lock()

    @classmethod
    def connect(cls, host: str, port: int, timeout: float = 30.0,
                secret: Optional[bytes] = None):
        s = socket.create_connection((host, port), timeout=timeout)
        s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        conn = cls(s)
        conn._client_auth(secret if secret is not None else cluster_secret())
        return conn

    # -- authentication ----------------------------------------------------
    def _client_auth(self, secret: bytes) -> None:
        nonce = self._recv_exact(32)
        mac = hmac.new(secret, nonce, hashlib.sha256).digest()
        self.sock.sendall(mac)
        ok = self._recv_exact(1)
        if ok != b"\x01":
            raise ConnectionError(
                "cluster auth rejected — set the same SCALERL_CLUSTER_SECRET "
                "on every node")

    def server_auth(self, secret: bytes) -> bool:
        """Server side of the handshake; returns False on bad MAC."""
        nonce = _secrets.token_bytes(32)
        self.sock.sendall(nonce)
        want = hmac.new(secret, nonce, hashlib.sha256).digest()
        try:
            got = self._recv_exact(32)
        except (ConnectionResetError, OSError):
            return False
        if not hmac.compare_digest(want, got):
            try:
                self.sock.sendall(b"\x00")
            except OSError:
                pass
            return False
        self.sock.sendall(b"\x01")
        return True

    # -- framing -----------------------------------------------------------
    def _recv_exact(self, n: int) -> bytes:
        buf = bytearray()
        while len(buf) < n:
            chunk = self.sock.recv(n - len(buf))
            if not chunk:
                raise ConnectionResetError("peer closed")
            buf.extend(chunk)
        return bytes(buf)

    def send(self, header: Dict[str, Any],
             tensors: Optional[List[torch.Tensor]] = None) -> None:
        tensors = tensors or []
        meta = {"h": header,
                "t": [(list(t.shape), str(t.dtype).replace("torch.", ""))
                      for t in tensors]}
        mb = json.dumps(meta).encode()
        with self.lock:
            self.sock.sendall(_HDR.pack(len(mb)) + mb)
            for t in tensors:
                b = t.contiguous().cpu().numpy().tobytes()
                self.sock.sendall(_HDR.pack(len(b)) + b)

    def recv(self) -> Tuple[Dict[str, Any], List[torch.Tensor]]:
        n = _HDR.unpack(self._recv_exact(_HDR.size))[0]
        if n > _MAX_META:
            raise ConnectionError(f"metadata frame too large ({n} B)")
        meta = json.loads(self._recv_exact(n).decode())
        tensors = []
        for shape, dtype in meta["t"]:
            if dtype not in _DTYPES:
                raise ConnectionError(f"disallowed tensor dtype {dtype!r}")
            nb = _HDR.unpack(self._recv_exact(_HDR.size))[0]
            if nb > _MAX_TENSOR:
                raise ConnectionError(f"tensor frame too large ({nb} B)")
            raw = self._recv_exact(nb)
            t = torch.frombuffer(bytearray(raw),
                                 dtype=_DTYPES[dtype]).reshape(shape)
            tensors.append(t)
        return meta["h"], tensors

    def close(self) -> None:
        try:
            self.sock.close()
        except OSError:
            pass


class ParameterServer:
    """Versioned flat-weight blob (parameter_server.py:4-33 semantics)."""

    def __init__(self):
        self._weights: Optional[torch.Tensor] = None
        self._version = 0
        self._lock = threading.Lock()

    def push(self, flat_weights: torch.Tensor) -> int:
        with self._lock:
            self._weights = flat_weights.detach().cpu().clone()
            self._version += 1
            return self._version

    def pull(self, have_version: int = -1):
        with self._lock:
            if self._weights is None or have_version == self._version:
                return None, self._version
            return self._weights, self._version


class WorkerServer:
    """Learner-side server: hands out worker ids + config on the entry
    port, then serves weight pulls and accepts episode payloads
    (worker.py:269-297 structure, one port, typed frames)."""

    def __init__(self, config: Dict[str, Any], port: int = 9999,
                 episode_callback: Optional[Callable] = None,
                 retain_episodes: bool = True, bind: Optional[str] = None,
                 secret: Optional[bytes] = None):
        self.config = config
        self.param_server = ParameterServer()
        self.episode_callback = episode_callback
        self.retain_episodes = retain_episodes
        self.episodes: List[Tuple[Dict, List[torch.Tensor]]] = []
        self._next_worker_id = 0
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._secret = secret if secret is not None else cluster_secret()
        # default bind is loopback; cross-node runs opt in with
        # bind="0.0.0.0" (or SCALERL_CLUSTER_BIND) + a shared secret
        bind = bind or os.environ.get("SCALERL_CLUSTER_BIND", "127.0.0.1")
        if bind not in ("127.0.0.1", "localhost", "::1") and not self._secret:
            import warnings
            warnings.warn(
                "cluster server binding a non-loopback interface with an "
                "EMPTY shared secret — any host that can reach the port "
                "can pull weights and push rollouts; set "
                "SCALERL_CLUSTER_SECRET on every node", stacklevel=2)
        self._srv = socket.create_server((bind, port), backlog=64)
        self._srv.settimeout(0.5)
        self.port = self._srv.getsockname()[1]
        self._threads: List[threading.Thread] = []
        self._accept_thread = threading.Thread(target=self._accept_loop,
                                               daemon=True)
        self._accept_thread.start()

    def _accept_loop(self):
        while not self._stop.is_set():
            try:
                sock, _ = self._srv.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            t = threading.Thread(target=self._auth_and_serve,
                                 args=(FrameConnection(sock),), daemon=True)
            t.start()
            self._threads.append(t)

    def _auth_and_serve(self, conn: FrameConnection):
        if not conn.server_auth(self._secret):
            conn.close()
            return
        self._serve(conn)

    def _serve(self, conn: FrameConnection):
        try:
            while not self._stop.is_set():
                header, tensors = conn.recv()
                kind = header.get("kind")
                if kind == "entry":
                    with self._lock:
                        wid = self._next_worker_id
                        self._next_worker_id += 1
                    conn.send({"kind": "entry_ack", "worker_id": wid,
                               "config": self.config})
                elif kind == "pull_weights":
                    w, v = self.param_server.pull(header.get("have_version", -1))
                    conn.send({"kind": "weights", "version": v},
                              [w] if w is not None else [])
                elif kind == "episode":
                    if self.retain_episodes:
                        with self._lock:
                            self.episodes.append((header, tensors))
                    if self.episode_callback is not None:
                        self.episode_callback(header, tensors)
                    conn.send({"kind": "episode_ack"})
                elif kind == "bye":
                    break
        except (ConnectionResetError, OSError, EOFError):
            pass
        finally:
            conn.close()

    def publish_weights(self, flat: torch.Tensor) -> int:
        return self.param_server.push(flat)

    def close(self):
        self._stop.set()
        try:
            self._srv.close()
        except OSError:
            pass


class RemoteWorkerCluster:
    """Remote-node client: entry handshake, then a pull/push session
    (worker.py:300-341 role).  Drives local actor work via a user-provided
    ``generate`` callable: generate(config, weights) → (header, tensors)."""

    def __init__(self, host: str, port: int):
        self.host, self.port = host, port
        self.conn = FrameConnection.connect(host, port)
        self.conn.send({"kind": "entry"})
        ack, _ = self.conn.recv()
        assert ack["kind"] == "entry_ack"
        self.worker_id = ack["worker_id"]
        self.config = ack["config"]
        self._weights: Optional[torch.Tensor] = None
        self._version = -1

    def pull_weights(self) -> Optional[torch.Tensor]:
        self.conn.send({"kind": "pull_weights", "have_version": self._version})
        h, tensors = self.conn.recv()
        if tensors:
            self._weights = tensors[0]
            self._version = h["version"]
        return self._weights

    def push_episode(self, header: Dict[str, Any],
                     tensors: List[torch.Tensor]) -> None:
        self.conn.send(dict(header, kind="episode",
                            worker_id=self.worker_id), tensors)
        ack, _ = self.conn.recv()
        assert ack["kind"] == "episode_ack"

    def run(self, generate: Callable, iterations: int = 0) -> None:
        i = 0
        while iterations == 0 or i < iterations:
            weights = self.pull_weights()
            header, tensors = generate(self.config, weights)
            self.push_episode(header, tensors)
            i += 1

    def close(self):
        try:
            self.conn.send({"kind": "bye"})
        except OSError:
            pass
        self.conn.close()
