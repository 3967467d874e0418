"""N-stream TCP bulk weight transfer for the multi-node split.

Reference capability: rlboost/weight_transfer/transfer_engine.py
(TCPTransferEngine, SURVEY.md §2.1 'Transfer engine' row): the receiver
pre-registers a buffer and runs one listener thread per stream doing
``recv_into`` slices (zero-copy into the registered memory); the sender
splits the buffer into per-stream spans, each prefixed with a 16-byte
(offset, length) header, and pushes them over parallel sockets.  Async
submit/poll API mirrors transfer_submit_write / transfer_check_status
(transfer_engine.py:195-274).

On one MI355X node this plane is never used (transfer/collective.py moves
bytes over xGMI); it exists for elastic REMOTE instances on other machines.
"""
from __future__ import annotations

import socket
import struct
import threading
from typing import Dict, List, Optional, Tuple

import torch

_HDR = struct.Struct("<QQ")          # offset, length


def addr_allowed(host: Optional[str], cidrs: Optional[List[str]]) -> bool:
    """CIDR allow-list for transfer peers (the reference's
    allowed_sender_ips filter, utils.rs:303-339 / weight_transfer
    utils.py:12-53).  No list (or unknown host) => allow."""
    if not cidrs or host is None:
        return True
    import ipaddress
    try:
        ip = ipaddress.ip_address(host)
    except ValueError:
        return False
    for c in cidrs:
        try:
            if ip in ipaddress.ip_network(c, strict=False):
                return True
        except ValueError:
            continue
    return False
SOCK_BUF = 16 << 20                  # 16 MB socket buffers (config.toml tune)
CHUNK = 4 << 20


class TcpWeightReceiver:
    """Listens on ``num_streams`` ports; every incoming span lands directly
    in the registered buffer via recv_into."""

    def __init__(self, buffer: torch.Tensor, host: str = "127.0.0.1",
                 num_streams: int = 4):
        assert buffer.dtype == torch.uint8 and buffer.is_contiguous()
        assert not buffer.is_cuda, "register a CPU (shm/pinned) buffer"
        self.buffer = buffer
        self._mv = memoryview(buffer.numpy())
        self.host = host
        self.num_streams = num_streams
        self._listeners: List[socket.socket] = []
        self.ports: List[int] = []
        self._threads: List[threading.Thread] = []
        self._received = 0
        self._lock = threading.Lock()
        self._done = threading.Event()
        self._expected: Optional[int] = None
        self._failed: Optional[str] = None
        for _ in range(num_streams):
            s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            s.setsockopt(socket.SOL_SOCKET, socket.SO_RCVBUF, SOCK_BUF)
            s.bind((host, 0))
            s.listen(1)
            self._listeners.append(s)
            self.ports.append(s.getsockname()[1])

    def expect(self, total_bytes: Optional[int] = None):
        """Arm for one transfer of ``total_bytes`` (default: whole buffer)."""
        with self._lock:
            self._received = 0
            self._failed = None
        self._done.clear()
        self._expected = total_bytes if total_bytes is not None \
            else self.buffer.numel()
        self._threads = []
        for ls in self._listeners:
            t = threading.Thread(target=self._serve_one, args=(ls,),
                                 daemon=True)
            t.start()
            self._threads.append(t)

    def _serve_one(self, ls: socket.socket):
        try:
            conn, _ = ls.accept()
        except OSError:
            return                        # listener closed
        conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        try:
            while True:
                hdr = self._recv_exact(conn, _HDR.size)
                if hdr is None:
                    return
                off, length = _HDR.unpack(hdr)
                if length == 0:      # stream end marker
                    return
                if off < 0 or length < 0 or off + length > len(self._mv):
                    # Fail fast instead of hanging until the install
                    # timeout: flag the transfer failed and wake wait().
                    with self._lock:
                        self._failed = (
                            f"span ({off}, {length}) exceeds buffer "
                            f"size {len(self._mv)}")
                    self._done.set()
                    return                # error signaled via _failed
                got = 0
                while got < length:
                    n = conn.recv_into(self._mv[off + got: off + length],
                                       min(length - got, CHUNK))
                    if n == 0:
                        return            # peer closed mid-span; the
                                          # byte count will not complete
                                          # and wait() times out cleanly
                    got += n
                with self._lock:
                    self._received += length
                    if self._expected is not None and \
                            self._received >= self._expected:
                        self._done.set()
        finally:
            conn.close()

    @staticmethod
    def _recv_exact(conn, n) -> Optional[bytes]:
        buf = b""
        while len(buf) < n:
            d = conn.recv(n - len(buf))
            if not d:
                return None
            buf += d
        return buf

    def wait(self, timeout: float = 300.0) -> bool:
        ok = self._done.wait(timeout)
        if self._failed is not None:
            raise RuntimeError(f"tcp weight receive failed: {self._failed}")
        for t in self._threads:
            t.join(timeout=5.0)
        return ok

    def close(self):
        for s in self._listeners:
            s.close()


class TcpWeightSender:
    """Pushes a CPU buffer to a receiver's (host, ports) over parallel
    streams; async submit + poll (the reference's transfer_submit_write /
    transfer_check_status surface)."""

    def __init__(self, num_streams: int = 4):
        self.num_streams = num_streams
        self._status: Dict[int, str] = {}
        self._batch = 0
        self._lock = threading.Lock()

    def submit(self, buffer: torch.Tensor, host: str,
               ports: List[int]) -> int:
        assert buffer.dtype == torch.uint8 and buffer.is_contiguous()
        with self._lock:
            self._batch += 1
            bid = self._batch
            self._status[bid] = "running"
        mv = memoryview(buffer.numpy())
        total = buffer.numel()
        n = min(len(ports), self.num_streams)
        spans: List[Tuple[int, int]] = []
        per = (total + n - 1) // n
        for i in range(n):
            off = i * per
            spans.append((off, min(per, total - off)))

        state = {"left": n, "err": None}

        def push(port, off, length):
            try:
                s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
                s.setsockopt(socket.SOL_SOCKET, socket.SO_SNDBUF, SOCK_BUF)
                s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                s.connect((host, port))
                s.sendall(_HDR.pack(off, length))
                sent = 0
                while sent < length:
                    sent += s.send(mv[off + sent: off + min(sent + CHUNK,
                                                            length)])
                s.sendall(_HDR.pack(0, 0))   # end marker
                s.close()
            except Exception as e:           # noqa: BLE001
                state["err"] = e
            finally:
                with self._lock:
                    state["left"] -= 1
                    if state["left"] == 0:
                        self._status[bid] = \
                            "failed" if state["err"] else "done"

        for i, (off, length) in enumerate(spans):
            if length <= 0:
                with self._lock:
                    state["left"] -= 1
                continue
            threading.Thread(target=push, args=(ports[i], off, length),
                             daemon=True).start()
        return bid

    def check_status(self, bid: int) -> str:
        with self._lock:
            return self._status.get(bid, "unknown")

    def wait(self, bid: int, timeout: float = 300.0) -> str:
        import time
        t0 = time.monotonic()
        while time.monotonic() - t0 < timeout:
            st = self.check_status(bid)
            if st in ("done", "failed"):
                return st
            time.sleep(0.005)
        return "timeout"


def push_state_dict_tcp(state_dict, client, host: str,
                        num_streams: int = 4, version: int = 1,
                        timeout_s: float = 600.0,
                        compress: Optional[str] = None) -> bool:
    """Sender-agent data path to ONE remote instance (sender_agent.py:390-427
    capability): handshake -> N-stream TCP push of the flattened state dict
    -> install call.  ``client`` is an httpx.Client/AsyncClient-compatible
    SYNC client bound to the instance's base URL; ``host`` is the address
    its TCP ports are reachable at.

    ``compress="fp8"`` quantizes float tensors to e4m3 with a per-tensor
    scale before the push — HALF the bytes of bf16 on the wire (the
    reference's own roadmap left 'weight compression before transfer'
    unchecked).  The receiver dequantizes on install; rollout-side only
    (the trainer never reads these weights back)."""
    import torch
    metas = []
    flats = []
    for name in sorted(state_dict):
        t = state_dict[name]
        t = t.full_tensor() if hasattr(t, "full_tensor") else t
        t = t.detach().cpu().contiguous()
        if compress == "fp8" and t.is_floating_point():
            scale = float(t.abs().amax().clamp(min=1e-12)) / 448.0
            q = (t.float() / scale).clamp(-448.0, 448.0) \
                .to(torch.float8_e4m3fn)
            metas.append((name, list(t.shape), "float8_e4m3fn", scale))
            flats.append(q.reshape(-1).view(torch.uint8))
        else:
            metas.append((name, list(t.shape),
                          str(t.dtype).replace("torch.", "")))
            flats.append(t.reshape(-1).view(torch.uint8).reshape(-1)
                         if t.dtype == torch.uint8
                         else t.reshape(-1).contiguous().view(torch.uint8))
    buf = torch.cat([f.reshape(-1) for f in flats])
    r = client.post("/weights_handshake",
                    json={"metas": metas, "num_streams": num_streams},
                    timeout=60.0)
    if r.status_code != 200:
        return False
    ports = r.json()["ports"]
    tx = TcpWeightSender(num_streams=num_streams)
    bid = tx.submit(buf, host, ports)
    if tx.wait(bid, timeout=timeout_s) != "done":
        return False
    r = client.post("/update_weights_from_tcp",
                    json={"version": version, "timeout_s": timeout_s},
                    timeout=timeout_s)
    return r.status_code == 200 and r.json().get("success", False)
