"""Torch-free TCP rendezvous for multi-process GPU runs.

A tiny lockstep key-exchange store: rank 0 listens, every other rank keeps
one connected socket.  All ranks call the same operations in the same order
(SPMD), so rank 0 can serve each operation synchronously inside its own
call — no server thread, no external dependencies (stdlib socket/struct
only).  Used to exchange the 128-byte ncclUniqueId, for host barriers and
for the bench's max-over-ranks reduction; after bootstrap RCCL runs
natively over xGMI with nothing in the communication path.

This makes torch.distributed strictly optional for GPU runs (it remains
the transport for multi-process *CPU* testing via the gloo allreduce
callback).
"""
import os
import socket
import struct
import time

_OP_BARRIER = 1
_OP_BCAST = 2
_OP_MAX = 3

_HDR = struct.Struct("!BI")  # opcode, payload length


def _send_msg(sock, op, payload=b""):
    sock.sendall(_HDR.pack(op, len(payload)) + payload)


def _recv_exact(sock, n):
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("rendezvous peer closed the connection")
        buf += chunk
    return buf


def _recv_msg(sock):
    op, ln = _HDR.unpack(_recv_exact(sock, _HDR.size))
    return op, _recv_exact(sock, ln) if ln else b""


class TcpStore:
    """Lockstep rendezvous over one TCP socket per non-zero rank."""

    def __init__(self, addr, port, rank, world, timeout=120.0):
        self.rank = rank
        self.world = world
        self.timeout = timeout
        self._socks = {}   # rank 0: peer rank -> socket
        self._sock = None  # rank > 0: socket to rank 0
        if world == 1:
            return
        if rank == 0:
            srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            srv.bind((addr, port))
            srv.listen(world)
            srv.settimeout(timeout)
            deadline = time.monotonic() + timeout
            while len(self._socks) < world - 1:
                if time.monotonic() > deadline:
                    raise TimeoutError(
                        f"rendezvous: only {len(self._socks)}/{world - 1} "
                        f"peers connected within {timeout}s")
                conn, _ = srv.accept()
                conn.settimeout(timeout)
                conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                (peer,) = struct.unpack("!I", _recv_exact(conn, 4))
                self._socks[peer] = conn
            srv.close()
        else:
            deadline = time.monotonic() + timeout
            last_err = None
            while True:
                try:
                    s = socket.create_connection((addr, port), timeout=5.0)
                    break
                except OSError as e:
                    last_err = e
                    if time.monotonic() > deadline:
                        raise TimeoutError(
                            f"rendezvous: rank {rank} could not reach "
                            f"{addr}:{port} within {timeout}s: {last_err}")
                    time.sleep(0.2)
            s.settimeout(timeout)
            s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            s.sendall(struct.pack("!I", rank))
            self._sock = s

    # -- collective ops (all ranks must call each in the same order) -------
    def barrier(self):
        if self.world == 1:
            return
        if self.rank == 0:
            for s in self._socks.values():
                op, _ = _recv_msg(s)
                assert op == _OP_BARRIER
            for s in self._socks.values():
                _send_msg(s, _OP_BARRIER)
        else:
            _send_msg(self._sock, _OP_BARRIER)
            op, _ = _recv_msg(self._sock)
            assert op == _OP_BARRIER

    def broadcast_bytes(self, data=None):
        """Rank 0 provides `data`; every rank returns it."""
        if self.world == 1:
            return data
        if self.rank == 0:
            assert data is not None
            for s in self._socks.values():
                op, _ = _recv_msg(s)       # request
                assert op == _OP_BCAST
                _send_msg(s, _OP_BCAST, data)
            return data
        _send_msg(self._sock, _OP_BCAST)
        op, payload = _recv_msg(self._sock)
        assert op == _OP_BCAST
        return payload

    def all_max(self, value):
        """Max of a float over all ranks."""
        if self.world == 1:
            return value
        if self.rank == 0:
            vals = [value]
            for s in self._socks.values():
                op, payload = _recv_msg(s)
                assert op == _OP_MAX
                vals.append(struct.unpack("!d", payload)[0])
            out = max(vals)
            for s in self._socks.values():
                _send_msg(s, _OP_MAX, struct.pack("!d", out))
            return out
        _send_msg(self._sock, _OP_MAX, struct.pack("!d", value))
        op, payload = _recv_msg(self._sock)
        assert op == _OP_MAX
        return struct.unpack("!d", payload)[0]

    def close(self):
        for s in self._socks.values():
            s.close()
        if self._sock:
            self._sock.close()
        self._socks = {}
        self._sock = None


def from_env(rank, world, timeout=120.0):
    """Build a TcpStore from the standard torchrun env (MASTER_ADDR/PORT).

    Uses MASTER_PORT+1 (or MEGBA_STORE_PORT) so it never collides with a
    torchrun elastic store that may own MASTER_PORT itself.
    """
    addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = int(os.environ.get("MEGBA_STORE_PORT",
                              int(os.environ.get("MASTER_PORT", "29500")) + 1))
    return TcpStore(addr, port, rank, world, timeout)
