"""Device-resident connection-state table (BASELINE config 5).

The reference scales concurrent connections with goroutine-per-conn
(reference httpServer.go:35 — the Go runtime owns all conn state in host
memory). The MI355X design keeps per-connection state in HBM3E instead:
one row per connection (protocol, flags, counters, last activity) plus a
per-connection receive window where partial requests park between
batches, so a 100k-conn (and beyond) working set never touches host
memory on the hot path.

Sizing (the "288 GB HBM conn-state" requirement): a connection costs
``NS*8`` state bytes + ``window`` staging bytes — 16 KiB + 48 B at the
default window. 100k connections ≈ 1.6 GB ≈ 0.6% of one MI355X's
288 GB; `sized_for_hbm()` computes the actual ceiling from the device's
free memory (≈ 8.8M connections per GPU at a 50% reservation), and the
table refuses to oversubscribe what the device reports free.

Host-side id management (open/close free list) is control-plane work and
stays on the CPU, mirroring the reference's accept-loop split; the data
plane (`record_batch`) is pure device ops (index_add_ scatter updates on
the current stream; bench.py exercises it via --grpc-frac/--conns).

Without a GPU the same table runs on numpy so every semantic is testable
on the CPU box (tests/test_connstate.py).
"""

from __future__ import annotations

import numpy as np

# state row layout (int64 [capacity, NS])
CS_PROTO = 0       # 0 = http/1.1, 1 = grpc (http/2)
CS_FLAGS = 1       # bit0 = open
CS_REQS = 2        # requests served
CS_BYTES_IN = 3
CS_BYTES_OUT = 4
CS_LAST_BATCH = 5  # batch number of last activity (engine batch clock)
NS = 6

PROTO_HTTP = 0
PROTO_GRPC = 1

DEFAULT_WINDOW = 16384


def per_conn_bytes(window: int = DEFAULT_WINDOW) -> int:
    """HBM footprint of one connection: state row + receive window."""
    return NS * 8 + window


class ConnStateTable:
    """Fixed-capacity per-connection state, resident in HBM (or numpy on
    the CPU box)."""

    def __init__(self, capacity: int, device=None,
                 window: int = DEFAULT_WINDOW):
        self.capacity = int(capacity)
        self.window = int(window)
        self.device = None
        self.torch = None
        if device is not None and str(device) != "cpu":
            import torch
            if torch.cuda.is_available():
                self.torch = torch
                self.device = torch.device(device)
        if self.device is not None:
            t = self.torch
            free, _total = t.cuda.mem_get_info(self.device)
            need = self.capacity * per_conn_bytes(self.window)
            if need > free:
                raise MemoryError(
                    f"conn table needs {need >> 20} MiB but device has "
                    f"{free >> 20} MiB free — lower capacity/window")
            self.state = t.zeros((self.capacity, NS), dtype=t.int64,
                                 device=self.device)
            self.rx = t.zeros((self.capacity, self.window), dtype=t.uint8,
                              device=self.device)
            # staged partial-request byte counts (device-side, so the
            # harvest kernel path can append without host round-trips)
            self.rx_len = t.zeros(self.capacity, dtype=t.int32,
                                  device=self.device)
        else:
            self.state = np.zeros((self.capacity, NS), np.int64)
            self.rx = None  # CPU mirror doesn't reserve staging windows
            self.rx_len = np.zeros(self.capacity, np.int32)
        self._free = list(range(self.capacity - 1, -1, -1))
        self.n_open = 0

    # ---- sizing -------------------------------------------------------------
    @classmethod
    def sized_for_hbm(cls, device=None, frac: float = 0.5,
                      window: int = DEFAULT_WINDOW,
                      hbm_bytes: int | None = None) -> int:
        """Connection capacity that fits in `frac` of the device's free
        HBM (or of `hbm_bytes` when given — e.g. 288e9 for planning)."""
        if hbm_bytes is None:
            import torch
            free, _ = torch.cuda.mem_get_info(device)
            hbm_bytes = free
        return int(hbm_bytes * frac) // per_conn_bytes(window)

    # ---- control plane (host) -----------------------------------------------
    def open(self, n: int, proto: int = PROTO_HTTP) -> np.ndarray:
        """Allocate n connection ids (LIFO free list — hot ids reuse warm
        HBM rows). Marks rows open with the given protocol."""
        if n > len(self._free):
            raise RuntimeError(
                f"conn table full: {self.n_open}/{self.capacity} open")
        ids = np.asarray([self._free.pop() for _ in range(n)], np.int64)
        self.n_open += n
        if self.device is not None:
            t = self.torch
            idx = t.from_numpy(ids).to(self.device)
            self.state[idx, CS_PROTO] = int(proto)
            self.state[idx, CS_FLAGS] = 1
        else:
            self.state[ids, CS_PROTO] = int(proto)
            self.state[ids, CS_FLAGS] = 1
        return ids

    def close(self, ids: np.ndarray) -> None:
        if self.device is not None:
            idx = self.torch.from_numpy(np.asarray(ids)).to(self.device)
            self.state[idx] = 0
            self.rx_len[idx] = 0
        else:
            self.state[np.asarray(ids)] = 0
            self.rx_len[np.asarray(ids)] = 0
        self._free.extend(int(i) for i in ids)
        self.n_open -= len(ids)

    # ---- data plane (device, capturable) ------------------------------------
    def record_batch(self, ids_t, bytes_in_t, bytes_out_t,
                     batch_no: int) -> None:
        """Scatter-update per-conn counters for one processed batch.
        ids/bytes_* are device tensors (int64/int64); runs entirely on
        the current stream — safe inside hipGraph capture."""
        if self.device is not None:
            t = self.torch
            ones = t.ones_like(ids_t)
            self.state[:, CS_REQS].index_add_(0, ids_t, ones)
            self.state[:, CS_BYTES_IN].index_add_(0, ids_t, bytes_in_t)
            self.state[:, CS_BYTES_OUT].index_add_(0, ids_t, bytes_out_t)
            self.state[:, CS_LAST_BATCH].index_fill_(0, ids_t,
                                                     int(batch_no))
        else:
            ids = np.asarray(ids_t)
            np.add.at(self.state[:, CS_REQS], ids, 1)
            np.add.at(self.state[:, CS_BYTES_IN], ids,
                      np.asarray(bytes_in_t))
            np.add.at(self.state[:, CS_BYTES_OUT], ids,
                      np.asarray(bytes_out_t))
            self.state[ids, CS_LAST_BATCH] = int(batch_no)

    # ---- introspection ------------------------------------------------------
    def stats(self, ids: np.ndarray) -> np.ndarray:
        """Rows for the given ids, as numpy [len(ids), NS]."""
        if self.device is not None:
            idx = self.torch.from_numpy(np.asarray(ids)).to(self.device)
            return self.state[idx].cpu().numpy()
        return self.state[np.asarray(ids)].copy()

    def hbm_bytes(self) -> int:
        return self.capacity * per_conn_bytes(self.window)
