"""GPU request-batch engine.

The MI355X data plane (SURVEY.md §7 phase 2): request bytes are staged
into pinned ring buffers, one H2D copy moves the batch onto the device,
k_parse_route + k_respond (native/hip/gofr_kernels.hip) parse, route and
serialize the whole batch, and one D2H copy brings the response bytes
back. Handlers with a GPU spec (gofr_amd/handlers.py) never touch the
host; the rest run through the host trampoline between the two kernels
(only for the requests that need it — a device-side counter tells the
host whether any do, so the pure-GPU path never blocks on Python).

Without a GPU the engine runs the byte-exact CPU mirrors
(gofr_amd/ops), so every test of this module runs on the CPU box.
"""

from __future__ import annotations

import time

import numpy as np

from .. import ops
from ..http.request import parse_request_bytes
from ..server import dispatch

_CT_IDS = {"application/json": 0, "image/x-icon": 1,
           "application/octet-stream": 2, "text/plain": 3}

INVALID_BODY_ENV = b'{"error":{"message":"invalid JSON body"}}'
NOTFOUND_ENV = b'{"error":{"message":"http: no such file"}}'


class RouteProgram:
    """Compiled device image of an app's route table.

    blob layout: [int32 len][invalid-body envelope][static args...]
    handler_tab: int32 [n_routes, 4] = (kind, arg_off, arg_len, status)
    """

    def __init__(self, app):
        app.install_default_routes()
        self.app = app
        self.trie = app.router.compile()
        blob = bytearray()
        blob += len(INVALID_BODY_ENV).to_bytes(4, "little")
        blob += INVALID_BODY_ENV
        rows = []
        self.py_handlers = []
        for route in app.router.routes:
            spec = getattr(route.handler, "__gofr_gpu__", None)
            self.py_handlers.append(route.handler)
            if route.is_prefix and spec is None:
                # default catch-all: 404 envelope served on-device
                spec = ("static", NOTFOUND_ENV, 404)
            if spec is None:
                rows.append((ops.HK_HOST, 0, 0, 200))
            elif spec[0] == "echo_json":
                rows.append((ops.HK_ECHO_JSON, 0, 0, 200))
            elif spec[0] == "static":
                off = len(blob)
                body = spec[1]
                status = spec[2] if len(spec) > 2 else 200
                blob += body
                rows.append((ops.HK_STATIC, off, len(body), status))
            else:
                rows.append((ops.HK_HOST, 0, 0, 200))
        self.handler_tab = np.asarray(rows, np.int32).reshape(-1)
        self.n_routes = len(rows)
        self.blob = bytes(blob)


class _Lane:
    """One pipeline stage: its own HIP stream + device/pinned buffer set.

    The serving loop round-robins lanes so the H2D of batch i+1 and the
    D2H of batch i-1 overlap the kernels of batch i (separate streams,
    pinned staging both ways).
    """

    def __init__(self, t, dev, nb, slot, rslot, host_blob_cap=4 << 20):
        self.stream = t.cuda.Stream(device=dev)
        self.event = t.cuda.Event()
        self.d_reqs = t.empty(nb * slot, dtype=t.uint8, device=dev)
        self.d_req_len = t.empty(nb, dtype=t.int32, device=dev)
        self.d_fields = t.zeros(nb * ops.NF, dtype=t.int32, device=dev)
        self.d_resp = t.empty(nb * rslot, dtype=t.uint8, device=dev)
        self.d_resp_len = t.empty(nb, dtype=t.int32, device=dev)
        self.d_host_needed = t.zeros(1, dtype=t.int32, device=dev)
        self.d_host_tab = t.zeros(nb * 4, dtype=t.int32, device=dev)
        self.d_host_blob = t.zeros(host_blob_cap, dtype=t.uint8, device=dev)
        self.p_reqs = t.empty(nb * slot, dtype=t.uint8).pin_memory()
        self.p_req_len = t.empty(nb, dtype=t.int32).pin_memory()
        self.p_resp = t.empty(nb * rslot, dtype=t.uint8).pin_memory()
        self.p_resp_len = t.empty(nb, dtype=t.int32).pin_memory()
        self.p_fields = t.empty(nb * ops.NF, dtype=t.int32).pin_memory()
        self.p_host_needed = t.zeros(1, dtype=t.int32).pin_memory()
        self.n = 0
        self.seed = 0


class BatchEngine:
    """Processes request batches through the GPU kernels (or CPU mirrors)."""

    def __init__(self, app, device=None, slot: int = 2048,
                 rslot: int = 0, max_batch: int = 65536,
                 require_gpu: bool = False, pipeline: int = 1):
        self.app = app
        self.slot = slot
        # response slot must hold worst-case: headers(~260) + envelope(9) +
        # a body as large as the request slot
        self.rslot = rslot if rslot else slot + 512
        assert self.rslot >= slot + 512, "rslot too small for worst case"
        self.max_batch = max_batch
        self.pipeline = max(1, pipeline)
        self.program = RouteProgram(app)
        self._seed = 0x6F667247414D4421  # advanced per batch
        self.device = None
        self.torch = None
        try:
            import torch
            if device is None and torch.cuda.is_available():
                device = "cuda"
            if device is not None and str(device).startswith("cuda") \
                    and torch.cuda.is_available():
                self.torch = torch
                self.device = torch.device(device)
        except ImportError:
            pass
        if self.device is not None:
            self.hip = ops.HipOps()  # raises if extension missing
            self._alloc_device()
        elif require_gpu:
            raise RuntimeError(
                "BatchEngine: GPU required but torch.cuda unavailable")

    # -- device state --------------------------------------------------------
    def _alloc_device(self):
        t, dev = self.torch, self.device
        nb, slot, rslot = self.max_batch, self.slot, self.rslot
        tr = {}
        for k, v in self.program.trie.items():
            if k == "n_nodes":
                continue
            tr[k] = t.as_tensor(np.ascontiguousarray(v)).to(dev)
        self.d_trie = tr
        self.d_handler_tab = t.as_tensor(self.program.handler_tab).to(dev)
        self.d_blob = t.as_tensor(
            np.frombuffer(self.program.blob, np.uint8).copy()).to(dev)
        self.lanes = [_Lane(t, dev, nb, slot, rslot)
                      for _ in range(self.pipeline)]
        # lane-0 aliases: the synchronous API and the multi-GPU path
        ln = self.lanes[0]
        self.d_reqs, self.d_req_len = ln.d_reqs, ln.d_req_len
        self.d_fields, self.d_resp = ln.d_fields, ln.d_resp
        self.d_resp_len = ln.d_resp_len
        self.d_host_needed = ln.d_host_needed
        self.d_host_tab, self.d_host_blob = ln.d_host_tab, ln.d_host_blob
        self.p_reqs, self.p_req_len = ln.p_reqs, ln.p_req_len
        self.p_resp, self.p_resp_len = ln.p_resp, ln.p_resp_len
        self.p_fields = ln.p_fields

    def _next_seed(self) -> int:
        self._seed = ops.splitmix64(self._seed)
        return self._seed

    # -- main entry ----------------------------------------------------------
    def process(self, reqs: np.ndarray, req_len: np.ndarray):
        """reqs: uint8 [n*slot], req_len: int32 [n].
        Returns (resp uint8 [n*rslot], resp_len int32 [n])."""
        n = len(req_len)
        assert n <= self.max_batch
        if self.device is None:
            return self._process_cpu(reqs, req_len)
        return self._process_gpu(reqs, req_len, n)

    # CPU fallback: byte-exact mirrors (never used on a GPU box)
    def _process_cpu(self, reqs, req_len):
        seed = self._next_seed()
        fields = ops.cpu_parse_route(reqs, req_len, self.slot,
                                     self.program.trie,
                                     self.program.handler_tab)
        host_blob, host_tab = self._run_host_rows(fields, reqs, req_len)
        return ops.cpu_respond(
            reqs, fields, self.slot, self.rslot, self.program.handler_tab,
            self.program.blob, host_blob, host_tab, seed)

    def _process_gpu(self, reqs, req_len, n):
        t = self.torch
        nb_req = n * self.slot
        self.p_reqs[:nb_req] = t.from_numpy(reqs[:nb_req].view(np.uint8))
        self.p_req_len[:n] = t.from_numpy(req_len.astype(np.int32,
                                                         copy=False))
        p_resp, p_rlen = self.process_pinned(n)
        return p_resp.numpy().copy(), p_rlen.numpy().copy()

    # -- pipelined API --------------------------------------------------------
    def submit(self, n: int, lane_idx: int = 0) -> None:
        """Enqueue one batch (already staged in lane.p_reqs/p_req_len[:n])
        on the lane's stream: H2D -> parse -> respond (optimistic) -> D2H.
        Never blocks. complete() finishes it."""
        t = self.torch
        ln = self.lanes[lane_idx]
        slot, rslot = self.slot, self.rslot
        ln.n = n
        ln.seed = self._next_seed()
        with t.cuda.stream(ln.stream):
            ln.d_reqs[:n * slot].copy_(ln.p_reqs[:n * slot],
                                       non_blocking=True)
            ln.d_req_len[:n].copy_(ln.p_req_len[:n], non_blocking=True)
            ln.d_host_needed.zero_()
            cs = ln.stream.cuda_stream
            self.hip.parse_route(cs, ln.d_reqs, ln.d_req_len, ln.d_fields,
                                 n, slot, self.d_trie, self.d_handler_tab,
                                 self.program.n_routes, ln.d_host_needed)
            ln.p_host_needed.copy_(ln.d_host_needed, non_blocking=True)
            # optimistic respond: host rows render a 500 fallback that the
            # fixup pass overwrites before the responses are released
            self.hip.respond(cs, ln.d_reqs, ln.d_fields, ln.d_resp,
                             ln.d_resp_len, n, slot, rslot,
                             self.d_handler_tab, self.program.n_routes,
                             self.d_blob, ln.d_host_blob, ln.d_host_tab,
                             ln.seed)
            ln.p_resp[:n * rslot].copy_(ln.d_resp[:n * rslot],
                                        non_blocking=True)
            ln.p_resp_len[:n].copy_(ln.d_resp_len[:n], non_blocking=True)
            ln.event.record(ln.stream)

    def complete(self, lane_idx: int = 0):
        """Wait for the lane's in-flight batch; run the host fixup pass if
        any row needed the trampoline. Returns pinned (resp, resp_len)."""
        t = self.torch
        ln = self.lanes[lane_idx]
        n, slot, rslot = ln.n, self.slot, self.rslot
        ln.event.synchronize()
        if int(ln.p_host_needed[0]):
            # fixup: run Python handlers for HK_HOST rows, re-serialize
            with t.cuda.stream(ln.stream):
                ln.p_fields[:n * ops.NF].copy_(ln.d_fields[:n * ops.NF],
                                               non_blocking=True)
            ln.stream.synchronize()
            fields = ln.p_fields[:n * ops.NF].numpy().reshape(n, ops.NF)
            host_reqs = ln.p_reqs[:n * slot].numpy()
            host_req_len = ln.p_req_len[:n].numpy()
            host_blob, host_tab = self._run_host_rows(
                fields, host_reqs, host_req_len)
            hb = np.frombuffer(host_blob, np.uint8)
            with t.cuda.stream(ln.stream):
                if len(hb):
                    ln.d_host_blob[:len(hb)].copy_(
                        t.from_numpy(hb.copy()), non_blocking=True)
                ln.d_host_tab[:n * 4].copy_(
                    t.from_numpy(host_tab.reshape(-1).copy()),
                    non_blocking=True)
                self.hip.respond(ln.stream.cuda_stream, ln.d_reqs,
                                 ln.d_fields, ln.d_resp, ln.d_resp_len,
                                 n, slot, rslot, self.d_handler_tab,
                                 self.program.n_routes, self.d_blob,
                                 ln.d_host_blob, ln.d_host_tab, ln.seed)
                ln.p_resp[:n * rslot].copy_(ln.d_resp[:n * rslot],
                                            non_blocking=True)
                ln.p_resp_len[:n].copy_(ln.d_resp_len[:n],
                                        non_blocking=True)
            ln.stream.synchronize()
        return ln.p_resp[:n * rslot], ln.p_resp_len[:n]

    def process_pinned(self, n, lane_idx: int = 0):
        """Synchronous one-batch pipeline on a lane (requests staged in
        lane.p_reqs/p_req_len). The socket layer recv()s directly into the
        pinned ring, so H2D + kernels + D2H is the whole per-batch path."""
        self.submit(n, lane_idx)
        return self.complete(lane_idx)

    def process_device(self, d_reqs, d_req_len, n, host_reqs=None,
                       host_req_len=None):
        """Run the kernel pipeline on request bytes already resident on the
        device (the multi-GPU all-to-all path hands exchanged slabs in
        directly). Runs on the CALLER's current stream; returns
        (d_resp, d_resp_len) device tensors, no D2H."""
        t = self.torch
        ln = self.lanes[0]
        slot, rslot = self.slot, self.rslot
        seed = self._next_seed()
        stream = t.cuda.current_stream(self.device).cuda_stream
        ln.d_host_needed.zero_()
        self.hip.parse_route(stream, d_reqs, d_req_len,
                             ln.d_fields, n, slot, self.d_trie,
                             self.d_handler_tab, self.program.n_routes,
                             ln.d_host_needed)
        # host trampoline only when some row needs it (4-byte D2H + sync)
        host_needed = int(ln.d_host_needed.item())
        if host_needed:
            ln.p_fields[:n * ops.NF].copy_(ln.d_fields[:n * ops.NF])
            t.cuda.synchronize(self.device)
            fields = ln.p_fields[:n * ops.NF].numpy().reshape(n, ops.NF)
            if host_reqs is None:
                host_reqs = d_reqs[:n * slot].cpu().numpy()
                host_req_len = d_req_len[:n].cpu().numpy()
            host_blob, host_tab = self._run_host_rows(
                fields, host_reqs, host_req_len)
            hb = np.frombuffer(host_blob, np.uint8)
            if len(hb):
                ln.d_host_blob[:len(hb)].copy_(
                    t.from_numpy(hb.copy()), non_blocking=True)
            ln.d_host_tab[:n * 4].copy_(
                t.from_numpy(host_tab.reshape(-1).copy()),
                non_blocking=True)
        self.hip.respond(stream, d_reqs, ln.d_fields, ln.d_resp,
                         ln.d_resp_len, n, slot, rslot,
                         self.d_handler_tab, self.program.n_routes,
                         self.d_blob, ln.d_host_blob, ln.d_host_tab,
                         seed)
        return ln.d_resp, ln.d_resp_len


    # -- host trampoline ------------------------------------------------------
    def _run_host_rows(self, fields, reqs, req_len):
        """Run Python handlers for HK_HOST rows; returns (blob, tab)."""
        n = len(req_len)
        host_tab = np.zeros((n, 4), np.int32)
        blob = bytearray()
        for r in range(n):
            if fields[r][ops.FI_KIND] != ops.HK_HOST:
                continue
            raw = reqs[r * self.slot:r * self.slot + int(req_len[r])] \
                .tobytes()
            try:
                request = parse_request_bytes(raw)
                resp = dispatch(self.app, request)
                status, body = resp.status, resp.body
                ct = dict(resp.headers).get("Content-Type",
                                            "application/json")
            except (ValueError, KeyError):
                status = 400
                body = b'{"error":{"message":"malformed request"}}'
                ct = "application/json"
            off = len(blob)
            blob += body
            host_tab[r] = (off, len(body), status, _CT_IDS.get(ct, 0))
        return bytes(blob), host_tab


def make_batch(payloads: list[bytes], slot: int):
    """Pack raw request byte strings into the engine's slot layout."""
    n = len(payloads)
    reqs = np.zeros(n * slot, np.uint8)
    lens = np.zeros(n, np.int32)
    for i, p in enumerate(payloads):
        assert len(p) <= slot, "request exceeds slot size"
        reqs[i * slot:i * slot + len(p)] = np.frombuffer(p, np.uint8)
        lens[i] = len(p)
    return reqs, lens


class GPUServer:
    """Socket front-end serving through the batch engine.

    Accepts connections with the CPU listener machinery, forms batches
    with an adaptive deadline, processes them through BatchEngine, and
    writes responses back. This is the serving path of App.Run(engine=gpu);
    bench.py drives BatchEngine directly (synthetic in-memory load).
    """

    def __init__(self, app, port: int, batch_window_us: int = 200,
                 max_batch: int = 4096):
        self.app = app
        self.port = port
        self.engine = BatchEngine(app, max_batch=max_batch)
        self.batch_window_us = batch_window_us
        self._stop = None
        self._listener = None

    def start(self):
        import queue
        import socket
        import threading
        self._stop = threading.Event()
        self._q = queue.Queue()
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        sock.bind(("0.0.0.0", self.port))
        sock.listen(1024)
        self._listener = sock
        threading.Thread(target=self._accept_loop, daemon=True).start()
        threading.Thread(target=self._batch_loop, daemon=True).start()

    def _accept_loop(self):
        import threading
        while not self._stop.is_set():
            try:
                conn, _ = self._listener.accept()
            except OSError:
                return
            threading.Thread(target=self._conn_loop, args=(conn,),
                             daemon=True).start()

    def _conn_loop(self, conn):
        import socket as _s
        import threading
        conn.setsockopt(_s.IPPROTO_TCP, _s.TCP_NODELAY, 1)
        buf = b""
        try:
            while not self._stop.is_set():
                while b"\r\n\r\n" not in buf:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                he = buf.index(b"\r\n\r\n") + 4
                clen = 0
                for line in buf[:he].split(b"\r\n")[1:]:
                    if line[:15].lower() == b"content-length:":
                        clen = int(line.split(b":", 1)[1].strip() or b"0")
                        break
                while len(buf) < he + clen:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                raw, buf = buf[:he + clen], buf[he + clen:]
                done = threading.Event()
                slotref = {}
                self._q.put((raw, done, slotref))
                done.wait(timeout=30)
                resp = slotref.get("resp", b"")
                if resp:
                    conn.sendall(resp)
                else:
                    return
        except (OSError, ValueError):
            return
        finally:
            try:
                conn.close()
            except OSError:
                pass

    def _batch_loop(self):
        import queue
        while not self._stop.is_set():
            try:
                first = self._q.get(timeout=0.2)
            except queue.Empty:
                continue
            items = [first]
            deadline = time.perf_counter() + self.batch_window_us / 1e6
            while len(items) < self.engine.max_batch:
                remain = deadline - time.perf_counter()
                if remain <= 0:
                    break
                try:
                    items.append(self._q.get(timeout=remain))
                except queue.Empty:
                    break
            payloads = [it[0] for it in items]
            reqs, lens = make_batch(payloads, self.engine.slot)
            resp, resp_len = self.engine.process(reqs, lens)
            for i, (_, done, slotref) in enumerate(items):
                o = i * self.engine.rslot
                slotref["resp"] = resp[o:o + int(resp_len[i])].tobytes()
                done.set()

    def stop(self):
        if self._stop is not None:
            self._stop.set()
        if self._listener is not None:
            try:
                self._listener.close()
            except OSError:
                pass
