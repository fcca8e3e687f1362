"""gRPC unary server over the from-scratch HTTP/2 transport.

Reference: pkg/gofr/grpc.go:16-47 (grpcServer with panic-recovery +
logging interceptor chain, serving registered protoc services) and
pkg/gofr/grpc/log.go:15-50 (RPCLog with trace id, start time, µs
response time, method).

Services are registered via App.RegisterService(ServiceDesc, impl);
impl methods have the signature method(ctx, request_dict) -> dict.
gRPC handlers get a plain context (not *gofr.Context), matching the
reference (SURVEY.md §3.5 note).
"""

from __future__ import annotations

import socket
import struct
import threading
import time

from .codec import MessageDesc, decode_message, encode_message
from . import http2 as h2


class RPCLog:
    """Reference: grpc/log.go:15-25."""

    __slots__ = ("id", "start_time", "response_time_us", "method")

    def __init__(self, id_, start_time, response_time_us, method):
        self.id = id_
        self.start_time = start_time
        self.response_time_us = response_time_us
        self.method = method

    def to_dict(self):
        return {"id": self.id, "startTime": self.start_time,
                "responseTime": self.response_time_us,
                "method": self.method}

    def pretty(self) -> str:
        return f"RPC    {self.response_time_us:8.0f}µs  {self.method}"


class ServiceDesc:
    """methods: {name: (request MessageDesc, response MessageDesc)}.

    gpu_methods marks unary methods the batched GPU codec can serve
    end-to-end (currently the "hello_echo" shape: HelloRequest in,
    "Hello <name>!" HelloResponse out — k_varint_spans + k_grpc_echo).
    Marked methods are decoded/responded in whole batches; everything
    else takes the host path.
    """

    def __init__(self, name: str, methods: dict,
                 gpu_methods: dict | None = None):
        self.name = name
        self.methods = methods
        self.gpu_methods = gpu_methods or {}


class GRPCServer:
    """gRPC unary server. Two transports:

    - Python (default): thread-per-connection HTTP/2, GPU-marked
      methods batched through the codec worker.
    - native=True: the C++ epoll reactors speak h2c directly (HPACK
      incl. huffman decode, frame state machines per connection) and
      stage complete unary request messages for the batched GPU codec —
      config 3 measured socket-attached, not codec-only (VERDICT r1
      item 7). Reference: pkg/gofr/grpc.go:32-47 (gRPC on the
      production listener).
    """

    def __init__(self, app, port: int, batch_window_us: int = 200,
                 max_codec_batch: int = 8192, native: bool = False):
        self.native = native
        self.app = app
        self.port = port
        self._services: dict[str, tuple[ServiceDesc, object]] = {}
        self._gpu_methods: dict[tuple[str, str], str] = {}
        for service, impl in app._grpc_services:
            self._services[service.name] = (service, impl)
            for meth, kind in getattr(service, "gpu_methods", {}).items():
                self._gpu_methods[(service.name, meth)] = kind
        self._stop = threading.Event()
        self._sock = None
        # batched GPU codec worker (config-3 data plane in production):
        # marked unary methods are queued, decoded and responded in
        # whole batches by k_varint_spans + k_grpc_echo (CPU mirrors on
        # a CPU box)
        self.batch_window_us = batch_window_us
        self.max_codec_batch = max_codec_batch
        self.codec_batches = 0
        self.codec_msgs = 0
        self._codec_q = None
        if self._gpu_methods:
            import queue as _queue
            self._codec_q = _queue.Queue()
            threading.Thread(target=self._codec_worker,
                             daemon=True).start()

    def _codec_worker(self) -> None:
        import queue as _queue

        from .. import ops
        from ..engine import pack_batch
        gpu = None
        try:
            import torch
            if torch.cuda.is_available():
                hip = ops.HipOps()
                dev = torch.device("cuda")
                nb = self.max_codec_batch
                GR = 256
                gpu = {
                    "hip": hip, "t": torch, "dev": dev, "GR": GR,
                    "p_buf": torch.empty(nb * 512,
                                         dtype=torch.uint8).pin_memory(),
                    "d_buf": torch.empty(nb * 512, dtype=torch.uint8,
                                         device=dev),
                    "d_off": torch.empty(nb, dtype=torch.int64,
                                         device=dev),
                    "d_len": torch.empty(nb, dtype=torch.int32,
                                         device=dev),
                    "d_spans": torch.zeros(nb * ops.MAX_PB_FIELDS * 4,
                                           dtype=torch.int32, device=dev),
                    "d_span_n": torch.zeros(nb, dtype=torch.int32,
                                            device=dev),
                    "d_out": torch.empty(nb * GR, dtype=torch.uint8,
                                         device=dev),
                    "d_out_len": torch.empty(nb, dtype=torch.int32,
                                             device=dev),
                    "p_out": hip.host_alloc(nb * GR),
                    "p_out_len": torch.empty(
                        nb, dtype=torch.int32).pin_memory(),
                }
        except (ImportError, FileNotFoundError):
            gpu = None
        while not self._stop.is_set():
            try:
                first = self._codec_q.get(timeout=0.2)
            except _queue.Empty:
                continue
            batch = [first]
            deadline = time.perf_counter() + self.batch_window_us / 1e6
            while len(batch) < self.max_codec_batch and \
                    time.perf_counter() < deadline:
                try:
                    batch.append(self._codec_q.get_nowait())
                except _queue.Empty:
                    time.sleep(0)
            payloads = [b[0] for b in batch]
            buf, offs, lens = pack_batch(payloads)
            n = len(lens)
            if gpu is not None and \
                    all(len(p) <= 512 for p in payloads):
                t = gpu["t"]
                nbytes = len(buf)
                gpu["p_buf"][:nbytes] = t.from_numpy(buf)
                stream = t.cuda.current_stream().cuda_stream
                gpu["d_buf"][:nbytes].copy_(gpu["p_buf"][:nbytes],
                                            non_blocking=True)
                gpu["d_off"][:n].copy_(t.from_numpy(offs),
                                       non_blocking=True)
                gpu["d_len"][:n].copy_(t.from_numpy(lens),
                                       non_blocking=True)
                gpu["hip"].varint_spans(stream, gpu["d_buf"],
                                        gpu["d_off"], gpu["d_len"],
                                        gpu["d_spans"], gpu["d_span_n"],
                                        n)
                gpu["hip"].grpc_echo(stream, gpu["d_buf"], gpu["d_spans"],
                                     gpu["d_span_n"], gpu["d_out"],
                                     gpu["d_out_len"], n, gpu["GR"])
                gpu["p_out"][:n * gpu["GR"]].copy_(
                    gpu["d_out"][:n * gpu["GR"]], non_blocking=True)
                gpu["p_out_len"][:n].copy_(gpu["d_out_len"][:n],
                                           non_blocking=True)
                t.cuda.synchronize()
                out = gpu["p_out"].numpy()
                out_len = gpu["p_out_len"].numpy()
                GR = gpu["GR"]
            else:
                spans, span_n = ops.cpu_varint_spans(buf, offs, lens)
                out, out_len = ops.cpu_grpc_echo(buf, spans, span_n, 256)
                GR = 256
            self.codec_batches += 1
            self.codec_msgs += n
            for i, (_, rq) in enumerate(batch):
                ln = int(out_len[i])
                if ln < 0:
                    rq.put(None)  # malformed: host error path
                else:
                    rq.put(out[i * GR:i * GR + ln].tobytes())
        if gpu is not None:
            # worker owns the hipHostMalloc'd egress buffer: free it on
            # shutdown so server start/stop cycles don't leak pinned mem
            p_out = gpu.pop("p_out")
            gpu["hip"].host_free(p_out)

    def start(self) -> None:
        if self.native:
            self._start_native()
            return
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        sock.bind(("0.0.0.0", self.port))
        sock.listen(256)
        self.port = sock.getsockname()[1]  # resolve port-0 binds
        self._sock = sock
        threading.Thread(target=self._accept_loop, daemon=True).start()

    def stop(self) -> None:
        self._stop.set()
        if getattr(self, "_native_thread", None) is not None:
            self._native_thread.join(timeout=10)
            if not self._native_thread.is_alive():
                self._core.stop()
            return
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass

    # -- native h2c ingress (C++ reactors + batched codec) --------------------
    def _start_native(self) -> None:
        import os

        from .. import _core
        core = _core.EpollServer(
            self.port, 1 << 20, int(os.environ.get("GOFR_REACTORS",
                                                   "16")))
        # path registry: id -> (desc, impl, method, gpu fast path)
        self._paths = []
        for sname, (desc, impl) in self._services.items():
            for meth in desc.methods:
                pid = core.register_grpc_path(f"/{sname}/{meth}")
                assert pid == len(self._paths)
                self._paths.append(
                    (desc, impl, meth,
                     (sname, meth) in self._gpu_methods))
        import numpy as _np
        self._gpu_mask = _np.array([p[3] for p in self._paths],
                                   dtype=bool)
        core.set_grpc_blocks(
            h2.HpackEncoder.encode([(":status", "200"),
                                    ("content-type",
                                     "application/grpc")]),
            h2.HpackEncoder.encode([("grpc-status", "0")]),
            h2.HpackEncoder.encode([("grpc-status", "13"),
                                    ("grpc-message", "internal error")]))
        core.start()
        self._core = core
        self.port = core.port()
        from ..engine import _register_server
        _register_server(self)
        self._native_thread = threading.Thread(target=self._native_loop,
                                               daemon=True)
        self._native_thread.start()

    def _native_loop(self) -> None:
        import numpy as np

        from .. import ops
        core = self._core
        MB = self.max_codec_batch
        CAP = MB * 512
        GR = 256
        HOSTCAP = 1 << 20
        buf = np.zeros(CAP, np.uint8)
        offs = np.zeros(MB, np.int64)
        lens = np.zeros(MB, np.int32)
        cids = np.zeros(MB, np.uint64)
        sids = np.zeros(MB, np.uint32)
        pids = np.zeros(MB, np.int32)
        out = np.zeros(MB * GR + HOSTCAP, np.uint8)
        roffs = np.zeros(MB, np.int32)
        rlens = np.zeros(MB, np.int32)
        gpu = None
        try:
            import torch
            if torch.cuda.is_available():
                hip = ops.HipOps()
                dev = torch.device("cuda")
                gpu = {
                    "hip": hip, "t": torch, "dev": dev,
                    "p_buf": torch.empty(CAP,
                                         dtype=torch.uint8).pin_memory(),
                    "d_buf": torch.empty(CAP, dtype=torch.uint8,
                                         device=dev),
                    "d_off": torch.empty(MB, dtype=torch.int64,
                                         device=dev),
                    "d_len": torch.empty(MB, dtype=torch.int32,
                                         device=dev),
                    "d_spans": torch.zeros(MB * ops.MAX_PB_FIELDS * 4,
                                           dtype=torch.int32,
                                           device=dev),
                    "d_span_n": torch.zeros(MB, dtype=torch.int32,
                                            device=dev),
                    "d_out": torch.empty(MB * GR, dtype=torch.uint8,
                                         device=dev),
                    "d_out_len": torch.empty(MB, dtype=torch.int32,
                                             device=dev),
                    "p_out": hip.host_alloc(MB * GR),
                    "p_out_len": torch.empty(
                        MB, dtype=torch.int32).pin_memory(),
                }
        except (ImportError, FileNotFoundError):
            gpu = None
        logger = self.app.container.logger
        while not self._stop.is_set():
            n, nbytes = core.harvest_grpc(
                buf.ctypes.data, CAP, offs.ctypes.data,
                lens.ctypes.data, cids.ctypes.data, sids.ctypes.data,
                pids.ctypes.data, MB, self.batch_window_us)
            if n == 0:
                continue
            self.codec_batches += 1
            self.codec_msgs += n
            rlens[:n] = -1  # default: error trailers
            pids_n = pids[:n]
            valid = pids_n >= 0
            gmask = np.zeros(n, bool)
            gmask[valid] = self._gpu_mask[pids_n[valid]]
            gpu_rows = np.nonzero(gmask)[0]
            host_rows = np.nonzero(valid & ~gmask)[0]
            if len(gpu_rows):
                go = offs[gpu_rows]
                gl = lens[gpu_rows]
                m = len(gpu_rows)
                if gpu is not None:
                    t = gpu["t"]
                    gpu["p_buf"][:nbytes] = t.from_numpy(buf[:nbytes])
                    stream = t.cuda.current_stream().cuda_stream
                    gpu["d_buf"][:nbytes].copy_(gpu["p_buf"][:nbytes],
                                                non_blocking=True)
                    gpu["d_off"][:m].copy_(t.from_numpy(go),
                                           non_blocking=True)
                    gpu["d_len"][:m].copy_(t.from_numpy(gl),
                                           non_blocking=True)
                    gpu["hip"].varint_spans(
                        stream, gpu["d_buf"], gpu["d_off"],
                        gpu["d_len"], gpu["d_spans"], gpu["d_span_n"],
                        m)
                    gpu["hip"].grpc_echo(
                        stream, gpu["d_buf"], gpu["d_spans"],
                        gpu["d_span_n"], gpu["d_out"],
                        gpu["d_out_len"], m, GR)
                    gpu["p_out"][:m * GR].copy_(gpu["d_out"][:m * GR],
                                                non_blocking=True)
                    gpu["p_out_len"][:m].copy_(gpu["d_out_len"][:m],
                                               non_blocking=True)
                    t.cuda.synchronize()
                    gout = gpu["p_out"].numpy()
                    gout_len = gpu["p_out_len"].numpy()
                else:
                    spans, span_n = ops.cpu_varint_spans(buf, go, gl)
                    gout, gout_len = ops.cpu_grpc_echo(buf, spans,
                                                       span_n, GR)
                # vectorized row scatter (a per-row Python loop cost
                # ~2 us x batch and capped the wire at ~300k msg/s)
                ok = gout_len[:m] > 0
                gi = gpu_rows[ok]
                out[:MB * GR].reshape(MB, GR)[gi] = \
                    gout[:m * GR].reshape(m, GR)[ok]
                roffs[gi] = (gi * GR).astype(np.int32)
                rlens[gi] = gout_len[:m][ok]
            host_pos = MB * GR
            for i in host_rows:
                desc, impl, meth, _ = self._paths[int(pids[i])]
                try:
                    req_desc, resp_desc = desc.methods[meth]
                    msg = buf[int(offs[i]):int(offs[i]) +
                              int(lens[i])].tobytes()
                    req = decode_message(msg, req_desc)
                    resp = getattr(impl, meth)(None, req)
                    mb = encode_message(resp or {}, resp_desc)
                    frame = (b"\0" + struct.pack(">I", len(mb)) + mb)
                    if host_pos + len(frame) > len(out):
                        continue  # error trailers for this row
                    out[host_pos:host_pos + len(frame)] = \
                        np.frombuffer(frame, np.uint8)
                    roffs[i] = host_pos
                    rlens[i] = len(frame)
                    host_pos += len(frame)
                except Exception:  # noqa: BLE001 — recovery interceptor
                    pass  # rlens stays -1 -> grpc-status 13 trailers
            core.send_grpc(cids.ctypes.data, sids.ctypes.data, n,
                           out.ctypes.data, roffs.ctypes.data,
                           rlens.ctypes.data)
            # sampled RPCLog (the logging interceptor's record type;
            # per-RPC logging would serialize the batch)
            if logger is not None and (self.codec_batches % 64) == 1:
                from ..http.middleware import rfc3339nano
                pid0 = int(pids[0])
                path = ("/" + self._paths[pid0][0].name + "/" +
                        self._paths[pid0][2]) \
                    if 0 <= pid0 < len(self._paths) else "?"
                logger.info_record(RPCLog("", rfc3339nano(), 0.0,
                                          f"{path} x{n} (batched)"))

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _ = self._sock.accept()
            except OSError:
                return
            threading.Thread(target=self._serve_conn, args=(conn,),
                             daemon=True).start()

    # -- one HTTP/2 connection ------------------------------------------------
    def _serve_conn(self, conn: socket.socket) -> None:
        conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        buf = bytearray()

        def read_exact(n: int) -> bytes:
            while len(buf) < n:
                chunk = conn.recv(65536)
                if not chunk:
                    raise ConnectionError("closed")
                buf.extend(chunk)
            out = bytes(buf[:n])
            del buf[:n]
            return out

        try:
            preface = read_exact(len(h2.PREFACE))
            if preface != h2.PREFACE:
                return
            conn.sendall(h2.pack_frame(h2.FT_SETTINGS, 0, 0, b""))
            decoder = h2.HpackDecoder()
            streams: dict[int, dict] = {}
            while not self._stop.is_set():
                ftype, flags, sid, payload = h2.read_frame(read_exact)
                if ftype == h2.FT_SETTINGS:
                    if not flags & h2.FLAG_ACK:
                        conn.sendall(h2.pack_frame(h2.FT_SETTINGS,
                                                   h2.FLAG_ACK, 0, b""))
                elif ftype == h2.FT_PING:
                    if not flags & h2.FLAG_ACK:
                        conn.sendall(h2.pack_frame(h2.FT_PING, h2.FLAG_ACK,
                                                   0, payload))
                elif ftype == h2.FT_HEADERS:
                    pos = 0
                    if flags & h2.FLAG_PADDED:
                        pos += 1
                    if flags & h2.FLAG_PRIORITY:
                        pos += 5
                    headers = decoder.decode(payload[pos:])
                    streams[sid] = {"headers": dict(headers),
                                    "data": bytearray()}
                    if flags & h2.FLAG_END_STREAM:
                        self._dispatch(conn, sid, streams.pop(sid))
                elif ftype == h2.FT_DATA:
                    st = streams.get(sid)
                    if st is None:
                        continue
                    st["data"].extend(payload)
                    if flags & h2.FLAG_END_STREAM:
                        self._dispatch(conn, sid, streams.pop(sid))
                elif ftype == h2.FT_GOAWAY:
                    return
                elif ftype in (h2.FT_WINDOW_UPDATE, h2.FT_RST_STREAM,
                               h2.FT_CONTINUATION):
                    pass
        except (ConnectionError, OSError, ValueError):
            return
        finally:
            try:
                conn.close()
            except OSError:
                pass

    # -- unary dispatch with interceptors -------------------------------------
    def _dispatch(self, conn, sid: int, stream: dict) -> None:
        path = stream["headers"].get(":path", "")
        start = time.time()
        t0 = time.perf_counter_ns()
        tracer = self.app.tracer
        span = tracer.start_span(f"grpc{path}") if tracer else None
        grpc_status, msg_bytes, err_msg = 0, b"", ""
        try:
            service_name, _, method_name = path.lstrip("/").partition("/")
            entry = self._services.get(service_name)
            if entry is None or method_name not in entry[0].methods:
                grpc_status, err_msg = 12, f"unknown method {path}"  # UNIMPL.
            else:
                desc, impl = entry
                req_desc, resp_desc = desc.methods[method_name]
                data = bytes(stream["data"])
                if len(data) < 5:
                    raise ValueError("short gRPC frame")
                compressed = data[0]
                mlen = struct.unpack(">I", data[1:5])[0]
                if compressed:
                    grpc_status, err_msg = 12, "compression not supported"
                elif (self._codec_q is not None and
                      (service_name, method_name) in self._gpu_methods):
                    # batched GPU codec path (k_varint_spans +
                    # k_grpc_echo decode/respond the whole batch)
                    import queue as _queue
                    rq = _queue.Queue(1)
                    self._codec_q.put((data[5:5 + mlen], rq))
                    frame = rq.get(timeout=10)
                    if frame is None:
                        raise ValueError("malformed request message")
                    # frame already carries the 5-byte gRPC prefix
                    msg_bytes = frame[5:]
                else:
                    req = decode_message(data[5:5 + mlen], req_desc)
                    fn = getattr(impl, method_name)
                    resp = fn(None, req)
                    msg_bytes = encode_message(resp or {}, resp_desc)
        except Exception as e:  # noqa: BLE001 — recovery interceptor
            # reference: grpc_recovery -> codes.Internal (grpc.go:25)
            grpc_status, err_msg = 13, str(e) or "internal error"
        finally:
            if span is not None:
                span.End()
            dur_us = (time.perf_counter_ns() - t0) / 1000.0
            from ..http.middleware import rfc3339nano
            self.app.container.logger.info_record(RPCLog(
                span.trace_id if span else "", rfc3339nano(start),
                dur_us, path))

        # response: HEADERS + DATA + trailers
        hdr = h2.HpackEncoder.encode([
            (":status", "200"), ("content-type", "application/grpc")])
        conn.sendall(h2.pack_frame(h2.FT_HEADERS, h2.FLAG_END_HEADERS,
                                   sid, hdr))
        if grpc_status == 0:
            frame = bytes([0]) + struct.pack(">I", len(msg_bytes)) + msg_bytes
            conn.sendall(h2.pack_frame(h2.FT_DATA, 0, sid, frame))
        trailers = [("grpc-status", str(grpc_status))]
        if err_msg:
            trailers.append(("grpc-message", err_msg))
        conn.sendall(h2.pack_frame(
            h2.FT_HEADERS, h2.FLAG_END_HEADERS | h2.FLAG_END_STREAM,
            sid, h2.HpackEncoder.encode(trailers)))


class GRPCClient:
    """Minimal unary client (tests + inter-service calls)."""

    def __init__(self, host: str, port: int, timeout: float = 5.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self.sock.sendall(h2.PREFACE)
        self.sock.sendall(h2.pack_frame(h2.FT_SETTINGS, 0, 0, b""))
        self._buf = bytearray()
        self._next_sid = 1
        self._decoder = h2.HpackDecoder()

    def _read_exact(self, n: int) -> bytes:
        while len(self._buf) < n:
            chunk = self.sock.recv(65536)
            if not chunk:
                raise ConnectionError("closed")
            self._buf.extend(chunk)
        out = bytes(self._buf[:n])
        del self._buf[:n]
        return out

    def call(self, service: str, method: str, request: dict,
             req_desc: MessageDesc, resp_desc: MessageDesc):
        """Returns (response dict or None, grpc_status, message)."""
        sid = self._next_sid
        self._next_sid += 2
        headers = h2.HpackEncoder.encode([
            (":method", "POST"), (":scheme", "http"),
            (":path", f"/{service}/{method}"), (":authority", "localhost"),
            ("content-type", "application/grpc"), ("te", "trailers")])
        self.sock.sendall(h2.pack_frame(h2.FT_HEADERS, h2.FLAG_END_HEADERS,
                                        sid, headers))
        payload = encode_message(request, req_desc)
        frame = bytes([0]) + struct.pack(">I", len(payload)) + payload
        self.sock.sendall(h2.pack_frame(h2.FT_DATA, h2.FLAG_END_STREAM,
                                        sid, frame))
        resp_msg = None
        status = -1
        message = ""
        while True:
            ftype, flags, fsid, payload = h2.read_frame(self._read_exact)
            if ftype == h2.FT_SETTINGS:
                if not flags & h2.FLAG_ACK:
                    self.sock.sendall(h2.pack_frame(h2.FT_SETTINGS,
                                                    h2.FLAG_ACK, 0, b""))
                continue
            if fsid != sid:
                continue
            if ftype == h2.FT_HEADERS:
                hdrs = dict(self._decoder.decode(payload))
                if "grpc-status" in hdrs:
                    status = int(hdrs["grpc-status"])
                    message = hdrs.get("grpc-message", "")
                if flags & h2.FLAG_END_STREAM:
                    return resp_msg, status, message
            elif ftype == h2.FT_DATA:
                if len(payload) >= 5:
                    mlen = struct.unpack(">I", payload[1:5])[0]
                    resp_msg = decode_message(payload[5:5 + mlen], resp_desc)
                if flags & h2.FLAG_END_STREAM:
                    return resp_msg, status, message

    def close(self):
        try:
            self.sock.close()
        except OSError:
            pass
