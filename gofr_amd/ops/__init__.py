"""Device op bindings + CPU golden mirrors.

`HipOps` drives the gfx950 kernels in gofr_amd/_gofr_hip.so (built from
native/hip/gofr_kernels.hip) through ctypes on raw torch device pointers.
`cpu_parse_route` / `cpu_respond` are byte-exact Python mirrors of the
kernels: the GPU tests (tests/test_gpu_engine.py) assert kernel output ==
mirror output on the same buffers, and the engine uses the mirrors as its
CPU fallback so every code path runs (slowly) without a GPU.

On a GPU box the extension is REQUIRED: HipOps raises if the .so is
missing rather than silently falling back (driver policy: native code
must be what actually runs).
"""

from __future__ import annotations

import ctypes
import os

import numpy as np

# ---- field table layout (mirror of gofr_kernels.hip) ------------------------
FI_METHOD = 0
FI_PATH_OFF = 1
FI_PATH_LEN = 2
FI_QUERY_OFF = 3
FI_QUERY_LEN = 4
FI_BODY_OFF = 5
FI_BODY_LEN = 6
FI_CLEN = 7
FI_FLAGS = 8
FI_ROUTE = 9
FI_KIND = 10
FI_STATUS = 11
FI_RESP_LEN = 12
FI_AUTH_OFF = 13
FI_AUTH_LEN = 14
FI_RESP_OFF = 15
FI_PARAM0 = 16
FI_INM_OFF = 24
FI_INM_LEN = 25
NF = 26

FL_ERR_PARSE = 1
FL_NEEDS_HOST = 2
FL_KEEP_ALIVE = 4
FL_JSON_CT = 8
FL_IS_OPTIONS = 16
FL_BODY_INVALID = 32
FL_ACCEPT_GZIP = 64
FL_AUTH_FAIL = 128
FL_EMPTY = 256  # padding slot (len 0): emit no response bytes

HK_HOST = 0
HK_ECHO_JSON = 1
HK_STATIC = 2
HK_TEMPLATE = 3
HK_KV = 4

# template piece opcodes / splice modes (mirror of gofr_kernels.hip)
TP_LIT = 0
TP_PATH = 1
TP_QUERY = 2
TP_JFIELD = 3
TM_PCT = 1
TM_JESC = 2
TM_JSTR = 4
MAX_JSON_FIELDS = 8

MAX_PARAMS = 4
N_METHODS_PAD = 8
MAX_SLOT = 4096

_METHOD_NAMES = ["GET", "POST", "PUT", "DELETE", "PATCH", "OPTIONS", "HEAD"]

_SO_PATH = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "_gofr_hip.so")


class GofrSubmitArgs(ctypes.Structure):
    """Mirror of GofrSubmitArgs in gofr_kernels.hip (the one-call staged
    submit: ingress H2D -> kernel chain -> egress, all raw HIP)."""

    _fields_ = [
        ("s_in", ctypes.c_void_p), ("s_k", ctypes.c_void_p),
        ("s_out", ctypes.c_void_p),
        ("ev_in", ctypes.c_void_p), ("ev_k", ctypes.c_void_p),
        ("ev_done", ctypes.c_void_p),
        ("p_reqs", ctypes.c_void_p), ("d_reqs", ctypes.c_void_p),
        ("nbytes", ctypes.c_longlong),
        ("p_off", ctypes.c_void_p), ("d_off", ctypes.c_void_p),
        ("p_len", ctypes.c_void_p), ("d_len", ctypes.c_void_p),
        ("d_fields", ctypes.c_void_p),
        ("trie", ctypes.c_void_p * 9),
        ("handler_tab", ctypes.c_void_p), ("n_routes", ctypes.c_int),
        ("d_host_needed", ctypes.c_void_p),
        ("secret", ctypes.c_void_p), ("secret_len", ctypes.c_int),
        ("d_resp", ctypes.c_void_p),
        ("d_tables", ctypes.c_void_p), ("p_tables", ctypes.c_void_p),
        ("blob", ctypes.c_void_p), ("host_blob", ctypes.c_void_p),
        ("host_tab", ctypes.c_void_p),
        ("auth_env_off", ctypes.c_int), ("auth_env_len", ctypes.c_int),
        ("gzip_min", ctypes.c_int), ("etag_on", ctypes.c_int),
        ("p_out", ctypes.c_void_p),
        ("n", ctypes.c_int), ("rslot", ctypes.c_int),
        ("d_out", ctypes.c_void_p),
        ("egress_budget", ctypes.c_longlong),
        ("d_flag", ctypes.c_void_p),
        ("p_serial", ctypes.c_void_p),
        ("serial", ctypes.c_uint64),
        ("flagged", ctypes.c_int),
        ("d_date", ctypes.c_void_p),  # 29-byte IMF-fixdate in device mem
        ("d_kv_tab", ctypes.c_void_p),   # HK_KV store rows (int32 x6)
        ("d_kv_blob", ctypes.c_void_p),  # HK_KV key/value bytes
    ]


class PersistKernArgs(ctypes.Structure):
    """Mirror of PersistKernArgs in gofr_kernels.hip (the persistent
    serving kernel's argument block; passed by value at launch)."""

    _fields_ = [
        ("first", ctypes.c_uint64),
        ("nbatch", ctypes.c_int),
        ("n", ctypes.c_int), ("rslot", ctypes.c_int),
        ("hdr_bytes", ctypes.c_longlong),
        ("lens_off", ctypes.c_longlong),
        ("date_off", ctypes.c_int), ("egress_blocks", ctypes.c_int),
        ("d_ingress", ctypes.c_void_p * 2),
        ("d_fields", ctypes.c_void_p * 2),
        ("d_resp", ctypes.c_void_p * 2),
        ("d_tables", ctypes.c_void_p * 2),
        ("p_tables", ctypes.c_void_p * 2),
        ("p_out", ctypes.c_void_p * 2),
        ("host_blob", ctypes.c_void_p * 2),
        ("host_tab", ctypes.c_void_p * 2),
        ("trie", ctypes.c_void_p * 9),
        ("handler_tab", ctypes.c_void_p), ("n_routes", ctypes.c_int),
        ("blob", ctypes.c_void_p),
        ("d_kv_tab", ctypes.c_void_p), ("d_kv_blob", ctypes.c_void_p),
        ("secret", ctypes.c_void_p), ("secret_len", ctypes.c_int),
        ("auth_env_off", ctypes.c_int), ("auth_env_len", ctypes.c_int),
        ("etag_on", ctypes.c_int),
        ("d_state", ctypes.c_void_p), ("d_barrier", ctypes.c_void_p),
    ]


class HipOps:
    """ctypes driver for the device kernels."""

    def __init__(self, so_path: str = _SO_PATH):
        if not os.path.exists(so_path):
            raise FileNotFoundError(
                f"HIP extension not built: {so_path}. Run "
                f"`python setup.py build_hip` or __graft_entry__.build().")
        # Load torch FIRST so its libamdhip64 (same soname) is the HIP
        # runtime our .so binds to. Loading /opt/rocm's copy first and
        # torch's second puts two HIP runtimes in the process and kernel
        # launches fail with hipErrorNoDevice(100).
        try:
            import torch  # noqa: F401
        except ImportError:
            pass
        self.lib = ctypes.CDLL(so_path)
        self.lib.gofr_launch_parse_route.restype = ctypes.c_int
        self.lib.gofr_launch_parse_route.argtypes = \
            [ctypes.c_void_p] * 5 + [ctypes.c_int] + \
            [ctypes.c_void_p] * 9 + [ctypes.c_void_p, ctypes.c_int,
                                     ctypes.c_void_p]
        self.lib.gofr_launch_respond.restype = ctypes.c_int
        self.lib.gofr_launch_respond.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_int, ctypes.c_int,
            ctypes.c_void_p, ctypes.c_int,
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_void_p, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_void_p]
        self.lib.gofr_launch_compact.restype = ctypes.c_int
        self.lib.gofr_launch_compact.argtypes = \
            [ctypes.c_void_p] * 5 + [ctypes.c_int, ctypes.c_int]
        self.lib.gofr_launch_auth.restype = ctypes.c_int
        self.lib.gofr_launch_auth.argtypes = \
            [ctypes.c_void_p] * 4 + [ctypes.c_int, ctypes.c_void_p,
                                     ctypes.c_int]
        self.lib.gofr_launch_varint_spans.restype = ctypes.c_int
        self.lib.gofr_launch_varint_spans.argtypes = \
            [ctypes.c_void_p] * 6 + [ctypes.c_int]
        self.lib.gofr_launch_grpc_echo.restype = ctypes.c_int
        self.lib.gofr_launch_grpc_echo.argtypes = \
            [ctypes.c_void_p] * 6 + [ctypes.c_int, ctypes.c_int]
        self.lib.gofr_submit_staged.restype = ctypes.c_int
        self.lib.gofr_submit_staged.argtypes = \
            [ctypes.POINTER(GofrSubmitArgs)]
        self.lib.gofr_pump_start.restype = ctypes.c_int
        self.lib.gofr_pump_submit.restype = ctypes.c_uint64
        self.lib.gofr_pump_submit.argtypes = \
            [ctypes.POINTER(GofrSubmitArgs)]
        self.lib.gofr_pump_done_ptr.restype = ctypes.c_void_p
        self.lib.gofr_pump_err.restype = ctypes.c_int
        self.lib.gofr_host_alloc.restype = ctypes.c_void_p
        self.lib.gofr_host_alloc.argtypes = [ctypes.c_longlong,
                                             ctypes.c_uint]
        self.lib.gofr_host_free.restype = ctypes.c_int
        self.lib.gofr_host_free.argtypes = [ctypes.c_void_p]
        self.lib.gofr_wait_cell.restype = ctypes.c_int
        self.lib.gofr_wait_cell.argtypes = [
            ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
            ctypes.c_double]
        self.lib.gofr_persist_grid.restype = ctypes.c_int
        self.lib.gofr_persist_grid.argtypes = [
            ctypes.POINTER(ctypes.c_int)]
        self.lib.gofr_persist_launch.restype = ctypes.c_int
        self.lib.gofr_persist_launch.argtypes = [
            ctypes.POINTER(PersistKernArgs), ctypes.c_void_p,
            ctypes.c_int]
        self.lib.gofr_persist_submit.restype = ctypes.c_int
        self.lib.gofr_persist_submit.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_longlong, ctypes.c_void_p, ctypes.c_void_p]
        try:
            self.lib.gofr_src_hash.restype = ctypes.c_char_p
        except AttributeError:
            pass
        # data_ptr -> hipHostMalloc base (host_free needs the base addr)
        self._host_allocs: dict[int, int] = {}

    def host_alloc(self, nbytes: int, dtype=np.uint8):
        """hipHostMalloc'd host memory as a torch CPU tensor. Unlike
        torch's pin_memory (hipHostRegister'd), the runtime serves D2H
        SDMA into this memory — registered destinations fall back to a
        CU-hungry blit kernel (measured: benchmarks/overlap_probe.py
        --mix: 96.6 GB/s duplex vs ~65 engine-observed with blit).
        Pair with host_free (engine/lane close paths) — hipHostMalloc'd
        pinned memory is NOT garbage-collected with the tensor."""
        import torch
        addr = self.lib.gofr_host_alloc(int(nbytes), 0)
        if not addr:
            raise MemoryError(f"gofr_host_alloc({nbytes}) failed")
        buf = (ctypes.c_ubyte * int(nbytes)).from_address(addr)
        t = torch.from_numpy(np.frombuffer(buf, dtype=dtype))
        self._host_allocs[t.data_ptr()] = addr
        return t

    def host_free(self, tensor) -> None:
        """Free a host_alloc'd tensor's pinned backing store. The caller
        must drop every reference to the tensor (and views) first."""
        addr = self._host_allocs.pop(tensor.data_ptr(), None)
        if addr is not None:
            self.lib.gofr_host_free(ctypes.c_void_p(addr))

    def parse_route(self, stream, reqs_t, req_off_t, req_len_t, fields_t,
                    n, trie_t: dict, handler_tab_t, n_routes,
                    host_needed_t):
        rc = self.lib.gofr_launch_parse_route(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(reqs_t.data_ptr()),
            ctypes.c_void_p(req_off_t.data_ptr()),
            ctypes.c_void_p(req_len_t.data_ptr()),
            ctypes.c_void_p(fields_t.data_ptr()),
            n,
            ctypes.c_void_p(trie_t["seg_blob"].data_ptr()),
            ctypes.c_void_p(trie_t["node_child_first"].data_ptr()),
            ctypes.c_void_p(trie_t["node_child_count"].data_ptr()),
            ctypes.c_void_p(trie_t["child_seg_off"].data_ptr()),
            ctypes.c_void_p(trie_t["child_seg_len"].data_ptr()),
            ctypes.c_void_p(trie_t["child_node"].data_ptr()),
            ctypes.c_void_p(trie_t["node_param"].data_ptr()),
            ctypes.c_void_p(trie_t["node_prefix"].data_ptr()),
            ctypes.c_void_p(trie_t["node_route"].data_ptr()),
            ctypes.c_void_p(handler_tab_t.data_ptr()), n_routes,
            ctypes.c_void_p(host_needed_t.data_ptr()))
        if rc != 0:
            raise RuntimeError(f"k_parse_route launch failed: hipError {rc}")

    def respond(self, stream, reqs_t, req_off_t, fields_t, resp_t,
                resp_len_t, n, rslot, handler_tab_t, n_routes, blob_t,
                host_blob_t, host_tab_t, seed_t, auth_env=(0, 0),
                gzip_min=0, etag_on=0, date_ptr=0, kv_tab_t=None,
                kv_blob_t=None):
        rc = self.lib.gofr_launch_respond(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(reqs_t.data_ptr()),
            ctypes.c_void_p(req_off_t.data_ptr()),
            ctypes.c_void_p(fields_t.data_ptr()),
            ctypes.c_void_p(resp_t.data_ptr()),
            ctypes.c_void_p(resp_len_t.data_ptr()),
            n, rslot,
            ctypes.c_void_p(handler_tab_t.data_ptr()), n_routes,
            ctypes.c_void_p(blob_t.data_ptr()),
            ctypes.c_void_p(host_blob_t.data_ptr()),
            ctypes.c_void_p(host_tab_t.data_ptr()),
            ctypes.c_void_p(seed_t.data_ptr()),
            auth_env[0], auth_env[1], gzip_min, etag_on,
            ctypes.c_void_p(date_ptr),
            ctypes.c_void_p(kv_tab_t.data_ptr() if kv_tab_t is not None
                            else 0),
            ctypes.c_void_p(kv_blob_t.data_ptr() if kv_blob_t is not None
                            else 0))
        if rc != 0:
            raise RuntimeError(f"k_respond launch failed: hipError {rc}")

    def auth(self, stream, reqs_t, req_off_t, fields_t, n, secret_t,
             secret_len):
        rc = self.lib.gofr_launch_auth(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(reqs_t.data_ptr()),
            ctypes.c_void_p(req_off_t.data_ptr()),
            ctypes.c_void_p(fields_t.data_ptr()),
            n,
            ctypes.c_void_p(secret_t.data_ptr()), secret_len)
        if rc != 0:
            raise RuntimeError(f"k_auth launch failed: hipError {rc}")

    def varint_spans(self, stream, buf_t, msg_off_t, msg_len_t, out_t,
                     out_n_t, n):
        rc = self.lib.gofr_launch_varint_spans(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(buf_t.data_ptr()),
            ctypes.c_void_p(msg_off_t.data_ptr()),
            ctypes.c_void_p(msg_len_t.data_ptr()),
            ctypes.c_void_p(out_t.data_ptr()),
            ctypes.c_void_p(out_n_t.data_ptr()), n)
        if rc != 0:
            raise RuntimeError(
                f"k_varint_spans launch failed: hipError {rc}")

    def submit_staged(self, args: GofrSubmitArgs):
        rc = self.lib.gofr_submit_staged(ctypes.byref(args))
        if rc != 0:
            raise RuntimeError(f"gofr_submit_staged failed: hipError {rc}")

    def grpc_echo(self, stream, buf_t, spans_t, span_n_t, out_t, out_len_t,
                  n, rslot):
        rc = self.lib.gofr_launch_grpc_echo(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(buf_t.data_ptr()),
            ctypes.c_void_p(spans_t.data_ptr()),
            ctypes.c_void_p(span_n_t.data_ptr()),
            ctypes.c_void_p(out_t.data_ptr()),
            ctypes.c_void_p(out_len_t.data_ptr()), n, rslot)
        if rc != 0:
            raise RuntimeError(f"k_grpc_echo launch failed: hipError {rc}")

    def compact(self, stream, resp_slots_t, resp_len_t, resp_off_t, out_t,
                n, rslot):
        rc = self.lib.gofr_launch_compact(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(resp_slots_t.data_ptr()),
            ctypes.c_void_p(resp_len_t.data_ptr()),
            ctypes.c_void_p(resp_off_t.data_ptr()),
            ctypes.c_void_p(out_t.data_ptr()),
            n, rslot)
        if rc != 0:
            raise RuntimeError(f"k_compact launch failed: hipError {rc}")


# ---------------------------------------------------------------------------
# CPU golden mirrors (byte-exact models of the kernels)
# ---------------------------------------------------------------------------

M64 = (1 << 64) - 1


def imf_date(epoch_s: int) -> bytes:
    """IMF-fixdate (RFC 9110 §5.6.7), always 29 bytes — the Date
    header value Go's net/http attaches to every response (the
    reference inherits it; parity surface). Shared by the kernel
    (host-rendered, rides the ingress block) and the CPU mirrors."""
    import email.utils
    s = email.utils.formatdate(epoch_s, usegmt=True).encode()
    assert len(s) == 29, s
    return s


def splitmix64(x: int) -> int:
    x = (x + 0x9E3779B97F4A7C15) & M64
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & M64
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & M64
    return x ^ (x >> 31)


def _lower(c: int) -> int:
    return c | 0x20 if ord("A") <= c <= ord("Z") else c


def _ieq(buf, off, lit: bytes) -> bool:
    return all(_lower(buf[off + i]) == lit[i] for i in range(len(lit)))


def cpu_parse_route(reqs: np.ndarray, req_off: np.ndarray,
                    req_len: np.ndarray,
                    trie: dict, handler_tab: np.ndarray) -> np.ndarray:
    """Mirror of k_parse_route over a packed uint8 buffer + offsets."""
    n = len(req_len)
    fields = np.zeros((n, NF), np.int32)
    n_routes = len(handler_tab) // 4
    for r in range(n):
        base = int(req_off[r])
        ln = int(req_len[r])
        if ln == 0:
            F = fields[r]
            F[FI_FLAGS] = FL_EMPTY
            F[FI_KIND] = HK_STATIC
            F[FI_ROUTE] = -1
            continue
        oversized = ln > MAX_SLOT
        if oversized:
            ln = MAX_SLOT
        buf = reqs[base:base + ln].tobytes()
        F = fields[r]
        flags = 0
        if oversized:
            flags |= FL_NEEDS_HOST
        sp1 = buf.find(b" ")
        lf1 = buf.find(b"\n")
        if sp1 < 0 or lf1 < 0 or sp1 >= lf1:
            F[FI_FLAGS] = FL_ERR_PARSE
            F[FI_METHOD] = -1
            F[FI_KIND] = HK_HOST
            F[FI_STATUS] = 400
            F[FI_ROUTE] = -1
            continue
        method = -1
        head = buf[:sp1]
        for mi, mn in enumerate(_METHOD_NAMES):
            if head == mn.encode():
                method = mi
                break
        sp2 = buf.find(b" ", sp1 + 1)
        target_end = sp2 if (0 < sp2 < lf1) else lf1 - 1
        path_off = sp1 + 1
        qm = buf.find(b"?", path_off)
        path_end = qm if (0 <= qm < target_end) else target_end
        query_off = qm + 1 if (0 <= qm < target_end) else target_end
        query_len = target_end - query_off if (0 <= qm < target_end) else 0
        path_len = path_end - path_off
        # %XX in the path: decode IN PLACE before the trie walk (mirror
        # of the kernel; Go routes on the decoded URL.Path). Invalid
        # escapes fall to the host parser's unquote leniency.
        if b"%" in buf[path_off:path_end]:
            dec = pct_decode(buf[path_off:path_end])
            if dec is None:
                flags |= FL_NEEDS_HOST
            else:
                reqs[base + path_off:base + path_off + len(dec)] = \
                    np.frombuffer(dec, np.uint8)
                path_len = len(dec)
                buf = reqs[base:base + ln].tobytes()
        F[FI_METHOD] = method
        F[FI_PATH_OFF] = path_off
        F[FI_PATH_LEN] = path_len
        F[FI_QUERY_OFF] = query_off
        F[FI_QUERY_LEN] = query_len
        if method == 5:  # OPTIONS
            flags |= FL_IS_OPTIONS
        if method < 0:
            flags |= FL_ERR_PARSE

        clen = 0
        # HTTP/1.1 defaults keep-alive; HTTP/1.0 defaults close
        # (kernel checks the same version-token bytes)
        keep_alive = True
        if sp2 > 0 and (lf1 - 1) - (sp2 + 1) == 8 and \
                buf[lf1 - 2:lf1 - 1] == b"0" and \
                buf[lf1 - 4:lf1 - 3] == b"1":
            keep_alive = False
        body_off = ln
        auth_off = auth_len = 0
        inm_off = inm_len = 0
        # mirror of the kernel's one-line-per-lane header pass: LF
        # position table, first blank line among lines 1..63, then the
        # header lines before it (last occurrence of a repeated header
        # wins — the reductions are max-by-line-start). The kernel's
        # classification stops at the 64B chunk containing the first
        # \r\n\r\n, so the LF table is chunk-granular-truncated there.
        bnd = buf.find(b"\r\n\r\n")
        region_end = ln if bnd < 0 else min(ln, ((bnd + 3) // 64 + 1) * 64)
        lfs = []
        p = -1
        while True:
            p = buf.find(b"\n", p + 1, region_end)
            if p < 0:
                break
            lfs.append(p)
        nlf = len(lfs)
        j_empty = -1
        for j in range(1, min(nlf - 1, 63) + 1):
            if lfs[j] <= lfs[j - 1] + 2:
                j_empty = j
                break
        if j_empty > 0:
            body_off = lfs[j_empty] + 1
            hdr_last = j_empty - 1
        elif nlf - 1 > 63:
            flags |= FL_NEEDS_HOST  # header-line budget: host serves
            hdr_last = 63
        else:
            flags |= FL_ERR_PARSE  # LFs ran out before a blank line
            hdr_last = min(nlf - 1, 63)
        for j in range(1, hdr_last + 1):
            ls = lfs[j - 1] + 1
            le = lfs[j] - 1
            co = buf.find(b":", ls, le)
            if co < 0:
                continue
            nlen = co - ls
            vs = co + 1
            while vs < le and buf[vs] == 0x20:
                vs += 1
            vlen = le - vs
            if nlen == 14 and _ieq(buf, ls, b"content-length"):
                v = 0
                for i in range(vlen):
                    d = buf[vs + i]
                    if 0x30 <= d <= 0x39:
                        v = v * 10 + d - 0x30
                clen = v
            elif nlen == 10 and _ieq(buf, ls, b"connection"):
                if vlen == 5 and _ieq(buf, vs, b"close"):
                    keep_alive = False
                elif vlen == 10 and _ieq(buf, vs, b"keep-alive"):
                    keep_alive = True  # HTTP/1.0 opt-in
            elif nlen == 12 and _ieq(buf, ls, b"content-type"):
                if vlen >= 16 and _ieq(buf, vs, b"application/json"):
                    flags |= FL_JSON_CT
            elif nlen == 13 and _ieq(buf, ls, b"authorization"):
                auth_off, auth_len = vs, vlen
            elif nlen == 13 and _ieq(buf, ls, b"if-none-match"):
                inm_off, inm_len = vs, vlen
            elif nlen == 17 and _ieq(buf, ls, b"transfer-encoding"):
                flags |= FL_NEEDS_HOST
            elif nlen == 15 and _ieq(buf, ls, b"accept-encoding"):
                if b"gzip" in buf[vs:vs + vlen]:
                    flags |= FL_ACCEPT_GZIP
        body_len = max(0, ln - body_off)
        if 0 < clen < body_len:
            body_len = clen
        if keep_alive:
            flags |= FL_KEEP_ALIVE
        F[FI_BODY_OFF] = body_off
        F[FI_BODY_LEN] = body_len
        F[FI_CLEN] = clen
        F[FI_AUTH_OFF] = auth_off
        F[FI_AUTH_LEN] = auth_len
        F[FI_INM_OFF] = inm_off
        F[FI_INM_LEN] = inm_len

        # trie walk
        node = 0
        best_prefix = int(trie["node_prefix"][0])
        nparams = 0
        pspans = []
        pos, pend = path_off, path_off + int(F[FI_PATH_LEN])
        # StrictSlash(false) parity: trailing slash never matches a
        # non-prefix route (kernel does the same)
        dead = (int(F[FI_PATH_LEN]) > 1 and
                buf[path_off + int(F[FI_PATH_LEN]) - 1:
                    path_off + int(F[FI_PATH_LEN])] == b"/")
        while pos < pend and not dead:
            while pos < pend and buf[pos] == ord("/"):
                pos += 1
            if pos >= pend:
                break
            se = pos
            while se < pend and buf[se] != ord("/"):
                se += 1
            slen = se - pos
            nxt = -1
            cf = int(trie["node_child_first"][node])
            cc = int(trie["node_child_count"][node])
            for ci in range(cf, cf + cc):
                if int(trie["child_seg_len"][ci]) != slen:
                    continue
                so = int(trie["child_seg_off"][ci])
                if bytes(trie["seg_blob"][so:so + slen]) == buf[pos:se]:
                    nxt = int(trie["child_node"][ci])
                    break
            if nxt < 0:
                pc = int(trie["node_param"][node])
                if pc >= 0:
                    if nparams < MAX_PARAMS:
                        pspans.append((pos, slen))
                        nparams += 1
                    else:
                        flags |= FL_NEEDS_HOST
                    nxt = pc
            if nxt < 0:
                dead = True
                break
            node = nxt
            if int(trie["node_prefix"][node]) >= 0:
                best_prefix = int(trie["node_prefix"][node])
            pos = se
        route = -1
        if not dead:
            m = method if method >= 0 else 0
            route = int(trie["node_route"][node * N_METHODS_PAD + m])
        if route < 0:
            route = best_prefix
        F[FI_ROUTE] = route
        for k in range(MAX_PARAMS):
            if k < len(pspans):
                F[FI_PARAM0 + 2 * k] = pspans[k][0]
                F[FI_PARAM0 + 2 * k + 1] = pspans[k][1]
        kind, status = HK_STATIC, 404
        if 0 <= route < n_routes:
            kind = int(handler_tab[route * 4 + 0])
            status = int(handler_tab[route * 4 + 3])
        if flags & FL_IS_OPTIONS:
            kind, status = HK_STATIC, 200
        # kernel parity: large echo bodies go to the host so envelope
        # responses stay inside the LDS working set
        if kind == HK_ECHO_JSON and int(F[FI_BODY_LEN]) > MAX_SLOT - 600:
            kind = HK_HOST
        if flags & (FL_ERR_PARSE | FL_NEEDS_HOST):
            kind = HK_HOST
        F[FI_KIND] = kind
        F[FI_STATUS] = status
        F[FI_FLAGS] = flags
    return fields


_REASONS_DEV = {200: b"OK", 404: b"Not Found", 405: b"Method Not Allowed",
                401: b"Unauthorized"}
_CT_STRS = [b"Content-Type: application/json\r\n",
            b"Content-Type: image/x-icon\r\n",
            b"Content-Type: application/octet-stream\r\n",
            b"Content-Type: text/plain\r\n"]
_CORS = (b"Access-Control-Allow-Origin: *\r\n"
         b"Access-Control-Allow-Methods: POST, GET, OPTIONS, PUT, DELETE\r\n")


def _dev_reason(st: int) -> bytes:
    return _REASONS_DEV.get(st, b"Internal Server Error")


def _json_body_valid(body: bytes) -> bool:
    """Mirror of the kernel's structural validation (NOT full JSON)."""
    depth = 0
    in_str = False
    i = 0
    while i < len(body):
        c = body[i:i + 1]
        if in_str:
            if c == b"\\":
                i += 1
            elif c == b'"':
                in_str = False
        elif c == b'"':
            in_str = True
        elif c in (b"{", b"["):
            depth += 1
        elif c in (b"}", b"]"):
            depth -= 1
            if depth < 0:
                return False
        i += 1
    return depth == 0 and not in_str and len(body) > 0


# ---- template / KV / JSON-bind mirrors (gofr_kernels.hip helpers) ----------

_HEXV = {c: i for i, c in enumerate(b"0123456789abcdef")}
_HEXV.update({c: i for i, c in enumerate(b"0123456789ABCDEF")})


def pct_decode(s: bytes):
    """Mirror of pct_decode_inplace: strict %XX decode ('+' untouched).
    Returns None on an invalid escape (kernel: host fallback). STRICT
    hex only — Python's int(x, 16) accepts ' 1'/'+1', the kernel's
    hexval does not (divergence caught by the 20k fuzz campaign)."""
    out = bytearray()
    i = 0
    n = len(s)
    while i < n:
        c = s[i]
        if c == 0x25:  # %
            if i + 2 >= n:
                return None
            hi = _HEXV.get(s[i + 1])
            lo = _HEXV.get(s[i + 2])
            if hi is None or lo is None:
                return None
            c = (hi << 4) | lo
            i += 2
        out.append(c)
        i += 1
    return bytes(out)


def fnv1a64(data: bytes) -> int:
    h = 0xcbf29ce484222325
    for b in data:
        h = ((h ^ b) * 0x100000001b3) & M64
    return h


def q_find_py(query: bytes, key: bytes):
    """Mirror of q_find: raw value span of `key` in the query string."""
    i = 0
    n = len(query)
    while i < n:
        ke = i
        while ke < n and query[ke] not in (0x3D, 0x26):  # '=' '&'
            ke += 1
        if query[i:ke] == key:
            if ke < n and query[ke] == 0x3D:
                ve = ke + 1
                while ve < n and query[ve] != 0x26:
                    ve += 1
                return query[ke + 1:ve]
            return b""
        while ke < n and query[ke] != 0x26:
            ke += 1
        i = ke + 1
    return None


def splice_py(src: bytes, mode: int) -> bytes:
    """Mirror of splice_bytes: TM_PCT decode then TM_JESC escaping."""
    out = bytearray()
    i = 0
    n = len(src)
    while i < n:
        c = src[i]
        if mode & TM_PCT:
            if c == 0x2B:  # '+'
                c = 0x20
            elif c == 0x25 and i + 2 < n:
                hi = _HEXV.get(src[i + 1])
                lo = _HEXV.get(src[i + 2])
                if hi is not None and lo is not None:
                    c = (hi << 4) | lo
                    i += 2
        if mode & TM_JESC:
            if c in (0x22, 0x5C):  # '"' '\\'
                out.append(0x5C)
                out.append(c)
                i += 1
                continue
            if c < 0x20:
                out += b"\\u00" + f"{c:02x}".encode()
                i += 1
                continue
        out.append(c)
        i += 1
    return bytes(out)


def json_top_fields_py(body: bytes):
    """Mirror of json_top_fields: [(key, val_off, val_len)] of a JSON
    object's top-level entries (value spans raw, ws-trimmed); [] when
    not an object / malformed / > MAX_JSON_FIELDS semantics."""
    n = len(body)
    i = 0
    while i < n and body[i] <= 0x20:
        i += 1
    if i >= n or body[i] != 0x7B:  # '{'
        return []
    i += 1
    fields = []
    while len(fields) < MAX_JSON_FIELDS:
        while i < n and (body[i] <= 0x20 or body[i] == 0x2C):
            i += 1
        if i >= n:
            return []
        if body[i] == 0x7D:  # '}'
            return fields
        if body[i] != 0x22:  # '"'
            return []
        i += 1
        ks = i
        while i < n and body[i] != 0x22:
            if body[i] == 0x5C:
                i += 1
            i += 1
        if i >= n:
            return []
        ke = i
        i += 1
        while i < n and body[i] <= 0x20:
            i += 1
        if i >= n or body[i] != 0x3A:  # ':'
            return []
        i += 1
        while i < n and body[i] <= 0x20:
            i += 1
        if i >= n:
            return []
        vs = i
        if body[i] == 0x22:
            i += 1
            while i < n and body[i] != 0x22:
                if body[i] == 0x5C:
                    i += 1
                i += 1
            if i >= n:
                return []
            i += 1
        elif body[i] in (0x7B, 0x5B):  # '{' '['
            depth = 0
            in_str = False
            while i < n:
                c = body[i]
                if in_str:
                    if c == 0x5C:
                        i += 1
                    elif c == 0x22:
                        in_str = False
                elif c == 0x22:
                    in_str = True
                elif c in (0x7B, 0x5B):
                    depth += 1
                elif c in (0x7D, 0x5D):
                    depth -= 1
                    if depth == 0:
                        i += 1
                        break
                i += 1
            if depth != 0:
                return []
        else:
            while i < n and body[i] not in (0x2C, 0x7D) and body[i] > 0x20:
                i += 1
        fields.append((body[ks:ke], vs, i - vs))
        while i < n and body[i] <= 0x20:
            i += 1
        if i < n and body[i] == 0x2C:
            continue
        if i < n and body[i] == 0x7D:
            return fields
        return []
    return fields


def render_template(blob: bytes, prog_off: int, req: bytes, F,
                    cap: int = MAX_SLOT - 1024):
    """Mirror of the HK_TEMPLATE branch: render the blob-encoded piece
    program against one parsed request. Returns bytes, or None on
    overflow (kernel renders the 500 envelope)."""
    prog = np.frombuffer(blob, np.int32, offset=prog_off,
                         count=1).item()
    pieces = np.frombuffer(blob, np.int32, offset=prog_off + 4,
                           count=prog * 4).reshape(prog, 4)
    jfields = None
    out = bytearray()
    body = req[int(F[FI_BODY_OFF]):int(F[FI_BODY_OFF]) +
               int(F[FI_BODY_LEN])]
    for op, a, b, mode in pieces:
        op, a, b, mode = int(op), int(a), int(b), int(mode)
        if op == TP_LIT:
            out += blob[a:a + b]
        elif op == TP_PATH:
            if a >= MAX_PARAMS:
                continue
            po = int(F[FI_PARAM0 + 2 * a])
            pl = int(F[FI_PARAM0 + 2 * a + 1])
            out += splice_py(req[po:po + pl], mode & ~TM_PCT)
        elif op == TP_QUERY:
            q = req[int(F[FI_QUERY_OFF]):int(F[FI_QUERY_OFF]) +
                    int(F[FI_QUERY_LEN])]
            val = q_find_py(q, blob[a:a + b])
            if val is not None:
                out += splice_py(val, mode)
        elif op == TP_JFIELD:
            if jfields is None:
                jfields = json_top_fields_py(body)
            for key, vs, vl in jfields:
                if key == blob[a:a + b]:
                    if mode & TM_JSTR and vl >= 2 and body[vs] == 0x22:
                        vs, vl = vs + 1, vl - 2
                    out += splice_py(body[vs:vs + vl],
                                     (mode & ~TM_PCT) & ~TM_JSTR)
                    break
    if len(out) > cap:
        return None
    return bytes(out)


def build_kv_table(store: dict, blob: bytearray, slot0: int):
    """Compile one KV store into open-addressing rows + key/value bytes
    appended to `blob` (shared kv_blob). Row layout (int32 x6, the
    kernel's probe layout): [h_lo, h_hi, key_off, key_len, val_off,
    val_len]; empty slots have val_len == -1. Values are pre-wrapped
    {"data":...} envelopes. Returns (rows int32[nslots*6], nslots)."""
    import json as _json
    n = len(store)
    nslots = 1
    while nslots < max(4, 2 * n):
        nslots *= 2
    rows = np.zeros((nslots, 6), np.int32)
    rows[:, 5] = -1
    for key, value in store.items():
        kb = key.encode("utf-8") if isinstance(key, str) else bytes(key)
        vb = (b'{"data":' +
              _json.dumps(value, separators=(",", ":"),
                          ensure_ascii=False).encode("utf-8") + b"}")
        h = fnv1a64(kb)
        koff = len(blob)
        blob += kb
        voff = len(blob)
        blob += vb
        idx = h % nslots
        while rows[idx, 5] >= 0:
            idx = (idx + 1) % nslots
        rows[idx] = (np.int32(h & 0xFFFFFFFF) if (h & 0xFFFFFFFF) < 2**31
                     else np.int32((h & 0xFFFFFFFF) - 2**32),
                     np.int32(h >> 32) if (h >> 32) < 2**31
                     else np.int32((h >> 32) - 2**32),
                     koff, len(kb), voff, len(vb))
    return rows.reshape(-1), nslots


def kv_lookup_mirror(kv_tab, kv_blob: bytes, slot0: int, nslots: int,
                     key: bytes):
    """Mirror of the HK_KV probe loop."""
    if nslots <= 0 or not key:
        return None
    h = fnv1a64(key)
    h_lo = np.int32(h & 0xFFFFFFFF) if (h & 0xFFFFFFFF) < 2**31 \
        else np.int32((h & 0xFFFFFFFF) - 2**32)
    h_hi = np.int32(h >> 32) if (h >> 32) < 2**31 \
        else np.int32((h >> 32) - 2**32)
    idx = h % nslots
    for _ in range(nslots):
        row = kv_tab[(slot0 + idx) * 6:(slot0 + idx) * 6 + 6]
        if row[5] < 0:
            return None
        if row[0] == h_lo and row[1] == h_hi and row[3] == len(key) and \
                kv_blob[int(row[2]):int(row[2]) + len(key)] == key:
            return kv_blob[int(row[4]):int(row[4]) + int(row[5])]
        idx = (idx + 1) % nslots
    return None


_ETAG_B = ((np.arange(64)[:, None] * 17 + np.arange(16)[None, :] * 29
            + 3) % 251 - 125).astype(np.int8)
_ETAG_SALT = (np.arange(256, dtype=np.uint32) * 2 + 1).reshape(16, 16)


def etag_u32(body: bytes) -> int:
    """Mirror of mfma_etag_wave (the MFMA batched body hash): tiles of
    16x64 i8 against the fixed coefficient matrix, uint32 fold."""
    n = len(body)
    pad = (-n) % 1024
    a = np.frombuffer(body + b"\x00" * pad, np.uint8).astype(np.int8)
    tiles = a.reshape(-1, 16, 64)
    state = np.zeros((16, 16), np.uint32)
    for t in range(tiles.shape[0]):
        d = (tiles[t].astype(np.int32) @ _ETAG_B.astype(np.int32))
        state = state * np.uint32(33) + d.astype(np.uint32)
    h = np.uint32(0)
    for v in (state * _ETAG_SALT).reshape(-1):
        h ^= v
    return int(h ^ np.uint32(n))


def cpu_respond(reqs: np.ndarray, req_off: np.ndarray,
                fields: np.ndarray, rslot: int,
                handler_tab: np.ndarray, blob: bytes,
                host_blob: bytes, host_tab: np.ndarray,
                seed: int, auth_env=(0, 0), gzip_min: int = 0,
                etag_on: bool = False, date29: bytes = b"",
                kv_tab=None, kv_blob: bytes = b""):
    """Mirror of k_respond. Returns (resp uint8 [n*rslot], resp_len int32)."""
    n = len(fields)
    host_tab = np.asarray(host_tab, np.int32).reshape(-1)
    n_routes = len(handler_tab) // 4
    resp = np.zeros(n * rslot, np.uint8)
    resp_len = np.zeros(n, np.int32)
    for r in range(n):
        F = fields[r]
        base = int(req_off[r])
        kind = int(F[FI_KIND])
        status = int(F[FI_STATUS])
        flags = int(F[FI_FLAGS])
        if flags & FL_EMPTY:
            resp_len[r] = 0
            fields[r][FI_RESP_LEN] = 0
            fields[r][FI_RESP_OFF] = r * rslot
            continue
        keep = bool(flags & FL_KEEP_ALIVE)
        is_options = bool(flags & FL_IS_OPTIONS)
        body_src = b""
        env = False
        ct_id = 0
        if is_options:
            pass
        elif kind == HK_ECHO_JSON:
            bo, bl = int(F[FI_BODY_OFF]), int(F[FI_BODY_LEN])
            body_src = reqs[base + bo:base + bo + bl].tobytes()
            env = True
            if not _json_body_valid(body_src):
                status = 500
                elen = int.from_bytes(blob[:4], "little")
                body_src = blob[4:4 + elen]
                env = False
        elif kind == HK_STATIC:
            route = int(F[FI_ROUTE])
            if 0 <= route < n_routes:
                off = int(handler_tab[route * 4 + 1])
                ln = int(handler_tab[route * 4 + 2])
                body_src = blob[off:off + ln]
        elif kind == HK_TEMPLATE and not (flags & FL_AUTH_FAIL):
            route = int(F[FI_ROUTE])
            prog_off = int(handler_tab[route * 4 + 1])
            raw = reqs[base:base + MAX_SLOT].tobytes()
            rendered = render_template(blob, prog_off, raw, F)
            if rendered is None:
                status = 500
                elen = int.from_bytes(blob[:4], "little")
                body_src = blob[4:4 + elen]
            else:
                body_src = rendered
        elif kind == HK_KV and not (flags & FL_AUTH_FAIL):
            route = int(F[FI_ROUTE])
            slot0 = int(handler_tab[route * 4 + 1])
            nslots = int(handler_tab[route * 4 + 2])
            po = int(F[FI_PARAM0])
            pl = int(F[FI_PARAM0 + 1])
            key = reqs[base + po:base + po + pl].tobytes()
            val = kv_lookup_mirror(kv_tab, kv_blob, slot0, nslots, key)
            if val is not None:
                body_src = val
            else:
                status = 404
                elen = int.from_bytes(blob[:4], "little")
                mlen = int.from_bytes(blob[4 + elen:8 + elen], "little")
                body_src = blob[8 + elen:8 + elen + mlen]
        if kind == HK_HOST and not is_options and \
                not (flags & FL_AUTH_FAIL):
            off, ln, status, ct_id = (int(host_tab[r * 4 + i])
                                      for i in range(4))
            body_src = host_blob[off:off + ln]
            if status == 0:
                status = 500
                elen = int.from_bytes(blob[:4], "little")
                body_src = blob[4:4 + elen]
        if (flags & FL_AUTH_FAIL) and not is_options:
            status = 401
            body_src = blob[auth_env[0]:auth_env[0] + auth_env[1]]
            env = False
            ct_id = 0

        body_total = len(body_src) + (9 if env else 0)
        content_enc = False
        if (gzip_min > 0 and (flags & FL_ACCEPT_GZIP) and not is_options
                and gzip_min <= body_total <= MAX_SLOT):
            plain = (b'{"data":' + body_src + b"}") if env else body_src
            gz = gzip_static_mirror(plain)
            if gz is not None:
                content_enc = True
                body_src = gz
                body_total = len(gz)
                env = False
        reason = _dev_reason(status)
        h1 = splitmix64(seed ^ r)
        h2 = splitmix64(h1 ^ 0xD1B54A32D192ED03)
        corr = f"{h1:016x}{h2:016x}".encode()
        final_body = (b'{"data":' + body_src + b"}") if env else body_src
        etag_hdr = b""
        not_modified = False
        if etag_on:
            tag = f"{etag_u32(final_body):08x}"
            etag_hdr = b'ETag: "' + tag.encode() + b'"\r\n'
            io, il = int(F[FI_INM_OFF]), int(F[FI_INM_LEN])
            if status == 200 and il == 10:
                inm = reqs[base + io:base + io + il].tobytes()
                if inm == b'"' + tag.encode() + b'"':
                    not_modified = True
        head = (b"HTTP/1.1 " + f"{status:03d}".encode() + b" " + reason +
                b"\r\n" +
                (b"Date: " + date29 + b"\r\n" if date29 else b"") +
                _CT_STRS[ct_id] +
                (b"Content-Encoding: gzip\r\n" if content_enc else b"") +
                etag_hdr +
                _CORS +
                b"X-Correlation-ID: " + corr + b"\r\n" +
                b"Content-Length: " + str(body_total).encode() + b"\r\n" +
                (b"Connection: keep-alive\r\n\r\n" if keep
                 else b"Connection: close\r\n\r\n"))
        if not_modified:
            # in-place 304 rewrite quirks mirrored from the kernel:
            # status digits patched, reason kept, CL zero-padded
            cl = b"Content-Length: " + str(body_total).encode()
            head = head.replace(
                cl, b"Content-Length: " + b"0" * len(str(body_total)), 1)
            head = head[:9] + b"304" + head[12:]
            payload = head
        elif int(F[FI_METHOD]) == 6:  # HEAD (mirror of M_HEAD): no body
            payload = head
        elif env:
            payload = head + b'{"data":' + body_src + b"}"
        else:
            payload = head + body_src
        resp[r * rslot:r * rslot + len(payload)] = np.frombuffer(
            payload, np.uint8)
        resp_len[r] = len(payload)
        fields[r][FI_RESP_LEN] = len(payload)
        fields[r][FI_RESP_OFF] = r * rslot
    return resp, resp_len


GZ_HASH_BITS = 9
GZ_MIN_MATCH = 3
GZ_MAX_DIST = 2048
MAX_SLOT = 4096

_LEN_BASE = [3,4,5,6,7,8,9,10,11,13,15,17,19,23,27,31,
             35,43,51,59,67,83,99,115,131,163,195,227,258]
_LEN_EXTRA = [0,0,0,0,0,0,0,0,1,1,1,1,2,2,2,2,3,3,3,3,4,4,4,4,5,5,5,5,0]
_DIST_BASE = [1,2,3,4,5,7,9,13,17,25,33,49,65,97,129,193,257,385,513,769,
              1025,1537,2049,3073,4097,6145,8193,12289,16385,24577]
_DIST_EXTRA = [0,0,0,0,1,1,2,2,3,3,4,4,5,5,6,6,7,7,8,8,9,9,10,10,11,11,
               12,12,13,13]

import zlib as _zlib


def _bitrev(v, ln):
    r = 0
    for _ in range(ln):
        r = (r << 1) | (v & 1)
        v >>= 1
    return r


def _fixed_lit(sym):
    if sym < 144:
        return _bitrev(0x30 + sym, 8), 8
    if sym < 256:
        return _bitrev(0x190 + sym - 144, 9), 9
    if sym < 280:
        return _bitrev(sym - 256, 7), 7
    return _bitrev(0xC0 + sym - 280, 8), 8


def gzip_static_mirror(data: bytes, cap: int = MAX_SLOT - 512):
    """Byte-exact model of deflate_gzip_wave (greedy LZ77 + static
    huffman + gzip framing). Returns None when the output would exceed
    `cap` (the kernel falls back to uncompressed)."""
    n = len(data)
    out = bytearray(b"\x1f\x8b\x08\x00\x00\x00\x00\x00\x00\xff")
    bitbuf = 0
    nbits = 0

    def put(bits, ln):
        nonlocal bitbuf, nbits
        bitbuf |= bits << nbits
        nbits += ln
        while nbits >= 8:
            out.append(bitbuf & 0xFF)
            bitbuf >>= 8
            nbits -= 8

    put(1, 1)
    put(1, 2)
    hash_tab = [0] * (1 << GZ_HASH_BITS)
    pos = 0
    overflow = False
    while pos < n:
        cand = -1
        if pos + GZ_MIN_MATCH <= n:
            h = (((data[pos] | (data[pos + 1] << 8) |
                   (data[pos + 2] << 16)) * 0x9E3779B1) & 0xFFFFFFFF) \
                >> (32 - GZ_HASH_BITS)
            stored = hash_tab[h] - 1
            if stored >= 0 and pos - stored <= GZ_MAX_DIST and stored < pos:
                cand = stored
            hash_tab[h] = pos + 1
        mlen = 0
        if cand >= 0:
            while mlen < 258 and pos + mlen < n and \
                    data[cand + mlen] == data[pos + mlen]:
                mlen += 1
        if mlen >= GZ_MIN_MATCH:
            # largest i with mlen >= base[i] (mirror of the C loop)
            idx = max(i for i in range(29) if mlen >= _LEN_BASE[i])
            code, clen = _fixed_lit(257 + idx)
            put(code, clen)
            if _LEN_EXTRA[idx]:
                put(mlen - _LEN_BASE[idx], _LEN_EXTRA[idx])
            dist = pos - cand
            didx = max(i for i in range(30) if dist >= _DIST_BASE[i])
            put(_bitrev(didx, 5), 5)
            if _DIST_EXTRA[didx]:
                put(dist - _DIST_BASE[didx], _DIST_EXTRA[didx])
            for i in range(1, mlen):
                q = pos + i
                if q + GZ_MIN_MATCH <= n:
                    h2 = (((data[q] | (data[q + 1] << 8) |
                            (data[q + 2] << 16)) * 0x9E3779B1)
                          & 0xFFFFFFFF) >> (32 - GZ_HASH_BITS)
                    hash_tab[h2] = q + 1
            pos += mlen
        else:
            code, clen = _fixed_lit(data[pos])
            put(code, clen)
            pos += 1
        if len(out) - 10 > cap - 24:
            overflow = True
            break
    if overflow:
        return None
    code, clen = _fixed_lit(256)
    put(code, clen)
    if nbits > 0:
        out.append(bitbuf & 0xFF)
    crc = _zlib.crc32(data) & 0xFFFFFFFF
    out += crc.to_bytes(4, "little")
    out += (n & 0xFFFFFFFF).to_bytes(4, "little")
    if len(out) > cap:
        return None
    return bytes(out)


def cpu_auth(reqs: np.ndarray, req_off: np.ndarray, fields: np.ndarray,
             secret: bytes) -> None:
    """Mirror of k_auth: HMAC-SHA256 bearer check; sets FL_AUTH_FAIL."""
    import hashlib
    import hmac as hmac_mod
    for r in range(len(fields)):
        F = fields[r]
        flags = int(F[FI_FLAGS])
        if flags & (FL_ERR_PARSE | FL_IS_OPTIONS | FL_EMPTY):
            continue
        base = int(req_off[r])
        aoff, alen = int(F[FI_AUTH_OFF]), int(F[FI_AUTH_LEN])
        ok = False
        val = reqs[base + aoff:base + aoff + alen].tobytes()
        if alen == 69 and val[:5] == b"HMAC ":
            raw = reqs[base:base + 64].tobytes()
            sp = raw.index(b" ")
            msg = raw[:sp + 1] + reqs[
                base + F[FI_PATH_OFF]:
                base + F[FI_PATH_OFF] + F[FI_PATH_LEN]].tobytes()
            want = hmac_mod.new(secret, msg, hashlib.sha256).hexdigest()
            ok = hmac_mod.compare_digest(val[5:].decode("latin-1").lower(),
                                         want)
        if not ok:
            F[FI_FLAGS] = flags | FL_AUTH_FAIL
            F[FI_STATUS] = 401


MAX_PB_FIELDS = 16


def cpu_varint_spans(buf: np.ndarray, msg_off: np.ndarray,
                     msg_len: np.ndarray):
    """Mirror of k_varint_spans."""
    n = len(msg_len)
    out = np.zeros((n, MAX_PB_FIELDS, 4), np.int32)
    out_n = np.zeros(n, np.int32)
    data = buf.tobytes()
    for m in range(n):
        base = int(msg_off[m])
        ln = int(msg_len[m])
        pos = 0
        nf = 0
        bad = False
        while pos < ln and nf < MAX_PB_FIELDS:
            tag = 0
            shift = 0
            while pos < ln:
                b = data[base + pos]
                pos += 1
                tag |= (b & 0x7F) << shift
                if not b & 0x80:
                    break
                shift += 7
                if shift > 63:
                    bad = True
                    break
            if bad:
                break
            fno, wt = tag >> 3, tag & 7
            if wt == 0:
                v = 0
                shift = 0
                while pos < ln:
                    b = data[base + pos]
                    pos += 1
                    v |= (b & 0x7F) << shift
                    if not b & 0x80:
                        break
                    shift += 7
                    if shift > 63:
                        bad = True
                        break
                if bad:
                    break
                out[m, nf] = (fno, 0,
                              np.int32(v & 0xFFFFFFFF) if v & 0xFFFFFFFF < (1 << 31)
                              else np.int32((v & 0xFFFFFFFF) - (1 << 32)),
                              np.int32((v >> 32) & 0xFFFFFFFF)
                              if (v >> 32) < (1 << 31)
                              else np.int32((v >> 32) - (1 << 32)))
            elif wt == 2:
                l2 = 0
                shift = 0
                while pos < ln:
                    b = data[base + pos]
                    pos += 1
                    l2 |= (b & 0x7F) << shift
                    if not b & 0x80:
                        break
                    shift += 7
                if pos + l2 > ln:
                    bad = True
                    break
                out[m, nf] = (fno, 2, base + pos, l2)
                pos += l2
            elif wt == 1:
                if pos + 8 > ln:
                    bad = True
                    break
                out[m, nf] = (fno, 1, base + pos, 8)
                pos += 8
            elif wt == 5:
                if pos + 4 > ln:
                    bad = True
                    break
                out[m, nf] = (fno, 5, base + pos, 4)
                pos += 4
            else:
                bad = True
                break
            nf += 1
        out_n[m] = -1 if bad else nf
    return out, out_n


def cpu_grpc_echo(buf: np.ndarray, spans: np.ndarray, span_n: np.ndarray,
                  rslot: int):
    """Mirror of k_grpc_echo: gRPC length-prefixed HelloResponse frames
    {message: "Hello <name>!"} from the varint span tables ("World" when
    the name field is empty — examples/grpc-server semantics)."""
    n = len(span_n)
    out = np.zeros(n * rslot, np.uint8)
    out_len = np.zeros(n, np.int32)
    data = buf.tobytes()
    for m in range(n):
        nf = int(span_n[m])
        name = b""
        for i in range(max(0, nf)):
            if spans[m, i, 0] == 1 and spans[m, i, 1] == 2:
                off, ln = int(spans[m, i, 2]), int(spans[m, i, 3])
                name = data[off:off + ln]
        if not name:
            name = b"World"
        payload = b"Hello " + name + b"!"
        vlen = 1 if len(payload) < 128 else 2
        msg_len = 1 + vlen + len(payload)
        if nf < 0 or 5 + msg_len > rslot:
            out_len[m] = -1
            continue
        frame = bytearray()
        frame.append(0)
        frame += msg_len.to_bytes(4, "big")
        frame.append(0x0A)
        if vlen == 1:
            frame.append(len(payload))
        else:
            frame.append((len(payload) & 0x7F) | 0x80)
            frame.append(len(payload) >> 7)
        frame += payload
        out[m * rslot:m * rslot + len(frame)] = np.frombuffer(
            bytes(frame), np.uint8)
        out_len[m] = len(frame)
    return out, out_len
