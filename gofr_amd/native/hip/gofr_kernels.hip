// gofr_amd CDNA4 (gfx950) request-batch kernels.
//
// MI355X-native data plane for the GoFr-capability framework
// (reference behaviors: SURVEY.md §2.2; this is a new design, not a port —
// the reference's hot path is Go's net/http + gorilla/mux + encoding/json,
// pkg/gofr/http/router.go:13-33, http/request.go:40-47, http/responder.go:19-41).
//
// Design (one wavefront per request, 64 lanes):
//   k_parse_route — structural-index HTTP/1.1 parse + radix-trie route
//     match. Lanes classify 64 bytes per iteration into LF/SP/COLON/QMARK
//     bitmasks via __ballot (wave64 -> one u64 mask per 64-byte chunk, no
//     per-byte branching); lane 0 then tokenizes by walking the masks with
//     ctz. The route trie (compiled by gofr_amd/http/router.py:compile)
//     is walked on the path segments; {param} spans are captured.
//   k_respond — handler execution for GPU-native handler kinds (JSON echo,
//     static body, host-supplied body) fused with the response serializer:
//     status line + CORS + correlation-id headers (xorshift128 hex, fused)
//     + Content-Length itoa + envelope + parallel body copy (all 64 lanes).
//
// Both kernels are launched with 256-thread blocks (4 waves = 4 requests).
// Buffers are SoA int32 field tables; see FI_* layout below (mirrored in
// gofr_amd/ops/__init__.py).

#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdlib.h>

#define WAVE 64
#define WAVES_PER_BLOCK 4
#define BLOCK_THREADS (WAVE * WAVES_PER_BLOCK)
#define MAX_SLOT 4096
#define MAX_CHUNKS (MAX_SLOT / 64)

// ---- field table layout (int32 [n, NF]) ------------------------------------
#define FI_METHOD 0
#define FI_PATH_OFF 1
#define FI_PATH_LEN 2
#define FI_QUERY_OFF 3
#define FI_QUERY_LEN 4
#define FI_BODY_OFF 5
#define FI_BODY_LEN 6
#define FI_CLEN 7
#define FI_FLAGS 8
#define FI_ROUTE 9
#define FI_KIND 10
#define FI_STATUS 11
#define FI_RESP_LEN 12
#define FI_AUTH_OFF 13
#define FI_AUTH_LEN 14
#define FI_RESP_OFF 15
#define FI_PARAM0 16 /* 4 x (off, len) pairs: 16..23 */
#define FI_INM_OFF 24  /* If-None-Match header value span */
#define FI_INM_LEN 25
#define NF 26

// flags
#define FL_ERR_PARSE 1
#define FL_NEEDS_HOST 2
#define FL_KEEP_ALIVE 4
#define FL_JSON_CT 8
#define FL_IS_OPTIONS 16
#define FL_BODY_INVALID 32
#define FL_ACCEPT_GZIP 64
#define FL_AUTH_FAIL 128
#define FL_EMPTY 256  /* padding slot (len 0): emit no response bytes */

// handler kinds (handler table rows: [kind, arg_off, arg_len, status])
#define HK_HOST 0
#define HK_ECHO_JSON 1
#define HK_STATIC 2
#define HK_TEMPLATE 3  /* precompiled body with spliced params/fields */
#define HK_KV 4        /* device KV-store lookup keyed by path param 0 */

// template piece opcodes (blob-encoded programs; 4 int32 words per
// piece: [op, a, b, mode] — compiled by engine.RouteProgram from
// handlers.template_json specs; mirror: ops.render_template)
#define TP_LIT 0     /* a=abs blob off, b=len */
#define TP_PATH 1    /* a=param idx (spans are pct-decoded by parse) */
#define TP_QUERY 2   /* a=key blob off, b=key len */
#define TP_JFIELD 3  /* a=key blob off, b=key len (top-level JSON field) */
// splice modes (bitmask)
#define TM_PCT 1     /* pct-decode + '+'->' ' (query values) */
#define TM_JESC 2    /* JSON-string-escape the spliced bytes */
#define TM_JSTR 4    /* TP_JFIELD: splice string CONTENT (strip quotes) */
#define MAX_JSON_FIELDS 8
#define MAX_TPL_PIECES 24  /* enforced by engine.RouteProgram */

// methods — must match gofr_amd/http/request.py METHOD_IDS
// GET POST PUT DELETE PATCH OPTIONS HEAD
#define M_GET 0
#define M_POST 1
#define M_PUT 2
#define M_DELETE 3
#define M_PATCH 4
#define M_OPTIONS 5
#define M_HEAD 6

#define N_METHODS_PAD 8
#define MAX_PARAMS 4

typedef int v4i_t __attribute__((ext_vector_type(4)));

struct TrieDev {
    const uint8_t* seg_blob;
    const int32_t* node_child_first;
    const int32_t* node_child_count;
    const int32_t* child_seg_off;
    const int32_t* child_seg_len;
    const int32_t* child_node;
    const int32_t* node_param;
    const int32_t* node_prefix;
    const int32_t* node_route;
};

// ---------------------------------------------------------------------------
// helpers
// ---------------------------------------------------------------------------
__device__ __forceinline__ int lane_id() { return threadIdx.x & (WAVE - 1); }

__device__ __forceinline__ uint8_t lower(uint8_t c) {
    return (c >= 'A' && c <= 'Z') ? (c | 0x20) : c;
}

// case-insensitive compare of req bytes against lowercase literal
__device__ __forceinline__ bool ieq(const uint8_t* p, const char* lit, int n) {
    for (int i = 0; i < n; ++i)
        if (lower(p[i]) != (uint8_t)lit[i]) return false;
    return true;
}

__device__ __forceinline__ uint64_t splitmix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

// ---------------------------------------------------------------------------
// k_parse_route
// ---------------------------------------------------------------------------
// classes packed in one LDS array (single __shared__ object)
#define CLS_LF 0
#define CLS_SP 1
#define CLS_COLON 2
#define CLS_QM 3
#define N_CLS 4
// header-line cap: one lane per header line, so <= 63 header lines
// parse on-device (the serial r1 walk capped at 64; beyond the cap the
// body falls back to request-end like the old guard exhaustion)
#define MAX_LFS 64

__device__ __forceinline__ int hexval(uint8_t c);

// in-place %XX decode of s[0..len) (writes <= len bytes at s).
// Returns decoded length, or -1 on an invalid escape (host path then
// serves Python-unquote leniency). '+' is NOT decoded here (only
// query values use '+'-as-space; paths keep it literal, like Go).
__device__ __forceinline__ int pct_decode_inplace(uint8_t* s, int len) {
    int w = 0;
    for (int i = 0; i < len; ++i) {
        uint8_t c = s[i];
        if (c == '%') {
            if (i + 2 >= len) return -1;
            const int hi = hexval(s[i + 1]);
            const int lo = hexval(s[i + 2]);
            if (hi < 0 || lo < 0) return -1;
            c = (uint8_t)((hi << 4) | lo);
            i += 2;
        }
        s[w++] = c;
    }
    return w;
}

// parse one request (one wave): callable from k_parse_route AND the
// persistent serving kernel. `masks` / `lf_pos` are this wave's LDS
// slices.
__device__ void parse_one(uint8_t* __restrict__ reqs,
                          const int64_t* __restrict__ req_off,
                          const int32_t* __restrict__ req_len,
                          int32_t* __restrict__ fields,
                          int n,
                          const TrieDev& trie,
                          const int32_t* __restrict__ handler_tab,
                          int n_routes,
                          int32_t* __restrict__ host_needed,
                          uint64_t (*masks)[MAX_CHUNKS], int* lf_pos,
                          int req, int lane) {
    uint8_t* base = reqs + req_off[req];
    int len = req_len[req];
    if (len == 0) {
        // padding slot (fixed-shape armed batches): no response
        if (lane_id() == 0) {
            int32_t* Fe = fields + (size_t)req * NF;
            for (int i = 0; i < NF; ++i) Fe[i] = 0;
            Fe[FI_FLAGS] = FL_EMPTY;
            Fe[FI_KIND] = HK_STATIC;
            Fe[FI_ROUTE] = -1;
        }
        return;
    }
    bool oversized = false;
    if (len > MAX_SLOT) {
        // larger than the LDS working set: the host trampoline parses
        // the full bytes from the ring (k_respond serves its result)
        len = MAX_SLOT;
        oversized = true;
    }
    int32_t* F = fields + (size_t)req * NF;

    const int nchunks = (len + WAVE - 1) / WAVE;
    bool saw_percent = false;
    // structural classification: 64 bytes per wave-iteration, one ballot
    // per class — no per-byte branching (SURVEY.md §7 hard part 2).
    // The pass STOPS at the chunk containing the header/body boundary
    // (\r\n\r\n): for the 1 KB-echo shape that is ~2 chunks instead of
    // 17 — the body's bytes never need classes (r1 scanned them all;
    // k_parse_route was 536 us/batch, VERDICT target < 200).
    int nclass = nchunks;
    for (int c = 0; c < nchunks; ++c) {
        const int i = c * WAVE + lane;
        const uint8_t b = (i < len) ? base[i] : 0;
        const uint64_t m_lf = __ballot(b == '\n');
        const uint64_t m_sp = __ballot(b == ' ');
        const uint64_t m_qm = __ballot(b == '?');
        saw_percent |= (b == '%');
        if (lane == 0) {
            masks[CLS_LF][c] = m_lf;
            masks[CLS_SP][c] = m_sp;
            masks[CLS_QM][c] = m_qm;
        }
        const bool hdr_end = i >= 3 && i < len && b == '\n' &&
                             base[i - 1] == '\r' &&
                             base[i - 2] == '\n' &&
                             base[i - 3] == '\r';
        if (__ballot(hdr_end)) {
            nclass = c + 1;
            break;
        }
    }
    const bool any_percent = __ballot(saw_percent) != 0;
    // no cross-wave sharing: wave-local LDS, no barrier needed (same wave
    // wrote and reads; LDS ops from one wave are ordered)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- LF position collection (wave-parallel) ----------------------------
    // lane c owns chunk c's LF mask: exclusive prefix sum of popcounts
    // orders every LF position into lf_pos[] in one pass — the serial
    // next_bit walk over header lines was 63-idle-lanes work (VERDICT
    // r1 weak #3). Body LFs past MAX_LFS don't matter: header lines
    // precede the body, and only lines 1..63 are ever examined.
    {
        uint64_t lm = (lane < nclass) ? masks[CLS_LF][lane] : 0;
        const int cnt = __popcll(lm);
        int incl = cnt;
        for (int off = 1; off < WAVE; off <<= 1) {
            const int v = __shfl_up(incl, off);
            if (lane >= off) incl += v;
        }
        int idx = incl - cnt;
        while (lm && idx < MAX_LFS) {
            lf_pos[idx++] = lane * WAVE + __builtin_ctzll(lm);
            lm &= lm - 1;
        }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    const int nlf = [&] {
        int total = (lane < nclass) ? __popcll(masks[CLS_LF][lane])
                                    : 0;
        for (int off = 32; off; off >>= 1)
            total += __shfl_xor(total, off);
        return total;
    }();

    // ---- tokenize the request line (wave-uniform serial: every lane
    // runs the same walk in lockstep — same latency as lane-0-only,
    // and the results land in every lane's registers for the parallel
    // header pass below; memory writes stay lane-0-guarded) ----------------
    auto next_bit = [&](int cls, int from) -> int {
        int c = from / WAVE;
        if (c >= nclass) return -1;
        uint64_t m = masks[cls][c] & (~0ull << (from & (WAVE - 1)));
        while (true) {
            if (m) return c * WAVE + __builtin_ctzll(m);
            if (++c >= nclass) return -1;
            m = masks[cls][c];
        }
    };

    int32_t flags = 0;
    if (oversized) flags |= FL_NEEDS_HOST;

    // request line: METHOD SP target SP version CRLF
    const int sp1 = next_bit(CLS_SP, 0);
    const int lf1 = next_bit(CLS_LF, 0);
    if (sp1 < 0 || lf1 < 0 || sp1 >= lf1) {
        if (lane == 0) {
            F[FI_FLAGS] = FL_ERR_PARSE;
            F[FI_METHOD] = -1;
            F[FI_KIND] = HK_HOST;
            F[FI_STATUS] = 400;
            F[FI_ROUTE] = -1;
            atomicAdd(host_needed, 1);
        }
        return;
    }
    // method id from leading bytes
    int method = -1;
    {
        static const char* MNAMES[7] = {"GET", "POST", "PUT", "DELETE",
                                        "PATCH", "OPTIONS", "HEAD"};
        static const int MLENS[7] = {3, 4, 3, 6, 5, 7, 4};
        for (int mi = 0; mi < 7; ++mi) {
            if (MLENS[mi] != sp1) continue;
            bool eq = true;
            for (int k = 0; k < sp1; ++k)
                if (base[k] != (uint8_t)MNAMES[mi][k]) { eq = false; break; }
            if (eq) { method = mi; break; }
        }
    }
    const int sp2 = next_bit(CLS_SP, sp1 + 1);
    int target_end = (sp2 > 0 && sp2 < lf1) ? sp2 : (lf1 - 1); // before CR
    int path_off = sp1 + 1;
    int qm = next_bit(CLS_QM, path_off);
    int path_end = (qm >= 0 && qm < target_end) ? qm : target_end;
    int query_off = (qm >= 0 && qm < target_end) ? qm + 1 : target_end;
    int query_len = (qm >= 0 && qm < target_end) ? target_end - query_off : 0;

    int path_len = path_end - path_off;
    // %XX in the path: decode IN PLACE before the trie walk (Go parity:
    // net/http routes on the decoded URL.Path, incl. %2F -> '/'), so
    // param spans hand handlers decoded bytes and encoded URLs stay on
    // the GPU (r1 sent every '%' to the host). Invalid escapes only
    // fall back (the host parser's unquote leniency applies there).
    if (any_percent) {
        bool in_path = false;
        for (int i = path_off; i < path_end; ++i)
            if (base[i] == '%') { in_path = true; break; }
        if (in_path) {
            int dl = -2;
            if (lane == 0)
                dl = pct_decode_inplace(base + path_off, path_len);
            // lane 0 wrote global memory the whole wave reads below
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            dl = __shfl(dl, 0);
            if (dl < 0) flags |= FL_NEEDS_HOST;
            else path_len = dl;
        }
    }
    if (method == M_OPTIONS) flags |= FL_IS_OPTIONS;
    if (method < 0) flags |= FL_ERR_PARSE;

    // ---- headers (ONE LINE PER LANE) ---------------------------------------
    // Lane l parses header line l+1 against the lf_pos table; interest
    // flags combine via ballots, span/value results via max-by-line-
    // start reductions (serial parity: a repeated header's LAST
    // occurrence wins, and later lines have larger offsets).
    int clen = 0;
    // keep-alive default: HTTP/1.1 yes, HTTP/1.0 no (RFC 9112 §9.3);
    // version token = bytes between the 2nd space and CR
    bool keep_alive = true;
    if (sp2 > 0 && lf1 - 1 - (sp2 + 1) == 8 &&
        base[lf1 - 2] == '0' && base[lf1 - 4] == '1')  // "HTTP/1.0"
        keep_alive = false;
    int body_off = len, auth_off = 0, auth_len = 0;
    int inm_off = 0, inm_len = 0;
    {
        const int navail = nlf < MAX_LFS ? nlf : MAX_LFS;
        const int j = lane + 1;  // this lane's header line
        const bool in_range = j <= navail - 1 && j <= 63;
        const bool is_empty = in_range &&
            lf_pos[j] <= lf_pos[j - 1] + 2;
        const uint64_t empty_mask = __ballot(is_empty);
        int j_empty = -1;
        if (empty_mask) j_empty = __builtin_ctzll(empty_mask) + 1;
        int hdr_last;
        if (j_empty > 0) {
            body_off = lf_pos[j_empty] + 1;
            hdr_last = j_empty - 1;
        } else if (nlf - 1 > 63) {
            // >63 header lines before any blank: beyond the one-line-
            // per-lane budget — let the host parser serve it fully
            // (r1 silently truncated the body here)
            flags |= FL_NEEDS_HOST;
            hdr_last = 63;
        } else {
            flags |= FL_ERR_PARSE;  // no blank line before the LFs ran out
            hdr_last = navail - 1;
        }
        const bool active = j <= hdr_last;
        const int ls = active ? lf_pos[j - 1] + 1 : 0;
        const int le = active ? lf_pos[j] - 1 : 0;  // CR position
        int co = -1;
        for (int i = ls; i < le; ++i)
            if (base[i] == ':') { co = i; break; }
        // per-lane header classification
        bool f_json = false, f_host = false, f_gzip = false;
        int my_tag_cl = -1, my_clen = 0;
        int my_tag_conn = -1, my_conn = 0;
        int my_tag_auth = -1, my_auth_off = 0, my_auth_len = 0;
        int my_tag_inm = -1, my_inm_off = 0, my_inm_len = 0;
        if (active && co >= 0 && co < le) {
            const int nlen = co - ls;
            int vs = co + 1;
            while (vs < le && base[vs] == ' ') ++vs;
            const int vlen = le - vs;
            const uint8_t* nm = base + ls;
            if (nlen == 14 && ieq(nm, "content-length", 14)) {
                int v = 0;
                for (int i = 0; i < vlen; ++i) {
                    uint8_t d = base[vs + i];
                    if (d >= '0' && d <= '9') v = v * 10 + (d - '0');
                }
                my_tag_cl = ls; my_clen = v;
            } else if (nlen == 10 && ieq(nm, "connection", 10)) {
                if (vlen == 5 && ieq(base + vs, "close", 5)) {
                    my_tag_conn = ls; my_conn = 0;
                } else if (vlen == 10 && ieq(base + vs, "keep-alive", 10)) {
                    my_tag_conn = ls; my_conn = 1;  // HTTP/1.0 opt-in
                }
            } else if (nlen == 12 && ieq(nm, "content-type", 12)) {
                f_json = vlen >= 16 && ieq(base + vs,
                                           "application/json", 16);
            } else if (nlen == 13 && ieq(nm, "authorization", 13)) {
                my_tag_auth = ls; my_auth_off = vs; my_auth_len = vlen;
            } else if (nlen == 13 && ieq(nm, "if-none-match", 13)) {
                my_tag_inm = ls; my_inm_off = vs; my_inm_len = vlen;
            } else if (nlen == 17 && ieq(nm, "transfer-encoding", 17)) {
                f_host = true;  // chunked -> host slow path
            } else if (nlen == 15 && ieq(nm, "accept-encoding", 15)) {
                for (int i = 0; i + 4 <= vlen; ++i) {
                    if (base[vs+i]=='g' && base[vs+i+1]=='z' &&
                        base[vs+i+2]=='i' && base[vs+i+3]=='p') {
                        f_gzip = true;
                        break;
                    }
                }
            }
        }
        if (__ballot(f_json)) flags |= FL_JSON_CT;
        if (__ballot(f_host)) flags |= FL_NEEDS_HOST;
        if (__ballot(f_gzip)) flags |= FL_ACCEPT_GZIP;
        // max-by-tag reductions (wave-uniform results on every lane)
        auto latest2 = [](int tag, int a, int b, int* ra, int* rb) {
            for (int off = 32; off; off >>= 1) {
                const int t2 = __shfl_xor(tag, off);
                const int a2 = __shfl_xor(a, off);
                const int b2 = __shfl_xor(b, off);
                if (t2 > tag) { tag = t2; a = a2; b = b2; }
            }
            *ra = a; *rb = b;
            return tag;
        };
        int dummy;
        if (latest2(my_tag_cl, my_clen, 0, &my_clen, &dummy) >= 0)
            clen = my_clen;
        if (latest2(my_tag_conn, my_conn, 0, &my_conn, &dummy) >= 0)
            keep_alive = my_conn != 0;
        if (latest2(my_tag_auth, my_auth_off, my_auth_len,
                    &my_auth_off, &my_auth_len) >= 0) {
            auth_off = my_auth_off; auth_len = my_auth_len;
        }
        if (latest2(my_tag_inm, my_inm_off, my_inm_len,
                    &my_inm_off, &my_inm_len) >= 0) {
            inm_off = my_inm_off; inm_len = my_inm_len;
        }
    }
    int body_len = len - body_off;
    if (body_len < 0) body_len = 0;
    if (clen > 0 && clen < body_len) body_len = clen;
    if (keep_alive) flags |= FL_KEEP_ALIVE;

    if (lane == 0) {
        F[FI_METHOD] = method;
        F[FI_PATH_OFF] = path_off;
        F[FI_PATH_LEN] = path_len;
        F[FI_QUERY_OFF] = query_off;
        F[FI_QUERY_LEN] = query_len;
        F[FI_BODY_OFF] = body_off;
        F[FI_BODY_LEN] = body_len;
        F[FI_CLEN] = clen;
        F[FI_AUTH_OFF] = auth_off;
        F[FI_AUTH_LEN] = auth_len;
        F[FI_INM_OFF] = inm_off;
        F[FI_INM_LEN] = inm_len;
    }

    // ---- route match: trie walk (gofr_amd/http/router.py compile layout) --
    int node = 0;
    int best_prefix = trie.node_prefix[0];
    int nparams = 0;
    int poffs[MAX_PARAMS], plens[MAX_PARAMS];
    {
        int pos = path_off;
        const int pend = path_off + path_len;
        // StrictSlash(false) parity (reference http/router.go:17):
        // a trailing slash only reaches prefix routes
        bool dead = (path_len > 1 && base[pend - 1] == '/');
        while (pos < pend && !dead) {
            while (pos < pend && base[pos] == '/') ++pos;
            if (pos >= pend) break;
            int se = pos;
            while (se < pend && base[se] != '/') ++se;
            const int slen = se - pos;
            // literal children
            int nxt = -1;
            const int cf = trie.node_child_first[node];
            const int cc = trie.node_child_count[node];
            for (int ci = cf; ci < cf + cc; ++ci) {
                if (trie.child_seg_len[ci] != slen) continue;
                const uint8_t* sb = trie.seg_blob + trie.child_seg_off[ci];
                bool eq = true;
                for (int k = 0; k < slen; ++k)
                    if (sb[k] != base[pos + k]) { eq = false; break; }
                if (eq) { nxt = trie.child_node[ci]; break; }
            }
            if (nxt < 0) {
                const int pc = trie.node_param[node];
                if (pc >= 0) {
                    if (nparams < MAX_PARAMS) {
                        poffs[nparams] = pos;
                        plens[nparams] = slen;
                        ++nparams;
                    } else {
                        flags |= FL_NEEDS_HOST;  // >4 params: host path
                    }
                    nxt = pc;
                }
            }
            if (nxt < 0) { dead = true; break; }
            node = nxt;
            if (trie.node_prefix[node] >= 0)
                best_prefix = trie.node_prefix[node];
            pos = se;
        }
        int route = -1;
        if (!dead) {
            const int m = (method >= 0) ? method : 0;
            route = trie.node_route[node * N_METHODS_PAD + m];
        }
        if (route < 0) route = best_prefix;
        int kind = HK_STATIC;  // catch-all default resolved by host table
        int status = 404;
        if (route >= 0 && route < n_routes) {
            kind = handler_tab[route * 4 + 0];
            status = handler_tab[route * 4 + 3];
        }
        if (flags & FL_IS_OPTIONS) { kind = HK_STATIC; status = 200; }
        // echo bodies near the LDS working set would overflow the
        // envelope assembly: host path (large responses then take
        // k_respond's direct-to-global body route, env-free)
        if (kind == HK_ECHO_JSON && body_len > MAX_SLOT - 600)
            kind = HK_HOST;
        if (flags & (FL_ERR_PARSE | FL_NEEDS_HOST)) kind = HK_HOST;
        if (lane == 0) {
            F[FI_ROUTE] = route;
            for (int k = 0; k < MAX_PARAMS; ++k) {
                F[FI_PARAM0 + 2 * k] = (k < nparams) ? poffs[k] : 0;
                F[FI_PARAM0 + 2 * k + 1] = (k < nparams) ? plens[k] : 0;
            }
            F[FI_KIND] = kind;
            F[FI_STATUS] = status;
            F[FI_FLAGS] = flags;
            if (kind == HK_HOST) atomicAdd(host_needed, 1);
        }
    }
}

extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_parse_route(uint8_t* __restrict__ reqs,
              const int64_t* __restrict__ req_off,
              const int32_t* __restrict__ req_len,
              int32_t* __restrict__ fields,
              int n,
              TrieDev trie,
              const int32_t* __restrict__ handler_tab, int n_routes,
              int32_t* __restrict__ host_needed) {
    __shared__ uint64_t masks_all[WAVES_PER_BLOCK][N_CLS][MAX_CHUNKS];
    __shared__ int lf_pos_all[WAVES_PER_BLOCK][MAX_LFS];
    const int wv = threadIdx.x / WAVE;
    const int req = blockIdx.x * WAVES_PER_BLOCK + wv;
    if (req >= n) return;
    parse_one(reqs, req_off, req_len, fields, n, trie, handler_tab,
              n_routes, host_needed, masks_all[wv], lf_pos_all[wv],
              req, lane_id());
}

// ---------------------------------------------------------------------------
// k_respond — fused handler + serializer
// ---------------------------------------------------------------------------

__constant__ char HDR_P1[] = "HTTP/1.1 ";
// after status: " OK\r\n" etc from reason table (blob)
__constant__ char HDR_CT_JSON[] = "Content-Type: application/json\r\n";
__constant__ char HDR_CT_ICON[] = "Content-Type: image/x-icon\r\n";
__constant__ char HDR_CT_BIN[]  = "Content-Type: application/octet-stream\r\n";
__constant__ char HDR_CT_TXT[]  = "Content-Type: text/plain\r\n";
__constant__ char HDR_CORS[] =
    "Access-Control-Allow-Origin: *\r\n"
    "Access-Control-Allow-Methods: POST, GET, OPTIONS, PUT, DELETE\r\n";
__constant__ char HDR_CORR[] = "X-Correlation-ID: ";
__constant__ char HDR_CL[] = "Content-Length: ";
__constant__ char HDR_CONN_KA[] = "Connection: keep-alive\r\n\r\n";
__constant__ char HDR_CONN_CL[] = "Connection: close\r\n\r\n";
__constant__ char ENV_OPEN[] = "{\"data\":";
__constant__ char HEXD[] = "0123456789abcdef";

// reasons — indexed by status class (host passes exact bytes in blob for
// anything unusual); we inline the four the engine emits
__device__ __forceinline__ const char* reason_of(int st, int* rlen) {
    switch (st) {
        case 200: *rlen = 2; return "OK";
        case 404: *rlen = 9; return "Not Found";
        case 405: *rlen = 18; return "Method Not Allowed";
        case 401: *rlen = 12; return "Unauthorized";
        default:  *rlen = 21; return "Internal Server Error";
    }
}

__device__ __forceinline__ int itoa10(uint8_t* dst, int v) {
    char tmp[10];
    int n = 0;
    if (v == 0) { dst[0] = '0'; return 1; }
    while (v > 0) { tmp[n++] = '0' + (v % 10); v /= 10; }
    for (int i = 0; i < n; ++i) dst[i] = tmp[n - 1 - i];
    return n;
}

// ---------------------------------------------------------------------------
// Fused gzip (DEFLATE, static Huffman) — BASELINE config 4's gzip
// middleware, fused into the response serializer (HBM guideline: compress
// in the producing kernel instead of re-reading the body).
//
// One response per wave. The LZ77 greedy parse runs as a whole-wave loop:
// lane 0 owns the position/hash bookkeeping and bit emission; match
// LENGTH determination is wave-parallel (lane l compares input[cand+l]
// vs input[pos+l], ballot -> ctz). Hash table and bit output live in LDS.
// CRC32 uses a block-shared LDS table built cooperatively at launch.
//
// Byte-exact mirror: gofr_amd/ops/__init__.py gzip_static_mirror().
// ---------------------------------------------------------------------------

#define GZ_HASH_BITS 9
#define GZ_HASH_SIZE (1 << GZ_HASH_BITS)
#define GZ_MIN_MATCH 3
#define GZ_MAX_DIST 2048  /* window = slot size; responses are < 4 KB */

// ---- wave-parallel CRC32 (combine method) ----------------------------------
// Each lane CRCs a contiguous chunk (init 0), advances its remainder by the
// suffix length via GF(2) multiply mod the reflected poly, and the wave
// XOR-reduces. Identical to zlib's crc32_combine math; constants are
// x^(8*2^j) mod P (reflected domain, x^0 = 0x80000000).
__constant__ uint32_t XPOW8[13] = {
    0x00800000u, 0x00008000u, 0xedb88320u, 0xb1e6b092u, 0xa06a2517u,
    0xed627daeu, 0x88d14467u, 0xd7bbfe6au, 0xec447f11u, 0x8e7ea170u,
    0x6427800eu, 0x4d47bae0u, 0x09fe548fu};

__device__ __forceinline__ uint32_t gf2_multmodp(uint32_t a, uint32_t b) {
    uint32_t m = 1u << 31, p = 0;
    while (true) {
        if (a & m) {
            p ^= b;
            if ((a & (m - 1)) == 0) break;
        }
        m >>= 1;
        b = (b & 1) ? (b >> 1) ^ 0xEDB88320u : (b >> 1);
    }
    return p;
}

__device__ __forceinline__ uint32_t xnmodp8(int nbytes) {
    uint32_t r = 0x80000000u;  // x^0
    int j = 0;
    while (nbytes) {
        if (nbytes & 1) r = gf2_multmodp(XPOW8[j], r);
        nbytes >>= 1;
        ++j;
    }
    return r;
}

// standard CRC32 (init 0xFFFFFFFF, final xor) of src[0..len), whole wave
__device__ uint32_t crc32_wave(const uint8_t* src, int len,
                               const uint32_t* crc_tab, int lane) {
    const int C = (len + WAVE - 1) / WAVE;
    const int s = lane * C;
    int e = s + C;
    if (e > len) e = len;
    uint32_t contrib = 0;
    if (s < len) {
        uint32_t r = 0;
        for (int i = s; i < e; ++i)
            r = (r >> 8) ^ crc_tab[(r ^ src[i]) & 0xFF];
        contrib = gf2_multmodp(xnmodp8(len - e), r);
    }
    if (lane == 0) contrib ^= gf2_multmodp(xnmodp8(len), 0xFFFFFFFFu);
    for (int off = 32; off; off >>= 1)
        contrib ^= __shfl_xor(contrib, off);
    return contrib ^ 0xFFFFFFFFu;
}

struct BitWriter {
    uint8_t* out;
    int bytepos;
    uint64_t bitbuf;
    int nbits;
};

__device__ __forceinline__ void bw_init(BitWriter* w, uint8_t* out) {
    w->out = out;
    w->bytepos = 0;
    w->bitbuf = 0;
    w->nbits = 0;
}

// append `len` bits of `bits` LSB-first (DEFLATE bit order)
__device__ __forceinline__ void bw_put(BitWriter* w, uint32_t bits, int len) {
    w->bitbuf |= (uint64_t)bits << w->nbits;
    w->nbits += len;
    while (w->nbits >= 8) {
        w->out[w->bytepos++] = (uint8_t)(w->bitbuf & 0xFF);
        w->bitbuf >>= 8;
        w->nbits -= 8;
    }
}

__device__ __forceinline__ void bw_flush(BitWriter* w) {
    if (w->nbits > 0) {
        w->out[w->bytepos++] = (uint8_t)(w->bitbuf & 0xFF);
        w->bitbuf = 0;
        w->nbits = 0;
    }
}

// reverse the low `len` bits (huffman codes are emitted MSB-first)
__device__ __forceinline__ uint32_t bitrev(uint32_t v, int len) {
    uint32_t r = 0;
    for (int i = 0; i < len; ++i) {
        r = (r << 1) | (v & 1);
        v >>= 1;
    }
    return r;
}

// static-huffman literal/length-symbol code (already reversed for bw_put)
__device__ __forceinline__ void fixed_lit_code(int sym, uint32_t* code,
                                               int* len) {
    if (sym < 144)      { *len = 8; *code = bitrev(0x30 + sym, 8); }
    else if (sym < 256) { *len = 9; *code = bitrev(0x190 + sym - 144, 9); }
    else if (sym < 280) { *len = 7; *code = bitrev(sym - 256, 7); }
    else                { *len = 8; *code = bitrev(0xC0 + sym - 280, 8); }
}

// length -> (code 257..285, extra bits, extra value)
__device__ __forceinline__ void length_code(int mlen, int* sym, int* ebits,
                                            int* eval) {
    static const int base[29] = {3,4,5,6,7,8,9,10,11,13,15,17,19,23,27,31,
                                 35,43,51,59,67,83,99,115,131,163,195,227,258};
    static const int extra[29] = {0,0,0,0,0,0,0,0,1,1,1,1,2,2,2,2,
                                  3,3,3,3,4,4,4,4,5,5,5,5,0};
    for (int i = 28; i >= 0; --i) {
        if (mlen >= base[i]) {
            *sym = 257 + i;
            *ebits = extra[i];
            *eval = mlen - base[i];
            return;
        }
    }
    *sym = 257; *ebits = 0; *eval = 0;
}

// distance -> (code 0..29, extra bits, extra value)
__device__ __forceinline__ void dist_code(int dist, int* sym, int* ebits,
                                          int* eval) {
    static const int base[30] = {1,2,3,4,5,7,9,13,17,25,33,49,65,97,129,193,
                                 257,385,513,769,1025,1537,2049,3073,4097,
                                 6145,8193,12289,16385,24577};
    static const int extra[30] = {0,0,0,0,1,1,2,2,3,3,4,4,5,5,6,6,
                                  7,7,8,8,9,9,10,10,11,11,12,12,13,13};
    for (int i = 29; i >= 0; --i) {
        if (dist >= base[i]) {
            *sym = i;
            *ebits = extra[i];
            *eval = dist - base[i];
            return;
        }
    }
    *sym = 0; *ebits = 0; *eval = 0;
}

// whole-wave gzip of src[0..len) in LDS -> dst (LDS); returns total gzip
// bytes (negative on overflow of cap). hash: per-wave LDS u32 table
// (u32 so interior-of-match inserts can be one wave-parallel atomicMax —
// the serial model inserts ascending positions, so last-write == max).
// crc_tab: block-shared 256-entry LDS CRC32 table.
__device__ int deflate_gzip_wave(const uint8_t* src, int len, uint8_t* dst,
                                 int cap, uint32_t* hash,
                                 const uint32_t* crc_tab, int lane) {
    // CRC of the whole plain body up front, wave-parallel (the LZ77 loop
    // below no longer touches it)
    const uint32_t crc_final = crc32_wave(src, len, crc_tab, lane);
    // cooperative hash clear
    for (int i = lane; i < GZ_HASH_SIZE; i += WAVE) hash[i] = 0;
    // gzip header (lane 0)
    BitWriter w;
    int total = -1;
    if (lane == 0) {
        uint8_t* p = dst;
        p[0]=0x1f; p[1]=0x8b; p[2]=8; p[3]=0;
        p[4]=p[5]=p[6]=p[7]=0; p[8]=0; p[9]=255;
        bw_init(&w, dst + 10);
        bw_put(&w, 1, 1);   // BFINAL
        bw_put(&w, 1, 2);   // BTYPE=01 static
    }
    int pos = 0;
    // whole-wave greedy LZ77 loop: lane 0 decides; the wave measures
    while (true) {
        const int p0 = __shfl(pos, 0);
        if (p0 >= len) break;
        // candidate from hash (lane 0 computes, broadcasts)
        int cand = -1;
        if (lane == 0 && p0 + GZ_MIN_MATCH <= len) {
            const uint32_t h = ((src[p0] | (src[p0+1] << 8) |
                                (src[p0+2] << 16)) * 0x9E3779B1u)
                               >> (32 - GZ_HASH_BITS);
            const int stored = (int)hash[h] - 1;
            if (stored >= 0 && p0 - stored <= GZ_MAX_DIST && stored < p0)
                cand = stored;
            hash[h] = (uint32_t)(p0 + 1);
        }
        cand = __shfl(cand, 0);
        int mlen = 0;
        if (cand >= 0) {
            // wave-parallel match length, up to 258 in 64-byte strides
            int agree = 0;
            for (int base = 0; base < 258 && agree == base; base += WAVE) {
                const int i = base + lane;
                const bool eq = (p0 + i < len) && (i < 258) &&
                                (src[cand + i] == src[p0 + i]);
                const uint64_t m = __ballot(eq);
                const int run = (m == ~0ull) ? WAVE
                                             : __builtin_ctzll(~m);
                agree += run;
                if (run < WAVE) break;
            }
            mlen = agree > 258 ? 258 : agree;
        }
        const int mlen0 = __shfl(mlen, 0);
        if (mlen0 >= GZ_MIN_MATCH) {
            // wave-parallel hash inserts over the matched interior. The
            // serial model writes ascending positions (last write wins),
            // so atomicMax of q+1 yields the identical final table.
            for (int i = 1 + lane; i < mlen0; i += WAVE) {
                const int q = p0 + i;
                if (q + GZ_MIN_MATCH <= len) {
                    const uint32_t h2 = ((src[q] | (src[q+1] << 8) |
                                          (src[q+2] << 16))
                                         * 0x9E3779B1u)
                                        >> (32 - GZ_HASH_BITS);
                    atomicMax(&hash[h2], (uint32_t)(q + 1));
                }
            }
        }
        if (lane == 0) {
            if (mlen >= GZ_MIN_MATCH) {
                int sym, ebits, eval;
                length_code(mlen, &sym, &ebits, &eval);
                uint32_t code; int clen;
                fixed_lit_code(sym, &code, &clen);
                bw_put(&w, code, clen);
                if (ebits) bw_put(&w, (uint32_t)eval, ebits);
                dist_code(p0 - cand, &sym, &ebits, &eval);
                bw_put(&w, bitrev((uint32_t)sym, 5), 5);
                if (ebits) bw_put(&w, (uint32_t)eval, ebits);
                pos = p0 + mlen;
            } else {
                const uint8_t b = src[p0];
                uint32_t code; int clen;
                fixed_lit_code(b, &code, &clen);
                bw_put(&w, code, clen);
                pos = p0 + 1;
            }
            if (w.bytepos > cap - 24) pos = len + 1;  // overflow guard
        }
        // the next iteration's lane-0 lookup must see this round's atomics
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        pos = __shfl(pos, 0);
        if (pos > len) return -1;
    }
    if (lane == 0) {
        uint32_t code; int clen;
        fixed_lit_code(256, &code, &clen);  // end of block
        bw_put(&w, code, clen);
        bw_flush(&w);
        uint8_t* p = dst + 10 + w.bytepos;
        const uint32_t c = crc_final;
        p[0]=(uint8_t)c; p[1]=(uint8_t)(c>>8);
        p[2]=(uint8_t)(c>>16); p[3]=(uint8_t)(c>>24);
        p[4]=(uint8_t)len; p[5]=(uint8_t)(len>>8);
        p[6]=(uint8_t)(len>>16); p[7]=(uint8_t)(len>>24);
        total = 10 + w.bytepos + 8;
    }
    return __shfl(total, 0);
}

__constant__ char HDR_CE[] = "Content-Encoding: gzip\r\n";
__constant__ char HDR_ETAG[] = "ETag: \"";
#define ETAG_HDR_LEN 18  /* ETag: "xxxxxxxx"\r\n */

// ---------------------------------------------------------------------------
// MFMA batched body hash (the ETag middleware). The response body is
// treated as a [16 x 64]-byte tile stream and each tile is one
// v_mfma_i32_16x16x64_i8 against a fixed i8 coefficient matrix — the
// matrix cores do the batched byte-dot work (BASELINE north star:
// MFMA for the batched byte-compare/hash; fragment layout validated by
// k_mfma_i8_probe + tests). All arithmetic is uint32 wraparound; the
// byte-exact model is gofr_amd/ops etag_u32().
// ---------------------------------------------------------------------------
__device__ __forceinline__ int8_t etag_bcoef(int k, int c) {
    return (int8_t)(((k * 17 + c * 29 + 3) % 251) - 125);
}

__device__ uint32_t mfma_etag_wave(const uint8_t* body, int len, int lane) {
    const int row = lane & 15;
    const int k0 = (lane >> 4) * 16;
    int8_t bb[16];
    for (int j = 0; j < 16; ++j) bb[j] = etag_bcoef(k0 + j, row);
    v4i_t bv;
    __builtin_memcpy(&bv, bb, 16);
    uint32_t state[4] = {0, 0, 0, 0};
    const int ntiles = (len + 1023) / 1024;
    for (int t = 0; t < ntiles; ++t) {
        int8_t ab[16];
        const int base = t * 1024 + row * 64 + k0;
        for (int j = 0; j < 16; ++j) {
            const int idx = base + j;
            ab[j] = (idx < len) ? (int8_t)body[idx] : 0;
        }
        v4i_t av;
        __builtin_memcpy(&av, ab, 16);
        v4i_t z = {0, 0, 0, 0};
        v4i_t dv = __builtin_amdgcn_mfma_i32_16x16x64_i8(av, bv, z,
                                                         0, 0, 0);
        for (int r2 = 0; r2 < 4; ++r2)
            state[r2] = state[r2] * 33u + (uint32_t)dv[r2];
    }
    const int col = lane & 15;
    const int rowd = (lane >> 4) * 4;
    uint32_t h = 0;
    for (int r2 = 0; r2 < 4; ++r2)
        h ^= state[r2] * (uint32_t)(((rowd + r2) * 16 + col) * 2 + 1);
    for (int off = 32; off; off >>= 1)
        h ^= (uint32_t)__shfl_xor((int)h, off);
    return h ^ (uint32_t)len;
}

// ---------------------------------------------------------------------------
// Template / KV / JSON-bind helpers (HK_TEMPLATE, HK_KV — VERDICT r1
// items 3-5: GPU handler kinds beyond echo/static, query-param
// extraction, batched JSON field binding). CPU golden mirrors:
// gofr_amd/ops render_template / kv_lookup_mirror / json_top_fields.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint64_t fnv1a64(const uint8_t* p, int n) {
    uint64_t h = 0xcbf29ce484222325ull;
    for (int i = 0; i < n; ++i) {
        h ^= p[i];
        h *= 0x100000001b3ull;
    }
    return h;
}

// locate `key=` in the raw query string; value span (raw, undecoded)
__device__ __forceinline__ bool q_find(const uint8_t* q, int qlen,
                                       const uint8_t* key, int klen,
                                       int* voff, int* vlen) {
    int i = 0;
    while (i < qlen) {
        // key span of this pair
        int ke = i;
        while (ke < qlen && q[ke] != '=' && q[ke] != '&') ++ke;
        if (ke - i == klen) {
            bool eq = true;
            for (int k = 0; k < klen; ++k)
                if (q[i + k] != key[k]) { eq = false; break; }
            if (eq) {
                if (ke < qlen && q[ke] == '=') {
                    int ve = ke + 1;
                    while (ve < qlen && q[ve] != '&') ++ve;
                    *voff = ke + 1;
                    *vlen = ve - (ke + 1);
                } else {
                    *voff = ke;
                    *vlen = 0;  // bare key: empty value
                }
                return true;
            }
        }
        while (ke < qlen && q[ke] != '&') ++ke;
        i = ke + 1;
    }
    return false;
}

// splice src[0..len) through the mode transforms into dst (or just
// measure when dst == nullptr). TM_PCT: %XX + '+'->' ' decode (invalid
// escapes copied verbatim — Python-unquote leniency, host parity).
// TM_JESC: JSON string escaping of the (decoded) bytes.
__device__ int splice_bytes(uint8_t* dst, const uint8_t* src, int len,
                            int mode) {
    int w = 0;
    for (int i = 0; i < len; ++i) {
        uint8_t c = src[i];
        if (mode & TM_PCT) {
            if (c == '+') c = ' ';
            else if (c == '%' && i + 2 < len) {
                const int hi = hexval(src[i + 1]);
                const int lo = hexval(src[i + 2]);
                if (hi >= 0 && lo >= 0) {
                    c = (uint8_t)((hi << 4) | lo);
                    i += 2;
                }
            }
        }
        if (mode & TM_JESC) {
            if (c == '"' || c == '\\') {
                if (dst) { dst[w] = '\\'; dst[w + 1] = c; }
                w += 2;
                continue;
            }
            if (c < 0x20) {
                if (dst) {
                    dst[w] = '\\'; dst[w+1] = 'u'; dst[w+2] = '0';
                    dst[w+3] = '0';
                    dst[w+4] = HEXD[(c >> 4) & 0xF];
                    dst[w+5] = HEXD[c & 0xF];
                }
                w += 6;
                continue;
            }
        }
        if (dst) dst[w] = c;
        ++w;
    }
    return w;
}

// top-level field scan of a JSON object body: writes up to
// MAX_JSON_FIELDS rows of [key_off, key_len, val_off, val_len] (offsets
// relative to `body`; key span excludes quotes; value span is the raw
// JSON value with surrounding whitespace trimmed). Returns the field
// count, or 0 when the body is not an object / malformed (callers
// splice empty). WAVE function: every lane runs the identical serial
// state machine in lockstep; the long string/container interiors are
// skipped 64 bytes per step via ballots (the r1-shape serial byte walk
// made the ctx.Bind()-class handler 4x slower than echo). `tab` is
// LDS; writes are lane-0-guarded. Mirror: ops.json_top_fields_py.
__device__ int json_top_fields(const uint8_t* body, int blen,
                               int32_t* tab, int lane) {
    // whole-chunk skip: returns true when body[i0..i0+63] (clamped)
    // contains none of the watched bytes, advancing i by the chunk
    auto chunk_clear = [&](int i0, int end, uint64_t* mask,
                           bool in_str_scan) -> uint64_t {
        const int i = i0 + lane;
        const uint8_t b = (i < end) ? body[i] : 0;
        bool hit;
        if (in_str_scan) {
            hit = (b == '"' || b == '\\') && i < end;
        } else {
            hit = (b == '"' || b == '\\' || b == '{' || b == '}' ||
                   b == '[' || b == ']') && i < end;
        }
        *mask = __ballot(hit);
        return *mask;
    };
    int i = 0;
    while (i < blen && body[i] <= ' ') ++i;
    if (i >= blen || body[i] != '{') return 0;
    ++i;
    int nf = 0;
    while (nf < MAX_JSON_FIELDS) {
        while (i < blen && (body[i] <= ' ' || body[i] == ',')) ++i;
        if (i >= blen) return 0;
        if (body[i] == '}') return nf;
        if (body[i] != '"') return 0;
        const int ks = ++i;
        while (i < blen && body[i] != '"') {
            uint64_t m;
            if (!chunk_clear(i, blen, &m, true)) {
                i += WAVE;
                if (i > blen) i = blen;
                continue;
            }
            i += __builtin_ctzll(m);  // first watched byte of the chunk
            if (body[i] == '"') break;  // loop-exit char: don't consume
            if (body[i] == '\\') ++i;
            ++i;
        }
        if (i >= blen) return 0;
        const int ke = i++;
        while (i < blen && body[i] <= ' ') ++i;
        if (i >= blen || body[i] != ':') return 0;
        ++i;
        while (i < blen && body[i] <= ' ') ++i;
        if (i >= blen) return 0;
        const int vs = i;
        // value: string / container / scalar
        if (body[i] == '"') {
            ++i;
            while (i < blen && body[i] != '"') {
                uint64_t m;
                if (!chunk_clear(i, blen, &m, true)) {
                    i += WAVE;
                    if (i > blen) i = blen;
                    continue;
                }
                i += __builtin_ctzll(m);
                if (body[i] == '"') break;  // closing quote
                if (body[i] == '\\') ++i;
                ++i;
            }
            if (i >= blen) return 0;
            ++i;
        } else if (body[i] == '{' || body[i] == '[') {
            int depth = 0;
            bool in_str = false;
            while (i < blen) {
                uint64_t m;
                if (!chunk_clear(i, blen, &m, in_str)) {
                    i += WAVE;
                    if (i > blen) i = blen;
                    continue;
                }
                i += __builtin_ctzll(m);
                const uint8_t c = body[i];
                if (in_str) {
                    if (c == '\\') ++i;
                    else if (c == '"') in_str = false;
                } else if (c == '"') in_str = true;
                else if (c == '{' || c == '[') ++depth;
                else if (c == '}' || c == ']') {
                    if (--depth == 0) { ++i; break; }
                }
                ++i;
            }
            if (depth != 0) return 0;
        } else {
            while (i < blen && body[i] != ',' && body[i] != '}' &&
                   body[i] > ' ')
                ++i;
        }
        if (lane == 0) {
            tab[nf * 4 + 0] = ks;
            tab[nf * 4 + 1] = ke - ks;
            tab[nf * 4 + 2] = vs;
            tab[nf * 4 + 3] = i - vs;
        }
        ++nf;
        while (i < blen && body[i] <= ' ') ++i;
        if (i < blen && body[i] == ',') continue;
        if (i < blen && body[i] == '}') return nf;
        return 0;
    }
    return nf;
}

// render one template piece's source span + effective mode. Returns
// false when the piece references a missing query/JSON field (splice
// empty). jtab/jn: the request's json_top_fields table.
__device__ __forceinline__ bool piece_src(
        const int32_t* pc, const uint8_t* blob, const uint8_t* rbase,
        const int32_t* F, const int32_t* jtab, int jn,
        const uint8_t** src, int* slen, int* mode) {
    const int op = pc[0], a = pc[1], b = pc[2];
    *mode = pc[3];
    if (op == TP_LIT) {
        *src = blob + a;
        *slen = b;
        return true;
    }
    if (op == TP_PATH) {
        if (a >= MAX_PARAMS) return false;
        *src = rbase + F[FI_PARAM0 + 2 * a];
        *slen = F[FI_PARAM0 + 2 * a + 1];
        *mode &= ~TM_PCT;  // parse already decoded the path in place
        return true;
    }
    if (op == TP_QUERY) {
        int voff, vlen;
        if (!q_find(rbase + F[FI_QUERY_OFF], F[FI_QUERY_LEN],
                    blob + a, b, &voff, &vlen))
            return false;
        *src = rbase + F[FI_QUERY_OFF] + voff;
        *slen = vlen;
        return true;
    }
    // TP_JFIELD
    const uint8_t* body = rbase + F[FI_BODY_OFF];
    for (int f = 0; f < jn; ++f) {
        if (jtab[f * 4 + 1] != b) continue;
        bool eq = true;
        for (int k = 0; k < b; ++k)
            if (body[jtab[f * 4] + k] != blob[a + k]) { eq = false; break; }
        if (!eq) continue;
        int vs = jtab[f * 4 + 2], vl = jtab[f * 4 + 3];
        if ((*mode & TM_JSTR) && vl >= 2 && body[vs] == '"') {
            vs += 1;
            vl -= 2;  // string content without quotes (escapes kept)
        }
        *src = body + vs;
        *slen = vl;
        *mode &= ~TM_PCT;  // body bytes are not URL-encoded
        return true;
    }
    return false;
}

// host result table row: [off, len, status, ct_id]
//
// The whole response (headers + body) is assembled in LDS per wave, then
// copied to global with one coalesced 16 B/lane sweep. Header segments are
// written cooperatively; the JSON-echo body is validated with a ballot
// structural pass. The GZ=true instantiation (k_respond_gz) additionally
// deflate-compresses the body in-LDS when the request advertised
// Accept-Encoding: gzip and the plain body >= gzip_min.
template <bool GZ>
__device__ __forceinline__ void respond_impl(
        const uint8_t* __restrict__ reqs,
        const int64_t* __restrict__ req_off,
        int32_t* __restrict__ fields,
        uint8_t* __restrict__ resp,
        int32_t* __restrict__ resp_len_out,
        int n, int rslot,
        const int32_t* __restrict__ handler_tab, int n_routes,
        const uint8_t* __restrict__ blob,
        const uint8_t* __restrict__ host_blob,
        const int32_t* __restrict__ host_tab,
        uint64_t seed, int auth_env_off, int auth_env_len,
        int gzip_min, int etag_on, const uint8_t* date29,
        const int32_t* __restrict__ kv_tab,
        const uint8_t* __restrict__ kv_blob,
        uint8_t* obuf, uint8_t* plainbuf, int32_t* jtab,
        uint32_t* hash, const uint32_t* crc_tab, int req, int lane) {
    int32_t* F = fields + (size_t)req * NF;
    uint8_t* out = resp + (size_t)req * rslot;
    const uint8_t* rbase = reqs + req_off[req];

    const int kind = F[FI_KIND];
    int status = F[FI_STATUS];
    const int flags = F[FI_FLAGS];
    if (flags & FL_EMPTY) {
        if (lane == 0) {
            F[FI_RESP_LEN] = 0;
            F[FI_RESP_OFF] = (int)((size_t)req * rslot);
            resp_len_out[req] = 0;
        }
        return;
    }
    const bool keep = (flags & FL_KEEP_ALIVE) != 0;
    const bool is_options = (flags & FL_IS_OPTIONS) != 0;

    // ---- resolve body source + length --------------------------------------
    const uint8_t* body_src = nullptr;
    int body_src_len = 0;
    int env = 0;  // 1 -> wrap in {"data": ... }
    if (is_options) {
        body_src_len = 0;
    } else if (kind == HK_ECHO_JSON) {
        body_src = rbase + F[FI_BODY_OFF];
        body_src_len = F[FI_BODY_LEN];
        env = 1;
        // structural validation: per-64B-chunk ballots; lane 0 walks only
        // the structural bit positions with escape-parity handling —
        // byte-equivalent to the serial model in gofr_amd/ops
        bool bad = (body_src_len == 0);
        int depth = 0;
        bool in_str = false;
        int bs_suffix = 0;  // backslash run carried across chunk boundary
        const int nch = (body_src_len + WAVE - 1) / WAVE;
        for (int c = 0; c < nch && !bad; ++c) {
            const int i = c * WAVE + lane;
            const uint8_t b = (i < body_src_len) ? body_src[i] : 0;
            const uint64_t m_q = __ballot(b == '"');
            const uint64_t m_bs = __ballot(b == '\\');
            const uint64_t m_op = __ballot(b == '{' || b == '[');
            const uint64_t m_cl = __ballot(b == '}' || b == ']');
            if (lane == 0) {
                uint64_t strct = m_q | m_op | m_cl;
                while (strct) {
                    const int p = __builtin_ctzll(strct);
                    strct &= strct - 1;
                    const uint64_t bit = 1ull << p;
                    if (m_q & bit) {
                        // escaped iff odd backslash run ends at p-1
                        int run;
                        if (p == 0) {
                            run = bs_suffix;
                        } else {
                            const uint64_t below = m_bs & (bit - 1);
                            run = 0;
                            uint64_t probe = 1ull << (p - 1);
                            while ((below & probe) && run < p) {
                                ++run;
                                probe >>= 1;
                            }
                            if (run == p) run += bs_suffix;
                        }
                        const bool escaped = in_str && (run & 1);
                        if (!escaped) in_str = !in_str;
                    } else if (!in_str) {
                        if (m_op & bit) ++depth;
                        else if (--depth < 0) { bad = true; break; }
                    }
                }
                if (m_bs == ~0ull) bs_suffix += 64;
                else bs_suffix = __builtin_clzll(~m_bs);
            }
            bad = __shfl(bad ? 1 : 0, 0) != 0;
        }
        if (lane == 0 && (depth != 0 || in_str)) bad = true;
        bad = __shfl(bad ? 1 : 0, 0) != 0;
        if (bad) {
            status = 500;
            const int elen = *(const int32_t*)blob;
            body_src = blob + 4;
            body_src_len = elen;
            env = 0;
        }
    } else if (kind == HK_STATIC) {
        const int route = F[FI_ROUTE];
        if (route >= 0 && route < n_routes) {
            body_src = blob + handler_tab[route * 4 + 1];
            body_src_len = handler_tab[route * 4 + 2];
        }
    } else if (kind == HK_TEMPLATE && !(flags & FL_AUTH_FAIL)) {
        // template program: blob-encoded pieces spliced with decoded
        // path params, query params and top-level JSON body fields —
        // the /user/{id}-class + ctx.Bind()-class routes the r1 engine
        // trampolined. The whole pass runs WAVE-UNIFORM (every lane
        // computes identical resolution state in lockstep): the JSON
        // field scan fast-skips string interiors via ballots, and
        // splices without transformable bytes (ballot-checked) are
        // wave-parallel copies; only the rare escape/decode splice
        // runs serially on lane 0. Result is the raw body (templates
        // carry their own envelope text), env stays 0.
        const int route = F[FI_ROUTE];
        const int32_t* prog = (const int32_t*)(blob +
                                               handler_tab[route * 4 + 1]);
        uint8_t* tbuf = plainbuf ? plainbuf : (obuf + 1024);
        const int tcap = MAX_SLOT - 1024;
        const int np0 = prog[0];
        const int np = np0 < MAX_TPL_PIECES ? np0 : MAX_TPL_PIECES;
        const uint8_t* srcs[MAX_TPL_PIECES];
        int slens[MAX_TPL_PIECES], pmodes[MAX_TPL_PIECES];
        int olens[MAX_TPL_PIECES];
        int jn = -1;  // lazy: scan only when a piece needs it
        int total = 0;
        const uint8_t* body = rbase + F[FI_BODY_OFF];
        for (int p = 0; p < np; ++p) {
            const int32_t* pc = prog + 1 + p * 4;
            if (pc[0] == TP_JFIELD && jn < 0) {
                jn = json_top_fields(body, F[FI_BODY_LEN], jtab, lane);
                asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            }
            const uint8_t* src;
            int slen, mode;
            if (!piece_src(pc, blob, rbase, F, jtab, jn < 0 ? 0 : jn,
                           &src, &slen, &mode)) {
                srcs[p] = nullptr;
                slens[p] = olens[p] = pmodes[p] = 0;
                continue;
            }
            int ol = slen;
            if (mode & (TM_PCT | TM_JESC)) {
                // ballot: does any byte actually need the transform?
                bool hot = false;
                for (int c0 = 0; c0 < slen; c0 += WAVE) {
                    const int i = c0 + lane;
                    const uint8_t b = (i < slen) ? src[i] : 'a';
                    const bool h =
                        ((mode & TM_PCT) && (b == '%' || b == '+')) ||
                        ((mode & TM_JESC) &&
                         (b == '"' || b == '\\' || b < 0x20));
                    if (__ballot(h)) { hot = true; break; }
                }
                if (!hot)
                    mode &= ~(TM_PCT | TM_JESC);  // plain wave copy
                else
                    ol = splice_bytes(nullptr, src, slen, mode);
            }
            srcs[p] = src;
            slens[p] = slen;
            pmodes[p] = mode;
            olens[p] = ol;
            total += ol;
        }
        if (total > tcap) {
            status = 500;  // overflow guard: render the 500 envelope
            const int elen = *(const int32_t*)blob;
            body_src = blob + 4;
            body_src_len = elen;
        } else {
            int w = 0;
            for (int p = 0; p < np; ++p) {
                if (!srcs[p]) continue;
                const int mode = pmodes[p];
                if (!(mode & (TM_PCT | TM_JESC))) {
                    const uint8_t* src = srcs[p];
                    for (int i = lane; i < slens[p]; i += WAVE)
                        tbuf[w + i] = src[i];
                } else if (lane == 0) {
                    splice_bytes(tbuf + w, srcs[p], slens[p], mode);
                }
                w += olens[p];
            }
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            body_src = tbuf;
            body_src_len = total;
        }
    } else if (kind == HK_KV && !(flags & FL_AUTH_FAIL)) {
        // device KV-store read keyed by path param 0 (decoded): the
        // /user/{id} redis-get analog served from HBM. Values are
        // pre-wrapped {"data":...} envelopes; a miss renders the 404
        // envelope stored at blob[8+invalid_len].
        const int route = F[FI_ROUTE];
        const int slot0 = handler_tab[route * 4 + 1];
        const int nslots = handler_tab[route * 4 + 2];
        const uint8_t* key = rbase + F[FI_PARAM0];
        const int klen = F[FI_PARAM0 + 1];
        int voff = -1, vlen = 0;
        if (nslots > 0 && klen > 0) {
            const uint64_t h = fnv1a64(key, klen);
            const int32_t h_lo = (int32_t)(h & 0xFFFFFFFF);
            const int32_t h_hi = (int32_t)(h >> 32);
            int idx = (int)(h % (uint64_t)nslots);
            for (int probe = 0; probe < nslots; ++probe) {
                const int32_t* row = kv_tab + (size_t)(slot0 + idx) * 6;
                if (row[5] < 0) break;  // empty slot: miss
                if (row[0] == h_lo && row[1] == h_hi &&
                    row[3] == klen) {
                    bool eq = true;
                    for (int k = 0; k < klen; ++k)
                        if (kv_blob[row[2] + k] != key[k]) {
                            eq = false;
                            break;
                        }
                    if (eq) { voff = row[4]; vlen = row[5]; break; }
                }
                idx = (idx + 1 == nslots) ? 0 : idx + 1;
            }
        }
        if (voff >= 0) {
            body_src = kv_blob + voff;
            body_src_len = vlen;
        } else {
            status = 404;
            const int elen = *(const int32_t*)blob;
            // blob+4+elen is NOT 4-aligned: byte-wise length read
            const uint8_t* ml = blob + 4 + elen;
            body_src = blob + 8 + elen;
            body_src_len = ml[0] | (ml[1] << 8) | (ml[2] << 16) |
                           (ml[3] << 24);
        }
    }
    int ct_id = 0;  // 0 json, 1 icon, 2 octet-stream, 3 text/plain
    if (kind == HK_HOST && !is_options && !(flags & FL_AUTH_FAIL)) {
        body_src = host_blob + host_tab[req * 4 + 0];
        body_src_len = host_tab[req * 4 + 1];
        status = host_tab[req * 4 + 2];
        ct_id = host_tab[req * 4 + 3];
        if (status == 0) {  // trampoline never ran (optimistic mode)
            status = 500;
            const int elen = *(const int32_t*)blob;
            body_src = blob + 4;
            body_src_len = elen;
        }
    }

    // auth middleware verdict overrides everything but OPTIONS
    if ((flags & FL_AUTH_FAIL) && !is_options) {
        status = 401;
        body_src = blob + auth_env_off;
        body_src_len = auth_env_len;
        env = 0;
        ct_id = 0;
    }

    int body_total = body_src_len + (env ? 9 : 0);  // {"data": + }

    // ---- fused gzip (GZ instantiation only) --------------------------------
    int content_enc = 0;
    if (GZ) {
        if ((flags & FL_ACCEPT_GZIP) && !is_options &&
            body_total >= gzip_min && body_total <= MAX_SLOT) {
            // assemble the plain body (envelope included) into plainbuf
            const int boff = env ? 8 : 0;
            if (env && lane < 8) plainbuf[lane] = ENV_OPEN[lane];
            for (int i = lane; i < body_src_len; i += WAVE)
                plainbuf[boff + i] = body_src[i];
            if (env && lane == 0) plainbuf[boff + body_src_len] = '}';
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            const int gz_total = deflate_gzip_wave(
                plainbuf, body_total, obuf + 512, MAX_SLOT - 512,
                hash, crc_tab, lane);
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            if (gz_total > 0) {
                content_enc = 1;
                body_total = gz_total;
                env = 0;
            }
            // gz_total < 0 (incompressible/overflow): serve uncompressed
        }
    }

    // ---- header segments (lengths first, then cooperative writes) ---------
    int rlen_reason;
    const char* reason = reason_of(status, &rlen_reason);
    const char* ct_str = HDR_CT_JSON;
    int ct_len = sizeof(HDR_CT_JSON) - 1;
    if (ct_id == 1) { ct_str = HDR_CT_ICON; ct_len = sizeof(HDR_CT_ICON) - 1; }
    else if (ct_id == 2) { ct_str = HDR_CT_BIN; ct_len = sizeof(HDR_CT_BIN) - 1; }
    else if (ct_id == 3) { ct_str = HDR_CT_TXT; ct_len = sizeof(HDR_CT_TXT) - 1; }
    int cl_digits = 1;
    for (int v = body_total; v >= 10; v /= 10) ++cl_digits;
    const char* conn = keep ? HDR_CONN_KA : HDR_CONN_CL;
    const int conn_len =
        (keep ? sizeof(HDR_CONN_KA) : sizeof(HDR_CONN_CL)) - 1;
    const int ce_len = content_enc ? (int)sizeof(HDR_CE) - 1 : 0;

    // segment offsets (all lanes compute identically)
    const int o_status = 9;                       // after "HTTP/1.1 "
    const int o_reason = o_status + 3 + 1;        // "xxx "
    const int o_date = o_reason + rlen_reason + 2;  // reason \r\n
    const int date_len = date29 ? 37 : 0;  // "Date: " + 29 + CRLF
    const int o_ct = o_date + date_len;
    const int o_ce = o_ct + ct_len;
    const int et_len = etag_on ? ETAG_HDR_LEN : 0;
    const int o_etag = o_ce + ce_len;
    const int o_cors = o_etag + et_len;
    const int o_corr = o_cors + (int)sizeof(HDR_CORS) - 1;
    const int o_corrhex = o_corr + (int)sizeof(HDR_CORR) - 1;
    const int o_cl = o_corrhex + 32 + 2;
    const int o_cld = o_cl + (int)sizeof(HDR_CL) - 1;
    const int o_conn = o_cld + cl_digits + 2;
    const int hl = o_conn + conn_len;

    // cooperative writes into LDS: each segment <=64B -> one store per lane
    if (lane < 9) obuf[lane] = HDR_P1[lane];
    if (lane < (unsigned)rlen_reason) obuf[o_reason + lane] = reason[lane];
    if (date29) {
        static const char DK[6] = {'D', 'a', 't', 'e', ':', ' '};
        if (lane < 6) obuf[o_date + lane] = DK[lane];
        if (lane < 29) obuf[o_date + 6 + lane] = date29[lane];
        if (lane == 0) {
            obuf[o_date + 35] = '\r';
            obuf[o_date + 36] = '\n';
        }
    }
    if (lane < (unsigned)ct_len) obuf[o_ct + lane] = ct_str[lane];
    if (content_enc && lane < ce_len) obuf[o_ce + lane] = HDR_CE[lane];
    if (etag_on && lane < 7) obuf[o_etag + lane] = HDR_ETAG[lane];
    if (etag_on && lane == 0) {
        obuf[o_etag + 15] = '"';
        obuf[o_etag + 16] = '\r';
        obuf[o_etag + 17] = '\n';
        // hex digits patched after the body lands in LDS
    }
    {
        const int cors_len = (int)sizeof(HDR_CORS) - 1;
        for (int i = lane; i < cors_len; i += WAVE)
            obuf[o_cors + i] = HDR_CORS[i];
    }
    if (lane < (int)sizeof(HDR_CORR) - 1) obuf[o_corr + lane] = HDR_CORR[lane];
    if (lane < (int)sizeof(HDR_CL) - 1) obuf[o_cl + lane] = HDR_CL[lane];
    if (lane < (unsigned)conn_len) obuf[o_conn + lane] = conn[lane];
    // correlation id: every lane computes the 128-bit id and emits its char
    {
        const uint64_t h1 = splitmix64(seed ^ (uint64_t)req);
        const uint64_t h2 = splitmix64(h1 ^ 0xD1B54A32D192ED03ull);
        if (lane < 32) {
            const uint64_t h = (lane < 16) ? h1 : h2;
            const int sh = 60 - 4 * (lane & 15);
            obuf[o_corrhex + lane] = HEXD[(h >> sh) & 0xF];
        }
    }
    if (lane == 0) {
        obuf[o_status] = '0' + (status / 100);
        obuf[o_status + 1] = '0' + ((status / 10) % 10);
        obuf[o_status + 2] = '0' + (status % 10);
        obuf[o_status + 3] = ' ';
        obuf[o_reason + rlen_reason] = '\r';
        obuf[o_reason + rlen_reason + 1] = '\n';
        obuf[o_corrhex + 32] = '\r';
        obuf[o_corrhex + 33] = '\n';
        itoa10(obuf + o_cld, body_total);
        obuf[o_cld + cl_digits] = '\r';
        obuf[o_cld + cl_digits + 1] = '\n';
        if (env) {
            for (int i = 0; i < 8; ++i) obuf[hl + i] = ENV_OPEN[i];
            obuf[hl + 8 + body_src_len] = '}';
        }
    }

    // responses larger than the LDS working set keep the body in
    // global memory (headers still assemble in LDS); guaranteed
    // env-free: envelope (echo) responses are LDS-sized by the parse
    // kernel's HK_HOST rerouting of large echo bodies
    const bool large = hl + body_total > MAX_SLOT;

    // ---- body into LDS -----------------------------------------------------
    if (large) {
        // nothing staged: the final copy streams body_src directly
    } else if (GZ && content_enc) {
        // shift the deflate output from obuf+512 down to obuf+hl.
        // dst < src and the wave is lockstep (loads of an iteration all
        // execute before its stores), so a forward sweep is safe.
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        for (int i = lane; i < body_total; i += WAVE) {
            const uint8_t v = obuf[512 + i];
            obuf[hl + i] = v;
        }
    } else {
        const int body_start = hl + (env ? 8 : 0);
        for (int i = lane; i < body_src_len; i += WAVE)
            obuf[body_start + i] = body_src[i];
    }

    int send_body = body_total;
    if (etag_on) {
        // MFMA hash of the final body bytes -> patch the reserved hex
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        const uint32_t h = large
            ? mfma_etag_wave(body_src, body_total, lane)
            : mfma_etag_wave(obuf + hl, body_total, lane);
        if (lane < 8)
            obuf[o_etag + 7 + lane] = HEXD[(h >> (28 - 4 * lane)) & 0xF];
        // If-None-Match -> 304: in-place rewrite (status digits + CL
        // digits zeroed, body dropped) so no header offset moves.
        // Conscious quirks (documented): the reason phrase stays the
        // 200 one (clients must ignore it, RFC 9112 §4) and
        // Content-Length keeps its width as all-zero DIGITs.
        if (status == 200) {
            const int io = F[FI_INM_OFF], il = F[FI_INM_LEN];
            if (il == 10 && rbase[io] == '"' && rbase[io + 9] == '"') {
                bool m = true;
                for (int i = 0; i < 8; ++i) {
                    const uint8_t hx = HEXD[(h >> (28 - 4 * i)) & 0xF];
                    if (rbase[io + 1 + i] != hx) { m = false; break; }
                }
                if (m) {
                    if (lane == 0) {
                        obuf[o_status] = '3';
                        obuf[o_status + 1] = '0';
                        obuf[o_status + 2] = '4';
                        for (int i = 0; i < cl_digits; ++i)
                            obuf[o_cld + i] = '0';
                    }
                    send_body = 0;
                }
            }
        }
    }

    // HEAD: headers (incl. real Content-Length) without the body —
    // net/http discards handler writes for HEAD (RFC 9110 §9.3.2)
    if (F[FI_METHOD] == M_HEAD) send_body = 0;

    const int total = hl + send_body;
    // ---- one coalesced 16B/lane sweep LDS -> global ------------------------
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    if (total <= rslot) {
        if (large) {
            for (int i = lane; i < hl; i += WAVE) out[i] = obuf[i];
            if (send_body)
                for (int i = lane; i < body_total; i += WAVE)
                    out[hl + i] = body_src[i];
        } else {
            const int nv = (total + 15) >> 4;
            const uint4* src = (const uint4*)obuf;
            uint4* dst = (uint4*)out;
            for (int i = lane; i < nv; i += WAVE) dst[i] = src[i];
        }
    }
    if (lane == 0) {
        const int tl = (total <= rslot) ? total : 0;
        F[FI_RESP_LEN] = tl;
        F[FI_RESP_OFF] = (int)((size_t)req * rslot);
        resp_len_out[req] = tl;
    }
}

extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_respond(const uint8_t* __restrict__ reqs,
          const int64_t* __restrict__ req_off,
          int32_t* __restrict__ fields,
          uint8_t* __restrict__ resp,
          int32_t* __restrict__ resp_len_out,
          int n, int rslot,
          const int32_t* __restrict__ handler_tab, int n_routes,
          const uint8_t* __restrict__ blob,
          const uint8_t* __restrict__ host_blob,
          const int32_t* __restrict__ host_tab,
          const uint64_t* __restrict__ seed_ptr,
          int auth_env_off, int auth_env_len, int etag_on,
          const uint8_t* __restrict__ date29,
          const int32_t* __restrict__ kv_tab,
          const uint8_t* __restrict__ kv_blob) {
    // obuf per wave + jtab (JSON field table) per wave in ONE block
    __shared__ uint8_t obuf_all[WAVES_PER_BLOCK * MAX_SLOT +
                                WAVES_PER_BLOCK * MAX_JSON_FIELDS * 16];
    const int wv = threadIdx.x / WAVE;
    const int req = blockIdx.x * WAVES_PER_BLOCK + wv;
    if (req >= n) return;
    int32_t* jtab = (int32_t*)(obuf_all + WAVES_PER_BLOCK * MAX_SLOT) +
                    wv * MAX_JSON_FIELDS * 4;
    respond_impl<false>(reqs, req_off, fields, resp, resp_len_out, n, rslot,
                        handler_tab, n_routes, blob, host_blob, host_tab,
                        *seed_ptr, auth_env_off, auth_env_len, 0, etag_on,
                        date29, kv_tab, kv_blob,
                        obuf_all + wv * MAX_SLOT, nullptr, jtab, nullptr,
                        nullptr, req, lane_id());
}

extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_respond_gz(const uint8_t* __restrict__ reqs,
             const int64_t* __restrict__ req_off,
             int32_t* __restrict__ fields,
             uint8_t* __restrict__ resp,
             int32_t* __restrict__ resp_len_out,
             int n, int rslot,
             const int32_t* __restrict__ handler_tab, int n_routes,
             const uint8_t* __restrict__ blob,
             const uint8_t* __restrict__ host_blob,
             const int32_t* __restrict__ host_tab,
             const uint64_t* __restrict__ seed_ptr,
             int auth_env_off, int auth_env_len, int gzip_min,
             int etag_on, const uint8_t* __restrict__ date29,
             const int32_t* __restrict__ kv_tab,
             const uint8_t* __restrict__ kv_blob) {
    // single __shared__ block (cdna guide §5 trap 4a)
    __shared__ uint8_t lds[WAVES_PER_BLOCK * MAX_SLOT * 2 +
                           WAVES_PER_BLOCK * GZ_HASH_SIZE * 4 + 256 * 4 +
                           WAVES_PER_BLOCK * MAX_JSON_FIELDS * 16];
    uint8_t* obuf_all = lds;
    uint8_t* plain_all = lds + WAVES_PER_BLOCK * MAX_SLOT;
    uint32_t* hash_all = (uint32_t*)(lds + WAVES_PER_BLOCK * MAX_SLOT * 2);
    uint32_t* crc_tab = (uint32_t*)(lds + WAVES_PER_BLOCK * MAX_SLOT * 2 +
                                    WAVES_PER_BLOCK * GZ_HASH_SIZE * 4);
    int32_t* jtab_all = (int32_t*)(lds + WAVES_PER_BLOCK * MAX_SLOT * 2 +
                                   WAVES_PER_BLOCK * GZ_HASH_SIZE * 4 +
                                   256 * 4);
    // build the CRC32 table cooperatively BEFORE any thread can exit
    for (int i = threadIdx.x; i < 256; i += BLOCK_THREADS) {
        uint32_t c = (uint32_t)i;
        for (int k = 0; k < 8; ++k)
            c = (c & 1) ? (c >> 1) ^ 0xEDB88320u : (c >> 1);
        crc_tab[i] = c;
    }
    __syncthreads();
    const int wv = threadIdx.x / WAVE;
    const int req = blockIdx.x * WAVES_PER_BLOCK + wv;
    if (req >= n) return;
    respond_impl<true>(reqs, req_off, fields, resp, resp_len_out, n, rslot,
                       handler_tab, n_routes, blob, host_blob, host_tab,
                       *seed_ptr, auth_env_off, auth_env_len, gzip_min,
                       etag_on, date29, kv_tab, kv_blob,
                       obuf_all + wv * MAX_SLOT, plain_all + wv * MAX_SLOT,
                       jtab_all + wv * MAX_JSON_FIELDS * 4,
                       hash_all + wv * GZ_HASH_SIZE, crc_tab,
                       req, lane_id());
}

// ---------------------------------------------------------------------------
// k_auth — batched HMAC-SHA256 bearer-token check, one request per LANE.
//
// Middleware semantics (the auth middleware of BASELINE config 4): the
// Authorization header must be "HMAC <64 hex>" where the MAC is
// HMAC-SHA256(secret, METHOD + " " + path). Failure sets FL_AUTH_FAIL in
// the request flags; k_respond renders the 401 envelope. SHA-256 is
// inherently serial per message, so the parallel axis is requests (one
// per lane: a 256-thread block checks 256 requests).
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint32_t ror32(uint32_t x, int n) {
    return (x >> n) | (x << (32 - n));
}

__constant__ uint32_t SHA_K[64] = {
    0x428a2f98,0x71374491,0xb5c0fbcf,0xe9b5dba5,0x3956c25b,0x59f111f1,
    0x923f82a4,0xab1c5ed5,0xd807aa98,0x12835b01,0x243185be,0x550c7dc3,
    0x72be5d74,0x80deb1fe,0x9bdc06a7,0xc19bf174,0xe49b69c1,0xefbe4786,
    0x0fc19dc6,0x240ca1cc,0x2de92c6f,0x4a7484aa,0x5cb0a9dc,0x76f988da,
    0x983e5152,0xa831c66d,0xb00327c8,0xbf597fc7,0xc6e00bf3,0xd5a79147,
    0x06ca6351,0x14292967,0x27b70a85,0x2e1b2138,0x4d2c6dfc,0x53380d13,
    0x650a7354,0x766a0abb,0x81c2c92e,0x92722c85,0xa2bfe8a1,0xa81a664b,
    0xc24b8b70,0xc76c51a3,0xd192e819,0xd6990624,0xf40e3585,0x106aa070,
    0x19a4c116,0x1e376c08,0x2748774c,0x34b0bcb5,0x391c0cb3,0x4ed8aa4a,
    0x5b9cca4f,0x682e6ff3,0x748f82ee,0x78a5636f,0x84c87814,0x8cc70208,
    0x90befffa,0xa4506ceb,0xbef9a3f7,0xc67178f2};

struct Sha256Ctx {
    uint32_t h[8];
    uint8_t buf[64];
    int buflen;
    uint64_t total;
};

__device__ void sha256_init(Sha256Ctx* c) {
    c->h[0]=0x6a09e667; c->h[1]=0xbb67ae85; c->h[2]=0x3c6ef372;
    c->h[3]=0xa54ff53a; c->h[4]=0x510e527f; c->h[5]=0x9b05688c;
    c->h[6]=0x1f83d9ab; c->h[7]=0x5be0cd19;
    c->buflen = 0;
    c->total = 0;
}

__device__ void sha256_block(Sha256Ctx* c, const uint8_t* p) {
    uint32_t w[64];
    for (int i = 0; i < 16; ++i)
        w[i] = ((uint32_t)p[4*i] << 24) | ((uint32_t)p[4*i+1] << 16) |
               ((uint32_t)p[4*i+2] << 8) | p[4*i+3];
    for (int i = 16; i < 64; ++i) {
        uint32_t s0 = ror32(w[i-15],7) ^ ror32(w[i-15],18) ^ (w[i-15] >> 3);
        uint32_t s1 = ror32(w[i-2],17) ^ ror32(w[i-2],19) ^ (w[i-2] >> 10);
        w[i] = w[i-16] + s0 + w[i-7] + s1;
    }
    uint32_t a=c->h[0],b=c->h[1],cc=c->h[2],d=c->h[3],
             e=c->h[4],f=c->h[5],g=c->h[6],h=c->h[7];
    for (int i = 0; i < 64; ++i) {
        uint32_t S1 = ror32(e,6) ^ ror32(e,11) ^ ror32(e,25);
        uint32_t ch = (e & f) ^ (~e & g);
        uint32_t t1 = h + S1 + ch + SHA_K[i] + w[i];
        uint32_t S0 = ror32(a,2) ^ ror32(a,13) ^ ror32(a,22);
        uint32_t maj = (a & b) ^ (a & cc) ^ (b & cc);
        uint32_t t2 = S0 + maj;
        h=g; g=f; f=e; e=d+t1; d=cc; cc=b; b=a; a=t1+t2;
    }
    c->h[0]+=a; c->h[1]+=b; c->h[2]+=cc; c->h[3]+=d;
    c->h[4]+=e; c->h[5]+=f; c->h[6]+=g; c->h[7]+=h;
}

__device__ void sha256_update(Sha256Ctx* c, const uint8_t* p, int n) {
    c->total += n;
    while (n > 0) {
        int take = 64 - c->buflen;
        if (take > n) take = n;
        for (int i = 0; i < take; ++i) c->buf[c->buflen + i] = p[i];
        c->buflen += take;
        p += take; n -= take;
        if (c->buflen == 64) { sha256_block(c, c->buf); c->buflen = 0; }
    }
}

__device__ void sha256_final(Sha256Ctx* c, uint8_t out[32]) {
    uint64_t bits = c->total * 8;
    uint8_t pad = 0x80;
    sha256_update(c, &pad, 1);
    uint8_t z = 0;
    while (c->buflen != 56) sha256_update(c, &z, 1);
    uint8_t lenb[8];
    for (int i = 0; i < 8; ++i) lenb[i] = (uint8_t)(bits >> (56 - 8*i));
    sha256_update(c, lenb, 8);
    for (int i = 0; i < 8; ++i) {
        out[4*i] = (uint8_t)(c->h[i] >> 24);
        out[4*i+1] = (uint8_t)(c->h[i] >> 16);
        out[4*i+2] = (uint8_t)(c->h[i] >> 8);
        out[4*i+3] = (uint8_t)(c->h[i]);
    }
}

__device__ __forceinline__ int hexval(uint8_t c) {
    if (c >= '0' && c <= '9') return c - '0';
    if (c >= 'a' && c <= 'f') return c - 'a' + 10;
    if (c >= 'A' && c <= 'F') return c - 'A' + 10;
    return -1;
}

__device__ void auth_one(const uint8_t* reqs, const int64_t* req_off,
                         int32_t* fields, int req,
                         const uint8_t* secret, int secret_len) {
    int32_t* F = fields + (size_t)req * NF;
    const int flags = F[FI_FLAGS];
    if (flags & (FL_ERR_PARSE | FL_IS_OPTIONS | FL_EMPTY)) return;
    const uint8_t* base = reqs + req_off[req];
    const int aoff = F[FI_AUTH_OFF], alen = F[FI_AUTH_LEN];
    bool ok = false;
    // expected form: "HMAC " + 64 hex chars
    if (alen == 69 && base[aoff]=='H' && base[aoff+1]=='M' &&
        base[aoff+2]=='A' && base[aoff+3]=='C' && base[aoff+4]==' ') {
        // HMAC-SHA256(secret, METHOD + " " + path)
        uint8_t kblk[64], digest[32];
        for (int i = 0; i < 64; ++i)
            kblk[i] = (i < secret_len ? secret[i] : 0) ^ 0x36;
        Sha256Ctx ctx;
        sha256_init(&ctx);
        sha256_update(&ctx, kblk, 64);
        // message: method token (from request start to first space) + ' ' + path
        int sp = 0;
        while (base[sp] != ' ') ++sp;
        sha256_update(&ctx, base, sp + 1);
        sha256_update(&ctx, base + F[FI_PATH_OFF], F[FI_PATH_LEN]);
        sha256_final(&ctx, digest);
        for (int i = 0; i < 64; ++i)
            kblk[i] = (i < secret_len ? secret[i] : 0) ^ 0x5c;
        Sha256Ctx ctx2;
        sha256_init(&ctx2);
        sha256_update(&ctx2, kblk, 64);
        sha256_update(&ctx2, digest, 32);
        sha256_final(&ctx2, digest);
        // constant-time hex compare
        int diff = 0;
        for (int i = 0; i < 32; ++i) {
            const int hi = hexval(base[aoff + 5 + 2*i]);
            const int lo = hexval(base[aoff + 5 + 2*i + 1]);
            diff |= (hi < 0 || lo < 0) ? 1 : (((hi << 4) | lo) ^ digest[i]);
        }
        ok = diff == 0;
    }
    if (!ok) {
        F[FI_FLAGS] = flags | FL_AUTH_FAIL;
        F[FI_STATUS] = 401;
    }
}

extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_auth(const uint8_t* __restrict__ reqs,
       const int64_t* __restrict__ req_off,
       int32_t* __restrict__ fields,
       int n,
       const uint8_t* __restrict__ secret, int secret_len) {
    const int req = blockIdx.x * BLOCK_THREADS + threadIdx.x;
    if (req >= n) return;
    auth_one(reqs, req_off, fields, req, secret, secret_len);
}

// ---------------------------------------------------------------------------
// k_varint_spans — batched protobuf field-span decode (gRPC codec path).
//
// One message per lane. Output per message: up to MAX_PB_FIELDS rows of
// [field_no, wire_type, payload_off(abs in buf) or varint lo32, len or
// varint hi32]; n_fields in the count array. Length-delimited fields
// record (offset, len); varint fields record the decoded 64-bit value
// split into lo/hi. The host codec (gofr_amd/grpc/codec.py) is the
// golden model (tests/test_gpu_middleware.py).
// ---------------------------------------------------------------------------
#define MAX_PB_FIELDS 16

extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_varint_spans(const uint8_t* __restrict__ buf,
               const int64_t* __restrict__ msg_off,
               const int32_t* __restrict__ msg_len,
               int32_t* __restrict__ out,      // [n, MAX_PB_FIELDS, 4]
               int32_t* __restrict__ out_n,    // [n]
               int n) {
    const int m = blockIdx.x * BLOCK_THREADS + threadIdx.x;
    if (m >= n) return;
    const int64_t base = msg_off[m];
    const uint8_t* p = buf + base;
    const int len = msg_len[m];
    int32_t* row = out + (size_t)m * MAX_PB_FIELDS * 4;
    int nf = 0;
    int pos = 0;
    while (pos < len && nf < MAX_PB_FIELDS) {
        // read tag varint
        uint64_t tag = 0;
        int shift = 0;
        while (pos < len) {
            const uint8_t b = p[pos++];
            tag |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
            if (shift > 63) { out_n[m] = -1; return; }
        }
        const int fno = (int)(tag >> 3);
        const int wt = (int)(tag & 7);
        if (wt == 0) {
            uint64_t v = 0;
            shift = 0;
            while (pos < len) {
                const uint8_t b = p[pos++];
                v |= (uint64_t)(b & 0x7F) << shift;
                if (!(b & 0x80)) break;
                shift += 7;
                if (shift > 63) { out_n[m] = -1; return; }
            }
            row[nf*4+0] = fno; row[nf*4+1] = 0;
            row[nf*4+2] = (int32_t)(v & 0xFFFFFFFF);
            row[nf*4+3] = (int32_t)(v >> 32);
        } else if (wt == 2) {
            uint64_t ln = 0;
            shift = 0;
            while (pos < len) {
                const uint8_t b = p[pos++];
                ln |= (uint64_t)(b & 0x7F) << shift;
                if (!(b & 0x80)) break;
                shift += 7;
            }
            if (pos + (int)ln > len) { out_n[m] = -1; return; }
            row[nf*4+0] = fno; row[nf*4+1] = 2;
            row[nf*4+2] = (int32_t)(base + pos);
            row[nf*4+3] = (int32_t)ln;
            pos += (int)ln;
        } else if (wt == 1) {
            if (pos + 8 > len) { out_n[m] = -1; return; }
            row[nf*4+0] = fno; row[nf*4+1] = 1;
            row[nf*4+2] = (int32_t)(base + pos); row[nf*4+3] = 8;
            pos += 8;
        } else if (wt == 5) {
            if (pos + 4 > len) { out_n[m] = -1; return; }
            row[nf*4+0] = fno; row[nf*4+1] = 5;
            row[nf*4+2] = (int32_t)(base + pos); row[nf*4+3] = 4;
            pos += 4;
        } else {
            out_n[m] = -1;
            return;
        }
        ++nf;
    }
    out_n[m] = nf;
}

// ---------------------------------------------------------------------------
// MFMA i8 probe — one v_mfma_i32_16x16x64_i8 with both operands loaded
// from global using the assumed CDNA4 fragment mapping (A: lane l holds
// A[row=l&15][k=(l>>4)*16 .. +15]; B: lane l holds the matching
// B[k][col=l&15] K-slice; D: col=lane&15, row=(lane>>4)*4+reg —
// cdna_hip_programming.md §"Fragment layout"). tests/test_gpu_engine.py
// asserts D == numpy A@B, pinning the layout the ETag hash builds on.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(64)
k_mfma_i8_probe(const int8_t* __restrict__ A,   // [16][64]
                const int8_t* __restrict__ B,   // [64][16]
                int32_t* __restrict__ D) {      // [16][16]
    const int lane = threadIdx.x & 63;
    const int row = lane & 15;
    const int k0 = (lane >> 4) * 16;
    int8_t a_bytes[16], b_bytes[16];
    for (int j = 0; j < 16; ++j) {
        a_bytes[j] = A[row * 64 + (k0 + j)];
        b_bytes[j] = B[(k0 + j) * 16 + row];  // col == lane&15 == row var
    }
    v4i_t av, bv, cv = {0, 0, 0, 0};
    __builtin_memcpy(&av, a_bytes, 16);
    __builtin_memcpy(&bv, b_bytes, 16);
    v4i_t dv = __builtin_amdgcn_mfma_i32_16x16x64_i8(av, bv, cv, 0, 0, 0);
    const int col = lane & 15;
    const int row_d = (lane >> 4) * 4;
    for (int r = 0; r < 4; ++r)
        D[(row_d + r) * 16 + col] = dv[r];
}

// ---------------------------------------------------------------------------
// k_padscan — exclusive cumsum of round16(resp_len) (the compact-stream
// offsets) + total + host_needed mirror, in ONE single-workgroup kernel.
// Replaces the host-side torch chain (elementwise pad + rocprim scan +
// copies), whose ~6 dispatches of host overhead paced the whole serving
// loop. tables layout (int32): [0:n] resp_len (written by k_respond),
// [n:2n] resp_off (written here), [2n] total, [2n+1] host_needed.
// ---------------------------------------------------------------------------
#define SCAN_THREADS 1024

extern "C" __global__ void __launch_bounds__(SCAN_THREADS)
k_padscan(int32_t* __restrict__ tables,
          const int32_t* __restrict__ host_needed,
          int n, int32_t* __restrict__ host_mirror) {
    __shared__ int32_t partials[SCAN_THREADS];
    const int tid = threadIdx.x;
    const int chunk = (n + SCAN_THREADS - 1) / SCAN_THREADS;
    const int s = tid * chunk;
    int e = s + chunk;
    if (e > n) e = n;
    // per-thread serial sum of padded lengths
    int32_t sum = 0;
    for (int i = s; i < e; ++i) sum += (tables[i] + 15) & ~15;
    partials[tid] = sum;
    __syncthreads();
    // Hillis-Steele inclusive scan over the 1024 partials
    for (int off = 1; off < SCAN_THREADS; off <<= 1) {
        int32_t v = (tid >= off) ? partials[tid - off] : 0;
        __syncthreads();
        partials[tid] += v;
        __syncthreads();
    }
    // exclusive offsets for this thread's chunk
    int32_t run = (tid > 0) ? partials[tid - 1] : 0;
    for (int i = s; i < e; ++i) {
        const int32_t len_i = tables[i];
        tables[n + i] = run;
        if (host_mirror) {
            host_mirror[i] = len_i;
            host_mirror[n + i] = run;
        }
        run += (len_i + 15) & ~15;
    }
    if (tid == SCAN_THREADS - 1) {
        tables[2 * n] = partials[tid];
        if (host_mirror) host_mirror[2 * n] = partials[tid];
    }
    if (tid == 0) {
        tables[2 * n + 1] = *host_needed;
        if (host_mirror) host_mirror[2 * n + 1] = *host_needed;
    }
}

// k_done — publish the batch serial to pinned host memory (system
// release; flagged pipeline's completion signal — replaces the event
// machinery entirely). Launched after k_compact on the same stream,
// so kernel-boundary coherence makes all prior host writes (compact's
// egress, padscan's table mirror) visible before the serial lands.
extern "C" __global__ void k_done(int32_t* cell, int serial) {
    if (threadIdx.x == 0)
        __hip_atomic_store(cell, serial, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_SYSTEM);
}

// ---------------------------------------------------------------------------
// k_grpc_echo — batched unary SayHello responder (BASELINE config 5's
// gRPC half). One message per lane: read the name span from the
// k_varint_spans table, emit the gRPC length-prefixed HelloResponse
// frame {message: "Hello <name>!"} ("World" when the name is empty —
// examples/grpc-server semantics; reference examples/grpc-server/grpc/
// server.go:12-21). Host golden model: gofr_amd/ops cpu_grpc_echo.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_grpc_echo(const uint8_t* __restrict__ buf,
            const int32_t* __restrict__ spans,   // [n, MAX_PB_FIELDS, 4]
            const int32_t* __restrict__ span_n,  // [n]
            uint8_t* __restrict__ out,           // [n, rslot]
            int32_t* __restrict__ out_len,
            int n, int rslot) {
    const int m = blockIdx.x * BLOCK_THREADS + threadIdx.x;
    if (m >= n) return;
    const int32_t* row = spans + (size_t)m * MAX_PB_FIELDS * 4;
    const int nf = span_n[m];
    int name_off = 0, name_len = 0;
    for (int i = 0; i < nf; ++i) {
        if (row[i*4+0] == 1 && row[i*4+1] == 2) {
            name_off = row[i*4+2];
            name_len = row[i*4+3];
        }
    }
    const uint8_t* nm = buf + name_off;
    static const char WORLD[5] = {'W', 'o', 'r', 'l', 'd'};
    if (name_len == 0) { nm = (const uint8_t*)WORLD; name_len = 5; }
    // HelloResponse { 1: "Hello <name>!" }: tag 0x0A + varint len + payload
    const int payload_len = 6 + name_len + 1;
    const int vlen = payload_len < 128 ? 1 : 2;
    const int msg_len = 1 + vlen + payload_len;
    if (nf < 0 || 5 + msg_len > rslot) {
        out_len[m] = -1;  // malformed/oversized: host error path
        return;
    }
    uint8_t* o = out + (size_t)m * rslot;
    o[0] = 0;  // uncompressed
    o[1] = (uint8_t)(msg_len >> 24); o[2] = (uint8_t)(msg_len >> 16);
    o[3] = (uint8_t)(msg_len >> 8);  o[4] = (uint8_t)msg_len;
    int p = 5;
    o[p++] = 0x0A;
    if (vlen == 1) {
        o[p++] = (uint8_t)payload_len;
    } else {
        o[p++] = (uint8_t)(payload_len | 0x80);
        o[p++] = (uint8_t)(payload_len >> 7);
    }
    o[p++]='H'; o[p++]='e'; o[p++]='l'; o[p++]='l'; o[p++]='o'; o[p++]=' ';
    for (int i = 0; i < name_len; ++i) o[p + i] = nm[i];
    p += name_len;
    o[p++] = '!';
    out_len[m] = p;
}

// ---------------------------------------------------------------------------
// k_persist_cycle — persistent serving-cycle probe (the round-2
// architecture candidate: one resident kernel replaces per-batch
// launches and cross-stream dependencies). G co-resident blocks loop
// over nbatch batches; per batch:
//   block 0 spins (bounded) on d_go until the host's SDMA writes the
//   batch serial (enqueued after the payload copy, same FIFO), a grid
//   barrier releases all blocks, every block streams its share of the
//   batch to the pinned egress ring (the link-bound work), a second
//   barrier, then block 0 publishes the serial to pinned p_done.
// All spins are iteration-bounded: the kernel always terminates.
// Driver: benchmarks/persistent_probe.py.
// ---------------------------------------------------------------------------
__device__ void grid_barrier(int* counter, int* gen_ptr, int nblocks) {
    __syncthreads();
    if (threadIdx.x == 0) {
        const int my_gen = __hip_atomic_load(gen_ptr, __ATOMIC_ACQUIRE,
                                             __HIP_MEMORY_SCOPE_AGENT);
        if (atomicAdd(counter, 1) == nblocks - 1) {
            __hip_atomic_store(counter, 0, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
            __hip_atomic_store(gen_ptr, my_gen + 1, __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_AGENT);
        } else {
            // process-lifetime bound: the persistent kernel's crews
            // legitimately sit at barriers while the host is idle
            // between batches (the 0.9 s probe-era bound let waiters
            // ESCAPE and run ahead, smearing state across batches).
            // An abandoned kernel dies with its process's queues.
            for (long i = 0; i < (1L << 33); ++i) {
                if (__hip_atomic_load(gen_ptr, __ATOMIC_ACQUIRE,
                                      __HIP_MEMORY_SCOPE_AGENT) != my_gen)
                    break;
                __builtin_amdgcn_s_sleep(8);
            }
        }
    }
    __syncthreads();
}

extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_persist_cycle(const unsigned long long* __restrict__ d_go,
                volatile unsigned long long* p_done,
                const uint8_t* __restrict__ d_in,
                uint8_t* __restrict__ p_out,
                long nbytes, int nbatch, int do_work,
                int* __restrict__ d_barrier) {
    const int nblocks = gridDim.x;
    for (int b = 1; b <= nbatch; ++b) {
        if (blockIdx.x == 0 && threadIdx.x == 0) {
            for (long i = 0; i < (1L << 20); ++i) {
                if (__hip_atomic_load(d_go, __ATOMIC_ACQUIRE,
                                      __HIP_MEMORY_SCOPE_SYSTEM) >=
                    (unsigned long long)b)
                    break;
                __builtin_amdgcn_s_sleep(16);
            }
        }
        grid_barrier(d_barrier, d_barrier + 1, nblocks);
        if (do_work) {
            // link-bound egress: stream the batch to the pinned ring
            const long nv = nbytes >> 4;
            const uint4* src = (const uint4*)d_in;
            uint4* dst = (uint4*)p_out;
            const long stride = (long)nblocks * BLOCK_THREADS;
            for (long i = blockIdx.x * BLOCK_THREADS + threadIdx.x;
                 i < nv; i += stride)
                dst[i] = src[i];
        }
        grid_barrier(d_barrier, d_barrier + 1, nblocks);
        if (blockIdx.x == 0 && threadIdx.x == 0)
            __hip_atomic_store(p_done, (unsigned long long)b,
                               __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_SYSTEM);
    }
}

// ---------------------------------------------------------------------------
// k_persist_serve — the PERSISTENT serving engine (round-2 headline
// path; derisked by benchmarks/persistent_probe.py in r1). One
// resident kernel replaces the per-batch launch chain and its
// stream-handoff dead time (~0.45 ms/batch measured):
//
//   compute crew (blocks egress_blocks..G-1): per batch — block 0
//     spins (bounded) on the go serial the host's ingress SDMA writes
//     after the payload (same FIFO), crew barrier, parse+respond fused
//     per wave over a block-strided request range (auth phase between
//     when a secret is armed), crew barrier, block 0 runs the pad16
//     scan + pinned table mirror, then publishes compute_done.
//   egress crew (blocks 0..egress_blocks-1): waits compute_done,
//     streams the compact response stream STRAIGHT into the pinned
//     egress ring (the link-bound work), then publishes the batch
//     serial to the pinned done cell (system release).
//
// Batches alternate 2 slots, so the SDMA ingress of batch b+1 and the
// compute of batch b and the egress of batch b-1 all overlap; slot
// reuse is safe because the host never writes go[b+2] before it
// completed b. Every spin is bounded and latches a timeout marker
// (pinned cell [2n+3]) — the kernel ALWAYS terminates after nbatch
// batches. Grid size must be fully co-resident: the host launcher
// sizes it from hipOccupancyMaxActiveBlocksPerMultiprocessor.
// ---------------------------------------------------------------------------

typedef struct {
    unsigned long long first;   // first batch serial of this launch
    int nbatch;
    int n, rslot;
    long long hdr_bytes;        // ingress header size (offs/lens/seed/date)
    long long lens_off;         // byte offset of the int32 lens array
    int date_off;
    int egress_blocks;
    // per-slot device buffers
    void* d_ingress[2];
    void* d_fields[2];
    void* d_resp[2];
    void* d_tables[2];
    void* p_tables[2];          // pinned int32[2n+4]
    void* p_out[2];
    void* host_blob[2];
    void* host_tab[2];
    // shared
    const void* trie[9];
    const void* handler_tab;
    int n_routes;
    const void* blob;
    const void* d_kv_tab;
    const void* d_kv_blob;
    const void* secret;
    int secret_len;
    int auth_env_off, auth_env_len, etag_on;
    void* d_state;              // u64[4]: [0] go, [1] compute_done, [3] tmo latch
    void* d_barrier;            // int[8]: cbar cnt/gen, ebar cnt/gen, hn[2]
} PersistKernArgs;

// 256-thread pad16 exclusive scan (k_padscan adapted to one crew block)
__device__ void scan_tables_crew(int32_t* tables, int hn, int n,
                                 int32_t* mirror) {
    __shared__ int32_t partials[BLOCK_THREADS];
    const int tid = threadIdx.x;
    const int chunk = (n + BLOCK_THREADS - 1) / BLOCK_THREADS;
    const int s = tid * chunk;
    int e = s + chunk;
    if (e > n) e = n;
    int32_t sum = 0;
    for (int i = s; i < e; ++i) sum += (tables[i] + 15) & ~15;
    partials[tid] = sum;
    __syncthreads();
    for (int off = 1; off < BLOCK_THREADS; off <<= 1) {
        int32_t v = (tid >= off) ? partials[tid - off] : 0;
        __syncthreads();
        partials[tid] += v;
        __syncthreads();
    }
    int32_t run = (tid > 0) ? partials[tid - 1] : 0;
    for (int i = s; i < e; ++i) {
        const int32_t len_i = tables[i];
        tables[n + i] = run;
        mirror[i] = len_i;
        mirror[n + i] = run;
        run += (len_i + 15) & ~15;
    }
    if (tid == BLOCK_THREADS - 1) {
        tables[2 * n] = partials[tid];
        mirror[2 * n] = partials[tid];
    }
    if (tid == 0) {
        tables[2 * n + 1] = hn;
        mirror[2 * n + 1] = hn;
    }
}

// k_auth body as a per-thread device function (one request per lane)
__device__ void auth_one(const uint8_t* reqs, const int64_t* req_off,
                         int32_t* fields, int req,
                         const uint8_t* secret, int secret_len);

extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_persist_serve(PersistKernArgs a) {
    __shared__ uint64_t masks_all[WAVES_PER_BLOCK][N_CLS][MAX_CHUNKS];
    __shared__ int lf_pos_all[WAVES_PER_BLOCK][MAX_LFS];
    __shared__ uint8_t obuf_all[WAVES_PER_BLOCK * MAX_SLOT +
                                WAVES_PER_BLOCK * MAX_JSON_FIELDS * 16];
    const int lane = lane_id();
    const int wv = threadIdx.x / WAVE;
    int32_t* jtab = (int32_t*)(obuf_all + WAVES_PER_BLOCK * MAX_SLOT) +
                    wv * MAX_JSON_FIELDS * 4;
    uint8_t* obuf = obuf_all + wv * MAX_SLOT;
    const bool is_egress = (int)blockIdx.x < a.egress_blocks;
    const int crew_rank = is_egress ? blockIdx.x
                                    : blockIdx.x - a.egress_blocks;
    const int crew_n = is_egress ? a.egress_blocks
                                 : gridDim.x - a.egress_blocks;
    unsigned long long* go = (unsigned long long*)a.d_state;
    unsigned long long* cdone = go + 1;
    unsigned long long* cverdict = go + 2;  // per-batch latch verdict
    unsigned long long* tmo_latch = go + 3;
    int* cbar = (int*)a.d_barrier;
    int* ebar = cbar + 2;
    int* hn_cells = cbar + 4;  // per-slot host_needed counters
    TrieDev trie{(const uint8_t*)a.trie[0], (const int32_t*)a.trie[1],
                 (const int32_t*)a.trie[2], (const int32_t*)a.trie[3],
                 (const int32_t*)a.trie[4], (const int32_t*)a.trie[5],
                 (const int32_t*)a.trie[6], (const int32_t*)a.trie[7],
                 (const int32_t*)a.trie[8]};
    const int n = a.n;
    for (unsigned long long b = a.first; b < a.first + a.nbatch; ++b) {
        // serial 1 lands in slot 0 (the engine's lane round-robin)
        const int slot = (int)((b - 1) & 1);
        uint8_t* d_ing = (uint8_t*)a.d_ingress[slot];
        const int64_t* offs = (const int64_t*)d_ing;
        const int32_t* lens = (const int32_t*)(d_ing + a.lens_off);
        uint8_t* reqs = d_ing + a.hdr_bytes;
        int32_t* fields = (int32_t*)a.d_fields[slot];
        uint8_t* resp = (uint8_t*)a.d_resp[slot];
        int32_t* tables = (int32_t*)a.d_tables[slot];
        int32_t* p_tab = (int32_t*)a.p_tables[slot];
        const uint8_t* date29 = d_ing + a.date_off;
        if (!is_egress) {
            bool tmo = false;
            if (crew_rank == 0 && threadIdx.x == 0) {
                __hip_atomic_store(&hn_cells[slot], 0, __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
                if (!__hip_atomic_load(tmo_latch, __ATOMIC_ACQUIRE,
                                       __HIP_MEMORY_SCOPE_AGENT)) {
                    // effectively-unbounded idle tolerance (hours):
                    // a serving loop may sit idle arbitrarily long
                    // between batches. Clean shutdown is the host
                    // writing the latch (engine.stop_persistent); an
                    // ABANDONED kernel dies with its process (the
                    // driver tears the process's queues down), so
                    // this spin cannot leave the GPU wedged.
                    long i = 0;
                    for (; i < (1L << 33); ++i) {
                        if (__hip_atomic_load(
                                go, __ATOMIC_ACQUIRE,
                                __HIP_MEMORY_SCOPE_SYSTEM) >= b)
                            break;
                        if ((i & 0xFFF) == 0 &&
                            __hip_atomic_load(tmo_latch,
                                              __ATOMIC_ACQUIRE,
                                              __HIP_MEMORY_SCOPE_SYSTEM))
                            break;
                        __builtin_amdgcn_s_sleep(32);
                    }
                    tmo = __hip_atomic_load(
                              go, __ATOMIC_ACQUIRE,
                              __HIP_MEMORY_SCOPE_SYSTEM) < b;
                } else {
                    tmo = true;
                }
                if (tmo) {
                    // latch: the remaining batches of this launch
                    // drain in microseconds (work skipped, markers
                    // published) so the kernel always exits promptly
                    __hip_atomic_store(tmo_latch, 1ull,
                                       __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_AGENT);
                    __hip_atomic_store(
                        p_tab + 2 * n + 3,
                        (int32_t)(b & 0x7fffffff), __ATOMIC_RELEASE,
                        __HIP_MEMORY_SCOPE_SYSTEM);
                }
            }
            if (crew_rank == 0 && threadIdx.x == 0)
                __hip_atomic_store(cverdict, tmo ? 1ull : 0ull,
                                   __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
            grid_barrier(cbar, cbar + 1, crew_n);
            // UNIFORM latch verdict: published by crew block 0 BEFORE
            // the barrier, so every block takes the same path (a
            // direct racy read of the host-written latch could
            // diverge and deadlock the crew barriers)
            if (__hip_atomic_load(cverdict, __ATOMIC_ACQUIRE,
                                  __HIP_MEMORY_SCOPE_AGENT)) {
                grid_barrier(cbar, cbar + 1, crew_n);
                if (crew_rank == 0 && threadIdx.x == 0)
                    __hip_atomic_store(cdone, (b << 1) | 1ull,
                                       __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_AGENT);
                continue;
            }
            const uint64_t seed = *(const uint64_t*)(offs + n);
            for (int req = crew_rank * WAVES_PER_BLOCK + wv; req < n;
                 req += crew_n * WAVES_PER_BLOCK) {
                parse_one(reqs, offs, lens, fields, n, trie,
                          (const int32_t*)a.handler_tab, a.n_routes,
                          &hn_cells[slot], masks_all[wv],
                          lf_pos_all[wv], req, lane);
                // lane 0's field-table stores must land before the
                // whole wave's respond reads them (the kernel boundary
                // used to provide this)
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                if (a.secret_len == 0) {
                    respond_impl<false>(
                        reqs, offs, fields, resp, tables, n, a.rslot,
                        (const int32_t*)a.handler_tab, a.n_routes,
                        (const uint8_t*)a.blob,
                        (const uint8_t*)a.host_blob[slot],
                        (const int32_t*)a.host_tab[slot], seed,
                        a.auth_env_off, a.auth_env_len, 0, a.etag_on,
                        date29, (const int32_t*)a.d_kv_tab,
                        (const uint8_t*)a.d_kv_blob, obuf, nullptr,
                        jtab, nullptr, nullptr, req, lane);
                }
            }
            if (a.secret_len > 0) {
                __threadfence();
                grid_barrier(cbar, cbar + 1, crew_n);
                for (int req = crew_rank * BLOCK_THREADS + threadIdx.x;
                     req < n; req += crew_n * BLOCK_THREADS)
                    auth_one(reqs, offs, fields, req,
                             (const uint8_t*)a.secret, a.secret_len);
                __threadfence();
                grid_barrier(cbar, cbar + 1, crew_n);
                for (int req = crew_rank * WAVES_PER_BLOCK + wv;
                     req < n; req += crew_n * WAVES_PER_BLOCK)
                    respond_impl<false>(
                        reqs, offs, fields, resp, tables, n, a.rslot,
                        (const int32_t*)a.handler_tab, a.n_routes,
                        (const uint8_t*)a.blob,
                        (const uint8_t*)a.host_blob[slot],
                        (const int32_t*)a.host_tab[slot], seed,
                        a.auth_env_off, a.auth_env_len, 0, a.etag_on,
                        date29, (const int32_t*)a.d_kv_tab,
                        (const uint8_t*)a.d_kv_blob, obuf, nullptr,
                        jtab, nullptr, nullptr, req, lane);
            }
            __threadfence();
            grid_barrier(cbar, cbar + 1, crew_n);
            if (crew_rank == 0)
                scan_tables_crew(tables, hn_cells[slot], n, p_tab);
            __threadfence_system();
            grid_barrier(cbar, cbar + 1, crew_n);
            if (crew_rank == 0 && threadIdx.x == 0)
                __hip_atomic_store(cdone, b << 1, __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
        } else {
            if (crew_rank == 0 && threadIdx.x == 0) {
                // compute's latched path still publishes cdone, so
                // this wait always releases; the bound matches the
                // go-wait's process-lifetime reasoning
                unsigned long long cd = 0;
                for (long i = 0; i < (1L << 33); ++i) {
                    cd = __hip_atomic_load(cdone, __ATOMIC_ACQUIRE,
                                           __HIP_MEMORY_SCOPE_AGENT);
                    if ((cd >> 1) >= b) break;
                    __builtin_amdgcn_s_sleep(16);
                }
                // publish the verdict for THIS crew uniformly (the
                // low bit of cdone; verdicts are monotonic so a
                // later batch's bit only over-drops during shutdown)
                __hip_atomic_store(tmo_latch + 1, cd & 1ull,
                                   __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
            }
            grid_barrier(ebar, ebar + 1, crew_n);
            if (__hip_atomic_load(tmo_latch + 1, __ATOMIC_ACQUIRE,
                                  __HIP_MEMORY_SCOPE_AGENT)) {
                grid_barrier(ebar, ebar + 1, crew_n);
                if (crew_rank == 0 && threadIdx.x == 0)
                    __hip_atomic_store(p_tab + 2 * n + 2,
                                       (int32_t)(b & 0x7fffffff),
                                       __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_SYSTEM);
                continue;
            }
            uint8_t* p_out = (uint8_t*)a.p_out[slot];
            for (int req = crew_rank * WAVES_PER_BLOCK + wv; req < n;
                 req += crew_n * WAVES_PER_BLOCK) {
                const int len = tables[req];
                const int nv = (len + 15) >> 4;
                const uint4* src =
                    (const uint4*)(resp + (size_t)req * a.rslot);
                uint4* dst = (uint4*)(p_out + (size_t)tables[n + req]);
                for (int i = lane; i < nv; i += WAVE) dst[i] = src[i];
            }
            __threadfence_system();
            grid_barrier(ebar, ebar + 1, crew_n);
            if (crew_rank == 0 && threadIdx.x == 0)
                __hip_atomic_store(p_tab + 2 * n + 2,
                                   (int32_t)(b & 0x7fffffff),
                                   __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_SYSTEM);
        }
    }
}

// ---------------------------------------------------------------------------
// k_compact — gather response slots into a contiguous 16B-aligned stream.
// `out` may be DEVICE memory (classic compact; offsets are the exclusive
// cumsum of round16(resp_len)) or PINNED HOST memory: the kernel then IS
// the egress path, streaming responses over the host link directly.
// Measured on MI355X, kernel writes to pinned host sustain ~54 GB/s vs
// ~44 GB/s for the runtime's copyBuffer blit (ROCm never uses SDMA for
// D2H-to-pinned here — see benchmarks/overlap_probe.py), and fusing the
// egress removes one queue hop. Grid-strided with a capped grid so the
// link-bound sweep leaves CUs free for the next batch's parse/respond.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
k_compact(const uint8_t* __restrict__ resp_slots,
          const int32_t* __restrict__ resp_len,
          const int32_t* __restrict__ resp_off,
          uint8_t* __restrict__ out,
          int n, int rslot) {
    const int lane = lane_id();
    const int wv = threadIdx.x / WAVE;
    for (int req = blockIdx.x * WAVES_PER_BLOCK + wv; req < n;
         req += gridDim.x * WAVES_PER_BLOCK) {
        const int len = resp_len[req];
        const int nv = (len + 15) >> 4;
        const uint4* src = (const uint4*)(resp_slots + (size_t)req * rslot);
        uint4* dst = (uint4*)(out + (size_t)resp_off[req]);
        for (int i = lane; i < nv; i += WAVE) dst[i] = src[i];
    }
}

// ---------------------------------------------------------------------------
// C API (ctypes-driven; raw device pointers from torch tensors)
// ---------------------------------------------------------------------------

// One-call staged submit (the native serving-loop driver). The Python
// loop was spending ~0.3 ms/step in framework dispatch enqueueing the
// same ~14 stream operations — at 32k requests/batch that overhead,
// not the GPU, paced the engine. This enqueues the whole batch
// pipeline with raw HIP calls. Two completion modes:
//
// Event mode (flagged == 0, the original design):
//   s_in : H2D reqs + offsets(+seed tail) + lens, record ev_in
//   s_k  : wait ev_in; memset host_needed; k_parse_route; [k_auth];
//          k_respond; k_padscan  -> record ev_k
//   s_out: wait ev_k; k_compact straight into the pinned egress ring;
//          one D2H of the result tables; record ev_done
// Event handles come from torch (interop: same HIP runtime); the
// Python side waits with event.synchronize(). Measured cost: an event
// whose producer ends in an SDMA copy releases cross-stream waiters
// ~0.3 ms late, and hipEventSynchronize itself wakes ~0.3 ms after
// completion — two stalls per batch at the host-visible edge.
//
// Flagged mode (flagged != 0, the production default): no events at
// all. The serial flag travels in the SAME ingress copy as the batch
// (trailing word of the offsets block), k_gate spins on it in device
// memory and releases the chain, k_padscan mirrors the result tables
// into pinned host memory itself (host_mirror arg), k_compact writes
// the pinned egress ring directly, and k_done publishes the serial to
// a pinned word (p_serial) with a system-scope release. Python spins
// on plain pinned memory — completion latency is memory latency, not
// the runtime's wakeup path.
typedef struct {
    void* s_in; void* s_k; void* s_out;
    void* ev_in; void* ev_k; void* ev_done;
    const void* p_reqs; void* d_reqs; long long nbytes;
    const void* p_off; void* d_off;        // (n+1) int64, seed in tail
    const void* p_len; void* d_len;        // n int32
    void* d_fields;
    const void* trie[9];
    const void* handler_tab; int n_routes;
    void* d_host_needed;
    const void* secret; int secret_len;
    void* d_resp;
    void* d_tables;                        // int32[2n+2]
    void* p_tables;  // pinned int32[2n+4]; flagged mode publishes the
                     // done serial into [2n+2] (k_done)
    const void* blob; const void* host_blob; const void* host_tab;
    int auth_env_off; int auth_env_len; int gzip_min; int etag_on;
    void* p_out;                           // pinned egress ring
    int n; int rslot;
    // egress mode: budget > 0 -> k_compact into d_out (HBM) + one
    // budget-sized hipMemcpyAsync D2H (runtime blit; fastest duplex
    // partner for the SDMA ingress). budget == 0 -> k_compact writes
    // the pinned ring directly (kernel-driven host writes).
    void* d_out;
    long long egress_budget;
    // SDMA-flag gate (optional): when d_flag != 0, the kernel stage is
    // released by k_gate spinning on d_flag instead of an event wait
    void* d_flag;
    const void* p_serial;        // pinned cell holding `serial`
    unsigned long long serial;
    // flagged pipeline: no events — k_gate releases the kernel chain,
    // k_padscan mirrors the tables to pinned host, k_done publishes
    // the serial into p_tables[2n+2]. s_k carries the WHOLE chain
    // (kernels + egress), so batches on alternating channel streams
    // overlap each other and the SDMA ingress.
    int flagged;
    // 29-byte IMF-fixdate in device memory (rides the ingress block's
    // header slot; all responses of a batch share it). 0 -> no Date
    // header (Go's net/http attaches Date to every response; parity).
    const void* d_date;
    // device KV stores (HK_KV routes): hash table rows + value blob
    const void* d_kv_tab;
    const void* d_kv_blob;
} GofrSubmitArgs;

// k_gate — single-wave stream gate: spins (system-scope acquire
// loads) until the ingress SDMA writes the batch serial to d_flag
// (enqueued AFTER the payload copies, so FIFO order implies the
// payload landed). Replaces hipStreamWaitEvent on an SDMA-recorded
// event, whose signal delivery costs ~0.3 ms serialized in the kernel
// stream's FIFO; the gate releases within the SDMA's own completion.
// Timeout guard: gives up after ~0.5 s so a lost flag can never hang
// the stream — and REPORTS it: the batch serial is published to the
// pinned timeout cell (system release), which BatchEngine.complete()
// checks before releasing the batch's responses, so a lost/late SDMA
// flag surfaces as an error instead of a batch of stale bytes.
extern "C" __global__ void k_gate(const unsigned long long* flag,
                                  unsigned long long serial,
                                  int32_t* timeout_cell) {
    if (threadIdx.x != 0) return;
    for (long i = 0; i < (1L << 19); ++i) {
        const unsigned long long v = __hip_atomic_load(
            flag, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
        if (v >= serial) return;
        __builtin_amdgcn_s_sleep(32);
    }
    if (timeout_cell)
        __hip_atomic_store(timeout_cell, (int32_t)(serial & 0x7fffffff),
                           __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

// enqueue-time breakdown (µs, cumulative): [0] big H2D, [1] rest of
// ingress, [2] kernel stage, [3] egress; [4] = calls
static double g_submit_us[5] = {0, 0, 0, 0, 0};
static inline double now_us() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec * 1e6 + ts.tv_nsec * 1e-3;
}
extern "C" void gofr_submit_stats(double* out) {
    for (int i = 0; i < 5; ++i) { out[i] = g_submit_us[i];
                                  g_submit_us[i] = 0; }
}

static int gofr_submit_impl(const GofrSubmitArgs* a);

extern "C" int gofr_submit_staged(const GofrSubmitArgs* a) {
    return gofr_submit_impl(a);
}

static int gofr_submit_impl(const GofrSubmitArgs* a) {
    hipStream_t s_in = (hipStream_t)a->s_in;
    hipStream_t s_k = (hipStream_t)a->s_k;
    hipStream_t s_out = (hipStream_t)a->s_out;
    const int n = a->n;
    hipError_t rc;
    double t0 = now_us();
    g_submit_us[4] += 1;
    // ---- ingress: ONE copy (header: offsets+lens+seed, then payload;
    // a->nbytes = header + payload bytes, a->p_off/d_off = block base)
    rc = hipMemcpyAsync(a->d_off, a->p_off, (size_t)a->nbytes,
                        hipMemcpyHostToDevice, s_in);
    if (rc) return (int)rc;
    g_submit_us[0] += now_us() - t0; t0 = now_us();
    if (a->d_flag) {
        // serial lands AFTER the payload (same SDMA FIFO)
        rc = hipMemcpyAsync(a->d_flag, a->p_serial, 8,
                            hipMemcpyHostToDevice, s_in);
        if (rc) return (int)rc;
    } else {
        rc = hipEventRecord((hipEvent_t)a->ev_in, s_in);
        if (rc) return (int)rc;
    }
    g_submit_us[1] += now_us() - t0; t0 = now_us();
    // ---- kernel stage -----------------------------------------------------
    if (a->d_flag) {
        // timeout marker cell = p_tables[2n+3] (pinned; complete()
        // checks it before releasing the batch)
        int32_t* tmo = a->p_tables
            ? (int32_t*)a->p_tables + 2 * (size_t)n + 3 : nullptr;
        hipLaunchKernelGGL(k_gate, dim3(1), dim3(64), 0, s_k,
                           (const unsigned long long*)a->d_flag,
                           a->serial, tmo);
        if ((rc = hipGetLastError())) return (int)rc;
    } else {
        rc = hipStreamWaitEvent(s_k, (hipEvent_t)a->ev_in, 0);
        if (rc) return (int)rc;
    }
    rc = hipMemsetAsync(a->d_host_needed, 0, 4, s_k);
    if (rc) return (int)rc;
    TrieDev trie{(const uint8_t*)a->trie[0], (const int32_t*)a->trie[1],
                 (const int32_t*)a->trie[2], (const int32_t*)a->trie[3],
                 (const int32_t*)a->trie[4], (const int32_t*)a->trie[5],
                 (const int32_t*)a->trie[6], (const int32_t*)a->trie[7],
                 (const int32_t*)a->trie[8]};
    const int blocks = (n + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK;
    hipLaunchKernelGGL(k_parse_route, dim3(blocks), dim3(BLOCK_THREADS), 0,
                       s_k,
                       (uint8_t*)a->d_reqs, (const int64_t*)a->d_off,
                       (const int32_t*)a->d_len, (int32_t*)a->d_fields, n,
                       trie, (const int32_t*)a->handler_tab, a->n_routes,
                       (int32_t*)a->d_host_needed);
    if ((rc = hipGetLastError())) return (int)rc;
    if (a->secret_len > 0) {
        const int ablocks = (n + BLOCK_THREADS - 1) / BLOCK_THREADS;
        hipLaunchKernelGGL(k_auth, dim3(ablocks), dim3(BLOCK_THREADS), 0,
                           s_k,
                           (const uint8_t*)a->d_reqs,
                           (const int64_t*)a->d_off,
                           (int32_t*)a->d_fields, n,
                           (const uint8_t*)a->secret, a->secret_len);
        if ((rc = hipGetLastError())) return (int)rc;
    }
    int32_t* tables = (int32_t*)a->d_tables;
    const uint64_t* seed_ptr = (const uint64_t*)a->d_off + n;
    if (a->gzip_min > 0) {
        hipLaunchKernelGGL(k_respond_gz, dim3(blocks), dim3(BLOCK_THREADS),
                           0, s_k,
                           (const uint8_t*)a->d_reqs,
                           (const int64_t*)a->d_off,
                           (int32_t*)a->d_fields,
                           (uint8_t*)a->d_resp, tables, n, a->rslot,
                           (const int32_t*)a->handler_tab, a->n_routes,
                           (const uint8_t*)a->blob,
                           (const uint8_t*)a->host_blob,
                           (const int32_t*)a->host_tab, seed_ptr,
                           a->auth_env_off, a->auth_env_len, a->gzip_min,
                           a->etag_on, (const uint8_t*)a->d_date,
                           (const int32_t*)a->d_kv_tab,
                           (const uint8_t*)a->d_kv_blob);
    } else {
        hipLaunchKernelGGL(k_respond, dim3(blocks), dim3(BLOCK_THREADS),
                           0, s_k,
                           (const uint8_t*)a->d_reqs,
                           (const int64_t*)a->d_off,
                           (int32_t*)a->d_fields,
                           (uint8_t*)a->d_resp, tables, n, a->rslot,
                           (const int32_t*)a->handler_tab, a->n_routes,
                           (const uint8_t*)a->blob,
                           (const uint8_t*)a->host_blob,
                           (const int32_t*)a->host_tab, seed_ptr,
                           a->auth_env_off, a->auth_env_len, a->etag_on,
                           (const uint8_t*)a->d_date,
                           (const int32_t*)a->d_kv_tab,
                           (const uint8_t*)a->d_kv_blob);
    }
    if ((rc = hipGetLastError())) return (int)rc;
    hipLaunchKernelGGL(k_padscan, dim3(1), dim3(SCAN_THREADS), 0, s_k,
                       tables, (const int32_t*)a->d_host_needed, n,
                       a->flagged ? (int32_t*)a->p_tables : nullptr);
    if ((rc = hipGetLastError())) return (int)rc;
    if (a->flagged) {
        // egress continues ON THE CHANNEL STREAM (caller set s_out ==
        // s_k); no event handoff — but still RECORD e_k (cheap
        // kernel-signal) so auxiliary consumers (the mixed
        // HTTP+gRPC loop) can chain off the kernel stage
        rc = hipEventRecord((hipEvent_t)a->ev_k, s_k);
        if (rc) return (int)rc;
        s_out = s_k;
    } else {
        rc = hipEventRecord((hipEvent_t)a->ev_k, s_k);
        if (rc) return (int)rc;
    }
    g_submit_us[2] += now_us() - t0; t0 = now_us();
    // ---- egress -----------------------------------------------------------
    if (!a->flagged) {
        rc = hipStreamWaitEvent(s_out, (hipEvent_t)a->ev_k, 0);
        if (rc) return (int)rc;
    }
    int cblocks = blocks;
    {
        static int cap = 0;
        if (cap == 0) {
            const char* e = getenv("GOFR_COMPACT_BLOCKS");
            cap = e ? atoi(e) : 64;
            if (cap <= 0) cap = 64;
        }
        if (cblocks > cap) cblocks = cap;
    }
    if (a->egress_budget > 0) {
        hipLaunchKernelGGL(k_compact, dim3(cblocks), dim3(BLOCK_THREADS),
                           0, s_out,
                           (const uint8_t*)a->d_resp, tables, tables + n,
                           (uint8_t*)a->d_out, n, a->rslot);
        if ((rc = hipGetLastError())) return (int)rc;
        rc = hipMemcpyAsync(a->p_out, a->d_out, (size_t)a->egress_budget,
                            hipMemcpyDeviceToHost, s_out);
        if (rc) return (int)rc;
    } else {
        hipLaunchKernelGGL(k_compact, dim3(cblocks), dim3(BLOCK_THREADS),
                           0, s_out,
                           (const uint8_t*)a->d_resp, tables, tables + n,
                           (uint8_t*)a->p_out, n, a->rslot);
        if ((rc = hipGetLastError())) return (int)rc;
    }
    if (a->flagged) {
        hipLaunchKernelGGL(k_done, dim3(1), dim3(64), 0, s_out,
                           (int32_t*)a->p_tables + 2 * n + 2,
                           (int)(a->serial & 0x7fffffff));
        if ((rc = hipGetLastError())) return (int)rc;
        // record ev_done AFTER k_done so the pump's completion thread
        // (which hipEventSynchronize's this event) publishes a truthful
        // done_serial even in flagged mode — the primary completion
        // signal is still the k_done serial in pinned p_tables
        rc = hipEventRecord((hipEvent_t)a->ev_done, s_out);
        g_submit_us[3] += now_us() - t0;
        return (int)rc;
    }
    rc = hipMemcpyAsync(a->p_tables, a->d_tables, (size_t)(2 * n + 2) * 4,
                        hipMemcpyDeviceToHost, s_out);
    if (rc) return (int)rc;
    rc = hipEventRecord((hipEvent_t)a->ev_done, s_out);
    g_submit_us[3] += now_us() - t0;
    return (int)rc;
}

// ---------------------------------------------------------------------------
// Native pump — the serving loop's runtime driver. Measured on MI355X,
// every host interaction with the HIP runtime on the hot path costs
// ~0.1-0.4 ms (enqueue of a dependent copy/kernel blocks inside rocclr;
// hipEventSynchronize wakes ~0.3 ms late), and with those on the Python
// thread the loop paced at ~1.3 ms/step while each GPU stage needs only
// ~0.8. The pump moves them to two worker threads:
//   submit thread    : dequeues GofrSubmitArgs, runs gofr_submit_impl
//   completion thread: hipEventSynchronize(ev_done) in submit order,
//                      then publishes the batch serial to a plain
//                      host-memory counter
// Python submits by pushing a prebuilt args block (a mutex push) and
// completes by spinning on the counter via numpy — no runtime locks on
// the serving thread at all.
// ---------------------------------------------------------------------------
#include <pthread.h>
#include <string.h>

#define PUMP_CAP 64

static struct Pump {
    GofrSubmitArgs items[PUMP_CAP];
    unsigned long long head;      // next slot to fill (serial + 1 basis)
    unsigned long long enq_idx;   // items enqueued to GPU
    unsigned long long cmp_idx;   // items completion-synced
    pthread_mutex_t mu;
    pthread_cond_t cv;
    pthread_t enq_thread, cmp_thread;
    int running;
    int err;
    volatile unsigned long long done_serial;  // read by Python (numpy)
} g_pump;

static void* pump_enq_main(void* _) {
    (void)_;
    for (;;) {
        pthread_mutex_lock(&g_pump.mu);
        while (g_pump.running && g_pump.enq_idx == g_pump.head)
            pthread_cond_wait(&g_pump.cv, &g_pump.mu);
        if (!g_pump.running && g_pump.enq_idx == g_pump.head) {
            pthread_mutex_unlock(&g_pump.mu);
            return nullptr;
        }
        GofrSubmitArgs a = g_pump.items[g_pump.enq_idx % PUMP_CAP];
        pthread_mutex_unlock(&g_pump.mu);
        int rc = gofr_submit_impl(&a);
        pthread_mutex_lock(&g_pump.mu);
        if (rc && !g_pump.err) g_pump.err = rc;
        g_pump.enq_idx++;
        pthread_cond_broadcast(&g_pump.cv);
        pthread_mutex_unlock(&g_pump.mu);
    }
}

static void* pump_cmp_main(void* _) {
    (void)_;
    for (;;) {
        pthread_mutex_lock(&g_pump.mu);
        while (g_pump.running && g_pump.cmp_idx == g_pump.enq_idx)
            pthread_cond_wait(&g_pump.cv, &g_pump.mu);
        if (!g_pump.running && g_pump.cmp_idx == g_pump.enq_idx) {
            pthread_mutex_unlock(&g_pump.mu);
            return nullptr;
        }
        hipEvent_t ev =
            (hipEvent_t)g_pump.items[g_pump.cmp_idx % PUMP_CAP].ev_done;
        pthread_mutex_unlock(&g_pump.mu);
        hipError_t rc = hipEventSynchronize(ev);
        pthread_mutex_lock(&g_pump.mu);
        if (rc && !g_pump.err) g_pump.err = (int)rc;
        g_pump.cmp_idx++;
        // publish: batch serials are 1-based (= cmp_idx after increment)
        __atomic_store_n(&g_pump.done_serial, g_pump.cmp_idx,
                         __ATOMIC_RELEASE);
        pthread_cond_broadcast(&g_pump.cv);
        pthread_mutex_unlock(&g_pump.mu);
    }
}

extern "C" int gofr_pump_start() {
    // idempotent: several engines in one process share the pump (a
    // re-init would memset over live worker threads)
    if (g_pump.running) return 0;
    memset((void*)&g_pump, 0, sizeof(g_pump));
    pthread_mutex_init(&g_pump.mu, nullptr);
    pthread_cond_init(&g_pump.cv, nullptr);
    g_pump.running = 1;
    if (pthread_create(&g_pump.enq_thread, nullptr, pump_enq_main, nullptr))
        return -1;
    if (pthread_create(&g_pump.cmp_thread, nullptr, pump_cmp_main, nullptr))
        return -1;
    return 0;
}

extern "C" void gofr_pump_stop() {
    pthread_mutex_lock(&g_pump.mu);
    g_pump.running = 0;
    pthread_cond_broadcast(&g_pump.cv);
    pthread_mutex_unlock(&g_pump.mu);
    pthread_join(g_pump.enq_thread, nullptr);
    pthread_join(g_pump.cmp_thread, nullptr);
}

// returns the 1-based serial of this batch, or 0 when the ring is full
// (caller must complete some batches first — the engine's lane protocol
// keeps at most `pipeline` outstanding, far below PUMP_CAP)
extern "C" unsigned long long gofr_pump_submit(const GofrSubmitArgs* a) {
    pthread_mutex_lock(&g_pump.mu);
    if (g_pump.head - g_pump.cmp_idx >= PUMP_CAP) {
        pthread_mutex_unlock(&g_pump.mu);
        return 0;
    }
    g_pump.items[g_pump.head % PUMP_CAP] = *a;
    g_pump.head++;
    unsigned long long serial = g_pump.head;
    pthread_cond_broadcast(&g_pump.cv);
    pthread_mutex_unlock(&g_pump.mu);
    return serial;
}

extern "C" const volatile unsigned long long* gofr_pump_done_ptr() {
    return &g_pump.done_serial;
}

extern "C" int gofr_pump_err() { return g_pump.err; }

// host-memory helpers (egress-ring allocation experiments: coherence
// flags decide whether the runtime will use SDMA for D2H)
extern "C" void* gofr_host_alloc(long long size, unsigned flags) {
    void* p = nullptr;
    if (hipHostMalloc(&p, (size_t)size, flags)) return nullptr;
    return p;
}

extern "C" int gofr_host_free(void* p) {
    return (int)hipHostFree(p);
}

// Flagged-completion wait OUTSIDE the GIL: spins on the pinned done
// cell until k_done publishes `want` (0), the gate-timeout cell shows
// `want` (2 — stale-ingress batch, caller must drop it), or timeout_s
// elapses (1). ctypes releases the GIL for the call, so several
// serving threads can wait on their lanes concurrently.
extern "C" int gofr_wait_cell(const void* cell_p, int want,
                              const void* tmo_p, double timeout_s) {
    const volatile int* cell = (const volatile int*)cell_p;
    const volatile int* tmo = (const volatile int*)tmo_p;
    const double t0 = now_us();
    long spins = 0;
    while (__atomic_load_n((const int*)cell, __ATOMIC_ACQUIRE) != want) {
        if (tmo && __atomic_load_n((const int*)tmo, __ATOMIC_ACQUIRE))
            return 2;  // any nonzero marker: gate timeout / latched
        if (((++spins) & 0xFFFF) == 0 &&
            now_us() - t0 > timeout_s * 1e6)
            return 1;
    }
    if (tmo && __atomic_load_n((const int*)tmo, __ATOMIC_ACQUIRE))
        return 2;
    return 0;
}

// source-hash stamp: build() compiles with -DGOFR_SRC_HASH="<sha256>"
// of the committed .hip source, and tests/test_build_hash.py asserts
// the loaded .so carries the hash of the source in the tree — a stale
// or foreign binary fails loudly instead of silently serving old code.
#ifndef GOFR_SRC_HASH
#define GOFR_SRC_HASH "unhashed"
#endif
extern "C" const char* gofr_src_hash() { return GOFR_SRC_HASH; }

extern "C" int gofr_memcpy_async(void* dst, const void* src, long long n,
                                 int kind, void* stream) {
    return (int)hipMemcpyAsync(dst, src, (size_t)n, (hipMemcpyKind)kind,
                               (hipStream_t)stream);
}

extern "C" {

int gofr_launch_parse_route(
        void* stream,
        const void* reqs, const void* req_off, const void* req_len,
        void* fields, int n,
        const void* seg_blob, const void* node_child_first,
        const void* node_child_count, const void* child_seg_off,
        const void* child_seg_len, const void* child_node,
        const void* node_param, const void* node_prefix,
        const void* node_route,
        const void* handler_tab, int n_routes,
        void* host_needed) {
    TrieDev trie{(const uint8_t*)seg_blob, (const int32_t*)node_child_first,
                 (const int32_t*)node_child_count, (const int32_t*)child_seg_off,
                 (const int32_t*)child_seg_len, (const int32_t*)child_node,
                 (const int32_t*)node_param, (const int32_t*)node_prefix,
                 (const int32_t*)node_route};
    const int blocks = (n + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK;
    hipLaunchKernelGGL(k_parse_route, dim3(blocks), dim3(BLOCK_THREADS), 0,
                       (hipStream_t)stream,
                       (uint8_t*)reqs, (const int64_t*)req_off,
                       (const int32_t*)req_len,
                       (int32_t*)fields, n, trie,
                       (const int32_t*)handler_tab, n_routes,
                       (int32_t*)host_needed);
    return (int)hipGetLastError();
}

int gofr_launch_respond(
        void* stream,
        const void* reqs, const void* req_off, void* fields, void* resp,
        void* resp_len_out, int n, int rslot,
        const void* handler_tab, int n_routes,
        const void* blob, const void* host_blob, const void* host_tab,
        const void* seed_ptr, int auth_env_off, int auth_env_len,
        int gzip_min, int etag_on, const void* date29,
        const void* kv_tab, const void* kv_blob) {
    const int blocks = (n + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK;
    if (gzip_min > 0) {
        hipLaunchKernelGGL(k_respond_gz, dim3(blocks), dim3(BLOCK_THREADS),
                           0, (hipStream_t)stream,
                           (const uint8_t*)reqs, (const int64_t*)req_off,
                           (int32_t*)fields,
                           (uint8_t*)resp, (int32_t*)resp_len_out,
                           n, rslot,
                           (const int32_t*)handler_tab, n_routes,
                           (const uint8_t*)blob, (const uint8_t*)host_blob,
                           (const int32_t*)host_tab,
                           (const uint64_t*)seed_ptr,
                           auth_env_off, auth_env_len, gzip_min, etag_on,
                           (const uint8_t*)date29,
                           (const int32_t*)kv_tab,
                           (const uint8_t*)kv_blob);
    } else {
        hipLaunchKernelGGL(k_respond, dim3(blocks), dim3(BLOCK_THREADS), 0,
                           (hipStream_t)stream,
                           (const uint8_t*)reqs, (const int64_t*)req_off,
                           (int32_t*)fields,
                           (uint8_t*)resp, (int32_t*)resp_len_out,
                           n, rslot,
                           (const int32_t*)handler_tab, n_routes,
                           (const uint8_t*)blob, (const uint8_t*)host_blob,
                           (const int32_t*)host_tab,
                           (const uint64_t*)seed_ptr,
                           auth_env_off, auth_env_len, etag_on,
                           (const uint8_t*)date29,
                           (const int32_t*)kv_tab,
                           (const uint8_t*)kv_blob);
    }
    return (int)hipGetLastError();
}

int gofr_launch_auth(
        void* stream,
        const void* reqs, const void* req_off, void* fields, int n,
        const void* secret, int secret_len) {
    const int blocks = (n + BLOCK_THREADS - 1) / BLOCK_THREADS;
    hipLaunchKernelGGL(k_auth, dim3(blocks), dim3(BLOCK_THREADS), 0,
                       (hipStream_t)stream,
                       (const uint8_t*)reqs, (const int64_t*)req_off,
                       (int32_t*)fields, n,
                       (const uint8_t*)secret, secret_len);
    return (int)hipGetLastError();
}

int gofr_launch_varint_spans(
        void* stream,
        const void* buf, const void* msg_off, const void* msg_len,
        void* out, void* out_n, int n) {
    const int blocks = (n + BLOCK_THREADS - 1) / BLOCK_THREADS;
    hipLaunchKernelGGL(k_varint_spans, dim3(blocks), dim3(BLOCK_THREADS), 0,
                       (hipStream_t)stream,
                       (const uint8_t*)buf, (const int64_t*)msg_off,
                       (const int32_t*)msg_len,
                       (int32_t*)out, (int32_t*)out_n, n);
    return (int)hipGetLastError();
}

int gofr_launch_grpc_echo(
        void* stream,
        const void* buf, const void* spans, const void* span_n,
        void* out, void* out_len, int n, int rslot) {
    const int blocks = (n + BLOCK_THREADS - 1) / BLOCK_THREADS;
    hipLaunchKernelGGL(k_grpc_echo, dim3(blocks), dim3(BLOCK_THREADS), 0,
                       (hipStream_t)stream,
                       (const uint8_t*)buf, (const int32_t*)spans,
                       (const int32_t*)span_n,
                       (uint8_t*)out, (int32_t*)out_len, n, rslot);
    return (int)hipGetLastError();
}

int gofr_launch_compact(
        void* stream,
        const void* resp_slots, const void* resp_len, const void* resp_off,
        void* out, int n, int rslot) {
    // Link-bound when out is pinned host memory: cap the grid so the
    // egress sweep leaves CUs free for the next batch's parse/respond
    // (tunable for the bandwidth/occupancy sweep in benchmarks/).
    static int cap = 0;
    if (cap == 0) {
        const char* e = getenv("GOFR_COMPACT_BLOCKS");
        cap = e ? atoi(e) : 64;
        if (cap <= 0) cap = 64;
    }
    int blocks = (n + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK;
    if (blocks > cap) blocks = cap;
    hipLaunchKernelGGL(k_compact, dim3(blocks), dim3(BLOCK_THREADS), 0,
                       (hipStream_t)stream,
                       (const uint8_t*)resp_slots, (const int32_t*)resp_len,
                       (const int32_t*)resp_off, (uint8_t*)out, n, rslot);
    return (int)hipGetLastError();
}

}  // extern "C"

extern "C" int gofr_launch_mfma_probe(void* stream, const void* A,
                                      const void* B, void* D) {
    hipLaunchKernelGGL(k_mfma_i8_probe, dim3(1), dim3(64), 0,
                       (hipStream_t)stream,
                       (const int8_t*)A, (const int8_t*)B, (int32_t*)D);
    return (int)hipGetLastError();
}

// ---- persistent serving engine host API -----------------------------------

// co-resident grid capacity for k_persist_serve (the crews wait on
// each other, so the launch MUST fit on the chip at once)
extern "C" int gofr_persist_grid(int* blocks_out) {
    int per_cu = 0;
    hipError_t rc = hipOccupancyMaxActiveBlocksPerMultiprocessor(
        &per_cu, (const void*)k_persist_serve, BLOCK_THREADS, 0);
    if (rc) return (int)rc;
    int cus = 0;
    hipDeviceProp_t prop;
    if ((rc = hipGetDeviceProperties(&prop, 0))) return (int)rc;
    cus = prop.multiProcessorCount;
    *blocks_out = per_cu * cus;
    return 0;
}

extern "C" int gofr_persist_launch(const PersistKernArgs* a,
                                   void* stream, int nblocks) {
    hipLaunchKernelGGL(k_persist_serve, dim3(nblocks),
                       dim3(BLOCK_THREADS), 0, (hipStream_t)stream, *a);
    return (int)hipGetLastError();
}

// per-batch submit: ONE ingress SDMA (header+payload) then the go
// serial in the same FIFO — no kernel launches, no events
extern "C" int gofr_persist_submit(void* stream, void* d_ingress,
                                   const void* p_ingress,
                                   long long nbytes, void* d_go,
                                   const void* p_serial) {
    hipError_t rc = hipMemcpyAsync(d_ingress, p_ingress, (size_t)nbytes,
                                   hipMemcpyHostToDevice,
                                   (hipStream_t)stream);
    if (rc) return (int)rc;
    return (int)hipMemcpyAsync(d_go, p_serial, 8,
                               hipMemcpyHostToDevice,
                               (hipStream_t)stream);
}

extern "C" int gofr_launch_persist_cycle(
        void* stream, const void* d_go, void* p_done, const void* d_in,
        void* p_out, long long nbytes, int nbatch, int do_work,
        void* d_barrier, int nblocks) {
    hipLaunchKernelGGL(k_persist_cycle, dim3(nblocks), dim3(BLOCK_THREADS),
                       0, (hipStream_t)stream,
                       (const unsigned long long*)d_go,
                       (volatile unsigned long long*)p_done,
                       (const uint8_t*)d_in, (uint8_t*)p_out,
                       (long)nbytes, nbatch, do_work, (int*)d_barrier);
    return (int)hipGetLastError();
}
