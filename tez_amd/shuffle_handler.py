"""HTTP ShuffleHandler compatibility — serve this
engine's spill files to ordered-shuffle fetchers with the reference wire
protocol (tez-plugins/tez-aux-services ShuffleHandler.java):

  GET /mapOutput?job=<jobId>&dag=<dagId>&reduce=<r|r0-r1>&map=<id1,id2,...>
  required request headers: name: mapreduce, version: 1.0.0
    (ShuffleHandler.java:1030-1036; ShuffleHeader.java:41-44)
  response body, per mapId (getContentLength, ShuffleHandler.java:1413-1433):
    vint(reduceCount) then, per reduce in range:
    ShuffleHeader{mapId, partLength, rawLength, reduce} + partLength segment
    bytes of file.out (via file.out.index)

Files are resolved under <local_dir>/output/<mapId>/file.out[.index] — the
layout tzs_sorter_write_files materializes (TezTaskOutputFiles.java:52-69).
The fetch client below is the FetcherOrderedGrouped.copyFromHost equivalent
(FetcherOrderedGrouped.java:265-337,456-601) for this ordered path.
"""
import threading
import urllib.request
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import urlparse, parse_qs

from . import ifile

HTTP_HEADER_NAME = "name"
DEFAULT_HTTP_HEADER_NAME = "mapreduce"
HTTP_HEADER_VERSION = "version"
DEFAULT_HTTP_HEADER_VERSION = "1.0.0"


def decode_index(blob: bytes):
    """file.out.index: 24B big-endian triples + 8B CRC32-as-long trailer
    (TezSpillRecord.java:112-147)."""
    import zlib
    body = blob[:-8]
    if zlib.crc32(body) != int.from_bytes(blob[-8:], "big"):
        raise ValueError("index CRC mismatch")
    return [(int.from_bytes(body[i:i + 8], "big"),
             int.from_bytes(body[i + 8:i + 16], "big"),
             int.from_bytes(body[i + 16:i + 24], "big"))
            for i in range(0, len(body), 24)]


def encode_shuffle_header(map_id: str, part_len: int, raw_len: int, reduce: int):
    mid = map_id.encode()
    return (ifile.vint_write(len(mid)) + mid + ifile.vint_write(part_len)
            + ifile.vint_write(raw_len) + ifile.vint_write(reduce))


def decode_shuffle_header(buf, pos):
    n, pos = ifile.vint_read(buf, pos)
    mid = bytes(buf[pos:pos + n]).decode()
    pos += n
    clen, pos = ifile.vint_read(buf, pos)
    rlen, pos = ifile.vint_read(buf, pos)
    reduce, pos = ifile.vint_read(buf, pos)
    return mid, clen, rlen, reduce, pos


class ShuffleHandlerServer:
    """Threaded HTTP server over a local_dir of map outputs."""

    def __init__(self, local_dir, host="127.0.0.1", port=0):
        self.local_dir = local_dir
        outer = self

        class H(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):
                pass

            def do_GET(self):
                u = urlparse(self.path)
                if u.path != "/mapOutput":
                    return self._err(404, "not found")
                # shuffle version check (ShuffleHandler.java:1030-1036)
                if (self.headers.get(HTTP_HEADER_NAME) != DEFAULT_HTTP_HEADER_NAME
                        or self.headers.get(HTTP_HEADER_VERSION)
                        != DEFAULT_HTTP_HEADER_VERSION):
                    return self._err(400, "Incompatible shuffle request version")
                q = parse_qs(u.query)
                if not all(k in q for k in ("job", "dag", "map", "reduce")):
                    return self._err(400, "Required param job, dag, map and reduce")
                maps = q["map"][0].split(",")
                rng = q["reduce"][0]
                if "-" in rng:
                    r0, r1 = (int(x) for x in rng.split("-"))
                else:
                    r0 = r1 = int(rng)
                chunks = []
                try:
                    for map_id in maps:
                        # a client-supplied id must not escape local_dir
                        # (the reference resolves ids through the NM's
                        # path allowlist)
                        if "/" in map_id or "\\" in map_id or ".." in map_id:
                            return self._err(400, "Invalid map id")
                        base = f"{outer.local_dir}/output/{map_id}/file.out"
                        idx = decode_index(open(base + ".index", "rb").read())
                        chunks.append(ifile.vint_write(r1 - r0 + 1))
                        with open(base, "rb") as f:
                            for r in range(r0, r1 + 1):
                                st, raw, cl = idx[r]
                                chunks.append(
                                    encode_shuffle_header(map_id, cl, raw, r))
                                f.seek(st)
                                chunks.append(f.read(cl))
                except FileNotFoundError:
                    return self._err(404, "map output not found")
                body = b"".join(chunks)
                self.send_response(200)
                self.send_header("Content-Length", str(len(body)))
                self.send_header(HTTP_HEADER_NAME, DEFAULT_HTTP_HEADER_NAME)
                self.send_header(HTTP_HEADER_VERSION, DEFAULT_HTTP_HEADER_VERSION)
                self.end_headers()
                self.wfile.write(body)

            def _err(self, code, msg):
                b = msg.encode()
                self.send_response(code)
                self.send_header("Content-Length", str(len(b)))
                self.end_headers()
                self.wfile.write(b)

        self._srv = ThreadingHTTPServer((host, port), H)
        self.port = self._srv.server_address[1]
        self._thread = threading.Thread(target=self._srv.serve_forever, daemon=True)

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._srv.shutdown()
        self._srv.server_close()


def fetch_map_outputs(host, port, job, dag, reduce, map_ids, timeout=30):
    """FetcherOrderedGrouped equivalent: returns
    [(map_id, reduce, raw_len, segment_bytes)] for the given partition."""
    url = (f"http://{host}:{port}/mapOutput?job={job}&dag={dag}"
           f"&reduce={reduce}&map={','.join(map_ids)}")
    req = urllib.request.Request(url, headers={
        HTTP_HEADER_NAME: DEFAULT_HTTP_HEADER_NAME,
        HTTP_HEADER_VERSION: DEFAULT_HTTP_HEADER_VERSION,
    })
    body = urllib.request.urlopen(req, timeout=timeout).read()
    out = []
    pos = 0
    for _ in map_ids:
        cnt, pos = ifile.vint_read(body, pos)
        for _ in range(cnt):
            mid, clen, rlen, red, pos = decode_shuffle_header(body, pos)
            seg = body[pos:pos + clen]
            pos += clen
            out.append((mid, red, rlen, seg))
    assert pos == len(body)
    return out
