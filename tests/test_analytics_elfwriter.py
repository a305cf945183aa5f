"""Coverage for the analytics remote-write encoder, the debug-only ELF
extractor, and the standalone frame resolver (reference capabilities:
analytics/, reporter/elfwriter/ OnlyKeepDebug, symbolization)."""

import http.server
import io
import os
import subprocess
import threading

import pytest

from parca_agent_amd.analytics.sender import (
    AnalyticsSender,
    encode_remote_write,
)
from parca_agent_amd.elf import ELFFile
from parca_agent_amd.pprof.proto import iter_fields
from parca_agent_amd.reporter.elfwriter import only_keep_debug


def test_remote_write_encoding_roundtrip():
    payload = encode_remote_write([
        ({"__name__": "parca_agent_info", "version": "v1"}, 1.0, 1234),
        ({"__name__": "parca_agent_cpu_cores"}, 256.0, 1234),
    ])
    series = [v for n, _, v in iter_fields(payload) if n == 1]
    assert len(series) == 2
    labels = []
    for n, wt, v in iter_fields(series[0]):
        if n == 1:
            kv = {fn: fv for fn, _, fv in iter_fields(v)}
            labels.append((kv[1].decode(), kv[2].decode()))
    assert ("__name__", "parca_agent_info") in labels
    assert ("version", "v1") in labels


def test_analytics_sender_posts(tmp_path):
    hits = []

    class H(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            body = self.rfile.read(int(self.headers["Content-Length"]))
            hits.append((self.path, dict(self.headers), body))
            self.send_response(200)
            self.end_headers()

        def log_message(self, *a):
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), H)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        s = AnalyticsSender(
            "vtest", endpoint=f"http://127.0.0.1:{srv.server_port}/w")
        s.send_once()
    finally:
        srv.shutdown()
    assert len(hits) == 1
    path, headers, body = hits[0]
    assert headers.get("Content-Encoding") == "snappy"
    # Decode the literal-only snappy stream back to the protobuf.
    from parca_agent_amd.analytics.sender import snappy_block

    def snappy_decode(buf):
        ln = 0
        shift = 0
        i = 0
        while True:
            b = buf[i]
            i += 1
            ln |= (b & 0x7F) << shift
            shift += 7
            if not b & 0x80:
                break
        out = bytearray()
        while i < len(buf):
            tag = buf[i]
            i += 1
            assert tag & 3 == 0, "non-literal element"
            l6 = tag >> 2
            if l6 < 60:
                n = l6 + 1
            elif l6 == 60:
                n = buf[i] + 1
                i += 1
            else:
                n = int.from_bytes(buf[i:i + 2], "little") + 1
                i += 2
            out += buf[i:i + n]
            i += n
        assert len(out) == ln
        return bytes(out)

    raw = snappy_decode(body)
    assert snappy_block(raw) == body
    names = [v for n, _, v in iter_fields(raw) if n == 1]
    assert len(names) == 2  # two timeseries


@pytest.fixture
def debug_binary(tmp_path):
    src = tmp_path / "x.c"
    src.write_text("int helper(int v){return v*3;}\n"
                   "int main(void){return helper(2);}\n")
    out = tmp_path / "x"
    subprocess.run(["gcc", "-g", "-O0", str(src), "-o", str(out)],
                   check=True)
    return str(out)


def test_only_keep_debug(debug_binary, tmp_path):
    dst_path = tmp_path / "x.debug"
    with open(dst_path, "wb") as fh:
        n = only_keep_debug(debug_binary, fh)
    assert n == os.path.getsize(dst_path) > 0
    assert n < os.path.getsize(debug_binary) * 2
    with ELFFile.open(str(dst_path)) as elf:
        names = {s.name for s in elf.sections}
        assert any(nm.startswith(".debug_") for nm in names)
        assert ".symtab" in names
        # Text content must be dropped (header may remain, bytes gone).
        text = elf.section(".text")
        assert text is None or text.sh_type == 8 or \
            elf.section_data(text) == b"" or \
            set(elf.section_data(text)) == {0}
        syms = {s.name for s in elf.symbols()}
        assert "helper" in syms


def test_frame_resolver(debug_binary):
    """Resolve an IP inside a mapped executable of a live process."""
    from parca_agent_amd.model import FrameType
    from parca_agent_amd.procmaps import ExecutableCache, ProcessTable
    from parca_agent_amd.symbolize import FrameResolver

    procs = ProcessTable()
    resolver = FrameResolver(procs, ExecutableCache(load_symbols=True))
    me = os.getpid()
    proc = procs.ensure_maps(me)
    exe = next(m for m in proc.mappings
               if m.path.startswith("/") and "python" in m.path)
    f = resolver.resolve(me, exe.start + 64)
    assert f.kind == FrameType.NATIVE
    assert f.mapping is not None and "python" in f.mapping.path
    unknown = resolver.resolve(me, 0x41)
    assert unknown.kind == FrameType.UNKNOWN
    assert resolver.no_mapping == 1


def test_only_keep_debug_compress(tmp_path):
    """--debuginfo-compress: .debug_* sections become SHF_COMPRESSED
    with an Elf64_Chdr + zlib stream that round-trips to the
    uncompressed extraction's bytes."""
    import io
    import struct
    import subprocess
    import zlib

    from parca_agent_amd.elf import ELFFile
    from parca_agent_amd.reporter.elfwriter import (SHF_COMPRESSED,
                                                    only_keep_debug)

    src = tmp_path / "t.c"
    src.write_text(
        "#include <stdio.h>\n#include <stdlib.h>\n"
        "struct point { double x, y; int tags[16]; };\n"
        "int add(int a, int b) { return a + b; }\n"
        "double norm(struct point *p) { return p->x * p->x + p->y; }\n"
        "int main(void) { struct point p = {1, 2, {0}};\n"
        "  printf(\"%f\\n\", norm(&p)); return add(1, 2); }\n")
    binary = tmp_path / "t"
    subprocess.run(["gcc", "-g", "-O0", str(src), "-o", str(binary)],
                   check=True)

    plain = io.BytesIO()
    only_keep_debug(str(binary), plain)
    packed = io.BytesIO()
    only_keep_debug(str(binary), packed, compress=True)
    assert len(packed.getvalue()) < len(plain.getvalue())

    def section(data, name):
        elf = ELFFile.from_bytes(data)
        s = elf.section(name)
        assert s is not None
        return s, elf.section_data(s)

    s_plain, d_plain = section(plain.getvalue(), ".debug_info")
    s_comp, d_comp = section(packed.getvalue(), ".debug_info")
    assert not (s_plain.flags & SHF_COMPRESSED)
    assert s_comp.flags & SHF_COMPRESSED
    ch_type, _res, ch_size, ch_align = struct.unpack_from("<IIQQ", d_comp)
    assert ch_type == 1  # ELFCOMPRESS_ZLIB
    assert ch_size == len(d_plain)
    assert ch_align == 8
    assert zlib.decompress(d_comp[24:]) == d_plain


def test_maybe_compress_edge_cases():
    import struct
    import zlib

    from parca_agent_amd.reporter.elfwriter import (SHF_ALLOC,
                                                    SHF_COMPRESSED,
                                                    _maybe_compress)

    big = b"A" * 4096
    # disabled / wrong name / alloc / tiny: pass through untouched
    assert _maybe_compress(".debug_info", 0, big, False) == (0, big)
    assert _maybe_compress(".text", 0, big, True) == (0, big)
    fl, d = _maybe_compress(".debug_info", SHF_ALLOC, big, True)
    assert (fl, d) == (SHF_ALLOC, big)
    assert _maybe_compress(".debug_info", 0, b"x" * 64, True) == \
        (0, b"x" * 64)
    # already compressed stays
    fl, d = _maybe_compress(".debug_info", SHF_COMPRESSED, big, True)
    assert fl == SHF_COMPRESSED and d == big
    # effective case round-trips
    fl, d = _maybe_compress(".debug_str", 0, big, True)
    assert fl & SHF_COMPRESSED
    ch_type, _r, ch_size, ch_align = struct.unpack_from("<IIQQ", d)
    assert (ch_type, ch_size, ch_align) == (1, len(big), 8)
    assert zlib.decompress(d[24:]) == big
    # incompressible data is left alone
    import os as _os
    rnd = _os.urandom(4096)
    assert _maybe_compress(".debug_info", 0, rnd, True) == (0, rnd)
