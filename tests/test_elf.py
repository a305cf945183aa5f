"""ELF reader + FileID + procmaps unit tests."""

import os
import subprocess

import pytest

from parca_agent_amd.elf import (
    ELFFile,
    SymbolIndex,
    file_id,
    file_id_from_bytes,
)
from parca_agent_amd.procmaps import (
    ExecutableCache,
    Mapping,
    Process,
    ProcessTable,
    read_cgroup,
    read_comm,
)


@pytest.fixture(scope="module")
def sample_binary(tmp_path_factory):
    d = tmp_path_factory.mktemp("elf")
    src = d / "x.c"
    src.write_text(
        "int exported_fn(int x){return x*2;}\n"
        "static int hidden_fn(int x){return x+1;}\n"
        "int main(void){return exported_fn(hidden_fn(1));}\n")
    out = d / "x"
    subprocess.run(["gcc", "-g", str(src), "-o", str(out)], check=True)
    return str(out)


def test_parse_headers_and_sections(sample_binary):
    with ELFFile.open(sample_binary) as elf:
        assert elf.e_machine == 62  # EM_X86_64
        names = {s.name for s in elf.sections}
        assert ".text" in names
        assert ".symtab" in names
        assert elf.segments
        assert any(seg.p_type == 1 for seg in elf.segments)  # PT_LOAD


def test_build_id(sample_binary):
    with ELFFile.open(sample_binary) as elf:
        bid = elf.build_id()
    # Ubuntu gcc links with --build-id by default.
    assert bid is None or (len(bid) >= 16 and
                           all(c in "0123456789abcdef" for c in bid))


def test_symbols_and_index(sample_binary):
    with ELFFile.open(sample_binary) as elf:
        syms = elf.symbols()
        idx = SymbolIndex(syms)
    names = {s.name for s in syms}
    assert {"exported_fn", "hidden_fn", "main"} <= names
    target = next(s for s in syms if s.name == "exported_fn")
    assert idx.lookup(target.value).name == "exported_fn"
    assert idx.lookup(target.value + 1).name == "exported_fn"


def test_vaddr_file_offset_roundtrip(sample_binary):
    with ELFFile.open(sample_binary) as elf:
        text = elf.section(".text")
        off = elf.file_offset_for_vaddr(text.addr)
        assert off == text.offset
        assert elf.vaddr_for_file_offset(off) == text.addr


def test_stripped_and_debug_detection(sample_binary, tmp_path):
    with ELFFile.open(sample_binary) as elf:
        assert not elf.is_stripped()
        assert elf.has_debug_info()
    stripped = tmp_path / "stripped"
    subprocess.run(["strip", "-o", str(stripped), sample_binary], check=True)
    with ELFFile.open(str(stripped)) as elf:
        assert elf.is_stripped()


def test_file_id_stability(sample_binary, tmp_path):
    a = file_id(sample_binary)
    assert a == file_id(sample_binary)
    assert len(a) == 32
    copy = tmp_path / "copy"
    copy.write_bytes(open(sample_binary, "rb").read())
    assert file_id(str(copy)) == a  # content-derived
    other = tmp_path / "other"
    other.write_bytes(b"\x7fELF" + b"different" * 100)
    assert file_id(str(other)) != a


def test_file_id_from_bytes_large():
    data = bytes(range(256)) * 100  # 25600 bytes
    fid = file_id_from_bytes(data)
    # Only head/tail + size matter: middle mutation is invisible by
    # design (cheap identity for huge binaries)...
    mutated = bytearray(data)
    mutated[10000] ^= 0xFF
    assert file_id_from_bytes(bytes(mutated)) == fid
    # ...but size or edge changes are not.
    assert file_id_from_bytes(data + b"x") != fid
    assert file_id_from_bytes(b"y" + data[1:]) != fid


def test_not_an_elf(tmp_path):
    bad = tmp_path / "bad"
    bad.write_bytes(b"#!/bin/sh\necho hi\n")
    with pytest.raises(ValueError):
        ELFFile.open(str(bad))


# -- procmaps --------------------------------------------------------------


def test_process_mapping_resolution():
    p = Process(pid=1)
    p.add_mapping(Mapping(start=0x1000, end=0x2000, file_offset=0,
                          path="/a"))
    p.add_mapping(Mapping(start=0x3000, end=0x4000, file_offset=0x1000,
                          path="/b"))
    assert p.find_mapping(0x1000).path == "/a"
    assert p.find_mapping(0x1FFF).path == "/a"
    assert p.find_mapping(0x2000) is None
    assert p.find_mapping(0x3500).path == "/b"
    # Remap over an existing range replaces it.
    p.add_mapping(Mapping(start=0x800, end=0x2800, file_offset=0,
                          path="/c"))
    assert p.find_mapping(0x1000).path == "/c"


def test_process_table_events():
    class Ev:
        def __init__(self, kind, pid, tid=0, **kw):
            self.kind = kind
            self.pid = pid
            self.tid = tid or pid
            self.__dict__.update(kw)

    t = ProcessTable()
    t.handle_proc_event(Ev(0, 10, comm="worker"))
    t.handle_proc_event(Ev(1, 10, addr=0x1000, len=0x1000, pgoff=0,
                           filename="/bin/x", prot=4))
    p = t.get(10, create=False)
    assert p.comm == "worker"
    assert p.find_mapping(0x1800).path == "/bin/x"
    # Anonymous/non-file exec mappings are ignored.
    t.handle_proc_event(Ev(1, 10, addr=0x9000, len=0x1000, pgoff=0,
                           filename="//anon", prot=4))
    assert p.find_mapping(0x9000) is None
    # Thread exit does not drop the process; process exit does.
    t.handle_proc_event(Ev(2, 10, tid=11))
    assert t.get(10, create=False) is not None
    t.handle_proc_event(Ev(2, 10, tid=10))
    assert t.get(10, create=False) is None


def test_ensure_maps_self():
    t = ProcessTable()
    p = t.ensure_maps(os.getpid())
    assert p is not None and p.maps_loaded
    assert any("python" in m.path for m in p.mappings)


def test_executable_cache(sample_binary):
    c = ExecutableCache(load_symbols=True)
    info = c.get(sample_binary)
    assert info.file_id
    assert info.load_segments
    assert info.symbols is not None and len(info.symbols) > 0
    # Identity-mapped at its ELF vaddrs: normalize is the identity for a
    # mapping whose start equals the segment vaddr.
    seg_off, seg_vaddr, _ = info.load_segments[0]
    assert info.normalize(seg_vaddr + 10, seg_vaddr, seg_off) == \
        seg_vaddr + 10
    assert c.get(sample_binary) is info  # cached


def test_read_proc_helpers():
    assert read_comm(os.getpid())
    cg = read_cgroup(os.getpid())
    assert cg is None or cg.startswith("/")
