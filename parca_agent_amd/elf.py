"""Minimal ELF reader: build-id, symbols, section/program headers, FileID.

The reference gets ELF inspection from Go's debug/elf plus the ainur
compiler-detection library (reference: reporter/metadata/process.go:152-197,
reporter/elfwriter/). This is a from-scratch reader covering what the agent
needs: GNU build-id extraction, symbol tables for local symbolization,
section layout for debuginfo extraction, executable-address ranges, and the
stable FileID used to key executables across the fleet.

Also parses AMD GPU code objects (ELF, e_machine=EM_AMDGPU=224) — the
gfx950 analog of the reference's cubin parsing (parcagpu.go:231-277).
"""

from __future__ import annotations

import hashlib
import io
import os
import struct
from dataclasses import dataclass, field
from typing import BinaryIO, List, Optional, Tuple

EM_AMDGPU = 224
EM_X86_64 = 62
EM_AARCH64 = 183

PT_LOAD = 1
PT_NOTE = 4
SHT_SYMTAB = 2
SHT_DYNSYM = 11
SHT_NOTE = 7
SHT_NOBITS = 8
SHF_ALLOC = 0x2
SHF_EXECINSTR = 0x4
STT_FUNC = 2
NT_GNU_BUILD_ID = 3


@dataclass
class Section:
    name: str
    sh_type: int
    flags: int
    addr: int
    offset: int
    size: int
    link: int
    entsize: int
    addralign: int = 1
    info: int = 0


@dataclass
class Segment:
    p_type: int
    flags: int
    offset: int
    vaddr: int
    filesz: int
    memsz: int
    align: int = 1


@dataclass
class Symbol:
    name: str
    value: int
    size: int
    info: int

    @property
    def is_function(self) -> bool:
        return (self.info & 0xF) == STT_FUNC


@dataclass
class ELFFile:
    path: str
    e_type: int
    e_machine: int
    entry: int
    sections: List[Section] = field(default_factory=list)
    segments: List[Segment] = field(default_factory=list)
    _fh: Optional[BinaryIO] = None
    _symbols: Optional[List[Symbol]] = None

    # -- parsing -----------------------------------------------------------

    @classmethod
    def open(cls, path: str) -> "ELFFile":
        fh = open(path, "rb")
        try:
            return cls.from_file(fh, path=path)
        except Exception:
            fh.close()
            raise

    @classmethod
    def from_bytes(cls, data: bytes, path: str = "<memory>") -> "ELFFile":
        return cls.from_file(io.BytesIO(data), path=path)

    @classmethod
    def from_file(cls, fh: BinaryIO, path: str = "<file>") -> "ELFFile":
        fh.seek(0)
        ident = fh.read(16)
        if len(ident) < 16 or ident[:4] != b"\x7fELF":
            raise ValueError(f"{path}: not an ELF file")
        if ident[4] != 2:
            raise ValueError(f"{path}: only 64-bit ELF supported")
        if ident[5] != 1:
            raise ValueError(f"{path}: only little-endian ELF supported")
        (e_type, e_machine, _ver, entry, phoff, shoff, _flags, _ehsize,
         phentsize, phnum, shentsize, shnum, shstrndx) = struct.unpack(
            "<HHIQQQIHHHHHH", fh.read(48))

        elf = cls(path=path, e_type=e_type, e_machine=e_machine, entry=entry)
        elf._fh = fh

        fh.seek(phoff)
        for _ in range(phnum):
            raw = fh.read(phentsize)
            p_type, flags, offset, vaddr, _paddr, filesz, memsz, align = (
                struct.unpack("<IIQQQQQQ", raw[:56]))
            elf.segments.append(Segment(p_type, flags, offset, vaddr,
                                        filesz, memsz, align))

        raw_sections = []
        fh.seek(shoff)
        for _ in range(shnum):
            raw = fh.read(shentsize)
            (name_off, sh_type, flags, addr, offset, size, link, info,
             addralign, entsize) = struct.unpack("<IIQQQQIIQQ", raw[:64])
            raw_sections.append((name_off, sh_type, flags, addr, offset,
                                 size, link, info, addralign, entsize))

        shstrtab = b""
        if shstrndx < len(raw_sections):
            _, _, _, _, offset, size, _, _, _, _ = raw_sections[shstrndx]
            fh.seek(offset)
            shstrtab = fh.read(size)

        def cstr(tab: bytes, off: int) -> str:
            endi = tab.find(b"\x00", off)
            return tab[off:endi if endi >= 0 else None].decode(
                "utf-8", "replace")

        for (name_off, sh_type, flags, addr, offset, size, link, info,
             addralign, entsize) in raw_sections:
            elf.sections.append(Section(
                name=cstr(shstrtab, name_off), sh_type=sh_type, flags=flags,
                addr=addr, offset=offset, size=size, link=link,
                entsize=entsize, addralign=addralign, info=info))
        return elf

    def close(self) -> None:
        if self._fh is not None:
            self._fh.close()
            self._fh = None

    def __enter__(self) -> "ELFFile":
        return self

    def __exit__(self, *exc) -> None:
        self.close()

    # -- queries -----------------------------------------------------------

    def section(self, name: str) -> Optional[Section]:
        for s in self.sections:
            if s.name == name:
                return s
        return None

    def section_data(self, s: Section) -> bytes:
        if s.sh_type == SHT_NOBITS:
            return b""
        assert self._fh is not None
        self._fh.seek(s.offset)
        return self._fh.read(s.size)

    def build_id(self) -> Optional[str]:
        """GNU build-id from PT_NOTE segments / .note.gnu.build-id."""
        for s in self.sections:
            if s.sh_type == SHT_NOTE:
                bid = self._scan_notes(self.section_data(s))
                if bid:
                    return bid
        for seg in self.segments:
            if seg.p_type == PT_NOTE:
                assert self._fh is not None
                self._fh.seek(seg.offset)
                bid = self._scan_notes(self._fh.read(seg.filesz))
                if bid:
                    return bid
        return None

    @staticmethod
    def _scan_notes(data: bytes) -> Optional[str]:
        pos = 0
        while pos + 12 <= len(data):
            namesz, descsz, n_type = struct.unpack_from("<III", data, pos)
            pos += 12
            name = data[pos : pos + namesz].rstrip(b"\x00")
            pos += (namesz + 3) & ~3
            desc = data[pos : pos + descsz]
            pos += (descsz + 3) & ~3
            if name == b"GNU" and n_type == NT_GNU_BUILD_ID:
                return desc.hex()
        return None

    def symbols(self) -> List[Symbol]:
        """.symtab symbols, falling back to .dynsym. Cached."""
        if self._symbols is not None:
            return self._symbols
        out: List[Symbol] = []
        for want in (SHT_SYMTAB, SHT_DYNSYM):
            for s in self.sections:
                if s.sh_type != want:
                    continue
                strtab = self.section_data(self.sections[s.link])
                data = self.section_data(s)
                n = len(data) // 24
                for i in range(n):
                    name_off, info, _other, _shndx, value, size = (
                        struct.unpack_from("<IBBHQQ", data, i * 24))
                    if value == 0 and size == 0:
                        continue
                    endi = strtab.find(b"\x00", name_off)
                    name = strtab[name_off:endi if endi >= 0 else None].decode(
                        "utf-8", "replace")
                    if name:
                        out.append(Symbol(name, value, size, info))
            if out:
                break
        out.sort(key=lambda sym: sym.value)
        self._symbols = out
        return out

    def text_ranges(self) -> List[Tuple[int, int]]:
        """(vaddr, size) of executable ALLOC sections."""
        return [(s.addr, s.size) for s in self.sections
                if s.flags & SHF_EXECINSTR and s.flags & SHF_ALLOC]

    def is_stripped(self) -> bool:
        return self.section(".symtab") is None

    def has_debug_info(self) -> bool:
        return any(s.name.startswith(".debug_") for s in self.sections)

    def vaddr_for_file_offset(self, file_offset: int) -> Optional[int]:
        for seg in self.segments:
            if seg.p_type == PT_LOAD and \
               seg.offset <= file_offset < seg.offset + seg.filesz:
                return seg.vaddr + (file_offset - seg.offset)
        return None

    def file_offset_for_vaddr(self, vaddr: int) -> Optional[int]:
        for seg in self.segments:
            if seg.p_type == PT_LOAD and \
               seg.vaddr <= vaddr < seg.vaddr + seg.filesz:
                return seg.offset + (vaddr - seg.vaddr)
        return None


class SymbolIndex:
    """Sorted function-symbol index for address -> name lookups.

    Stored as parallel packed arrays, not Symbol objects: Tensile-scale
    code objects carry ~10^5 kernel symbols and per-object dataclasses
    cost ~250 B each — the arrays cut the index to addr/size u64 pairs
    plus the interned name strings."""

    __slots__ = ("_addrs", "_sizes", "_names")

    def __init__(self, symbols: List[Symbol]) -> None:
        import array

        funcs = [s for s in symbols if s.is_function]
        funcs.sort(key=lambda s: s.value)
        self._addrs = array.array("Q", (s.value for s in funcs))
        self._sizes = array.array("Q", (s.size for s in funcs))
        self._names = [s.name for s in funcs]

    def lookup(self, addr: int) -> Optional[Symbol]:
        import bisect

        i = bisect.bisect_right(self._addrs, addr) - 1
        if i < 0:
            return None
        value = self._addrs[i]
        size = self._sizes[i]
        if size and addr >= value + size:
            return None
        return Symbol(name=self._names[i], value=value, size=size,
                      info=STT_FUNC)

    def __len__(self) -> int:
        return len(self._names)


def file_id(path: str) -> str:
    """Stable 16-byte hex executable identity.

    Hash of (size, first 4 KiB, last 4 KiB) — cheap on large binaries but
    collision-resistant in practice; serves the role of the fork's
    libpf.FileID. Not byte-compatible with the fork's
    hash (we define our own fleet keyspace).
    """
    st = os.stat(path)
    h = hashlib.blake2b(digest_size=16)
    h.update(struct.pack("<Q", st.st_size))
    with open(path, "rb") as fh:
        h.update(fh.read(4096))
        if st.st_size > 4096:
            fh.seek(max(st.st_size - 4096, 4096))
            h.update(fh.read(4096))
    return h.hexdigest()


def file_id_from_bytes(data: bytes) -> str:
    h = hashlib.blake2b(digest_size=16)
    h.update(struct.pack("<Q", len(data)))
    h.update(data[:4096])
    if len(data) > 4096:
        h.update(data[max(len(data) - 4096, 4096):])
    return h.hexdigest()


def is_amdgpu_code_object(elf: ELFFile) -> bool:
    return elf.e_machine == EM_AMDGPU
