"""Debug-only ELF extraction (`objcopy --only-keep-debug` equivalent).

The reference streams a rewritten ELF keeping PT_NOTE program headers and
debug-relevant sections while nullifying the rest before upload
(reference: reporter/elfwriter/extract.go:14-39,
nullifying_elfwriter.go:23-62). This implementation rebuilds the file:
kept sections keep their bytes, stripped ALLOC sections become SHT_NOBITS
(headers preserved so addresses still resolve), and only PT_NOTE segments
survive in the program header table.
"""

from __future__ import annotations

import struct
import zlib
from typing import BinaryIO

from ..elf import ELFFile, PT_NOTE, SHT_NOBITS, SHT_NOTE

SHF_COMPRESSED = 0x800
ELFCOMPRESS_ZLIB = 1
SHF_ALLOC = 0x2

_KEEP_PREFIXES = (
    ".debug_", ".zdebug_", ".note", ".comment", ".gnu_debuglink",
    ".gosymtab", ".gopclntab",
)
_KEEP_EXACT = {".symtab", ".strtab", ".shstrtab", ".plt", ".plt.got",
               ".dynsym", ".dynstr"}
_NULL_TYPES = {8}  # already NOBITS


def _keep_section(name: str, sh_type: int) -> bool:
    if sh_type == SHT_NOTE:
        return True
    if name in _KEEP_EXACT:
        return True
    return any(name.startswith(p) for p in _KEEP_PREFIXES)


def _maybe_compress(name: str, flags: int, data: bytes,
                    enabled: bool):
    """SHF_COMPRESSED/zlib transform for DWARF sections (reference
    --debuginfo-compress, flags.go:378): Elf64_Chdr{ELFCOMPRESS_ZLIB,
    size, align} + deflate stream. Non-alloc .debug_* only; skipped
    when compression does not help."""
    if not enabled or not name.startswith(".debug_") or \
            (flags & (SHF_COMPRESSED | SHF_ALLOC)) or len(data) < 256:
        return flags, data
    packed = struct.pack("<IIQQ", ELFCOMPRESS_ZLIB, 0, len(data), 8) + \
        zlib.compress(data, 6)
    if len(packed) >= len(data):
        return flags, data
    return flags | SHF_COMPRESSED, packed


def only_keep_debug(src_path: str, dst: BinaryIO,
                    compress: bool = False) -> int:
    """Write the debug-only ELF for src_path into dst; returns bytes
    written."""
    with ELFFile.open(src_path) as elf:
        assert elf._fh is not None
        fh = elf._fh
        fh.seek(0)
        ehdr = bytearray(fh.read(64))

        note_segments = [s for s in elf.segments if s.p_type == PT_NOTE]

        # Layout: ehdr | phdrs | section data (kept only) | shdrs
        phoff = 64
        phentsize = 56
        phnum = len(note_segments)
        data_off = phoff + phentsize * phnum

        out_sections = []  # (name, type, flags, addr, offset, size, link,
        #                     info, align, entsize, data|None)
        cursor = data_off
        for s in elf.sections:
            keep = _keep_section(s.name, s.sh_type)
            if s.sh_type == 0:  # SHT_NULL
                out_sections.append((s, 0, None, s.flags))
                continue
            if keep and s.sh_type != SHT_NOBITS:
                data = elf.section_data(s)
                new_flags, data = _maybe_compress(
                    s.name, s.flags, data, compress)
                align = max(int(s.addralign) or 1, 1)
                if new_flags & SHF_COMPRESSED:
                    align = max(align, 8)
                cursor = (cursor + align - 1) & ~(align - 1)
                out_sections.append((s, cursor, data, new_flags))
                cursor += len(data)
            else:
                # Nullified: keep the header, drop the bytes.
                out_sections.append((s, 0, None, s.flags))

        shoff = (cursor + 7) & ~7
        shentsize = 64
        shnum = len(out_sections)

        # Patch the ELF header.
        struct.pack_into("<Q", ehdr, 0x20, phoff)
        struct.pack_into("<Q", ehdr, 0x28, shoff)
        struct.pack_into("<H", ehdr, 0x36, phentsize)
        struct.pack_into("<H", ehdr, 0x38, phnum)
        struct.pack_into("<H", ehdr, 0x3A, shentsize)
        struct.pack_into("<H", ehdr, 0x3C, shnum)
        # shstrndx unchanged (same section order).

        dst.write(bytes(ehdr))
        # PT_NOTE program headers: point at the kept copy of the note
        # bytes when the matching section was kept, else keep vaddr info
        # with zero file size.
        for seg in note_segments:
            new_off = 0
            new_filesz = 0
            for (s, off, data, _fl) in out_sections:
                if data is not None and s.addr == seg.vaddr and \
                        len(data) >= seg.filesz:
                    new_off = off
                    new_filesz = seg.filesz
                    break
            dst.write(struct.pack(
                "<IIQQQQQQ", seg.p_type, seg.flags, new_off, seg.vaddr,
                seg.vaddr, new_filesz, seg.memsz, seg.align))

        # Section data.
        pos = data_off
        for (s, off, data, _fl) in out_sections:
            if data is None:
                continue
            if off > pos:
                dst.write(b"\x00" * (off - pos))
                pos = off
            dst.write(data)
            pos += len(data)
        if shoff > pos:
            dst.write(b"\x00" * (shoff - pos))
            pos = shoff

        # Section headers: recompute name offsets are unchanged (same
        # shstrtab content), only offsets/types change.
        name_offsets = _name_offsets(elf)
        for (s, off, data, fl) in out_sections:
            if s.sh_type == 0:
                dst.write(b"\x00" * 64)
                pos += 64
                continue
            if data is not None:
                sh_type = s.sh_type
                sh_off = off
                sh_size = len(data)
            else:
                sh_type = SHT_NOBITS
                sh_off = 0
                sh_size = s.size
            align = s.addralign
            if fl & SHF_COMPRESSED:
                align = max(int(align) or 1, 8)
            dst.write(struct.pack(
                "<IIQQQQIIQQ", name_offsets.get(s.name, 0), sh_type,
                fl, s.addr, sh_off, sh_size, s.link, s.info,
                align, s.entsize))
            pos += 64
        return pos


def _name_offsets(elf: ELFFile):
    """Map section name -> offset in the original shstrtab."""
    shstr = None
    for s in elf.sections:
        if s.name == ".shstrtab":
            shstr = elf.section_data(s)
            break
    out = {}
    if shstr is None:
        return out
    for s in elf.sections:
        if not s.name:
            continue
        idx = shstr.find(s.name.encode() + b"\x00")
        if idx >= 0:
            out[s.name] = idx
    return out
