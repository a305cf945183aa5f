"""CRuby unwinder tests against synthetic struct images.

No ruby runtime ships in this container, so — like the CPython 3.11+
eras in test_python_unwinder.py — the invariant-driven calibrator is
exercised against faithful mocks of the CRuby 3.1 (56-byte control
frames, flag-packed embedded strings) and 3.2+ (long-len embedded
strings) layouts. The live template for this pattern is the perl/node
tests."""

import struct

import pytest

from parca_agent_amd.interp.python import RemoteMem
from parca_agent_amd.interp.ruby import (
    RubyCalibrator,
    RubyOffsets,
    RubyProcess,
    RubyStringReader,
    RubyUnwinder,
)
from tests.test_python_unwinder import Image

RSTRING_NOEMBED = 1 << 13


def build_ruby_image(era):
    """era: '3.1' (stride 56, embed len in flags) | '3.2' (stride 56,
    long embed len) — returns (mem, ractor, cstring, expected)."""
    img = Image()
    cstring = img.alloc(64)

    def mk_str(s, heap=False):
        if heap:
            buf = img.alloc(len(s) + 1)
            img.wbytes(buf, s.encode() + b"\x00")
            addr = img.alloc(40)
            img.w64(addr, RSTRING_NOEMBED)
            img.w64(addr + 8, cstring)
            img.w64(addr + 16, len(s))
            img.w64(addr + 24, buf)
            return addr
        addr = img.alloc(48)
        img.w64(addr + 8, cstring)
        if era == "3.1":
            img.w64(addr, (len(s) & 0x1F) << 15)  # RSTRING_EMBED_LEN
            img.wbytes(addr + 16, s.encode() + b"\x00")
        else:
            img.w64(addr, 0)
            img.w64(addr + 16, len(s))
            img.wbytes(addr + 24, s.encode() + b"\x00")
        return addr

    path = mk_str("/app/worker.rb", heap=True)
    names = ["handle_job", "dispatch", "<main>"]

    def mk_iseq(name):
        body = img.alloc(256)
        img.w64(body + 64, mk_str(name))
        img.w64(body + 72, path)
        iseq = img.alloc(40)
        img.w64(iseq + 16, body)  # iseq->body (3.x offset)
        return iseq

    iseqs = [mk_iseq(n) for n in names]

    stride = 56
    n_slots = 1024
    vm_stack = img.alloc(n_slots * 8)
    stack_end = vm_stack + n_slots * 8
    # control frames grow downward: leaf cfp sits lowest; walking is
    # cfp -> stack_end. Lay 3 iseq frames + 1 C frame.
    n_frames = 4
    cfp0 = stack_end - n_frames * stride
    for i in range(n_frames):
        f = cfp0 + i * stride
        if i < 3:
            img.w64(f + 0, vm_stack + 8)       # pc (plausible nonzero)
            img.w64(f + 8, vm_stack + 64 * (i + 1))  # sp inside stack
            img.w64(f + 16, iseqs[i])          # iseq
        else:
            img.w64(f + 0, 0)                  # C frame: pc NULL
            img.w64(f + 8, vm_stack + 8)
        img.w64(f + 32, vm_stack + 64 * (i + 1))   # ep inside stack

    ec = img.alloc(512)
    img.w64(ec + 0, vm_stack)
    img.w64(ec + 8, n_slots)
    img.w64(ec + 16, cfp0)

    ractor = img.alloc(4096)
    img.w64(ractor + 16, 0xDEAD)    # decoy non-pointer
    img.w64(ractor + 40, cstring)   # decoy pointer (fails ec check)
    img.w64(ractor + 72, ec)        # running_ec

    return RemoteMem(img.read), ractor, cstring, names


@pytest.mark.parametrize("era", ["3.1", "3.2"])
def test_ruby_calibration_and_walk(era):
    mem, ractor, cstring, names = build_ruby_image(era)
    strings = RubyStringReader(mem, cstring)
    off = RubyCalibrator(mem, ractor, strings).run()
    assert off is not None, f"ruby calibration failed for {era}"
    assert off.complete()
    assert off.ec_in_ractor == 72
    assert off.cfp_stride == 56
    assert off.iseq_body == 16
    assert off.body_label == 64
    assert off.body_path == 72

    u = RubyUnwinder()
    info = RubyProcess(pid=4321, ractor_ptr_addr=0, cstring_addr=cstring,
                       offsets=off, mem=mem)
    info.ractor = ractor
    u._procs.put(4321, info)
    frames = u.stack_for(4321, 4321)
    assert [f.function_name for f in frames] == names
    assert frames[0].source_file == "/app/worker.rb"
    assert u.stacks_resolved == 1
    # non-main thread: skipped (single-ractor walker)
    assert u.stack_for(4321, 9999) == []


def test_ruby_string_reader_forms():
    img = Image()
    cstring = img.alloc(64)
    mem = RemoteMem(img.read)
    r = RubyStringReader(mem, cstring)

    heap_buf = img.alloc(32)
    img.wbytes(heap_buf, b"heap_string_value")
    heap = img.alloc(40)
    img.w64(heap, RSTRING_NOEMBED)
    img.w64(heap + 8, cstring)
    img.w64(heap + 16, len("heap_string_value"))
    img.w64(heap + 24, heap_buf)
    assert r.is_string(heap)
    assert r.read(heap) == "heap_string_value"

    other = img.alloc(40)
    img.w64(other + 8, cstring + 8)  # wrong klass
    assert not r.is_string(other)
    assert r.read(0) == ""
