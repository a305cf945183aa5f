"""HotSpot JVM interpreter unwinder tests against synthetic struct
images for both Symbol-layout eras (JDK 8/11: length at +0, body at
+8; JDK 15+: hash word first, length at +4, body at +6). No JVM ships
in this container, so — like the CPython 3.11+/Ruby eras — the
invariant-driven calibrator is proven against faithful mocks
(VERDICT.md next#3)."""

import struct

import pytest

from parca_agent_amd.interp.jvm import (
    METHOD_SLOT,
    JvmCalibrator,
    JvmOffsets,
    JvmProcess,
    JvmUnwinder,
)
from parca_agent_amd.interp.python import RemoteMem
from tests.test_python_unwinder import Image

SP = 0x7FFE_0000_0000


def build_jvm_image(era):
    img = Image()
    libjvm = img.alloc(4096)  # stands in for libjvm.so's mapped range
    jvm_range = (libjvm, libjvm + 4096)
    method_vtbl = libjvm + 0x100
    cp_vtbl = libjvm + 0x200
    klass_vtbl = libjvm + 0x300

    def mk_symbol(s):
        addr = img.alloc(512)
        data = s.encode()
        if era == "17":
            img.w32(addr, 0xBEEF)                    # hash+refcount
            struct.pack_into("<H", img.buf, addr - img.BASE + 4,
                             len(data))
            img.wbytes(addr + 6, data)
        else:
            struct.pack_into("<H", img.buf, addr - img.BASE, len(data))
            img.wbytes(addr + 8, data)
        return addr

    cls_sym = mk_symbol("com/acme/Svc")
    klass = img.alloc(64)
    img.w64(klass, klass_vtbl)
    img.w64(klass + 16, cls_sym)

    cp = img.alloc(64 + 16 * 8)
    img.w64(cp, cp_vtbl)
    img.w64(cp + 24, klass)                          # _pool_holder
    img.w64(cp + 64 + 5 * 8, mk_symbol("doWork"))    # name idx 5
    img.w64(cp + 64 + 6 * 8, mk_symbol("(I)V"))      # signature idx 6
    img.w64(cp + 64 + 9 * 8, mk_symbol("helper"))    # name idx 9
    img.w64(cp + 64 + 10 * 8, mk_symbol("()J"))      # signature idx 10

    def mk_method(name_idx, sig_idx):
        cm = img.alloc(128)
        img.w64(cm + 8, cp)                          # _constants
        struct.pack_into("<H", img.buf, cm - img.BASE + 42, name_idx)
        struct.pack_into("<H", img.buf, cm - img.BASE + 44, sig_idx)
        m = img.alloc(64)
        img.w64(m, method_vtbl)
        img.w64(m + 16, cm)                          # _constMethod
        return m

    m1 = mk_method(5, 6)
    m2 = mk_method(9, 10)

    # Fake captured stack: leaf interpreter frame -> caller interpreter
    # frame -> native frame (garbage method slot) -> end.
    stack = bytearray(4096)

    def sw(addr, val):
        struct.pack_into("<Q", stack, addr - SP, val)

    fp0, fp1, fp2 = SP + 256, SP + 512, SP + 1024
    sw(fp0 + METHOD_SLOT, m1)
    sw(fp0, fp1)
    sw(fp1 + METHOD_SLOT, m2)
    sw(fp1, fp2)
    sw(fp2 + METHOD_SLOT, 0x1234)  # not a pointer
    sw(fp2, 0)

    return img, jvm_range, bytes(stack), fp0, m1


@pytest.mark.parametrize("era", ["8", "17"])
def test_jvm_calibration_and_walk(era):
    img, jvm_range, stack, fp0, m1 = build_jvm_image(era)
    mem = RemoteMem(img.read)

    cal = JvmCalibrator(mem, jvm_range)
    off = cal.run(m1)
    assert off is not None, f"jvm calibration failed for era {era}"
    assert off.complete()
    assert off.const_method == 16
    assert off.constants == 8
    assert off.name_index == 42
    assert off.cp_base == 64
    assert (off.sym_len, off.sym_body) == ((4, 6) if era == "17"
                                           else (0, 8))
    assert off.pool_holder == 24
    assert off.klass_name == 16

    u = JvmUnwinder()
    info = JvmProcess(pid=7777, jvm_lo=jvm_range[0], jvm_hi=jvm_range[1],
                      mem=mem)
    u._procs.put(7777, info)
    frames = u.stack_for(7777, fp0, SP, stack)
    names = [f.function_name for f in frames]
    assert names == ["com.acme.Svc.doWork", "com.acme.Svc.helper"]
    assert u.calibrations == 1
    # second walk uses the calibrated offsets without recalibrating
    frames2 = u.stack_for(7777, fp0, SP, stack)
    assert [f.function_name for f in frames2] == names
    assert u.calibrations == 1


def test_jvm_non_jvm_process_skipped():
    u = JvmUnwinder()
    import os

    frames = u.stack_for(os.getpid(), 0x1000, 0x800, b"\x00" * 64)
    assert frames == []  # no libjvm mapping in this process
