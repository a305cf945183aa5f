""".eh_frame unwinder tests: parse tables from real binaries and recover
stacks of frame-pointer-LESS binaries from DWARF-mode captures — the
case FP unwinding cannot handle (SURVEY.md §7 stage 4)."""

import subprocess
import textwrap
import time

import pytest

from parca_agent_amd.elf import ELFFile
from parca_agent_amd.procmaps import ExecutableCache, ProcessTable


def _native():
    from parca_agent_amd.native import sampler
    return sampler()


NOFP_C = textwrap.dedent("""
    #include <time.h>
    /* -fomit-frame-pointer: rbp is a scratch register; only .eh_frame
       can unwind these frames. Arg varies to defeat CSE. */
    __attribute__((noinline)) double dwarf_leaf(long n){
        double x=0; for(long i=0;i<n;i++) x+=i*0.5; return x; }
    __attribute__((noinline)) double dwarf_mid(long n){
        return dwarf_leaf(n)+1; }
    __attribute__((noinline)) double dwarf_top(long n){
        return dwarf_mid(n)+1; }
    int main(){
        struct timespec t0,t1; clock_gettime(CLOCK_MONOTONIC,&t0);
        double acc=0; long it=0;
        do { acc+=dwarf_top(1000000 + (++it & 7));
             clock_gettime(CLOCK_MONOTONIC,&t1); }
        while ((t1.tv_sec-t0.tv_sec)*1000000000L+(t1.tv_nsec-t0.tv_nsec)
               < 2500000000L);
        return acc > 1e308; }
""")


@pytest.fixture(scope="module")
def nofp_binary(tmp_path_factory):
    d = tmp_path_factory.mktemp("nofp")
    src = d / "nofp.c"
    src.write_text(NOFP_C)
    binary = d / "nofp"
    subprocess.run(
        ["gcc", "-O2", "-fomit-frame-pointer", str(src), "-o", str(binary)],
        check=True)
    return str(binary)


def test_parse_eh_frame_of_libc():
    """Table generation on a real-world big binary."""
    u = _native().Unwinder()
    libc = "/usr/lib/x86_64-linux-gnu/libc.so.6"
    with ELFFile.open(libc) as elf:
        sec = elf.section(".eh_frame")
        assert sec is not None
        data = elf.section_data(sec)
    t0 = time.perf_counter()
    mid = u.add_module_from_eh_frame(data, sec.addr)
    dt = time.perf_counter() - t0
    rows = u.module_rows(mid)
    assert rows > 10000, rows
    # Table generation must be fast enough for on-demand use.
    assert dt < 5.0, f"libc table took {dt:.1f}s"


def test_parse_eh_frame_of_own_binary(nofp_binary):
    u = _native().Unwinder()
    with ELFFile.open(nofp_binary) as elf:
        sec = elf.section(".eh_frame")
        data = elf.section_data(sec)
    mid = u.add_module_from_eh_frame(data, sec.addr)
    assert u.module_rows(mid) > 4


def _perf_available():
    try:
        s = _native().PerfSampler(freq=1, track_mmaps=False)
        s.start()
        s.stop()
        return True
    except Exception:
        return False


@pytest.mark.skipif(not _perf_available(), reason="perf unavailable")
def test_unwind_nofp_binary_end_to_end(nofp_binary):
    """Profile the FP-less binary in DWARF mode; FP unwinding yields
    nothing, the .eh_frame unwinder must recover dwarf_leaf<-dwarf_mid<-
    dwarf_top<-main chains."""
    from parca_agent_amd.cpu import CPUSamplerService
    from parca_agent_amd.cpu.unwind import EhFrameUnwinder
    from parca_agent_amd.elf import SymbolIndex
    from parca_agent_amd.reporter import Reporter

    class Dest:
        def __init__(self):
            self.samples = []

        def write_batch(self, batch):
            self.samples.extend(batch)

        def close(self):
            pass

    dest = Dest()
    rep = Reporter([dest], cpu_sampling_frequency=97)
    executables = ExecutableCache()
    unwinder = EhFrameUnwinder(executables)
    svc = CPUSamplerService(rep, freq=97, dwarf_stacks=True,
                            poll_interval=0.05, unwinder=unwinder)
    unwinder.processes = svc.processes
    svc.start()
    proc = subprocess.Popen([nofp_binary])
    proc.wait()
    time.sleep(0.3)
    svc.stop()
    rep.flush()

    assert svc.stacks_unwound_dwarf > 50, (
        f"dwarf unwinds: {svc.stacks_unwound_dwarf}, "
        f"tables: {unwinder.tables_built}, fails: {unwinder.table_failures}")

    with ELFFile.open(nofp_binary) as elf:
        idx = SymbolIndex(elf.symbols())

    def names(sample):
        out = []
        for f in sample.trace.frames:
            if f.mapping and f.mapping.path == nofp_binary:
                sym = idx.lookup(f.address)
                out.append(sym.name if sym else hex(f.address))
        return out

    mine = [s for s in dest.samples
            if any(f.mapping and f.mapping.path == nofp_binary
                   for f in s.trace.frames)]
    total = sum(s.value for s in mine)
    assert total > 30, total
    # Weight chains by aggregated sample counts.
    full = sum(s.value for s in mine
               if {"dwarf_leaf", "dwarf_mid", "dwarf_top", "main"} <=
               set(names(s)))
    # The hot leaf dominates; demand a healthy fraction of full chains.
    assert full > total // 4, (
        f"full chains {full}/{total}; example chains: "
        f"{[names(s) for s in mine[:5]]}")


def test_parse_garbage_eh_frame():
    """Corrupt CFI must fail cleanly (bounded), never crash or hang —
    the parser runs on every executable the fleet maps in."""
    from parca_agent_amd.native import sampler as native_sampler

    u = native_sampler().Unwinder()
    import random

    rng = random.Random(7)
    for size in (0, 1, 7, 64, 4096):
        blob = bytes(rng.randrange(256) for _ in range(size))
        try:
            mid = u.add_module_from_eh_frame(blob, 0x1000, max_rows=100000)
            assert u.module_rows(mid) >= 0
        except (ValueError, RuntimeError):
            pass  # rejecting is fine; crashing is not
    # Truncated real data: take libc's .eh_frame and cut it mid-record.
    from parca_agent_amd.elf import ELFFile

    libc = None
    with open("/proc/self/maps") as fh:
        for line in fh:
            if "libc.so" in line and line.split()[-1].startswith("/"):
                libc = line.split()[-1]
                break
    if libc:
        with ELFFile.open(libc) as elf:
            sec = elf.section(".eh_frame")
            data = elf.section_data(sec)
        for cut in (1, len(data) // 3, len(data) - 3):
            try:
                u.add_module_from_eh_frame(data[:cut], sec.addr,
                                           max_rows=100000)
            except (ValueError, RuntimeError):
                pass


def test_row_budget_scales_with_cgroup_limit():
    """automemlimit analog: the table budget adapts to the container's
    memory limit with a floor, defaulting to 24M rows unconstrained."""
    from parca_agent_amd.cpu.unwind import default_row_budget

    assert default_row_budget(1 << 62) == 24_000_000
    assert default_row_budget(2 << 30) == 24_000_000
    quarter_gib = default_row_budget(256 << 20)
    assert 1_000_000 < quarter_gib < 24_000_000  # 8 B/row packed
    assert quarter_gib == int((256 << 20) * 0.4) // 8
    assert default_row_budget(16 << 20) == 1_000_000  # floor
