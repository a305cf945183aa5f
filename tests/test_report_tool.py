"""tools/report.py: offline top-N over local-store pprof files."""

import os
import subprocess
import sys

from parca_agent_amd.model import (Frame, FrameType, MappingFile, Trace,
                                   TraceEventMeta)
from parca_agent_amd.reporter import LocalStoreDestination, Reporter

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_report_top_n(tmp_path):
    m = MappingFile(file_id="a" * 32, path="/usr/bin/app")
    dest = LocalStoreDestination(str(tmp_path))
    rep = Reporter([dest])
    hot = Trace(frames=(
        Frame(kind=FrameType.NATIVE, address=0x10, mapping=m,
              function_name="hot_leaf"),
        Frame(kind=FrameType.NATIVE, address=0x20, mapping=m,
              function_name="caller"),
    ))
    cold = Trace(frames=(Frame(kind=FrameType.NATIVE, address=0x30,
                               mapping=m, function_name="cold_leaf"),))
    for _ in range(9):
        rep.report_trace_event(hot, TraceEventMeta(pid=1, tid=1))
    rep.report_trace_event(cold, TraceEventMeta(pid=1, tid=1))
    rep.flush()

    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "report.py"),
         str(tmp_path), "--type", "samples", "-n", "5"],
        capture_output=True, text=True, timeout=60, cwd=REPO)
    assert out.returncode == 0, out.stderr[-800:]
    lines = out.stdout
    assert "hot_leaf" in lines and "cold_leaf" in lines
    assert lines.index("hot_leaf") < lines.index("cold_leaf")  # ranked

    cum = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "report.py"),
         str(tmp_path), "--type", "samples", "--cum"],
        capture_output=True, text=True, timeout=60, cwd=REPO)
    assert cum.returncode == 0, cum.stderr[-800:]
    assert "caller" in cum.stdout  # cumulative credits the caller
