"""OOM watcher tests: kmsg parse + report shape (live OOM kills are not
triggered in CI; the reference's analog is likewise a manual script,
SURVEY.md §4)."""

from parca_agent_amd.model import Frame, FrameType, Trace
from parca_agent_amd.oom.watcher import OOMWatcher, parse_oom_kill
from parca_agent_amd.reporter import Reporter


KMSG = ("6,1234,5678,-;Out of memory: Killed process 4242 (python3) "
        "total-vm:8388608kB, anon-rss:4194304kB, file-rss:1024kB, "
        "shmem-rss:0kB, UID:0 pgtables:8192kB oom_score_adj:0")


def test_parse_oom_kill():
    kill = parse_oom_kill(KMSG)
    assert kill is not None
    assert kill.pid == 4242
    assert kill.comm == "python3"
    assert kill.total_vm_kb == 8388608
    assert kill.anon_rss_kb == 4194304
    assert parse_oom_kill("normal log line") is None


class Dest:
    def __init__(self):
        self.samples = []

    def write_batch(self, batch):
        self.samples.extend(batch)

    def close(self):
        pass


def test_oom_report_without_stack():
    dest = Dest()
    rep = Reporter([dest])
    w = OOMWatcher(rep)
    w.handle_line(KMSG)
    rep.flush()
    [s] = dest.samples
    assert s.sample_type.sample_type == "inuse_space"
    assert s.value == 4194304 * 1024
    assert s.labels["job"] == "oom"
    assert "python3" in s.trace.frames[0].function_name


def test_oom_report_with_last_stack():
    dest = Dest()
    rep = Reporter([dest])
    stack = Trace(frames=(Frame(kind=FrameType.NATIVE, address=0x10,
                                function_name="allocate_all_the_things"),))
    w = OOMWatcher(rep, last_stack_lookup=lambda pid: stack)
    w.handle_line(KMSG)
    rep.flush()
    [s] = dest.samples
    assert s.trace.frames[0].function_name == "allocate_all_the_things"
    assert s.labels["job"] == "oom"
