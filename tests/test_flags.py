"""Flag/config precedence tests (reference: flags/flags_test.go — CLI over
YAML over defaults matrix)."""

import pytest

from parca_agent_amd import flags as flagsmod


def test_defaults():
    f = flagsmod.parse([])
    assert f.profiling.cpu_sampling_frequency == 19
    assert f.profiling.duration == 5.0
    assert f.http_address == "127.0.0.1:7071"
    assert f.remote_store.batch_write_interval == 10.0
    assert f.rocm.enable is True
    assert f.gpu_ring_bytes == 32 << 20


def test_cli_overrides():
    f = flagsmod.parse([
        "--profiling-cpu-sampling-frequency", "97",
        "--rocm-ring-scale-factor", "3",
        "--remote-store-address", "parca:7070",
        "--merge-gpu-profiles", "true",
    ])
    assert f.profiling.cpu_sampling_frequency == 97
    assert f.gpu_ring_bytes == 256 << 20
    assert f.remote_store.address == "parca:7070"
    assert f.merge_gpu_profiles is True


def test_duration_parsing():
    f = flagsmod.parse(["--profiling-duration", "10s"])
    assert f.profiling.duration == 10.0
    f = flagsmod.parse(["--clock-sync-interval", "3m"])
    assert f.clock_sync_interval == 180.0


def test_yaml_overlay_cli_wins(tmp_path):
    cfg = tmp_path / "agent.yaml"
    cfg.write_text(
        "profiling-cpu-sampling-frequency: 41\n"
        "node: yaml-node\n"
        "remote-store-address: yaml:7070\n"
    )
    f = flagsmod.parse([
        "--config-path", str(cfg),
        "--node", "cli-node",
    ])
    # YAML applied where CLI silent; CLI wins where both set.
    assert f.profiling.cpu_sampling_frequency == 41
    assert f.remote_store.address == "yaml:7070"
    assert f.node == "cli-node"


def test_yaml_unknown_key_rejected(tmp_path):
    cfg = tmp_path / "agent.yaml"
    cfg.write_text("definitely-not-a-flag: 1\n")
    with pytest.raises(ValueError):
        flagsmod.parse(["--config-path", str(cfg)])


def test_validate_scale_factor():
    with pytest.raises(ValueError):
        flagsmod.parse(["--rocm-ring-scale-factor", "5"])


def test_validate_offline_exclusive():
    with pytest.raises(ValueError):
        flagsmod.parse([
            "--offline-mode-storage-path", "/tmp/x",
            "--remote-store-address", "parca:7070",
        ])


def test_bearer_token_file(tmp_path):
    tok = tmp_path / "token"
    tok.write_text("s3cret\n")
    f = flagsmod.parse(["--remote-store-bearer-token-file", str(tok)])
    assert f.remote_store.bearer_token == "s3cret"


def test_trace_cache_size():
    f = flagsmod.parse([])
    # freq(19) x 5s x cores x 6, min 65536, pow2
    assert f.trace_cache_size(cores=8) == 65536
    size = f.trace_cache_size(cores=128)
    assert size == 131072  # 19*5*128*6 = 72960 -> next pow2


def test_external_labels_kv():
    f = flagsmod.parse([
        "--metadata-external-labels", "env=prod",
        "--metadata-external-labels", "region=eu",
    ])
    assert f.metadata.external_labels == {"env": "prod", "region": "eu"}


def test_reference_flag_aliases_and_gates():
    from parca_agent_amd import flags as flagsmod

    f = flagsmod.parse(["--java-unwinding-disable", "true",
                        "--enable-oom-prof", "true",
                        "--remote-store-use-v2-schema", "false"])
    assert f.jvm_unwinding_disable is True   # reference name aliases jvm
    assert f.enable_oom_watch is True        # --enable-oom-prof alias
    assert f.remote_store.write_format == "arrow_v1"  # v2 opt-out

    f2 = flagsmod.parse([])
    assert f2.remote_store.write_format == "arrow_v2"
    assert f2.profiling.label_ttl == 600.0   # reference 10m default
    assert f2.debuginfo.directories == ["/usr/lib/debug"]


def test_disable_label_flags_gate_patching():
    from parca_agent_amd.model import (Frame, FrameType, MappingFile,
                                       Trace, TraceEventMeta)
    from parca_agent_amd.reporter import Reporter

    class Dest:
        def __init__(self):
            self.samples = []

        def write_batch(self, b):
            self.samples.extend(b)

        def close(self):
            pass

    dest = Dest()
    rep = Reporter([dest], disable_cpu_label=True,
                   disable_thread_id_label=True,
                   disable_thread_comm_label=True)
    t = Trace(frames=(Frame(kind=FrameType.NATIVE, address=1,
                            mapping=MappingFile(path="/bin/x")),))
    rep.report_trace_event(t, TraceEventMeta(pid=9, tid=9, cpu=3,
                                             comm="worker"))
    rep.flush()
    labels = dest.samples[0].labels
    assert "cpu" not in labels
    assert "thread_id" not in labels
    assert "thread_comm" not in labels
