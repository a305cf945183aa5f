"""Relabel engine tests (reference behaviour: config/config_test.go,
prometheus relabel semantics applied in parca_reporter.go:779-841)."""

import pytest

from parca_agent_amd.config import parse_relabel_configs
from parca_agent_amd.relabel import RelabelConfig, relabel, strip_meta_labels


def test_keep_drop():
    cfgs = [RelabelConfig(source_labels=["comm"], regex="python.*", action="keep")]
    assert relabel({"comm": "python3"}, cfgs) == {"comm": "python3"}
    assert relabel({"comm": "bash"}, cfgs) is None

    cfgs = [RelabelConfig(source_labels=["comm"], regex="bash", action="drop")]
    assert relabel({"comm": "bash"}, cfgs) is None
    assert relabel({"comm": "python3"}, cfgs) is not None


def test_replace_with_capture():
    cfgs = [
        RelabelConfig(
            source_labels=["__meta_process_executable"],
            regex=".*/(.*)",
            target_label="binary",
            replacement="$1",
        )
    ]
    out = relabel({"__meta_process_executable": "/usr/bin/python3"}, cfgs)
    assert out["binary"] == "python3"


def test_replace_no_match_leaves_labels():
    cfgs = [
        RelabelConfig(source_labels=["a"], regex="nope", target_label="b",
                      replacement="x")
    ]
    out = relabel({"a": "value"}, cfgs)
    assert "b" not in out


def test_labelmap():
    cfgs = [RelabelConfig(regex="__meta_kubernetes_pod_label_(.+)",
                          replacement="$1", action="labelmap")]
    out = relabel({"__meta_kubernetes_pod_label_app": "web"}, cfgs)
    assert out["app"] == "web"


def test_labeldrop_labelkeep():
    cfgs = [RelabelConfig(regex="tmp_.*", action="labeldrop")]
    out = relabel({"tmp_x": "1", "keep": "2"}, cfgs)
    assert out == {"keep": "2"}

    cfgs = [RelabelConfig(regex="keep", action="labelkeep")]
    out = relabel({"tmp_x": "1", "keep": "2"}, cfgs)
    assert out == {"keep": "2"}


def test_lowercase_hashmod():
    cfgs = [RelabelConfig(source_labels=["comm"], target_label="lc",
                          action="lowercase")]
    assert relabel({"comm": "BASH"}, cfgs)["lc"] == "bash"

    cfgs = [RelabelConfig(source_labels=["comm"], target_label="shard",
                          modulus=4, action="hashmod")]
    out = relabel({"comm": "bash"}, cfgs)
    assert out["shard"] in {"0", "1", "2", "3"}


def test_keepequal_dropequal():
    cfgs = [RelabelConfig(source_labels=["a"], target_label="b",
                          action="keepequal")]
    assert relabel({"a": "x", "b": "x"}, cfgs) is not None
    assert relabel({"a": "x", "b": "y"}, cfgs) is None


def test_separator_join():
    cfgs = [RelabelConfig(source_labels=["a", "b"], regex="1;2", action="keep")]
    assert relabel({"a": "1", "b": "2"}, cfgs) is not None
    assert relabel({"a": "1", "b": "3"}, cfgs) is None


def test_regex_is_anchored():
    cfgs = [RelabelConfig(source_labels=["comm"], regex="bash", action="keep")]
    assert relabel({"comm": "bash2"}, cfgs) is None


def test_strip_meta():
    assert strip_meta_labels({"__meta_x": "1", "node": "n1"}) == {"node": "n1"}


def test_invalid_action():
    with pytest.raises(ValueError):
        RelabelConfig(action="explode")


def test_parse_yaml_doc():
    doc = {
        "relabel_configs": [
            {"source_labels": ["comm"], "regex": "python.*", "action": "keep"},
            {"source_labels": ["__meta_thread_comm"], "target_label": "thread",
             "action": "replace"},
        ]
    }
    cfgs = parse_relabel_configs(doc)
    assert len(cfgs) == 2
    assert cfgs[0].action == "keep"
    out = relabel({"comm": "python3", "__meta_thread_comm": "worker"}, cfgs)
    assert out["thread"] == "worker"
