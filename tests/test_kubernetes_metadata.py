"""Apiserver pod informer + docker client tests against fake servers
(reference capability: node-filtered shared informer and direct runtime
clients, containermetadata.go:250-296, 353-470, 481-525)."""

import http.server
import json
import os
import socket
import socketserver
import threading
import time

import pytest

from parca_agent_amd.metadata.container import ContainerMetadataProvider
from parca_agent_amd.metadata.kubernetes import (
    DockerClient,
    K8sPodInformer,
    pod_labelset,
    sanitize_label_name,
    strip_runtime_prefix,
)
from parca_agent_amd.relabel import RelabelConfig, relabel

CID = "a" * 64
POD_UID = "11111111-2222-3333-4444-555555555555"


def _pod(uid, name, node, cid=None, labels=None, rv="10"):
    doc = {
        "metadata": {
            "uid": uid, "name": name, "namespace": "prod",
            "resourceVersion": rv,
            "labels": labels or {"app": "web", "tier.kind/x": "backend"},
            "annotations": {"example.com/team": "infra"},
        },
        "spec": {"nodeName": node},
        "status": {"containerStatuses": [
            {"name": "main", "containerID": f"containerd://{cid or CID}"},
        ]},
    }
    return doc


class FakeApiserver(http.server.BaseHTTPRequestHandler):
    pods = []
    watch_events = []
    requests_seen = []

    def log_message(self, *a):
        pass

    def do_GET(self):
        FakeApiserver.requests_seen.append(self.path)
        assert self.headers.get("Authorization") == "Bearer test-token"
        if "watch=1" in self.path:
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.end_headers()
            for ev in FakeApiserver.watch_events:
                self.wfile.write(json.dumps(ev).encode() + b"\n")
                self.wfile.flush()
            time.sleep(0.3)  # keep the stream open briefly
            return
        body = json.dumps({
            "metadata": {"resourceVersion": "10"},
            "items": FakeApiserver.pods,
        }).encode()
        self.send_response(200)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)


@pytest.fixture
def apiserver():
    FakeApiserver.pods = [_pod(POD_UID, "web-abc", "node-1")]
    FakeApiserver.watch_events = [
        {"type": "ADDED",
         "object": _pod("99999999-0000-0000-0000-000000000000",
                        "late-pod", "node-1", cid="b" * 64,
                        labels={"app": "batch"}, rv="11")},
    ]
    FakeApiserver.requests_seen = []
    srv = socketserver.ThreadingTCPServer(("127.0.0.1", 0), FakeApiserver)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{srv.server_address[1]}"
    srv.shutdown()


def test_informer_list_watch_and_labelset(apiserver):
    inf = K8sPodInformer(node="node-1", api_base=apiserver,
                         token="test-token")
    inf.start()
    deadline = time.monotonic() + 10
    while time.monotonic() < deadline and inf.n_pods < 2:
        time.sleep(0.05)
    inf.stop()
    assert inf.n_pods == 2  # listed + watched
    assert any("fieldSelector=spec.nodeName%3Dnode-1" in p
               for p in FakeApiserver.requests_seen)

    info = inf.pod_by_uid(POD_UID)
    assert info.name == "web-abc"
    hit = inf.pod_by_container(CID)
    assert hit is not None and hit[1] == "main"

    labels = pod_labelset(info, "main")
    assert labels["namespace"] == "prod"
    assert labels["pod"] == "web-abc"
    assert labels["container"] == "main"
    assert labels["__meta_kubernetes_pod_label_app"] == "web"
    assert labels["__meta_kubernetes_pod_label_tier_kind_x"] == "backend"
    assert labels["__meta_kubernetes_pod_labelpresent_app"] == "true"
    assert labels["__meta_kubernetes_pod_annotation_example_com_team"] == \
        "infra"

    late = inf.pod_by_container("b" * 64)
    assert late is not None and late[0].name == "late-pod"


def test_provider_relabel_on_pod_labels(apiserver, monkeypatch):
    """The flagship relabeling workflow: rules keyed on
    __meta_kubernetes_pod_label_* must work off the informer data
    (reference kubernetes-config.yaml)."""
    inf = K8sPodInformer(node="node-1", api_base=apiserver,
                         token="test-token", watch=False)
    inf.list_once()
    prov = ContainerMetadataProvider(
        node="node-1", cri_client=False, docker_client=False,
        k8s_informer=inf)
    prov._cri = None
    prov._docker = None
    monkeypatch.setattr(
        "parca_agent_amd.metadata.container.procmaps.read_cgroup",
        lambda pid: f"/kubepods/burstable/pod{POD_UID}/"
                    f"cri-containerd-{CID}.scope")

    labels = {}
    assert prov.add_metadata(1234, labels)
    assert labels["pod"] == "web-abc"
    assert labels["container"] == "main"
    assert labels["__meta_kubernetes_pod_label_app"] == "web"

    keep = [RelabelConfig(action="keep",
                          source_labels=["__meta_kubernetes_pod_label_app"],
                          regex="web")]
    assert relabel(dict(labels), keep) is not None
    drop = [RelabelConfig(action="keep",
                          source_labels=["__meta_kubernetes_pod_label_app"],
                          regex="db")]
    assert relabel(dict(labels), drop) is None
    # label extraction into a target label
    extract = [RelabelConfig(action="replace",
                             source_labels=[
                                 "__meta_kubernetes_pod_label_app"],
                             target_label="app", regex="(.+)",
                             replacement="$1")]
    out = relabel(dict(labels), extract)
    assert out["app"] == "web"


def test_sanitize_and_strip_helpers():
    assert sanitize_label_name("app.kubernetes.io/name") == \
        "app_kubernetes_io_name"
    assert strip_runtime_prefix("containerd://" + CID) == CID
    assert strip_runtime_prefix("docker://abc") == "abc"
    assert strip_runtime_prefix("abc") == "abc"


class FakeDockerHandler(http.server.BaseHTTPRequestHandler):
    def log_message(self, *a):
        pass

    def do_GET(self):
        if self.path.startswith(f"/containers/{CID}/json"):
            body = json.dumps({
                "Name": "/web-1",
                "Config": {"Labels": {
                    "io.kubernetes.pod.name": "web-abc",
                    "io.kubernetes.pod.namespace": "prod",
                    "com.example.team": "infra",
                }},
            }).encode()
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)
        else:
            self.send_response(404)
            self.end_headers()


def test_docker_client_over_unix_socket(tmp_path):
    sock_path = str(tmp_path / "docker.sock")

    class UnixHTTPServer(socketserver.ThreadingUnixStreamServer):
        def get_request(self):
            request, _ = super().get_request()
            return request, ("127.0.0.1", 0)  # BaseHTTPRequestHandler addr

    srv = UnixHTTPServer(sock_path, FakeDockerHandler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        c = DockerClient(socket_path=sock_path)
        labels = c.container_labels(CID)
        assert labels["pod"] == "web-abc"
        assert labels["namespace"] == "prod"
        assert labels["container"] == "web-1"
        assert labels["__meta_docker_container_label_com_example_team"] == \
            "infra"
        assert c.container_labels("f" * 64) == {}
    finally:
        srv.shutdown()


def test_reference_kubernetes_config_workflow(apiserver, monkeypatch):
    """examples/kubernetes-config.yaml (mirroring the reference's
    flagship kubernetes-config.yaml) applied to informer-derived
    labels: pod labels are labelmapped in, machinery labels dropped."""
    import yaml

    from parca_agent_amd.config import parse_relabel_configs

    inf = K8sPodInformer(node="node-1", api_base=apiserver,
                         token="test-token", watch=False)
    inf.list_once()
    prov = ContainerMetadataProvider(
        node="node-1", cri_client=False, docker_client=False,
        k8s_informer=inf)
    prov._cri = None
    prov._docker = None
    monkeypatch.setattr(
        "parca_agent_amd.metadata.container.procmaps.read_cgroup",
        lambda pid: f"/kubepods/burstable/pod{POD_UID}/"
                    f"cri-containerd-{CID}.scope")
    labels = {}
    prov.add_metadata(4242, labels)
    labels["__meta_kubernetes_pod_label_pod_template_hash"] = "abc123"

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    with open(os.path.join(repo, "examples", "kubernetes-config.yaml")) as fh:
        doc = yaml.safe_load(fh)
    configs = parse_relabel_configs(doc)
    out = relabel(labels, configs)
    assert out is not None
    assert out["namespace"] == "prod"
    assert out["pod"] == "web-abc"
    assert out["container"] == "main"
    assert out["app"] == "web"                  # labelmapped pod label
    assert "pod_template_hash" not in out       # labeldropped


def test_informer_relists_after_watch_error(apiserver):
    """A watch ERROR (e.g. 410 Gone resourceVersion) must trigger a
    re-list instead of wedging the informer."""
    FakeApiserver.watch_events = [
        {"type": "ERROR",
         "object": {"kind": "Status", "code": 410,
                    "reason": "Expired"}},
    ]
    inf = K8sPodInformer(node="node-1", api_base=apiserver,
                         token="test-token")
    inf.start()
    deadline = time.monotonic() + 10
    while time.monotonic() < deadline and inf.lists < 2:
        time.sleep(0.05)
    inf.stop()
    assert inf.lists >= 2, (inf.lists, inf.errors)
    assert inf.errors >= 1
    assert inf.pod_by_uid(POD_UID) is not None  # state survives re-list
