"""Container metadata tests: cgroup container-id regexes (reference:
containermetadata_test.go:22) + CRI RuntimeService client against a fake
runtime over a unix socket."""

import os
from concurrent import futures

import grpc
import pytest

from parca_agent_amd.metadata.container import (
    ContainerMetadataProvider,
    extract_container_ids,
)
from parca_agent_amd.metadata.cri import (
    RUNTIME_SERVICE,
    ContainerInfo,
    CRIClient,
    SandboxInfo,
    encode_list_containers_response,
    encode_list_pod_sandbox_response,
)

CID = "a" * 64
POD_UID = "12345678-1234-1234-1234-123456789abc"


@pytest.mark.parametrize("cgroup,want_pod,want_cid", [
    # kubernetes + containerd (systemd driver)
    (f"/kubepods.slice/kubepods-burstable.slice/"
     f"kubepods-burstable-pod{POD_UID.replace('-', '_')}.slice/"
     f"cri-containerd-{CID}.scope", POD_UID, CID),
    # kubernetes + crio
    (f"/kubepods.slice/kubepods-pod{POD_UID.replace('-', '_')}.slice/"
     f"crio-{CID}.scope", POD_UID, CID),
    # kubernetes cgroupfs flat
    (f"/kubepods/besteffort/pod{POD_UID}/{CID}", POD_UID, CID),
    # plain docker
    (f"/docker/{CID}", None, CID),
    # lxc
    ("/lxc/mycontainer", None, "mycontainer"),
    # not containerized
    ("/user.slice/user-0.slice/session-1.scope", None, None),
])
def test_extract_container_ids(cgroup, want_pod, want_cid):
    pod, cid = extract_container_ids(cgroup)
    assert cid == want_cid
    if want_pod:
        assert pod == want_pod


@pytest.fixture
def fake_cri(tmp_path):
    containers = [ContainerInfo(
        id=CID, pod_sandbox_id="sb1", name="main",
        labels={"io.kubernetes.container.name": "main"})]
    sandboxes = [SandboxInfo(id="sb1", name="web-abc123",
                             namespace="prod", uid=POD_UID)]

    def list_containers(request, context):
        return encode_list_containers_response(containers)

    def list_sandboxes(request, context):
        return encode_list_pod_sandbox_response(sandboxes)

    ident = lambda b: b  # noqa: E731
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
    handler = grpc.method_handlers_generic_handler(RUNTIME_SERVICE, {
        "ListContainers": grpc.unary_unary_rpc_method_handler(
            list_containers, request_deserializer=ident,
            response_serializer=ident),
        "ListPodSandbox": grpc.unary_unary_rpc_method_handler(
            list_sandboxes, request_deserializer=ident,
            response_serializer=ident),
    })
    server.add_generic_rpc_handlers((handler,))
    sock = tmp_path / "cri.sock"
    server.add_insecure_port(f"unix://{sock}")
    server.start()
    yield f"unix://{sock}"
    server.stop(grace=None)


def test_cri_client_join(fake_cri):
    client = CRIClient(endpoint=fake_cri)
    containers = client.containers()
    info = containers[CID]
    assert info.name == "main"
    assert info.pod_name == "web-abc123"
    assert info.pod_namespace == "prod"
    assert info.pod_uid == POD_UID
    client.close()


def test_provider_with_cri(fake_cri, monkeypatch):
    client = CRIClient(endpoint=fake_cri)
    provider = ContainerMetadataProvider(cri_client=client)
    cgroup = f"/kubepods/besteffort/pod{POD_UID}/{CID}"
    labels = {"__meta_process_cgroup": cgroup}
    provider.add_metadata(os.getpid(), labels)
    assert labels["container_id"] == CID[:12]
    assert labels["container"] == "main"
    assert labels["pod"] == "web-abc123"
    assert labels["namespace"] == "prod"
    assert labels["__meta_kubernetes_pod_uid"] == POD_UID


def test_provider_without_cri():
    provider = ContainerMetadataProvider(cri_client=None)
    provider._cri = None  # force no runtime
    cgroup = f"/docker/{CID}"
    labels = {"__meta_process_cgroup": cgroup}
    provider.add_metadata(os.getpid(), labels)
    assert labels["container_id"] == CID[:12]
    assert "pod" not in labels or labels.get("pod")  # env may add one
