"""CRI RuntimeService client: pod/container metadata straight from the
node's container runtime.

Reference: reporter/metadata/cri_client.go:20-50 queries the CRI
RuntimeService directly to avoid apiserver load. Same here: hand-rolled
runtime.v1 codecs (ListContainers / ListPodSandbox) over the runtime's
unix socket; the container provider joins the cgroup-derived container
id against this to label samples with pod/namespace/container names.
"""

from __future__ import annotations

import logging
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..pprof.proto import Writer, iter_fields

log = logging.getLogger("parca_agent_amd.metadata.cri")

RUNTIME_SERVICE = "runtime.v1.RuntimeService"

DEFAULT_ENDPOINTS = (
    "/run/containerd/containerd.sock",
    "/run/crio/crio.sock",
    "/var/run/cri-dockerd.sock",
)

_identity = lambda b: b  # noqa: E731


@dataclass
class ContainerInfo:
    id: str
    pod_sandbox_id: str = ""
    name: str = ""
    labels: Dict[str, str] = field(default_factory=dict)
    # joined from the sandbox:
    pod_name: str = ""
    pod_namespace: str = ""
    pod_uid: str = ""


def _decode_map_entry(data: bytes):
    k = v = ""
    for f, _wt, val in iter_fields(data):
        if f == 1:
            k = val.decode("utf-8", "replace")
        elif f == 2:
            v = val.decode("utf-8", "replace")
    return k, v


def decode_list_containers_response(data: bytes) -> List[ContainerInfo]:
    out = []
    for f, _wt, v in iter_fields(data):
        if f != 1:
            continue
        c = ContainerInfo(id="")
        for cf, _cwt, cv in iter_fields(v):
            if cf == 1:
                c.id = cv.decode()
            elif cf == 2:
                c.pod_sandbox_id = cv.decode()
            elif cf == 3:  # ContainerMetadata{name=1}
                for mf, _mwt, mv in iter_fields(cv):
                    if mf == 1:
                        c.name = mv.decode()
            elif cf == 8:
                k, val = _decode_map_entry(cv)
                c.labels[k] = val
        if c.id:
            out.append(c)
    return out


@dataclass
class SandboxInfo:
    id: str
    name: str = ""
    namespace: str = ""
    uid: str = ""
    labels: Dict[str, str] = field(default_factory=dict)


def decode_list_pod_sandbox_response(data: bytes) -> List[SandboxInfo]:
    out = []
    for f, _wt, v in iter_fields(data):
        if f != 1:
            continue
        s = SandboxInfo(id="")
        for sf, _swt, sv in iter_fields(v):
            if sf == 1:
                s.id = sv.decode()
            elif sf == 2:  # PodSandboxMetadata{name=1,uid=2,namespace=3}
                for mf, _mwt, mv in iter_fields(sv):
                    if mf == 1:
                        s.name = mv.decode()
                    elif mf == 2:
                        s.uid = mv.decode()
                    elif mf == 3:
                        s.namespace = mv.decode()
            elif sf == 5:
                k, val = _decode_map_entry(sv)
                s.labels[k] = val
        if s.id:
            out.append(s)
    return out


def encode_container_metadata(c: ContainerInfo) -> Writer:
    w = Writer()
    w.string(1, c.id)
    w.string(2, c.pod_sandbox_id)
    if c.name:
        mw = Writer()
        mw.string(1, c.name)
        w.message(3, mw)
    for k, v in c.labels.items():
        ew = Writer()
        ew.string(1, k)
        ew.string(2, v)
        w.message(8, ew)
    return w


def encode_list_containers_response(containers: List[ContainerInfo]) -> bytes:
    w = Writer()
    for c in containers:
        w.message(1, encode_container_metadata(c))
    return w.getvalue()


def encode_sandbox(s: SandboxInfo) -> Writer:
    w = Writer()
    w.string(1, s.id)
    mw = Writer()
    mw.string(1, s.name)
    mw.string(2, s.uid)
    mw.string(3, s.namespace)
    w.message(2, mw)
    for k, v in s.labels.items():
        ew = Writer()
        ew.string(1, k)
        ew.string(2, v)
        w.message(5, ew)
    return w


def encode_list_pod_sandbox_response(sandboxes: List[SandboxInfo]) -> bytes:
    w = Writer()
    for s in sandboxes:
        w.message(1, encode_sandbox(s))
    return w.getvalue()


class CRIClient:
    def __init__(self, endpoint: Optional[str] = None,
                 timeout: float = 5.0) -> None:
        import grpc

        if endpoint is None:
            for cand in DEFAULT_ENDPOINTS:
                if os.path.exists(cand):
                    endpoint = f"unix://{cand}"
                    break
        if endpoint is None:
            raise FileNotFoundError("no CRI runtime socket found")
        self.endpoint = endpoint
        self.timeout = timeout
        channel = grpc.insecure_channel(endpoint)
        self._list_containers = channel.unary_unary(
            f"/{RUNTIME_SERVICE}/ListContainers",
            request_serializer=_identity, response_deserializer=_identity)
        self._list_sandboxes = channel.unary_unary(
            f"/{RUNTIME_SERVICE}/ListPodSandbox",
            request_serializer=_identity, response_deserializer=_identity)
        self._channel = channel

    def containers(self) -> Dict[str, ContainerInfo]:
        """Container id -> info, joined with sandbox pod identity."""
        resp = self._list_containers(b"", timeout=self.timeout)
        containers = decode_list_containers_response(resp)
        resp = self._list_sandboxes(b"", timeout=self.timeout)
        sandboxes = {s.id: s for s in
                     decode_list_pod_sandbox_response(resp)}
        out = {}
        for c in containers:
            sb = sandboxes.get(c.pod_sandbox_id)
            if sb is not None:
                c.pod_name = sb.name
                c.pod_namespace = sb.namespace
                c.pod_uid = sb.uid
            # kubelet also mirrors identity into labels
            c.pod_name = c.pod_name or c.labels.get(
                "io.kubernetes.pod.name", "")
            c.pod_namespace = c.pod_namespace or c.labels.get(
                "io.kubernetes.pod.namespace", "")
            c.pod_uid = c.pod_uid or c.labels.get(
                "io.kubernetes.pod.uid", "")
            c.name = c.name or c.labels.get(
                "io.kubernetes.container.name", "")
            out[c.id] = c
        return out

    def close(self) -> None:
        self._channel.close()
