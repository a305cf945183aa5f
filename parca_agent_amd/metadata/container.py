"""Container metadata provider: container-ID extraction from cgroup paths.

Covers the cgroup path shapes the reference recognizes — kubernetes
(+ containerd/crio/docker runtimes), plain docker, lxc, buildkit
(reference: reporter/metadata/containermetadata.go:80-96, 787-860). The
K8s API/CRI enrichment of the reference is replaced by environment-file
hints (downward API) plus the cgroup-derived pod UID/container ID, which
is what the relabeling rules key on.
"""

from __future__ import annotations

import os
import re
from typing import Dict, Optional, Tuple

from .. import procmaps
from ..lru import LRU

_HEX64 = r"[0-9a-f]{64}"

# Pod UID appears in kubepods cgroups with dashes or underscores.
_POD_RE = re.compile(
    r"kubepods[^/]*?(?:/|-(?:besteffort|burstable)[^/]*/|pod)"
    r".*?pod([0-9a-f-_]{36})", re.IGNORECASE)
_POD_SLICE_RE = re.compile(r"pod([0-9a-f_-]{36})\.slice")
_CONTAINER_RES = [
    # kubernetes w/ containerd or crio: .../cri-containerd-<id>.scope
    re.compile(r"(?:cri-containerd|crio|docker)-(" + _HEX64 + r")(?:\.scope)?"),
    # plain docker: /docker/<id>
    re.compile(r"/docker/(" + _HEX64 + r")"),
    # kubepods flat: .../<id> at the end
    re.compile(r"kubepods[^ ]*/(" + _HEX64 + r")$"),
    # buildkit
    re.compile(r"/buildkit/(\w+)$"),
    # lxc
    re.compile(r"/lxc/([^/]+)$"),
]


def extract_container_ids(cgroup_path: str) -> Tuple[Optional[str], Optional[str]]:
    """(pod_uid, container_id) from a cgroup path, either may be None."""
    if not cgroup_path:
        return None, None
    pod = None
    m = _POD_RE.search(cgroup_path)
    if not m:
        m = _POD_SLICE_RE.search(cgroup_path)
    if m:
        pod = m.group(1).replace("_", "-")
    for rx in _CONTAINER_RES:
        cm = rx.search(cgroup_path)
        if cm:
            return pod, cm.group(1)
    return pod, None


class ContainerMetadataProvider:
    name = "container"

    def __init__(self, node: str = "", cache_size: int = 4096,
                 cri_client=None, cri_refresh: float = 30.0,
                 k8s_informer=None, docker_client=None) -> None:
        self._node = node
        self._cache: LRU[int, Dict[str, str]] = LRU(cache_size, ttl_seconds=300)
        # Enrichment sources, in precedence order:
        #  1. apiserver informer (pod labels/annotations for relabeling —
        #     reference containermetadata.go:250-470),
        #  2. CRI RuntimeService (containerd/crio; no apiserver load —
        #     cri_client.go),
        #  3. docker engine socket (docker hosts without CRI, 481-525),
        #  4. downward-API env hints.
        self._k8s = k8s_informer
        self._docker = docker_client
        self._cri = cri_client
        self._cri_refresh = cri_refresh
        self._cri_containers: Dict[str, object] = {}
        self._cri_last = 0.0
        if cri_client is None:
            try:
                from .cri import CRIClient

                self._cri = CRIClient()
            except Exception:
                self._cri = None
        if docker_client is None:
            try:
                from .kubernetes import DockerClient

                self._docker = DockerClient()
            except Exception:
                self._docker = None

    def _cri_lookup(self, container_id: str):
        if self._cri is None:
            return None
        import time

        now = time.monotonic()
        if now - self._cri_last > self._cri_refresh:
            self._cri_last = now
            try:
                self._cri_containers = self._cri.containers()
            except Exception:
                self._cri_containers = {}
        return self._cri_containers.get(container_id)

    def _k8s_enrich(self, pod_uid, container_id,
                    cached: Dict[str, str]) -> None:
        """Apiserver-informer enrichment: full pod label/annotation set
        (the relabeling workflows' input) when the informer knows the
        pod, found via cgroup pod UID or container id."""
        if self._k8s is None:
            return
        from .kubernetes import pod_labelset

        info = None
        cname = ""
        if container_id:
            hit = self._k8s.pod_by_container(container_id)
            if hit is not None:
                info, cname = hit
        if info is None and pod_uid:
            info = self._k8s.pod_by_uid(pod_uid)
        if info is None:
            return
        for k, v in pod_labelset(info, cname).items():
            cached.setdefault(k, v)

    def add_metadata(self, pid: int, labels: Dict[str, str]) -> bool:
        cached = self._cache.get(pid)
        if cached is None:
            cached = {}
            cgroup = labels.get("__meta_process_cgroup") or \
                procmaps.read_cgroup(pid) or ""
            pod_uid, container_id = extract_container_ids(cgroup)
            if pod_uid:
                cached["__meta_kubernetes_pod_uid"] = pod_uid
            if container_id:
                cached["container_id"] = container_id[:12]
                cached["__meta_container_id"] = container_id
                info = self._cri_lookup(container_id)
                if info is not None:
                    if info.name:
                        cached["container"] = info.name
                    if info.pod_name:
                        cached["pod"] = info.pod_name
                    if info.pod_namespace:
                        cached["namespace"] = info.pod_namespace
                    if info.pod_uid:
                        cached["__meta_kubernetes_pod_uid"] = info.pod_uid
            self._k8s_enrich(pod_uid, container_id, cached)
            if container_id and self._docker is not None and \
                    "container" not in cached:
                try:
                    cached.update(
                        self._docker.container_labels(container_id))
                except Exception:
                    pass
            env = _downward_api_env(pid)
            for k, v in env.items():
                cached.setdefault(k, v)
            self._cache.put(pid, cached)
        labels.update(cached)
        return True


_ENV_LABELS = {
    "KUBERNETES_POD_NAME": "pod",
    "HOSTNAME": "__meta_hostname",
    "KUBERNETES_NAMESPACE": "namespace",
    "POD_NAME": "pod",
    "POD_NAMESPACE": "namespace",
}


def _downward_api_env(pid: int) -> Dict[str, str]:
    out: Dict[str, str] = {}
    try:
        with open(f"/proc/{pid}/environ", "rb") as fh:
            data = fh.read(65536)
    except OSError:
        return out
    for chunk in data.split(b"\x00"):
        try:
            k, _, v = chunk.decode("utf-8", "replace").partition("=")
        except ValueError:
            continue
        label = _ENV_LABELS.get(k)
        if label and v:
            out.setdefault(label, v)
    return out
