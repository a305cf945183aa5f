"""Node-level multi-GPU profile merge over RCCL/xGMI.

New MI355X-native component with no reference analog (SURVEY.md §2.2,
§5.8): when profiling runs as one rank per GPU (embedded mode, e.g.
torchrun workloads), each rank holds its own GPU's kernel-time totals and
PC-bucket histogram. This service all-gathers the per-rank summaries over
RCCL (xGMI point-to-point; payloads are <= a few MB so one all-gather per
report interval is bandwidth-trivial) and rank 0 emits ONE merged
node-level pprof with per-GPU labels.

Clock alignment: kernel timestamps are already in the shared
CLOCK_BOOTTIME domain on one node (rocprofiler timestamps), so per-rank
records need no re-basing; each rank still ships its (boottime, realtime)
pair so cross-node merges can re-base later.

Transports:
- RcclTransport: native ncclAllGather via csrc/gpu RcclMerger (GPU).
- TorchDistTransport: torch.distributed all_gather — gloo for CPU tests,
  "nccl" (=RCCL on ROCm) in GPU jobs that already have a process group.
"""

from __future__ import annotations

import json
import struct
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..pprof import FrameKey, MappingKey, ProfileBuilder, ValueType


@dataclass
class RankProfile:
    """One rank's contribution to the node profile."""

    rank: int
    gpu_index: int
    # (code_object file_id, kernel name) -> (total_ns, count). The
    # file_id keeps code-object identity through the merge — kernels
    # from different HSA code objects may share a demangled name, and
    # the merged pprof must attribute them to the right executable
    # upload.
    kernel_times: Dict[Tuple[str, str], Tuple[int, int]] = field(
        default_factory=dict)
    # (code_object file_id, bucket vaddr, symbol) -> count
    pc_buckets: Dict[Tuple[str, int, str], int] = field(default_factory=dict)
    # collective name -> (total_ns, count): RCCL kernels split out for
    # per-collective attribution (BASELINE config 3).
    collectives: Dict[str, Tuple[int, int]] = field(default_factory=dict)
    boottime_ns: int = 0
    realtime_ns: int = 0

    def to_bytes(self) -> bytes:
        doc = {
            "rank": self.rank,
            "gpu_index": self.gpu_index,
            "kernel_times": [[fid, name, t, n] for (fid, name), (t, n)
                             in self.kernel_times.items()],
            "pc_buckets": [[fid, addr, sym, n] for (fid, addr, sym), n
                           in self.pc_buckets.items()],
            "collectives": {k: list(v) for k, v in self.collectives.items()},
            "boottime_ns": self.boottime_ns,
            "realtime_ns": self.realtime_ns,
        }
        return json.dumps(doc, separators=(",", ":")).encode()

    @classmethod
    def from_bytes(cls, data: bytes) -> "RankProfile":
        doc = json.loads(data.decode())
        return cls(
            rank=doc["rank"],
            gpu_index=doc["gpu_index"],
            kernel_times={(fid, name): (t, n)
                          for fid, name, t, n in doc["kernel_times"]},
            pc_buckets={(fid, addr, sym): n
                        for fid, addr, sym, n in doc["pc_buckets"]},
            collectives={k: tuple(v) for k, v in doc["collectives"].items()},
            boottime_ns=doc["boottime_ns"],
            realtime_ns=doc["realtime_ns"],
        )


RCCL_KERNEL_PREFIXES = ("ncclDevKernel", "ncclKernel", "rcclKernel")


def is_collective_kernel(name: str) -> bool:
    return name.startswith(RCCL_KERNEL_PREFIXES)


def collective_op(name: str) -> str:
    """ncclDevKernel_AllReduce_Sum_bf16_RING_LL(...) -> AllReduce."""
    base = name.split("(")[0]
    for prefix in RCCL_KERNEL_PREFIXES:
        if base.startswith(prefix):
            base = base[len(prefix):].lstrip("_")
            break
    return base.split("_")[0] or name


def build_rank_profile(rank: int, gpu_index: int, samples,
                       boottime_ns: Optional[int] = None) -> RankProfile:
    """Fold a flush batch (PendingSample list) into a RankProfile."""
    rp = RankProfile(rank=rank, gpu_index=gpu_index,
                     boottime_ns=boottime_ns or time.clock_gettime_ns(
                         time.CLOCK_BOOTTIME),
                     realtime_ns=time.time_ns())
    for s in samples:
        st = s.sample_type.sample_type
        if st in ("gpu_kernel_time", "gpu_time"):
            leaf = s.trace.frames[0]
            name = leaf.function_name or "unknown_kernel"
            fid = leaf.mapping.file_id if leaf.mapping else ""
            if is_collective_kernel(name):
                op = collective_op(name)
                t, n = rp.collectives.get(op, (0, 0))
                rp.collectives[op] = (t + s.value, n + 1)
            key = (fid, name)
            t, n = rp.kernel_times.get(key, (0, 0))
            rp.kernel_times[key] = (t + s.value, n + 1)
        elif st == "gpu_pcsample":
            leaf = s.trace.frames[0]
            fid = leaf.mapping.file_id if leaf.mapping else ""
            key = (fid, leaf.address, leaf.function_name)
            rp.pc_buckets[key] = rp.pc_buckets.get(key, 0) + s.value
    return rp


# -- transports ------------------------------------------------------------


class TorchDistTransport:
    """all_gather over an existing torch.distributed process group
    (gloo on CPU; "nccl" IS RCCL on ROCm)."""

    def __init__(self, group=None) -> None:
        import torch.distributed as dist

        self._dist = dist
        self.group = group

    def allgather(self, payload: bytes) -> List[bytes]:
        dist = self._dist
        world = dist.get_world_size(self.group)
        obj_list: List = [None] * world
        dist.all_gather_object(obj_list, payload, group=self.group)
        return [bytes(o) for o in obj_list]

    @property
    def rank(self) -> int:
        return self._dist.get_rank(self.group)


class RcclTransport:
    """Native ncclAllGather via the _gpu extension. Payloads are padded
    to the max length (8-byte length prefix) because ncclAllGather is
    equal-size."""

    def __init__(self, merger) -> None:
        self.merger = merger  # native RcclMerger

    def allgather(self, payload: bytes) -> List[bytes]:
        world = self.merger.world_size
        framed = struct.pack("<Q", len(payload)) + payload
        # Exchange sizes first so every rank pads identically.
        sizes_blob = self.merger.allgather(struct.pack("<Q", len(framed)))
        sizes = [struct.unpack_from("<Q", sizes_blob, i * 8)[0]
                 for i in range(world)]
        maxlen = max(sizes)
        padded = framed + b"\x00" * (maxlen - len(framed))
        blob = self.merger.allgather(padded)
        out = []
        for i in range(world):
            chunk = blob[i * maxlen : (i + 1) * maxlen]
            (n,) = struct.unpack_from("<Q", chunk, 0)
            out.append(chunk[8 : 8 + n])
        return out

    @property
    def rank(self) -> int:
        return self.merger.rank


# -- merge -----------------------------------------------------------------


def merge_node_profile(profiles: List[RankProfile],
                       node: str = "") -> ProfileBuilder:
    """Rank 0's output: one pprof with gpu_time samples per kernel per
    GPU, gpu_pcsample per bucket per GPU, and collective pseudo-frames
    for per-collective attribution."""
    builder = ProfileBuilder(
        sample_types=[ValueType("gpu_time", "nanoseconds"),
                      ValueType("gpu_dispatches", "count")],
        period_type=ValueType("gpu_time", "nanoseconds"),
    )
    if node:
        builder.comments.append(f"node={node}")
    for rp in sorted(profiles, key=lambda p: p.rank):
        gpu_label = [("gpu", str(rp.gpu_index)), ("rank", str(rp.rank))]
        for (fid, name), (total_ns, count) in rp.kernel_times.items():
            mapping = None
            if fid:
                mapping = MappingKey(memory_start=0, memory_limit=0,
                                     file_offset=0,
                                     filename=f"codeobj-{fid[:16]}",
                                     build_id=fid)
            frames = [FrameKey(address=0, mapping=mapping,
                               function_name=name)]
            if is_collective_kernel(name):
                frames.append(FrameKey(
                    address=0,
                    function_name=f"rccl::{collective_op(name)}"))
            builder.add_sample(frames, [total_ns, count], labels=gpu_label)
        for (fid, addr, sym), count in rp.pc_buckets.items():
            mapping = MappingKey(memory_start=0, memory_limit=0,
                                 file_offset=0,
                                 filename=f"codeobj-{fid[:16]}",
                                 build_id=fid)
            builder.add_sample(
                [FrameKey(address=addr, mapping=mapping, function_name=sym)],
                [0, count],
                labels=gpu_label + [("view", "pc_sample")])
    return builder


def build_node_profile_from_batch(samples, node: str = "") -> Optional[bytes]:
    """Daemon-shape merge: ONE agent on the node drains every process's
    ring and sees events from all 8 GPUs; its flush batch carries the
    per-sample ``gpu`` label. Group the batch into per-GPU RankProfiles
    (gpu_index doubles as rank) and emit the same merged node pprof the
    RCCL path produces — no collective needed when a single process
    already holds all the data (SURVEY.md §2.2 RCCL component, daemon
    variant)."""
    by_gpu: Dict[int, list] = {}
    for s in samples:
        st = s.sample_type.sample_type
        if st not in ("gpu_kernel_time", "gpu_time", "gpu_pcsample"):
            continue
        try:
            gpu = int(s.labels.get("gpu", "-1"))
        except ValueError:
            gpu = -1
        by_gpu.setdefault(gpu, []).append(s)
    if not by_gpu:
        return None
    boot = time.clock_gettime_ns(time.CLOCK_BOOTTIME)
    profiles = [build_rank_profile(gpu, gpu, group, boottime_ns=boot)
                for gpu, group in sorted(by_gpu.items())]
    return merge_node_profile(profiles, node=node).serialize_gzip()


class DaemonNodeProfileDestination:
    """Destination that writes one merged node-level GPU pprof per flush
    (daemon shape; --merge-node-profiles)."""

    def __init__(self, directory: str, node: str = "") -> None:
        import os

        self.directory = directory
        self.node = node
        self._seq = 0
        os.makedirs(directory, exist_ok=True)

    def write_batch(self, samples) -> None:
        import os

        data = build_node_profile_from_batch(samples, node=self.node)
        if data is None:
            return
        ts = int(time.time())
        path = os.path.join(self.directory,
                            f"{ts}.{self._seq:06d}.node_gpu.pb.gz")
        tmp = path + ".tmp"
        with open(tmp, "wb") as fh:
            fh.write(data)
        os.replace(tmp, path)
        self._seq += 1

    def close(self) -> None:
        pass


class NodeMergeService:
    """Periodic merge driver for embedded per-rank agents."""

    def __init__(self, transport, rank: int, gpu_index: int,
                 node: str = "") -> None:
        self.transport = transport
        self.rank = rank
        self.gpu_index = gpu_index
        self.node = node

    def merge(self, samples) -> Optional[bytes]:
        """All ranks call this with their flush batch; rank 0 returns the
        merged node pprof, others None."""
        rp = build_rank_profile(self.rank, self.gpu_index, samples)
        gathered = self.transport.allgather(rp.to_bytes())
        if self.rank != 0:
            return None
        profiles = [RankProfile.from_bytes(b) for b in gathered]
        return merge_node_profile(profiles, node=self.node).serialize_gzip()
