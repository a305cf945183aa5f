"""Hand-rolled codecs for the Parca gRPC message types.

The reference links buf.build-generated Go stubs for
parca.profilestore.v1alpha1 / parca.debuginfo.v1alpha1 /
parca.telemetry.v1alpha1 (reference: main.go:402-403, 295). This image
has no protoc, so the messages are encoded with the in-repo protobuf
writer. Message/field shapes follow the public parca protos as exercised
by the reference call sites (parca_reporter.go:1704/2176,
parca_uploader.go:208-406, oom/oomprof.go:100-116); field numbers are the
upstream ones. A fake in-process server in tests uses the same codecs,
keeping both directions covered.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List

from ..pprof.proto import Writer, iter_fields

# -- profilestore ----------------------------------------------------------

PROFILE_STORE_SERVICE = "parca.profilestore.v1alpha1.ProfileStoreService"
AGENTS_SERVICE = "parca.profilestore.v1alpha1.AgentsService"


@dataclass
class Label:
    name: str
    value: str


def encode_write_arrow_request(ipc_buffer: bytes) -> bytes:
    w = Writer()
    w.bytes(1, ipc_buffer)
    return w.getvalue()


def decode_write_arrow_request(data: bytes) -> bytes:
    for f, _wt, v in iter_fields(data):
        if f == 1:
            return v
    return b""


def encode_write_request(record: bytes) -> bytes:
    w = Writer()
    w.bytes(1, record)
    return w.getvalue()


def decode_write_response(data: bytes) -> bytes:
    """WriteResponse.record: the server's request for unknown stacktrace
    IDs in the v1 two-phase protocol (parca_reporter.go:1667-1803)."""
    for f, _wt, v in iter_fields(data):
        if f == 1:
            return v
    return b""


decode_write_request = decode_write_arrow_request


def encode_write_raw_request(series_labels: List[List[Label]],
                             raw_profiles: List[List[bytes]]) -> bytes:
    """WriteRawRequest{series=2}; RawProfileSeries{labels=1, samples=2};
    LabelSet{labels=1}; Label{name=1,value=2}; RawSample{raw_profile=1}."""
    w = Writer()
    for labels, profiles in zip(series_labels, raw_profiles):
        series = Writer()
        labelset = Writer()
        for lbl in labels:
            lw = Writer()
            lw.string(1, lbl.name)
            lw.string(2, lbl.value)
            labelset.message(1, lw)
        series.message(1, labelset)
        for prof in profiles:
            sw = Writer()
            sw.bytes(1, prof)
            series.message(2, sw)
        w.message(2, series)
    return w.getvalue()


def decode_write_raw_request(data: bytes):
    out = []
    for f, _wt, v in iter_fields(data):
        if f != 2:
            continue
        labels, profiles = [], []
        for sf, _swt, sv in iter_fields(v):
            if sf == 1:
                for lf, _lwt, lv in iter_fields(sv):
                    if lf == 1:
                        name = value = ""
                        for llf, _llwt, llv in iter_fields(lv):
                            if llf == 1:
                                name = llv.decode()
                            elif llf == 2:
                                value = llv.decode()
                        labels.append(Label(name, value))
            elif sf == 2:
                for pf, _pwt, pv in iter_fields(sv):
                    if pf == 1:
                        profiles.append(pv)
        out.append((labels, profiles))
    return out


# -- debuginfo -------------------------------------------------------------

DEBUGINFO_SERVICE = "parca.debuginfo.v1alpha1.DebuginfoService"

# DebuginfoType enum
DEBUGINFO_TYPE_UNSPECIFIED = 0  # executable/debuginfo itself
DEBUGINFO_TYPE_EXECUTABLE = 1
DEBUGINFO_TYPE_SOURCES = 2

# UploadInstructions.upload_strategy enum
UPLOAD_STRATEGY_UNSPECIFIED = 0
UPLOAD_STRATEGY_GRPC = 1
UPLOAD_STRATEGY_SIGNED_URL = 2


@dataclass
class UploadInstructions:
    build_id: str = ""
    upload_id: str = ""
    upload_strategy: int = UPLOAD_STRATEGY_GRPC
    signed_url: str = ""
    type: int = DEBUGINFO_TYPE_UNSPECIFIED


def encode_should_initiate_upload_request(build_id: str, hash_: str = "",
                                          force: bool = False,
                                          type_: int = 0) -> bytes:
    w = Writer()
    w.string(1, build_id)
    w.string(2, hash_)
    w.bool(3, force)
    w.varint(4, type_)
    return w.getvalue()


def decode_should_initiate_upload_request(data: bytes):
    build_id = hash_ = ""
    force = False
    type_ = 0
    for f, _wt, v in iter_fields(data):
        if f == 1:
            build_id = v.decode()
        elif f == 2:
            hash_ = v.decode()
        elif f == 3:
            force = bool(v)
        elif f == 4:
            type_ = v
    return build_id, hash_, force, type_


def encode_should_initiate_upload_response(should: bool,
                                           reason: str = "") -> bytes:
    w = Writer()
    w.bool(1, should)
    w.string(2, reason)
    return w.getvalue()


def decode_should_initiate_upload_response(data: bytes):
    should = False
    reason = ""
    for f, _wt, v in iter_fields(data):
        if f == 1:
            should = bool(v)
        elif f == 2:
            reason = v.decode()
    return should, reason


def encode_initiate_upload_request(build_id: str, hash_: str, size: int,
                                   force: bool = False,
                                   type_: int = 0) -> bytes:
    w = Writer()
    w.string(1, build_id)
    w.string(2, hash_)
    w.varint(3, size)
    w.bool(4, force)
    w.varint(5, type_)
    return w.getvalue()


def decode_initiate_upload_request(data: bytes):
    build_id = hash_ = ""
    size = 0
    force = False
    type_ = 0
    for f, _wt, v in iter_fields(data):
        if f == 1:
            build_id = v.decode()
        elif f == 2:
            hash_ = v.decode()
        elif f == 3:
            size = v
        elif f == 4:
            force = bool(v)
        elif f == 5:
            type_ = v
    return build_id, hash_, size, force, type_


def encode_upload_instructions(ins: UploadInstructions) -> Writer:
    w = Writer()
    w.string(1, ins.build_id)
    w.string(2, ins.upload_id)
    w.varint(3, ins.upload_strategy)
    w.string(4, ins.signed_url)
    w.varint(5, ins.type)
    return w


def encode_initiate_upload_response(ins: UploadInstructions) -> bytes:
    w = Writer()
    w.message(1, encode_upload_instructions(ins))
    return w.getvalue()


def decode_initiate_upload_response(data: bytes) -> UploadInstructions:
    ins = UploadInstructions()
    for f, _wt, v in iter_fields(data):
        if f == 1:
            for sf, _swt, sv in iter_fields(v):
                if sf == 1:
                    ins.build_id = sv.decode()
                elif sf == 2:
                    ins.upload_id = sv.decode()
                elif sf == 3:
                    ins.upload_strategy = sv
                elif sf == 4:
                    ins.signed_url = sv.decode()
                elif sf == 5:
                    ins.type = sv
    return ins


def encode_upload_request_info(upload_id: str, type_: int = 0) -> bytes:
    """UploadRequest{ oneof data { UploadInfo info = 1; bytes chunk_data
    = 2; } }; UploadInfo{build_id=1? upload_id=2?} — upstream uses
    UploadInfo{upload_id=1, type=2}."""
    w = Writer()
    info = Writer()
    info.string(1, upload_id)
    info.varint(2, type_)
    w.message(1, info)
    return w.getvalue()


def encode_upload_request_chunk(chunk: bytes) -> bytes:
    w = Writer()
    w.bytes(2, chunk)
    return w.getvalue()


def decode_upload_request(data: bytes):
    """Returns ("info", upload_id) or ("chunk", bytes)."""
    for f, _wt, v in iter_fields(data):
        if f == 1:
            upload_id = ""
            for sf, _swt, sv in iter_fields(v):
                if sf == 1:
                    upload_id = sv.decode()
            return ("info", upload_id)
        if f == 2:
            return ("chunk", v)
    return ("empty", b"")


def encode_upload_response(build_id: str, size: int) -> bytes:
    w = Writer()
    w.string(1, build_id)
    w.varint(2, size)
    return w.getvalue()


def decode_upload_response(data: bytes):
    build_id = ""
    size = 0
    for f, _wt, v in iter_fields(data):
        if f == 1:
            build_id = v.decode()
        elif f == 2:
            size = v
    return build_id, size


def encode_mark_upload_finished_request(build_id: str, upload_id: str,
                                        type_: int = 0) -> bytes:
    w = Writer()
    w.string(1, build_id)
    w.string(2, upload_id)
    w.varint(3, type_)
    return w.getvalue()


def decode_mark_upload_finished_request(data: bytes):
    build_id = upload_id = ""
    type_ = 0
    for f, _wt, v in iter_fields(data):
        if f == 1:
            build_id = v.decode()
        elif f == 2:
            upload_id = v.decode()
        elif f == 3:
            type_ = v
    return build_id, upload_id, type_


# -- telemetry -------------------------------------------------------------

TELEMETRY_SERVICE = "parca.telemetry.v1alpha1.TelemetryService"


def encode_report_panic_request(stderr: str, metadata: dict) -> bytes:
    """ReportPanicRequest{stderr=1, metadata map<string,string>=2}."""
    w = Writer()
    w.string(1, stderr)
    for k, v in sorted(metadata.items()):
        entry = Writer()
        entry.string(1, k)
        entry.string(2, v)
        w.message(2, entry)
    return w.getvalue()


def decode_report_panic_request(data: bytes):
    stderr = ""
    metadata = {}
    for f, _wt, v in iter_fields(data):
        if f == 1:
            stderr = v.decode()
        elif f == 2:
            k = vv = ""
            for sf, _swt, sv in iter_fields(v):
                if sf == 1:
                    k = sv.decode()
                elif sf == 2:
                    vv = sv.decode()
            metadata[k] = vv
    return stderr, metadata
