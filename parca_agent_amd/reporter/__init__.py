from .reporter import (
    Destination,
    PendingSample,
    Reporter,
    ReporterMetrics,
    build_arrow_record,
)
from .destinations import (
    LocalStoreDestination,
    OfflineLogDestination,
    read_offline_log,
    samples_to_pprof,
)

__all__ = [
    "Destination",
    "PendingSample",
    "Reporter",
    "ReporterMetrics",
    "build_arrow_record",
    "LocalStoreDestination",
    "OfflineLogDestination",
    "read_offline_log",
    "samples_to_pprof",
]
