"""Metrics (reference: torchrec/metrics/__init__.py)."""

from torchrec_amd.metrics.metric_module import (  # noqa: F401
    RecMetricModule,
    ThroughputMetric,
)
from torchrec_amd.metrics.rec_metric import (  # noqa: F401
    AccuracyMetric,
    AUCMetric,
    CalibrationMetric,
    MSEMetric,
    NEMetric,
    PrecisionMetric,
    RecallMetric,
    RecTaskInfo,
)
