from .recorder import (InfluxEncoder, JsonEncoder, MetricsRecorder,
                       NodeMetrics, PoolMetrics, QosPricingTable,
                       SchedulerMetrics, WorkerMetrics)
from .tsdb import TSDB, parse_influx_line

__all__ = ["MetricsRecorder", "WorkerMetrics", "NodeMetrics", "PoolMetrics",
           "SchedulerMetrics", "QosPricingTable", "InfluxEncoder",
           "JsonEncoder", "TSDB", "parse_influx_line"]
