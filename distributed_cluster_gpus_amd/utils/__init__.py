from .logging import get_logger, LOGGER_NAME
from .csvlog import ClusterLogWriter, JobLogWriter, CLUSTER_COLUMNS, JOB_COLUMNS
from .timers import ThroughputMeter

__all__ = ["get_logger", "LOGGER_NAME", "ClusterLogWriter", "JobLogWriter",
           "CLUSTER_COLUMNS", "JOB_COLUMNS", "ThroughputMeter"]
