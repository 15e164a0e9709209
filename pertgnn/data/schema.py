"""Alibaba cluster-trace-microservices-v2021 schema constants.

Call-graph CSV columns (reference preprocess.py:296-298 example row) and
resource CSV columns (reference preprocess.py:227-242).
"""

CALL_COLUMNS = [
    "traceid", "timestamp", "rpcid", "um", "rpctype", "dm", "interface", "rt",
]

RESOURCE_COLUMNS = [
    "timestamp", "msname", "instance_cpu_usage", "instance_memory_usage",
]

# resource features after the (timestamp, msname) group-agg
# (preprocess.py:237-240): {cpu,mem} x {max,min,mean,median}
RESOURCE_FEATURE_COLUMNS = [
    f"{base}_{agg}"
    for base in ("instance_cpu_usage", "instance_memory_usage")
    for agg in ("max", "min", "mean", "median")
]

NUM_RESOURCE_FEATURES = len(RESOURCE_FEATURE_COLUMNS)  # 8
# +1 missing-indicator => in_channels = 9 (pert_gnn.py:330-334)
IN_CHANNELS = NUM_RESOURCE_FEATURES + 1

TS_BUCKET_MS = 30000  # trace timestamps floored to 30 s buckets (preprocess.py:39)
