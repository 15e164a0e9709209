from . import collate, dataset, graphs, ingest, schema, synthetic
from .collate import BatchLoader, GraphBatch, collate as collate_batch
from .dataset import TraceSample, build_data_list, split_60_20_20

__all__ = [
    "collate", "dataset", "graphs", "ingest", "schema", "synthetic",
    "BatchLoader", "GraphBatch", "collate_batch",
    "TraceSample", "build_data_list", "split_60_20_20",
]
