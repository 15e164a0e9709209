"""Offline ingest entrypoint — CLI-compatible with the reference preprocess.py
(`python preprocess.py` reads data/MSCallGraph + data/MSResource and writes
processed/ artifacts; see pertgnn/data/ingest.py for the pipeline)."""
import argparse

from pertgnn.data.ingest import run_ingest

if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--data_root", default="data")
    ap.add_argument("--processed_dir", default="processed")
    ap.add_argument("--min_occurence", type=int, default=100)
    args = ap.parse_args()
    run_ingest(args.data_root, args.processed_dir, min_occurence=args.min_occurence)
