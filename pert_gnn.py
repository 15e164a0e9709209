"""Training entrypoint — CLI-compatible with the reference pert_gnn.py.

Accepts every reference flag verbatim (reference pert_gnn.py:15-34; the dead
flags --use_sage/--runs/--log_steps are accepted-and-ignored for compat,
SURVEY.md §8 quirk 4) plus additive MI355X-framework flags.

Single process:  python pert_gnn.py --graph_type pert --epochs 100
Multi-GPU DDP:   python -m torch.distributed.run --nproc-per-node 8 \
                     --master-addr 127.0.0.1 pert_gnn.py ...
"""
from __future__ import annotations

import argparse
import os

import torch

from pertgnn.data.collate import BatchLoader
from pertgnn.data.dataset import build_data_list, split_60_20_20
from pertgnn.models import SAGEDeterministic
from pertgnn.parallel import Comm
from pertgnn.train.optim import FlatGradAllReduce, FusedAdam
from pertgnn.train import evaluate, load_checkpoint, save_checkpoint, train_epoch
from pertgnn.utils import JsonlLogger


def build_parser():
    parser = argparse.ArgumentParser(description="Alibaba traces")
    # --- reference flags, verbatim (pert_gnn.py:15-33) ---
    parser.add_argument("--device", type=int, default=0)
    parser.add_argument("--log_steps", type=int, default=1)        # dead (compat)
    parser.add_argument("--use_sage", action="store_true")         # dead (compat)
    parser.add_argument("--num_layers", type=int, default=1)
    parser.add_argument("--hidden_channels", type=int, default=32)
    parser.add_argument("--dropout", type=float, default=0)
    parser.add_argument("--lr", type=float, default=0.0003)
    parser.add_argument("--tau", type=float, default=0.5,
                        help="the quantile level, real number between 0 and 1, 0.5 (median) by default")
    parser.add_argument("--epochs", type=int, default=100)
    parser.add_argument("--runs", type=int, default=10)            # dead (compat)
    parser.add_argument("--batch_size", type=int, default=170)
    parser.add_argument("--graph_type", type=str, default="span", help="span or pert")
    # --- additive framework flags ---
    parser.add_argument("--processed_dir", type=str, default="processed")
    parser.add_argument("--max_traces", type=int, default=100000)
    parser.add_argument("--checkpoint", type=str, default=None,
                        help="path to write/read checkpoints (resume if exists)")
    parser.add_argument("--checkpoint_every", type=int, default=10)
    parser.add_argument("--metrics_jsonl", type=str, default=None)
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--synthetic", action="store_true",
                        help="generate a synthetic dataset in-place if processed/ is missing")
    parser.add_argument("--sync_bn", action="store_true",
                        help="exact-parity BatchNorm under DDP (statistics all-reduced)")
    parser.add_argument("--loss_scale", type=str, default="1.0",
                        help="loss scaling (fp16 mode): a float S for static "
                             "scaling (loss*S backward, optimizer unscales) or "
                             "'dynamic' for GradScaler-style scaling with "
                             "device-side overflow skip/backoff/growth")
    parser.add_argument("--precision", choices=["fp32", "bf16", "fp16"], default="fp32",
                        help="matmul compute precision on GPU")
    parser.add_argument("--hipgraph", action="store_true",
                        help="hipGraph-captured training over HBM-resident fixed-"
                             "composition batches (one seeded shuffle; only the "
                             "batch ORDER reshuffles per epoch — documented "
                             "deviation from the reference's per-epoch re-draw)")
    return parser


def load_artifacts(args, comm=None):
    import joblib
    import pandas as pd

    pdir = args.processed_dir
    if not os.path.isfile(os.path.join(pdir, "tr2data.pt")):
        if args.synthetic:
            # rank 0 generates, everyone else waits at the barrier — all
            # ranks racing to write the same processed/ dir corrupts it
            if comm is None or comm.rank == 0:
                from pertgnn.data.ingest import run_ingest
                from pertgnn.data.synthetic import SyntheticConfig, write_dataset

                root = os.path.dirname(pdir) or "."
                write_dataset(root, SyntheticConfig())
                run_ingest(data_root=os.path.join(root, "data"), processed_dir=pdir, verbose=False)
            if comm is not None:
                comm.barrier()
        else:
            raise FileNotFoundError(
                f"{pdir}/tr2data.pt not found — run `python preprocess.py` first "
                "(or pass --synthetic to generate synthetic data)"
            )
    tr2data = torch.load(os.path.join(pdir, "tr2data.pt"), weights_only=False)
    tr2data = {tr: tr2data[tr] for tr in list(tr2data.keys())[: args.max_traces]}
    runtime2graph = torch.load(
        os.path.join(pdir, f"runtime2{args.graph_type}graph_map.pt"), weights_only=False
    )
    entry2runtimes = joblib.load(os.path.join(pdir, "entry2runtimes.joblib"))
    resource_df = pd.read_csv(os.path.join(pdir, "processed_resource_df.csv"))
    resource_df["msname"] = resource_df["msname"].astype(int)
    return tr2data, entry2runtimes, runtime2graph, resource_df


def main(argv=None):
    args = build_parser().parse_args(argv)
    print(args)
    comm = Comm()
    torch.manual_seed(args.seed + comm.rank)

    if torch.cuda.is_available():
        # reference semantics: single process honors --device (pert_gnn.py:36);
        # under torchrun each rank uses its LOCAL_RANK device
        if comm.distributed:
            device = comm.device
        else:
            device = torch.device(f"cuda:{args.device % torch.cuda.device_count()}")
            torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    tr2data, entry2runtimes, runtime2graph, resource_df = load_artifacts(args, comm)

    cache_path = os.path.join(args.processed_dir, f"full_{args.graph_type}_data_list.pt")
    if os.path.exists(cache_path):
        # accepts our TraceSample caches AND the reference's PyG Data
        # pickles (no PyG needed — shim unpickler, pertgnn/data/pyg_compat)
        from pertgnn.data.pyg_compat import load_data_list_any
        data_list = load_data_list_any(cache_path)
    else:
        data_list = build_data_list(
            tr2data, entry2runtimes, runtime2graph, resource_df, limit=args.max_traces
        )
        if comm.rank == 0:
            torch.save(data_list, cache_path)

    train_list, valid_list, test_list = split_60_20_20(data_list)
    # per-rank shard of the training set; eval sets sharded too (metrics are
    # sum-all-reduced so global averages are exact)
    train_shard = comm.shard(train_list)
    valid_shard = comm.shard(valid_list)
    test_shard = comm.shard(test_list)

    on_gpu = torch.cuda.is_available()
    from pertgnn.data.collate import collate_native
    from pertgnn.data.prefetch import PrefetchLoader

    collate_fn = (lambda samples: collate_native(samples, pin=True)) if on_gpu else collate_native
    train_loader = BatchLoader(train_shard, args.batch_size, shuffle=True,
                               seed=args.seed + comm.rank, collate_fn=collate_fn)
    valid_loader = BatchLoader(valid_shard, args.batch_size, shuffle=False, collate_fn=collate_fn)
    test_loader = BatchLoader(test_shard, args.batch_size, shuffle=False, collate_fn=collate_fn)
    if on_gpu:
        from pertgnn.data.prefetch import ThreadedLoader

        train_loader = PrefetchLoader(ThreadedLoader(train_loader), comm.device)
        valid_loader = PrefetchLoader(ThreadedLoader(valid_loader), comm.device)
        test_loader = PrefetchLoader(ThreadedLoader(test_loader), comm.device)

    # vocab scans (pert_gnn.py:306-328)
    unique_ms_max = max(int(g["ms_id"].max()) for g in runtime2graph.values())
    entry_id_max = max(int(s.entry_id) for s in data_list)
    interface_id_max = max(int(s.edge_attr[:, 0].max()) for s in data_list)
    rpctype_id_max = max(int(s.edge_attr[:, 1].max()) for s in data_list)
    num_features = resource_df.shape[1] - 2  # minus timestamp, msname

    model = SAGEDeterministic(
        num_features + 1, [unique_ms_max + 1], entry_id_max, interface_id_max,
        rpctype_id_max, args.hidden_channels, args.num_layers, args.dropout,
    ).to(device)
    comm.broadcast_module_(model)
    if args.sync_bn and comm.distributed:
        model.enable_sync_bn(comm)
    if torch.cuda.is_available():
        from pertgnn.ops.functional import set_gemm_precision
        set_gemm_precision(args.precision)
    dynamic_scale = str(args.loss_scale).lower() == "dynamic"
    static_scale = 1.0 if dynamic_scale else float(args.loss_scale)
    args.loss_scale = static_scale  # loops use optimizer.scale_loss anyway
    optimizer = FusedAdam(model.parameters(), lr=args.lr,
                          grad_scale=static_scale,
                          dynamic_scale=dynamic_scale)
    engine = FlatGradAllReduce(optimizer, comm) if comm.distributed else None
    log = JsonlLogger(args.metrics_jsonl, rank=comm.rank)

    start_epoch = 1
    if args.checkpoint and os.path.exists(args.checkpoint):
        ep, _ = load_checkpoint(args.checkpoint, model, optimizer, map_location=device)
        start_epoch = ep + 1
        log.print0(f"resumed from {args.checkpoint} at epoch {ep}")

    stepper = None
    if args.hipgraph:
        from pertgnn.train.capture import GraphStepper, make_resident_batches

        resident = make_resident_batches(
            train_shard, args.batch_size, device, args.seed + comm.rank,
            collate_fn=collate_native,
        )
        stepper = GraphStepper(model, optimizer, engine, comm, args.tau,
                               device, resident, loss_scale=args.loss_scale,
                               seed=args.seed + comm.rank)
        mode = stepper.capture()
        log.print0(f"# hipgraph stepping mode: {mode} "
                   f"({len(resident)} resident batches)")
        # eval sets collated once and held resident too (order preserved)
        valid_loader = make_resident_batches(
            valid_shard, args.batch_size, device, 0, collate_fn=collate_native,
            shuffle=False)
        test_loader = make_resident_batches(
            test_shard, args.batch_size, device, 0, collate_fn=collate_native,
            shuffle=False)

    import time as _time

    def run_train_epoch(epoch_stats):
        if stepper is None:
            return train_epoch(
                model, train_loader, optimizer, args.tau, device, engine=engine,
                comm=comm, stats_out=epoch_stats, loss_scale=args.loss_scale,
            )
        t0 = _time.perf_counter()
        loss_sum, mape_sum, n = stepper.run_epoch_sums()
        elapsed = _time.perf_counter() - t0
        if comm.distributed:
            loss_sum = comm.all_reduce_scalar(loss_sum)
            mape_sum = comm.all_reduce_scalar(mape_sum)
            n = int(comm.all_reduce_scalar(float(n)))
            elapsed = comm.all_reduce_scalar(elapsed, op="max")
        epoch_stats.update({
            "epoch_s": elapsed,
            "graphs_per_s": n / max(elapsed, 1e-9),
        })
        n = max(n, 1)
        return loss_sum / n, mape_sum / n

    for epoch in range(start_epoch, args.epochs + 1):
        epoch_stats: dict = {}
        train_mae, train_mape = run_train_epoch(epoch_stats)
        valid_mae, valid_mape, valid_q = evaluate(model, valid_loader, args.tau, device, comm=comm)
        test_mae, test_mape, test_q = evaluate(model, test_loader, args.tau, device, comm=comm)
        # reference epoch line format (pert_gnn.py:348-350)
        log.print0(
            f"Epoch: {epoch}, Train: {train_mae}, Test mae: {test_mae}, "
            f"Train mape: {train_mape}, Test mape: {test_mape}, Test q95 loss: {test_q}"
        )
        log.log({
            "epoch": epoch, "train_loss": train_mae, "train_mape": train_mape,
            **epoch_stats,
            "valid_mae": valid_mae, "valid_mape": valid_mape, "valid_q": valid_q,
            "test_mae": test_mae, "test_mape": test_mape, "test_q": test_q,
        })
        if args.checkpoint and epoch % args.checkpoint_every == 0:
            save_checkpoint(args.checkpoint, model, optimizer, epoch, comm=comm)
    if args.checkpoint:
        save_checkpoint(args.checkpoint, model, optimizer, args.epochs, comm=comm)
    log.close()
    comm.finalize()


if __name__ == "__main__":
    main()
