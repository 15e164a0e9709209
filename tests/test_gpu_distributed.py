"""GPU DDP glue test: 2 ranks sharing cuda:0 (gloo transports the flat-grad
buckets; the compute path is the full HIP kernel stack).  Validates the
engine + sync-BN + FusedAdam end-to-end on device — the schedule the driver's
8-GPU RCCL run exercises, minus the RCCL transport itself."""
import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("no GPU available", allow_module_level=True)


def _worker(rank, world, port, q):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK="0",
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    import bench as bench_mod
    from pertgnn.models import SAGEDeterministic
    from pertgnn.ops import functional as F
    from pertgnn.parallel import Comm
    from pertgnn.train.optim import FlatGradAllReduce, FusedAdam

    comm = Comm(backend="gloo")  # both ranks share the one GPU on the box
    dev = torch.device("cuda:0")
    torch.manual_seed(7)
    batches, stats = bench_mod.build_synthetic_batches(1, 8, seed=50 + rank, device=dev)
    for key in ("cat_max", "entry_max", "ifc_max", "rpc_max"):
        stats[key] = int(comm.all_reduce_scalar(float(stats[key]), op="max"))
    torch.manual_seed(7)
    model = SAGEDeterministic(9, [stats["cat_max"] + 1], stats["entry_max"],
                              stats["ifc_max"], stats["rpc_max"], 64, 2, 0.0).to(dev)
    comm.broadcast_module_(model)
    model.enable_sync_bn(comm)
    opt = FusedAdam(model.parameters(), lr=1e-3)
    engine = FlatGradAllReduce(opt, comm)
    model.train()
    b = batches[0]
    for _ in range(2):
        opt.zero_grad()
        engine.reset()
        gp, _ = model(b.x, b.cat_X, b.edge_index, b.edge_attr,
                      b.pattern_num_nodes, b.rt_probs, b.entry_id, b.batch,
                      csr=b.csr, num_graphs=b.num_graphs)
        loss = F.quantile_loss(b.y, gp.flatten(), 0.5)
        loss.backward()
        engine.finalize()
        opt.step()
    torch.cuda.synchronize()
    q.put((rank, opt.flat_param.detach().cpu().numpy().copy(), float(loss.detach())))
    comm.barrier()
    comm.finalize()


@pytest.mark.timeout(300)
def test_rccl_ws1_allreduce_and_graph_capture():
    """RCCL transport on hardware + hipGraph x collective interaction: a
    world-size-1 nccl (=RCCL on ROCm) group runs a real all-reduce, then an
    all-reduce is CAPTURED inside a CUDA graph and replayed.  This is the
    mechanism bench.py's --graph-mode full relies on at world_size 8; the
    split mode needs only plain (uncaptured) collectives."""
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = "29541"
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        t = torch.ones(1 << 20, device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert float(t.sum()) == float(1 << 20)

        x = torch.ones(1 << 20, device="cuda:0")
        dist.all_reduce(x)  # collective path warm before capture
        x.fill_(1.0)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            dist.all_reduce(x)
            x.mul_(2.0)
        x.fill_(1.0)
        g.replay()
        torch.cuda.synchronize()
        assert float(x[0]) == 2.0
        g.replay()
        torch.cuda.synchronize()
        assert float(x[0]) == 4.0
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_bench_split_mode_gpu():
    """bench.py --graph-mode split on hardware: compute-only hipGraph capture
    with the eager comm/optimizer tail (the world_size>1 default)."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1",
         "--batch-size", "64", "--n-batches", "2", "--layers", "2",
         "--hidden", "256", "--graph-mode", "split", "--mae-epochs", "0",
         "--vocab", "small"],
        capture_output=True, text=True, timeout=540, cwd=root,
    )
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-2000:])
    rec = json.loads(out.stdout.strip().splitlines()[-1])
    assert rec["config"]["graph_mode"] == "split"
    assert rec["value"] > 0


@pytest.mark.timeout(300)
def test_two_rank_ddp_on_one_gpu():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, 29531, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    results.sort(key=lambda t: t[0])
    (_, p0, l0), (_, p1, l1) = results
    import numpy as np
    assert np.isfinite(l0) and np.isfinite(l1)
    assert np.array_equal(p0, p1), "ranks diverged after averaged-grad steps"
