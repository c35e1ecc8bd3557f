"""Multi-process CPU tests of the data-parallel engine (gloo, world_size=2)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from mine_amd.config import default_config


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _run_ranks(fn, world_size=2, timeout=300):
    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = []
    for rank in range(world_size):
        p = ctx.Process(target=_worker, args=(fn, rank, world_size, port))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout)
    for p in procs:
        assert p.exitcode == 0, f"rank exited with {p.exitcode}"


def _worker(fn, rank, world_size, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        fn(rank, world_size)
    finally:
        dist.destroy_process_group()


# ---------------------------------------------------------------------------

def _grad_engine_averages(rank, world_size):
    from mine_amd.parallel import GradAllReduceEngine
    torch.manual_seed(0)  # same init on both ranks
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    engine = GradAllReduceEngine([model], bucket_mb=0.0001)  # force many buckets
    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(4, 8)
    engine.zero_grad()
    model(x).pow(2).mean().backward()
    engine.finish_step()

    # expectation: average of per-rank gradients, computed locally
    ref_grads = []
    torch.manual_seed(0)
    ref = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                              torch.nn.Linear(16, 4))
    acc = [torch.zeros_like(p) for p in ref.parameters()]
    for r in range(world_size):
        for p in ref.parameters():
            p.grad = None
        torch.manual_seed(100 + r)
        xr = torch.randn(4, 8)
        ref(xr).pow(2).mean().backward()
        for a, p in zip(acc, ref.parameters()):
            a += p.grad / world_size
    for p, a in zip(model.parameters(), acc):
        torch.testing.assert_close(p.grad, a, rtol=1e-5, atol=1e-6)


def test_grad_engine_averages_across_ranks():
    _run_ranks(_grad_engine_averages)


def _params_stay_synced(rank, world_size):
    from mine_amd.data import SyntheticMPIDataset, collate_src_tgt
    from mine_amd.engine import SynthesisTask
    from mine_amd.config import RuntimeState, default_config as dc

    cfg = dc(**{
        "data.name": "synthetic", "data.img_h": 64, "data.img_w": 64,
        "mpi.num_bins_coarse": 4, "data.per_gpu_batch_size": 2,
        "data.visible_point_count": 16, "training.amp_dtype": "fp32",
    })
    torch.manual_seed(1234 + rank)  # DIFFERENT init; broadcast must fix it
    state = RuntimeState(global_rank=rank, world_size=world_size)
    task = SynthesisTask(cfg, state=state, device="cpu")

    # params identical after the construction-time broadcast
    w = task.decoder.dispconvs["0"].conv.weight.detach().clone()
    ws = [torch.zeros_like(w) for _ in range(world_size)]
    dist.all_gather(ws, w)
    torch.testing.assert_close(ws[0], ws[1])

    # one step with per-rank data; params must remain identical
    ds = SyntheticMPIDataset(cfg, length=4)
    items = collate_src_tgt([ds[2 * rank], ds[2 * rank + 1]])
    task.train_step(items)
    w = task.decoder.dispconvs["0"].conv.weight.detach().clone()
    ws = [torch.zeros_like(w) for _ in range(world_size)]
    dist.all_gather(ws, w)
    torch.testing.assert_close(ws[0], ws[1], rtol=1e-6, atol=1e-7)


def test_engine_step_keeps_ranks_in_sync():
    _run_ranks(_params_stay_synced, timeout=600)


def _bucket_boundaries(rank, world_size):
    from mine_amd.parallel import GradAllReduceEngine
    torch.manual_seed(0)
    model = torch.nn.Sequential(
        *[torch.nn.Linear(256, 256) for _ in range(8)])  # 8 x 256 KB weights
    engine = GradAllReduceEngine([model], bucket_mb=0.5)
    assert len(engine.buckets) >= 4
    total = sum(b.flat.numel() for b in engine.buckets)
    assert total == sum(p.numel() for p in model.parameters())
    # grads are views into the buckets
    engine.zero_grad()
    model(torch.randn(2, 256)).sum().backward()
    engine.finish_step()
    for b in engine.buckets:
        assert b.handle is None


def test_bucket_partitioning():
    _run_ranks(_bucket_boundaries)


def _bf16_allreduce_close_to_fp32(rank, world_size):
    from mine_amd.parallel import GradAllReduceEngine
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Tanh(),
                                torch.nn.Linear(32, 4))
    ref = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Tanh(),
                              torch.nn.Linear(32, 4))
    ref.load_state_dict(model.state_dict())

    eng = GradAllReduceEngine([model], bucket_mb=0.001,
                              allreduce_dtype=torch.bfloat16)
    eng_ref = GradAllReduceEngine([ref], bucket_mb=0.001,
                                  broadcast_params=False)
    torch.manual_seed(7 + rank)
    x = torch.randn(8, 16)
    for e, m in ((eng, model), (eng_ref, ref)):
        e.zero_grad()
        m(x).pow(2).mean().backward()
        e.finish_step()
    for p, q in zip(model.parameters(), ref.parameters()):
        # bf16 wire precision: ~0.4% relative
        torch.testing.assert_close(p.grad, q.grad, rtol=2e-2, atol=2e-3)


def test_bf16_gradient_compression():
    _run_ranks(_bf16_allreduce_close_to_fp32)


def _resume_state_broadcast(rank, world_size):
    """Rank 0 holds a checkpoint the other rank cannot see: after
    construction every rank must agree on the resume meta AND the Adam
    state (round-1 ADVICE items 1-2 — a mismatch desyncs epoch counts
    and collective schedules)."""
    import tempfile
    from mine_amd.config import default_config, RuntimeState
    from mine_amd.engine import SynthesisTask
    from mine_amd.engine.checkpoint import save_checkpoint

    base = {
        "data.name": "synthetic", "data.img_h": 64, "data.img_w": 64,
        "mpi.num_bins_coarse": 4, "data.per_gpu_batch_size": 1,
        "data.visible_point_count": 8, "training.amp_dtype": "fp32",
        "data.synthetic_length": 2,
    }
    # a rank-private path: only rank 0 writes/sees the file
    tmpdir = tempfile.mkdtemp(prefix=f"resume_r{rank}_")
    ckpt = os.path.join(tmpdir, "checkpoint_latest.pth")
    if rank == 0:
        # build the checkpoint WITHOUT SynthesisTask (its construction
        # runs collectives, which rank 1 would not mirror here)
        from mine_amd.models import MPIDecoder, ResNetEncoder
        backbone = ResNetEncoder(num_layers=50)
        decoder = MPIDecoder(num_ch_enc=backbone.num_ch_enc,
                             pos_encoding_multires=10)
        opt = torch.optim.Adam(
            [{"params": backbone.parameters(), "lr": 1e-3},
             {"params": decoder.parameters(), "lr": 1e-3}])
        loss = sum(p.sum() for p in decoder.dispconvs.parameters())
        loss.backward()
        opt.step()
        save_checkpoint(ckpt, backbone, decoder, opt,
                        meta={"epoch": 7, "global_step": 4321})
    dist.barrier()

    cfg = default_config(**base, **{
        "training.pretrained_checkpoint_path": ckpt if rank == 0 else
        os.path.join(tmpdir, "missing", "checkpoint_latest.pth")})
    state = RuntimeState(global_rank=rank, world_size=world_size)
    task = SynthesisTask(cfg, state=state)

    # meta agreed across ranks
    assert task._restored_meta.get("epoch") == 7
    assert task._restored_meta.get("global_step") == 4321
    # Adam moments broadcast to the rank that missed the file
    sd = task.optimizer.state_dict()
    n_state = len(sd["state"])
    t = torch.tensor([float(n_state)])
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    assert int(t.item()) == n_state and n_state > 0
    some = next(iter(sd["state"].values()))
    s0 = some["exp_avg"].flatten()[:8].clone()
    gathered = [torch.zeros_like(s0) for _ in range(world_size)]
    dist.all_gather(gathered, s0)
    torch.testing.assert_close(gathered[0], gathered[1])


def test_resume_state_broadcast():
    _run_ranks(_resume_state_broadcast)
