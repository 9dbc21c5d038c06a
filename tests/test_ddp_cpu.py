"""Data-parallel engine over gloo (2 ranks): bucketed all-reduce gradients
must equal single-process full-batch gradients (the reference left its DP
reducer as a stub — include/nn/reducer.hpp)."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from tnn_amd.nn import LayerBuilder, CrossEntropyLoss, AdamW


def _model(seed=3):
    torch.manual_seed(seed)
    # BN-free on purpose: per-shard batch statistics differ from
    # full-batch BN by construction (same as torch DDP), which would
    # break the exact single-process parity this test asserts
    return (LayerBuilder((6, 6, 2))
            .conv2d(4, 3, 3, 1, 1, 1, 1, True, "c1")
            .activation("relu", "r1")
            .flatten()
            .dense(3, True, "fc")
            .build("ddp_test"))


def _data(seed=11):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(16, 6, 6, 2, generator=g)
    y = torch.randint(0, 3, (16,), generator=g)
    return x, y


def _ddp_worker(rank, world, tmpdir):
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{tmpdir}/pg",
                            rank=rank, world_size=world)
    from tnn_amd.parallel import Communicator, DataParallelEngine
    comm = Communicator()
    model = _model(seed=3 + rank)  # different init; engine broadcasts rank 0's
    engine = DataParallelEngine(model, comm, bucket_bytes=256)
    crit = CrossEntropyLoss()
    opt = AdamW(model.parameters(), lr=1e-2)
    x, y = _data()
    shard_x = x.chunk(world)[rank]
    shard_y = y.chunk(world)[rank]
    for _ in range(3):
        loss = crit(model(shard_x), shard_y)
        opt.zero_grad()
        engine.train_step(loss, opt)
    torch.save({k: v.clone() for k, v in model.state_dict().items()},
               os.path.join(tmpdir, f"sd_{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_matches_full_batch(tmp_path):
    # reference: single process on the full batch (grad = mean over batch
    # == mean of shard means with equal shards)
    model = _model(seed=3)
    crit = CrossEntropyLoss()
    opt = AdamW(model.parameters(), lr=1e-2)
    x, y = _data()
    for _ in range(3):
        loss = crit(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()

    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_ddp_worker, args=(r, 2, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    sd0 = torch.load(tmp_path / "sd_0.pt", weights_only=False)
    sd1 = torch.load(tmp_path / "sd_1.pt", weights_only=False)
    ref = model.state_dict()
    for k in ref:
        assert torch.allclose(sd0[k], sd1[k], atol=1e-6), k  # ranks agree
        assert torch.allclose(sd0[k], ref[k], atol=1e-4), \
            (k, (sd0[k] - ref[k]).abs().max())
