"""Multi-process distributed paths over gloo (world_size=2, CPU) — the
correct-by-construction check for what RCCL runs on the GPU node."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _dist_worker(rank, world, fn_name, port, q):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch.distributed as dist
    from scalerl_amd.parallel.dist import (all_reduce_flat, broadcast_flat,
                                           init_distributed)
    init_distributed("gloo")
    try:
        if fn_name == "allreduce":
            g = torch.full((100,), float(rank + 1))
            all_reduce_flat(g, average=True)
            q.put((rank, g[0].item()))  # mean of 1,2 = 1.5
        elif fn_name == "broadcast":
            t = torch.full((10,), float(rank * 7))
            broadcast_flat(t, src=0)
            q.put((rank, t[0].item()))  # everyone gets rank0's 0.0
        elif fn_name == "a3cgpu":
            from scalerl_amd.config import A3CGpuArguments
            from scalerl_amd.runtime.a3c_gpu import A3CGpuTrainer
            args = A3CGpuArguments(num_actors=1, envs_per_actor=4,
                                   rollout_steps=6, slots_per_batch=1,
                                   device="cpu", seed=20 + rank,
                                   disable_checkpoint=True)
            tr = A3CGpuTrainer(args)
            tr.start_actors()
            tr.setup_learner()
            try:
                tr.train_iteration()
                q.put((rank, float(tr.flat.flat.sum())))
            finally:
                tr.shutdown()
        elif fn_name == "ddppo":
            from scalerl_amd.config import DDPPOArguments
            from scalerl_amd.runtime.ppo import DDPPOTrainer
            args = DDPPOArguments(rollout_length=6, num_envs=2, ppo_epochs=1,
                                  num_minibatches=1, device="cpu",
                                  seed=10 + rank)
            tr = DDPPOTrainer(args)
            s = tr.train_iteration()
            # ranks end bit-identical after synchronized updates
            q.put((rank, float(tr.flat.flat.sum())))
    finally:
        dist.destroy_process_group()


def _run(fn_name, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dist_worker, args=(r, 2, fn_name, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    out = {}
    for _ in range(2):
        rank, val = q.get(timeout=180)
        out[rank] = val
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
            pytest.fail("worker did not exit")
    return out


def test_allreduce_flat_gloo():
    out = _run("allreduce", 29611)
    assert out[0] == pytest.approx(1.5)
    assert out[1] == pytest.approx(1.5)


def test_broadcast_flat_gloo():
    out = _run("broadcast", 29612)
    assert out[0] == 0.0 and out[1] == 0.0


def test_ddppo_two_ranks_stay_in_sync():
    out = _run("ddppo", 29613)
    assert out[0] == pytest.approx(out[1], rel=1e-6)


def test_a3c_gpu_trainer_two_ranks_stay_in_sync():
    """A3CGpuTrainer's flat-grad all-reduce path keeps 2 CPU ranks
    bit-identical after a synchronized update (config-2 DP over gloo)."""
    out = _run("a3cgpu", 29731)
    assert out[0] == pytest.approx(out[1], rel=0, abs=0)
