"""Multi-process distributed tests (gloo backend, world_size 2, CPU).

These cover the code paths the 8-GPU RCCL runs exercise: fused flat gradient
all-reduce for CHSAC-AF DP training, metric reductions, module broadcast, and
replica sharding — correct by construction so the driver's multi-GPU scaling
bench works without a GPU here (brief: gloo, world_size>1, 127.0.0.1).
"""
import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    sys.path.insert(0, REPO)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _worker_allreduce_grads(rank, world, port, q):
    _init(rank, world, port)
    try:
        import torch.distributed as dist
        from distributed_cluster_gpus_amd.parallel.dist import (
            allreduce_gradients, allreduce_scalar, broadcast_module)
        torch.manual_seed(100 + rank)  # deliberately different per rank
        m = torch.nn.Linear(8, 4)
        broadcast_module(m)  # now equal across ranks
        x = torch.full((2, 8), float(rank + 1))
        m(x).sum().backward()
        allreduce_gradients(list(m.parameters()))
        g = m.weight.grad.clone()
        # expected: average of per-rank grads; rank r grad_w = sum_b x_b = 2*(r+1)
        expect = sum(2.0 * (r + 1) for r in range(world)) / world
        ok_grad = torch.allclose(g, torch.full_like(g, expect))
        s = allreduce_scalar(float(rank + 1), device=torch.device("cpu"))
        ok_scalar = abs(s - sum(r + 1 for r in range(world))) < 1e-9
        w0 = m.weight.detach().clone()
        q.put((rank, bool(ok_grad), bool(ok_scalar), w0.numpy()))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, False, False, str(e)))


def _worker_ddp_sac(rank, world, port, q):
    """Two ranks train CHSAC-AF with the all-reduce hook on DIFFERENT batches;
    parameters must remain identical across ranks after N updates."""
    _init(rank, world, port)
    try:
        import torch.distributed as dist
        from distributed_cluster_gpus_amd.rl.agent import (CHSACAgent,
                                                           CHSACAgentConfig)
        from distributed_cluster_gpus_amd.rl.replay import ReplayRing
        from distributed_cluster_gpus_amd.parallel.dist import broadcast_module
        torch.manual_seed(1234)  # same init
        agent = CHSACAgent(CHSACAgentConfig(
            obs_dim=13, n_dc=4, n_g_choices=4,
            constraints={"latency_p99": 100.0}, device="cpu"))
        for mod in (agent.encoder, agent.actor, agent.critic,
                    agent.algo.target_critic):
            broadcast_module(mod)
        agent.enable_ddp()
        # different data per rank
        torch.manual_seed(500 + rank)
        ring = ReplayRing(capacity=256, obs_dim=13, n_costs=1,
                          cost_names=["latency_p99"], n_dc=4, n_g=4,
                          seed=900 + rank)
        rng = np.random.default_rng(rank)
        for _ in range(256):
            ring.add(s=rng.normal(size=13).astype(np.float32),
                     s_next=rng.normal(size=13).astype(np.float32),
                     a_dc=int(rng.integers(4)), a_g=int(rng.integers(4)),
                     r=float(rng.normal()), costs={"latency_p99": 50.0},
                     done=True)
        # NOTE: sampling/action noise differs per rank; grads are synced, so
        # the OPTIMIZER steps stay identical only if the loss grads are the
        # only source of parameter change — they are (Adam on synced grads).
        for _ in range(3):
            agent.train_step(ring.sample(64))
        w = torch.cat([p.detach().reshape(-1)
                       for p in agent.critic.parameters()]).numpy()
        q.put((rank, w))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, str(e)))


def _worker_dp_sync(rank, world, port, q):
    """Ranks feed dp_sync_step deliberately UNEQUAL local state (different
    replay sizes, transition counts, done flags) and must still agree on the
    derived (err, all_done, min_replay, total_new) tuple — the control word
    that keeps per-launch SAC train-step counts identical across ranks."""
    _init(rank, world, port)
    try:
        import torch.distributed as dist
        from distributed_cluster_gpus_amd.parallel.dist import dp_sync_step
        # launch 1: rank 0 is done, rank 1 is not; different counters
        out1 = dp_sync_step(err=0, local_done=(rank == 0),
                            replay_size=100 + 50 * rank, n_new=10 * (rank + 1))
        # launch 2: rank 1 reports an error flag
        out2 = dp_sync_step(err=4 if rank == 1 else 0, local_done=True,
                            replay_size=1000, n_new=0)
        q.put((rank, out1, out2))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, str(e), None))


def _worker_dp_train_cadence(rank, world, port, q):
    """Both ranks must execute the SAME number of train_step collectives when
    their local transition streams differ (the advisor's round-1 mispairing
    scenario), derived from globally reduced counters."""
    _init(rank, world, port)
    try:
        import torch.distributed as dist
        from distributed_cluster_gpus_amd.parallel.dist import dp_sync_step
        interval, warmup = 4, 8
        backlog, steps_run = 0, []
        # rank 0 produces transitions twice as fast; rank 1's replay crosses
        # warmup later
        local_replay = 0
        for launch in range(6):
            n_new = (2 if rank == 0 else 1) * 3
            local_replay += n_new
            _, _, min_replay, tr_total = dp_sync_step(
                0, launch == 5, local_replay, n_new)
            backlog += tr_total
            steps = 0
            if min_replay >= warmup:
                steps = min(64, backlog // (interval * world))
                backlog -= steps * interval * world
            steps_run.append(steps)
        q.put((rank, steps_run))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, str(e)))


def _spawn(fn, world=2, port=29801):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=fn, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    return results


def test_allreduce_gradients_and_scalars():
    res = _spawn(_worker_allreduce_grads, port=29811)
    assert len(res) == 2
    weights = {}
    for rank, ok_grad, ok_scalar, w in sorted(res):
        assert ok_grad is True, f"rank {rank}: {w}"
        assert ok_scalar is True
        weights[rank] = w
    np.testing.assert_array_equal(weights[0], weights[1])  # broadcast worked


def test_ddp_sac_parameters_stay_synced():
    res = _spawn(_worker_ddp_sac, port=29821)
    ws = {}
    for rank, w in res:
        assert not isinstance(w, str), f"rank {rank} failed: {w}"
        ws[rank] = w
    # critic params must be bitwise identical after synced-grad training
    np.testing.assert_array_equal(ws[0], ws[1])


def test_dp_sync_step_agrees_across_ranks():
    res = _spawn(_worker_dp_sync, port=29831)
    outs = {}
    for rank, out1, out2 in res:
        assert not isinstance(out1, str), f"rank {rank} failed: {out1}"
        outs[rank] = (out1, out2)
    assert outs[0] == outs[1]
    out1, out2 = outs[0]
    assert out1 == (0, False, 100, 30)   # min replay, summed transitions
    assert out2[0] == 4 and out2[1] is True  # error surfaced everywhere


def test_dp_train_cadence_identical_across_ranks():
    res = _spawn(_worker_dp_train_cadence, port=29841)
    steps = {}
    for rank, s in res:
        assert not isinstance(s, str), f"rank {rank} failed: {s}"
        steps[rank] = s
    assert steps[0] == steps[1], "per-launch SAC step counts diverged"
    assert sum(steps[0]) > 0


def test_replica_shard_partition():
    from distributed_cluster_gpus_amd.parallel.sharding import replica_shard
    total = 65536
    got = []
    for world in (1, 2, 4, 8):
        shards = [replica_shard(total, r, world) for r in range(world)]
        assert sum(s.count for s in shards) == total
        # contiguous, non-overlapping, ordered
        pos = 0
        for s in shards:
            assert s.start == pos
            pos = s.end
        got.append([s.count for s in shards])
    # uneven division
    shards = [replica_shard(10, r, 3) for r in range(3)]
    assert [s.count for s in shards] == [4, 3, 3]
    assert [s.start for s in shards] == [0, 4, 7]


def test_dist_noop_without_init():
    from distributed_cluster_gpus_amd.parallel.dist import (
        allreduce_gradients, allreduce_scalar, barrier, is_distributed,
        rank, world_size)
    assert not is_distributed()
    assert world_size() == 1 and rank() == 0
    assert allreduce_scalar(3.5, device=torch.device("cpu")) == 3.5
    m = torch.nn.Linear(2, 2)
    m(torch.ones(1, 2)).sum().backward()
    allreduce_gradients(list(m.parameters()))  # no-op, no crash
    barrier()
