#!/usr/bin/env python3
"""Data-parallel CHSAC-AF training over the batched replica engine
(BASELINE.json config 4: RL scheduling policy trained DP on 8x MI355X,
RCCL all-reduce over xGMI).

Each rank owns a replica shard of the paper workload with the chsac_af
batched engine; the agent is replicated (broadcast at start), gradients are
all-reduced (one fused ~2.3 MB flat collective per optimizer step — the
latency-bound shape for the xGMI mesh) and the PID-lambda state is kept
identical via cross-rank cost averaging.

Launch:
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      scripts/train_rl_dp.py --replicas-per-gpu 1024 --duration 2000
Single GPU (no torchrun) also works.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--replicas-per-gpu", type=int, default=256)
    p.add_argument("--duration", type=float, default=1000.0)
    p.add_argument("--inf-rate", type=float, default=2.0)
    p.add_argument("--trn-rate", type=float, default=0.3)
    p.add_argument("--warmup", type=int, default=1000)
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--train-interval", type=int, default=256)
    p.add_argument("--seed", type=int, default=123)
    p.add_argument("--checkpoint", type=str, default="chsac_dp.pt")
    p.add_argument("--replay-npz", type=str, default=None,
                   help="also dump the replay ring as an offline .npz dataset")
    args = p.parse_args(argv)

    import torch

    from distributed_cluster_gpus_amd.configs.paper import paper_scenario
    from distributed_cluster_gpus_amd.engine.batched import BatchedEngine
    from distributed_cluster_gpus_amd.models.arrivals import ArrivalProcess
    from distributed_cluster_gpus_amd.parallel.dist import (broadcast_module,
                                                            init_distributed)

    rank, world = init_distributed()
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)

    sc = paper_scenario()
    inf = ArrivalProcess(mode="poisson", rate=args.inf_rate)
    trn = ArrivalProcess(mode="poisson", rate=args.trn_rate)
    eng = BatchedEngine(sc, inf, trn, algo="chsac_af",
                        replicas=args.replicas_per_gpu * world,
                        duration=args.duration, log_interval=5.0,
                        out_dir=None, seed=args.seed, device=device,
                        rank=rank, world=world, enable_logs=False,
                        rl_warmup=args.warmup, rl_batch=args.batch,
                        rl_train_interval=args.train_interval)
    # replicate the agent across ranks, then train data-parallel
    for mod in (eng.rl.encoder, eng.rl.actor, eng.rl.critic,
                eng.rl.algo.target_critic):
        broadcast_module(mod)
    if world > 1:
        eng.rl.enable_ddp()

    t0 = time.perf_counter()
    stats = eng.run()
    wall = time.perf_counter() - t0
    # per-rank timing breakdown (advance kernel+sync / policy serve+ingest /
    # DP control collectives / SAC train), gathered so rank 0 can report the
    # whole node's balance
    timing = {k: round(v, 3) if isinstance(v, float) else v
              for k, v in getattr(eng, "timing", {}).items()}
    timing["rank"] = rank
    timing["events"] = stats["events"]
    if world > 1:
        import torch.distributed as dist
        all_timing = [None] * world
        dist.all_gather_object(all_timing, timing)
    else:
        all_timing = [timing]
    if rank == 0:
        out = {
            "world": world,
            "replicas": args.replicas_per_gpu * world,
            "events": stats["events"],
            "events_per_sec": round(stats["events"] / wall),
            "jobs_completed": stats["jobs_completed"],
            "rl_updates": eng.rl_updates,
            "rl_updates_per_sec": round(eng.rl_updates / wall, 2),
            "replay_size": eng.replay.size,
            "wall_s": round(wall, 2),
            "lambda": {k: float(v) for k, v in eng.rl.cmdp.lmbda.items()},
            "per_rank_timing": all_timing,
        }
        print(json.dumps(out))
        eng.rl.save(args.checkpoint)
        print(f"checkpoint -> {args.checkpoint}")
        if args.replay_npz:
            eng.replay.save_npz(args.replay_npz)
            print(f"offline dataset -> {args.replay_npz}")


if __name__ == "__main__":
    main()
