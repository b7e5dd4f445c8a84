#!/usr/bin/env python3
"""Geo GPU cluster simulator CLI (MI355X-native framework).

Flag surface and output-directory rule are compatible with the reference CLI
(reference: run_sim_paper.py:11-160); outputs are the same cluster_log.csv /
job_log.csv / project.log triple.

Extensions over the reference:
  --engine {oracle,native,batched}   pick the scalar Python oracle, the C++
                                     DES core, or the batched MI355X engine
  --replicas N                       Monte-Carlo replicas (batched engine)
  --eco-objective is actually honored (the reference parses but drops it,
    SURVEY Appendix A.4), and --elastic-scaling is a store_true flag that
    really enables elastic scaling (the reference's type=bool flag could
    never be True).
"""
import argparse
import os


def parse_args(argv=None):
    p = argparse.ArgumentParser(
        description="Geo GPU Simulator (MI355X-native, multi-ingress)",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)

    # --- core ---
    p.add_argument("--duration", type=float, default=180.0,
                   help="Total simulated time (seconds).")
    p.add_argument("--policy", type=str, default="energy_aware",
                   choices=["energy_aware", "perf_first"],
                   help="In-DC heuristic allocation policy.")
    p.add_argument("--log-interval", type=float, default=5.0,
                   help="Cluster/job log cadence (seconds).")
    p.add_argument("--log-path", type=str, default=None, help="Log path")
    p.add_argument("--seed", type=int, default=123, help="Random seed.")
    p.add_argument("--progress", default=True,
                   help="Show a tqdm progress bar over simulated time.")

    # --- arrivals (inference) ---
    p.add_argument("--inf-mode", type=str, default="sinusoid",
                   choices=["poisson", "sinusoid", "off"])
    p.add_argument("--inf-rate", type=float, default=6.0)
    p.add_argument("--inf-amp", type=float, default=0.6)
    p.add_argument("--inf-period", type=float, default=300.0)

    # --- arrivals (training) ---
    p.add_argument("--trn-mode", type=str, default="poisson",
                   choices=["poisson", "sinusoid", "off"])
    p.add_argument("--trn-rate", type=float, default=0.3)

    # --- algorithm / controller ---
    p.add_argument("--algo", type=str, default="default_policy",
                   choices=["default_policy", "cap_uniform", "cap_greedy",
                            "joint_nf", "bandit", "carbon_cost",
                            "eco_route", "chsac_af", "debug"])
    p.add_argument("--elastic-scaling", action="store_true", default=False,
                   help="Enable elastic scaling (RL algo only).")
    p.add_argument("--power-cap", type=float, default=0.0,
                   help="Total power cap (W); only cap_uniform/cap_greedy, <=0 = off.")
    p.add_argument("--control-interval", type=float, default=5.0,
                   help="Controller cadence (seconds).")
    p.add_argument("--use-control-interval", action="store_true", default=False,
                   help="Actually fire the cap controller at --control-interval "
                        "(the reference parses the flag but fires at "
                        "log-interval; default keeps that parity).")
    p.add_argument("--eco-objective", type=str, default="energy",
                   choices=["energy", "carbon", "cost"])
    # debug params
    p.add_argument("--num_fixed_gpus", type=int, default=1)
    p.add_argument("--fixed_freq", type=float, default=None)

    # --- RL knobs ---
    p.add_argument("--upgr-buffer", type=int, default=200_000)
    p.add_argument("--upgr-batch", type=int, default=256)
    p.add_argument("--upgr-warmup", type=int, default=1_000)
    p.add_argument("--upgr-device", type=str, default="cuda", choices=["cuda", "cpu"])
    p.add_argument("--sla_p99_ms", type=float, default=500.0)
    p.add_argument("--energy_budget_j", type=float, default=None)

    # --- MI355X-framework extensions ---
    p.add_argument("--engine", type=str, default="oracle",
                   choices=["oracle", "native", "batched"],
                   help="oracle = Python DES; native = C++ DES core; "
                        "batched = MI355X HIP replica engine.")
    p.add_argument("--replicas", type=int, default=4096,
                   help="Monte-Carlo replicas (batched engine only).")
    p.add_argument("--rl-serve", type=str, default="device",
                   choices=["device", "host"],
                   help="chsac_af policy serving on the batched engine: "
                        "in-kernel actor (device) or host pause/resume.")
    p.add_argument("--rl-exact-p99", action="store_true", default=False,
                   help="Exact 2048-sample sliding-window p99 on the batched "
                        "engine (parity mode) instead of the histogram "
                        "approximation.")
    p.add_argument("--rl-target-ups", type=float, default=180.0,
                   help="chsac_af batched engine: SAC update-rate target for "
                        "the overlapped loop (<=0 = pure-throughput mode).")
    p.add_argument("--fp32-coeff-eval", action="store_true", default=False,
                   help="fp32 decision-score evaluation in the sim kernels "
                        "(batched engine; times/energies stay f64).")
    p.add_argument("--single-dc", action="store_true", default=False,
                   help="Use the single-DC debug topology.")
    p.add_argument("--rl-checkpoint", type=str, default=None,
                   help="Path to save the RL agent checkpoint at the end "
                        "(chsac_af only).")
    p.add_argument("--rl-resume", type=str, default=None,
                   help="Path to an RL agent checkpoint to load before running.")
    return p.parse_args(argv)


def resolve_out_dir(log_path, algo):
    """Reference output-dir rule (run_sim_paper.py:136-140): a bare name gets
    the algo appended; a path containing a separator is used as-is."""
    if log_path:
        norm = os.path.normpath(log_path)
        return os.path.join(norm, algo) if os.sep not in norm else norm
    return os.getcwd()


def main(argv=None):
    args = parse_args(argv)

    from distributed_cluster_gpus_amd.configs.paper import (
        build_arrivals, paper_scenario, single_dc_scenario)
    from distributed_cluster_gpus_amd.models.gputypes import validate_gpu_specs
    from distributed_cluster_gpus_amd.models.scenario import PolicyParams
    from distributed_cluster_gpus_amd.utils.logging import get_logger

    policy = PolicyParams(name=args.policy)
    sc = single_dc_scenario(policy=policy) if args.single_dc else paper_scenario(policy=policy)

    warnings = validate_gpu_specs((sc.gpu_specs[n] for n in sc.dc_names), strict=False)
    for m in warnings:
        print("[GPU VALIDATION]", m)

    arrival_inf, arrival_trn = build_arrivals(
        inf_mode=args.inf_mode, inf_rate=args.inf_rate, inf_amp=args.inf_amp,
        inf_period=args.inf_period, trn_mode=args.trn_mode, trn_rate=args.trn_rate)
    sc.arrival_inf, sc.arrival_trn = arrival_inf, arrival_trn

    out_dir = resolve_out_dir(args.log_path, args.algo)
    logger = get_logger(out_dir)

    rl_device = args.upgr_device
    if rl_device == "cuda":
        import torch
        if not torch.cuda.is_available():
            rl_device = "cpu"

    common = dict(
        algo=args.algo, duration=args.duration, log_interval=args.log_interval,
        out_dir=out_dir, seed=args.seed, power_cap=args.power_cap,
        control_interval=args.control_interval,
        elastic_scaling=args.elastic_scaling, eco_objective=args.eco_objective,
        use_control_interval=args.use_control_interval,
        num_fixed_gpus=args.num_fixed_gpus, fixed_freq=args.fixed_freq,
        sla_p99_ms=args.sla_p99_ms, energy_budget_j=args.energy_budget_j,
        rl_device=rl_device, rl_batch=args.upgr_batch,
        rl_warmup=args.upgr_warmup, rl_buffer=args.upgr_buffer,
        logger=logger,
        show_progress=(str(args.progress).lower() not in ("false", "0", "no")))

    if args.engine == "native":
        from distributed_cluster_gpus_amd.engine.native import NativeEngine
        eng = NativeEngine(sc, arrival_inf, arrival_trn, **common)
    elif args.engine == "batched":
        from distributed_cluster_gpus_amd.engine.batched import BatchedEngine
        eng = BatchedEngine(sc, arrival_inf, arrival_trn,
                            replicas=args.replicas, rl_serve=args.rl_serve,
                            rl_exact_p99=args.rl_exact_p99,
                            rl_target_updates_per_s=args.rl_target_ups,
                            fp32_coeff_eval=args.fp32_coeff_eval, **common)
    else:
        from distributed_cluster_gpus_amd.engine.oracle import OracleEngine
        eng = OracleEngine(sc, arrival_inf, arrival_trn, **common)

    if args.rl_resume and getattr(eng, "rl", None) is not None:
        eng.rl.load(args.rl_resume)

    stats = eng.run()
    if args.engine == "batched" and args.replicas > 1:
        # Monte-Carlo population report: distributions over the replica
        # ensemble (a capability the scalar reference cannot express)
        from distributed_cluster_gpus_amd.analysis.montecarlo import \
            population_report
        rep = population_report(eng, out_dir=os.path.join(out_dir, "population"))
        e = rep.get("total_energy_kJ")
        if e:
            print(f"[population] {rep['replicas']} replicas: total energy "
                  f"{e['mean']:.1f} kJ ± {e['stderr']:.2f} (95% CI "
                  f"[{e['ci_lo']:.1f}, {e['ci_hi']:.1f}])")
    if args.rl_checkpoint and getattr(eng, "rl", None) is not None:
        eng.rl.save(args.rl_checkpoint)
        print(f"RL checkpoint saved to {args.rl_checkpoint}")

    print(f"Done. ({args.algo}) Logs: cluster_log.csv, job_log.csv")
    print(f"[perf] events={stats['events']} wall_s={stats['wall_s']:.3f} "
          f"events_per_sec={stats['events_per_sec']:.1f} "
          f"rl_updates={stats.get('rl_updates', 0)} "
          f"jobs_completed={stats.get('jobs_completed', 0)}")
    return stats


if __name__ == "__main__":
    main()
