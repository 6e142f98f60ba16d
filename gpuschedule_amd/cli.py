"""Command-line interface.

    python -m gpuschedule_amd simulate --policy las --trace t.csv --gpus 8
    python -m gpuschedule_amd simulate --config configs/c1_fifo_sim.yaml
    python -m gpuschedule_amd replay   --policy srtf --trace t.csv
    python -m gpuschedule_amd gen-trace --jobs 64 --mix philly-slice -o t.csv

``simulate`` is the pure-CPU discrete-event path; ``replay`` executes
real training on GPUs (single process, or under torchrun for N>1).
"""
from __future__ import annotations

import argparse
import json

import yaml

from . import (Cluster, Scheduler, load_trace, make_policy, make_trace,
               save_trace, simulate)
from .trace.synth import make_elastic_trace


def _load_config(path):
    with open(path) as f:
        return yaml.safe_load(f)


def _policy_from_args(args, cfg):
    name = cfg.get("policy", args.policy)
    kw = cfg.get("policy_args", {})
    return make_policy(name, **kw)


def _specs_from_args(args, cfg):
    if cfg.get("philly_log"):
        from .trace.philly_log import load_philly_log
        pl = cfg["philly_log"]
        return load_philly_log(
            pl["file"], time_scale=pl.get("time_scale", 1.0),
            seed=pl.get("seed", 0), limit=pl.get("limit"),
            clamp_gpus=pl.get("clamp_gpus", 8))
    if cfg.get("trace_file") or args.trace:
        return load_trace(cfg.get("trace_file") or args.trace)
    t = cfg.get("synthetic", {})
    if t.get("kind") == "elastic":
        return make_elastic_trace(seed=t.get("seed", 0),
                                  total_iters=t.get("iters", 400))
    return make_trace(t.get("jobs", args.jobs), mix=t.get("mix", args.mix),
                      seed=t.get("seed", 0),
                      mean_interarrival=t.get(
                          "mean_interarrival",
                          getattr(args, "interarrival", None) or 10.0),
                      iters_range=tuple(t.get("iters_range", (50, 400))),
                      gang_sizes=t.get("gang_sizes"))


def cmd_simulate(args):
    cfg = _load_config(args.config) if args.config else {}
    pol = _policy_from_args(args, cfg)
    specs = _specs_from_args(args, cfg)
    n_gpus = cfg.get("gpus", args.gpus)
    warm = cfg.get("warm_pool", getattr(args, "warm_pool", False))
    log, summary = simulate(specs, pol, n_gpus=n_gpus, warm_pool=warm)
    if args.log:
        log.save(args.log)
    print(json.dumps({"policy": pol.name, "gpus": n_gpus,
                      **{k: round(v, 3) for k, v in summary.items()}}))


def cmd_replay(args):
    import torch
    from .exec import TorchBackend
    from .exec.backend import init_distributed
    cfg = _load_config(args.config) if args.config else {}
    rank, world = init_distributed("gloo")
    # the REAL backend needs one rank per GPU: the cluster is sized by
    # the launched world, not the config (configs size the SIMULATOR;
    # under torchrun the two should agree).
    n_gpus = world
    cfg_gpus = cfg.get("gpus", args.gpus)
    if rank == 0 and cfg_gpus != world:
        import sys
        print(f"replay: world={world} ranks; config wants {cfg_gpus} "
              f"GPUs — running on {world}", file=sys.stderr)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    # multi-node: ranks 0..world-1 span nodes of `gpus_per_node` GPUs
    # (torchrun --nnodes N); the cluster model bounds gang windows to
    # one node, so RCCL gangs never cross nodes and the gloo control
    # plane is the only cross-node traffic.
    gpn = cfg.get("gpus_per_node", getattr(args, "gpus_per_node", 0)) \
        or max(n_gpus, 8)
    sched = None
    if rank == 0:
        sched = Scheduler(Cluster(n_gpus, gpn),
                          _policy_from_args(args, cfg))
        sched.submit(_specs_from_args(args, cfg))
    be = TorchBackend(sched, rank=rank, world=world, device_type=device,
                      iters_per_tick=cfg.get("iters_per_tick", 2),
                      gang_backend="nccl" if device == "cuda" else "gloo")
    log = be.run(max_seconds=args.max_seconds)
    if rank == 0:
        if args.log:
            log.save(args.log)
        print(json.dumps({k: round(v, 3)
                          for k, v in log.summary().items()}))


def cmd_gen_trace(args):
    specs = make_trace(args.jobs, mix=args.mix, seed=args.seed,
                       mean_interarrival=args.interarrival)
    save_trace(specs, args.output)
    print(f"wrote {len(specs)} jobs -> {args.output}")


def cmd_compare(args):
    """Replay one trace through several policies in the simulator and
    print a comparison table (avg/p95 JCT, makespan, slowdown,
    preemptions)."""
    cfg = _load_config(args.config) if args.config else {}
    specs = _specs_from_args(args, cfg)
    n_gpus = cfg.get("gpus", args.gpus)
    names = [p.strip() for p in args.policies.split(",")]
    cols = ("avg_jct", "p95_jct", "makespan", "avg_slowdown",
            "p95_slowdown", "avg_queue_delay", "preemptions")
    print(f"{'policy':10s} " + " ".join(f"{c:>14s}" for c in cols))
    for name in names:
        _, s = simulate(specs, make_policy(name), n_gpus=n_gpus)
        print(f"{name:10s} " +
              " ".join(f"{s[c]:14.2f}" for c in cols))


def cmd_summarize(args):
    """Print the JCT/makespan/slowdown summary of a saved event log."""
    from .sched.events import EventLog
    log = EventLog.load(args.log)
    print(json.dumps({k: round(v, 3)
                      for k, v in log.summary().items()}, indent=1))


def cmd_export_trace(args):
    from .sched.events import EventLog
    log = EventLog.load(args.log)
    log.save_chrome_trace(args.output)
    n = len(log.to_chrome_trace()["traceEvents"])
    print(f"wrote {n} trace events -> {args.output} "
          "(open in chrome://tracing or ui.perfetto.dev)")


def main(argv=None):
    ap = argparse.ArgumentParser(prog="gpuschedule_amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    common = argparse.ArgumentParser(add_help=False)
    common.add_argument("--policy", default="fifo")
    common.add_argument("--trace", default=None)
    common.add_argument("--config", default=None)
    common.add_argument("--gpus", type=int, default=8)
    common.add_argument("--jobs", type=int, default=16)
    common.add_argument("--mix", default="mixed-cv-nlp")
    common.add_argument("--interarrival", type=float, default=None,
                        help="mean synthetic-trace interarrival seconds "
                             "(default 10; lower it to create "
                             "contention — iteration times are "
                             "hardware-calibrated and fast)")
    common.add_argument("--log", default=None, help="event log JSONL out")

    p = sub.add_parser("simulate", parents=[common])
    p.add_argument("--warm-pool", action="store_true",
                   help="model the backend's prewarmed runner pool: "
                        "first placements cost RESUME_HBM_S, not "
                        "COLD_START_S (see sim/simulator.py)")
    p.set_defaults(fn=cmd_simulate)
    p = sub.add_parser("replay", parents=[common])
    p.add_argument("--max-seconds", type=float, default=None)
    p.add_argument("--gpus-per-node", type=int, default=0,
                   help="GPUs per node for multi-node torchrun replays "
                        "(0 = single node); gangs never cross nodes")
    p.set_defaults(fn=cmd_replay)
    p = sub.add_parser("gen-trace")
    p.add_argument("--jobs", type=int, default=64)
    p.add_argument("--mix", default="philly-slice")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--interarrival", type=float, default=10.0)
    p.add_argument("-o", "--output", required=True)
    p.set_defaults(fn=cmd_gen_trace)
    p = sub.add_parser("compare", parents=[common],
                       help="replay one trace under several policies "
                            "(simulator) and print a metric table")
    p.add_argument("--policies", default="fifo,srtf,las,gandiva,elastic")
    p.set_defaults(fn=cmd_compare)
    p = sub.add_parser(
        "summarize",
        help="print JCT/makespan/slowdown metrics of a saved event log")
    p.add_argument("--log", required=True)
    p.set_defaults(fn=cmd_summarize)
    p = sub.add_parser(
        "export-trace",
        help="convert an event-log JSONL to a Chrome/Perfetto trace")
    p.add_argument("--log", required=True, help="event log JSONL in")
    p.add_argument("-o", "--output", required=True)
    p.set_defaults(fn=cmd_export_trace)

    args = ap.parse_args(argv)
    args.fn(args)


if __name__ == "__main__":
    main()
