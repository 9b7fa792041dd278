"""Node health agent entry point:
`python -m hivedscheduler_amd.agent --scheduler http://hived:9096`."""
import argparse
import logging


def main() -> None:
    ap = argparse.ArgumentParser(prog="hivedscheduler-amd-agent")
    ap.add_argument("--scheduler", default=None, help="scheduler base URL")
    ap.add_argument("--local", action="store_true",
                    help="one local health sweep (deep + sweep), print JSON, no posting")
    ap.add_argument("--node-name", default=None)
    ap.add_argument("--interval", type=float, default=60.0)
    ap.add_argument("--deep-every", type=int, default=10,
                    help="run HIP kernel probes every N sweeps")
    ap.add_argument("--probe-pairs", action="store_true",
                    help="RCCL-probe each xGMI pair during deep sweeps")
    ap.add_argument("--once", action="store_true")
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)

    from .health import NodeHealthAgent, collect_node_health

    if args.local:
        import json

        print(json.dumps(collect_node_health(deep=True, sweep=True,
                                             probe_pairs=args.probe_pairs), indent=2))
        return
    if not args.scheduler:
        ap.error("--scheduler is required (or use --local)")
    agent = NodeHealthAgent(args.scheduler, node_name=args.node_name,
                            interval_s=args.interval, deep_every=args.deep_every,
                            probe_pairs=args.probe_pairs)
    if args.once:
        import json

        print(json.dumps(agent.run_once(), indent=2))
    else:
        agent.run_forever()


if __name__ == "__main__":
    main()
