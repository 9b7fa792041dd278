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
    # pair probes default ON (the xGMI checks are the point of the agent)
    ap.add_argument("--probe-pairs", action="store_true", default=True)
    ap.add_argument("--no-probe-pairs", dest="probe_pairs", action="store_false",
                    help="skip the RCCL xGMI pair probes during deep sweeps")
    ap.add_argument("--p2p-matrix-every", type=int, default=30,
                    help="full 28-pair p2p link-matrix sweep every N deep sweeps")
    ap.add_argument("--p2p-matrix", action="store_true",
                    help="with --local: also measure the full p2p link matrix")
    ap.add_argument("--once", action="store_true")
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)

    from .health import NodeHealthAgent, collect_node_health, p2p_link_matrix

    if args.local:
        import json

        report = collect_node_health(deep=True, sweep=True,
                                     probe_pairs=args.probe_pairs)
        if args.p2p_matrix:
            try:
                m = p2p_link_matrix()
                report["p2p_matrix"] = m["matrix"]
                if m["links"]:
                    report["links"] = m["links"]
            except Exception as e:
                report["p2p_matrix_error"] = str(e)[:200]
        print(json.dumps(report, indent=2))
        return
    if not args.scheduler:
        ap.error("--scheduler is required (or use --local)")
    agent = NodeHealthAgent(args.scheduler, node_name=args.node_name,
                            interval_s=args.interval, deep_every=args.deep_every,
                            probe_pairs=args.probe_pairs,
                            p2p_matrix_every=args.p2p_matrix_every)
    if args.once:
        import json

        print(json.dumps(agent.run_once(), indent=2))
    else:
        agent.run_forever()


if __name__ == "__main__":
    main()
