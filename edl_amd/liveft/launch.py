"""liveft CLI: python -m edl_amd.liveft.launch --np 2 --job_id j -- train.py

Parity: reference liveft/launch.py (wait -> run -> watch loop with
ELASTIC_EXIT_CODE=101 restarts)."""
import argparse
import sys

from .elastic import launch


def main(argv=None):
    p = argparse.ArgumentParser("edl liveft launcher")
    p.add_argument("--job_id", default=None)
    p.add_argument("--np", type=int, default=None, help="desired node count")
    p.add_argument("--store_endpoints", default=None)
    p.add_argument("--nproc_per_node", type=int, default=1)
    p.add_argument("--log_dir", default="./edl_logs")
    p.add_argument("cmd", nargs=argparse.REMAINDER)
    args = p.parse_args(argv)
    cmd = args.cmd[1:] if args.cmd and args.cmd[0] == "--" else args.cmd
    if not cmd:
        print("liveft: no command", file=sys.stderr)
        return 2
    return launch(cmd, job_id=args.job_id, np=args.np,
                  store_endpoints=args.store_endpoints,
                  nproc_per_node=args.nproc_per_node, log_dir=args.log_dir)


if __name__ == "__main__":
    sys.exit(main())
