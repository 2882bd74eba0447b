"""Standalone async-PS task (used by bench.py --mode async and scripts):

    python -m rocps.cli.ps_server --cluster-json '{"ps": [...], ...}' \
        --task-index 0
"""

from __future__ import annotations

import argparse
import json
import sys


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--cluster-json", required=True)
    ap.add_argument("--task-index", type=int, required=True)
    args = ap.parse_args(argv)

    from rocps.cluster import ClusterSpec, Server
    from rocps.engine.async_engine import AsyncPSShard

    cluster = ClusterSpec(json.loads(args.cluster_json))
    server = Server(cluster, "ps", args.task_index)
    AsyncPSShard.serve_on(server, cluster)
    server.join()
    return 0


if __name__ == "__main__":
    sys.exit(main())
