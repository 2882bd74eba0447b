"""Test utilities — the reference stack's ``tf.test.create_local_cluster``
(N in-process Servers on localhost ports in one process,
multi-node semantics without a cluster)."""

from __future__ import annotations

from typing import List, Tuple

from rocps.cluster import ClusterSpec, Server

__all__ = ["create_local_cluster"]


def _free_ports(n: int) -> List[int]:
    import socket
    socks, ports = [], []
    for _ in range(n):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        socks.append(s)
        ports.append(s.getsockname()[1])
    for s in socks:
        s.close()
    return ports


def create_local_cluster(num_workers: int, num_ps: int = 0
                         ) -> Tuple[List[Server], List[Server], ClusterSpec]:
    """Start num_workers + num_ps in-process Servers on localhost.

    Returns (worker_servers, ps_servers, cluster_spec); callers stop()
    them when done (or let process exit handle it).
    """
    ports = _free_ports(num_workers + num_ps)
    spec = {}
    if num_ps:
        spec["ps"] = [f"127.0.0.1:{p}" for p in ports[:num_ps]]
    spec["worker"] = [f"127.0.0.1:{p}" for p in ports[num_ps:]]
    cluster = ClusterSpec(spec)
    ps = [Server(cluster, "ps", i) for i in range(num_ps)]
    workers = [Server(cluster, "worker", i) for i in range(num_workers)]
    return workers, ps, cluster
