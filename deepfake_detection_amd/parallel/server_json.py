"""Cluster topology from a server JSON file.

Parity: reference dfd/server_json.py:19-45 — the JSON maps hostnames to GPU
lists; `parse_server` matches the current hostname and returns
(hostname, gpus, world_size, local_size, start_rank). Layout:

    {"servers": [{"hostname": "node1", "gpus": "0,1,2,3,4,5,6,7"}, ...]}

start_rank = sum of gpu counts of the servers listed before this host.
"""

import json
import os
import socket


def _current_hostname():
    # reference reads /proc/sys/kernel/hostname (server_json.py:29)
    try:
        with open("/proc/sys/kernel/hostname") as f:
            return f.read().strip()
    except OSError:
        return socket.gethostname()


def load_server_json(json_file):
    with open(json_file) as f:
        return json.load(f)


def parse_server(json_file, hostname=None):
    """Return (hostname, gpus_str, world_size, local_size, start_rank)."""
    config = load_server_json(json_file)
    servers = config["servers"] if isinstance(config, dict) else config
    hostname = hostname or os.environ.get("DFD_HOSTNAME") or _current_hostname()

    world_size = 0
    start_rank = 0
    match = None
    for server in servers:
        gpus = str(server["gpus"])
        n = len([g for g in gpus.split(",") if g != ""])
        if server["hostname"] == hostname:
            match = (hostname, gpus, n)
            start_rank = world_size
        world_size += n
    if match is None:
        raise RuntimeError(f"hostname {hostname} not found in {json_file}")
    hostname, gpus, local_size = match
    return hostname, gpus, world_size, local_size, start_rank
