"""Swarm-wide constants (parity: reference constants.py:3-19).

The reference pins PUBLIC_INITIAL_PEERS (its hosted bootstrap swarm) and a
centralized reachability/health API. This build is offline-first: there is no
hosted swarm, so the defaults name the conventional self-hosted bootstrap
endpoint instead, and reachability is checked by peer dial-back
(server/reachability.py) rather than a central validator.
"""

import torch

# conventional bootstrap endpoint for a self-hosted swarm (cli/run_dht.py)
DEFAULT_DHT_PORT = 31337
PUBLIC_INITIAL_PEERS: list = []  # no public swarm in an offline build

DTYPE_MAP = {
    "bfloat16": torch.bfloat16,
    "float16": torch.float16,
    "float32": torch.float32,
    "auto": None,
}
