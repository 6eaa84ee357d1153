"""Cluster specification parsing (reference tools/cluster.py:48-91).

``cluster_parse`` accepts either a JSON spec ``{"workers": ["host:port",
...], ...}`` or a registered special value. The ``G5k`` parser reads
Grid5000's ``OAR_FILE_NODES`` host file and maps the first node to the
parameter-server role and the rest to workers on port 7000, exactly like
the reference -- on MI355X the "ps" host simply becomes rank 0's node.
"""

import json
import os

from . import UserException

DEFAULT_G5K_PORT = 7000


def _g5k_parser():
    path = os.environ.get("OAR_FILE_NODES")
    if not path or not os.path.exists(path):
        raise UserException(
            "G5k cluster requested but OAR_FILE_NODES is not set or missing")
    with open(path) as f:
        hosts = []
        for line in f:
            host = line.strip()
            if host and host not in hosts:
                hosts.append(host)
    if not hosts:
        raise UserException("OAR_FILE_NODES lists no hosts")
    return {
        "ps": [f"{hosts[0]}:{DEFAULT_G5K_PORT}"],
        "workers": [f"{h}:{DEFAULT_G5K_PORT}" for h in hosts[1:]],
    }


_special_parsers = {"G5k": _g5k_parser}

#: Help text fragment listing special cluster values (runner.py compat).
cluster_parsers = ", ".join(repr(k) for k in _special_parsers)


def cluster_parse(text):
    """Parse a cluster representation into {job: [host:port, ...]}."""
    text = text.strip()
    if text in _special_parsers:
        return _special_parsers[text]()
    try:
        spec = json.loads(text)
    except json.JSONDecodeError as e:
        raise UserException(f"Invalid cluster specification {text!r}: {e}")
    if not isinstance(spec, dict) or not spec:
        raise UserException(f"Invalid cluster specification {text!r}")
    return spec
