"""Device discovery and allocation.

Re-expression of the reference's ``cluster.Manager``
(/root/reference/cluster.py:46-221): where the reference queried a live TF
cluster for its devices and allocated them to worker/ps/eval roles with
type preference and reuse sets, this manager enumerates the node's ROCm
GPUs (plus the CPU), allocates devices to named roles with GPU preference
and optional reuse, and prints the same style of allocation report.

The MI355X deployment model makes most of the reference's machinery moot
(one process per GPU; no remote devices), but deploy.py and runner.py use
this to answer "how many ranks, which device per rank" and to report the
allocation.
"""

import torch

from . import tools


class Manager:
    """Node-local device allocation with role bookkeeping."""

    def __init__(self, use_gpu=True, reuse_gpu=False):
        self.devices = []
        if use_gpu and torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                props = torch.cuda.get_device_properties(i)
                self.devices.append({
                    "name": f"cuda:{i}",
                    "type": "GPU",
                    "model": props.name,
                    "memory_gb": round(props.total_memory / 1e9, 1),
                    "users": [],
                })
        self.devices.append({"name": "cpu", "type": "CPU", "model": "host",
                             "memory_gb": None, "users": []})
        self.reuse_gpu = reuse_gpu

    def _free(self, dev):
        return not dev["users"] or (dev["type"] == "GPU" and self.reuse_gpu) \
            or dev["type"] == "CPU"

    def allocate(self, name, count, prefer="GPU", partial=False):
        """Allocate `count` devices to role `name`, preferring `prefer` type.

        Returns the list of device name strings. With ``partial``, fewer
        than `count` may be returned (at least one).
        """
        picked = []
        # Preferred type first, least-used first (spread like cluster.py:171-192).
        for type_pass in (prefer, "GPU", "CPU"):
            pool = sorted((d for d in self.devices
                           if d["type"] == type_pass and self._free(d)),
                          key=lambda d: len(d["users"]))
            for dev in pool:
                if len(picked) >= count:
                    break
                if dev["name"] in (p["name"] for p in picked):
                    continue
                picked.append(dev)
            if len(picked) >= count:
                break
        if len(picked) < count and not partial:
            # CPU can host any number of logical workers (reuse semantics).
            cpu = next(d for d in self.devices if d["type"] == "CPU")
            while len(picked) < count:
                picked.append(cpu)
        if not picked:
            raise tools.UserException(
                f"Cannot allocate {count} device(s) for role {name!r}")
        for dev in picked:
            dev["users"].append(name)
        return [d["name"] for d in picked]

    def report(self):
        """Print the allocation table (cluster.py:134-145 style)."""
        with tools.Context("cluster", "info"):
            for dev in self.devices:
                users = ", ".join(dev["users"]) if dev["users"] else "<free>"
                mem = f" {dev['memory_gb']} GB" if dev["memory_gb"] else ""
                tools.info(f"{dev['name']:8s} [{dev['type']}]"
                           f" {dev['model']}{mem} -> {users}")
