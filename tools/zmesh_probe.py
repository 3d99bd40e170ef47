#!/usr/bin/env python3
"""Probe the current host for any way to obtain the real zmesh wheel
(the reference's meshing kernel) so parity could be re-anchored against
it. Records the attempt — VERDICT r01 asks for this probe on every GPU
lease. Writes gpurun_out/zmesh_probe.json when run under gpurun."""
import glob
import json
import os
import subprocess
import sys
import time


def probe() -> dict:
    out = {"when": time.strftime("%Y-%m-%d %H:%M:%S"),
           "host": os.uname().nodename}
    try:
        import zmesh  # noqa: F401
        out["importable"] = True
        out["version"] = getattr(zmesh, "__version__", "?")
        return out
    except ImportError as e:
        out["importable"] = False
        out["import_error"] = str(e)
    wheels = []
    for root in ("/opt", "/usr/share", "/root", "/tmp", "/var/cache"):
        wheels += glob.glob(os.path.join(root, "**", "zmesh*.whl"),
                            recursive=True)
    out["local_wheels"] = wheels
    # NOTE: no `pip install` attempt — the GPU hosts have no network and
    # a hung pip once consumed a whole gpurun budget slice. A cheap TCP
    # reachability check stands in for "could pip work at all".
    import socket
    try:
        socket.setdefaulttimeout(3)
        socket.create_connection(("pypi.org", 443), timeout=3).close()
        out["network"] = True
    except OSError as e:
        out["network"] = False
        out["network_error"] = str(e)
    return out


if __name__ == "__main__":
    result = probe()
    os.makedirs("gpurun_out", exist_ok=True)
    path = os.path.join("gpurun_out", "zmesh_probe.json")
    with open(path, "w") as f:
        json.dump(result, f, indent=1)
    print(json.dumps(result))
