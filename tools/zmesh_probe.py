#!/usr/bin/env python3
"""Probe the current host for any way to obtain the real zmesh wheel
(the reference's meshing kernel) so parity could be re-anchored against
it. Records the attempt — VERDICT r01 asks for this probe on every GPU
lease. Writes gpurun_out/zmesh_probe.json when run under gpurun."""
import glob
import json
import os
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
    # bounded search: shallow patterns only (a recursive glob over the
    # box's filesystem once stalled a suite run)
    wheels = []
    for pat in ("/opt/*.whl", "/opt/wheels/*.whl", "/opt/wheelhouse/*.whl",
                "/root/*.whl", "/tmp/*.whl", "/var/cache/pip/*.whl"):
        wheels += [w for w in glob.glob(pat)
                   if "zmesh" in os.path.basename(w)]
    out["local_wheels"] = wheels
    # NOTE: no `pip install` and no socket/DNS attempt — the GPU hosts
    # are airgapped and both a hung pip and a blackholed getaddrinfo
    # have eaten gpurun budget. Import + wheel search above is the
    # entire offline probe; 'network' is recorded as policy knowledge.
    out["network"] = False
    out["network_note"] = ("gpurun hosts are airgapped by policy; no "
                           "install path exists for the real zmesh")
    return out


if __name__ == "__main__":
    result = probe()
    os.makedirs("gpurun_out", exist_ok=True)
    path = os.path.join("gpurun_out", "zmesh_probe.json")
    with open(path, "w") as f:
        json.dump(result, f, indent=1)
    print(json.dumps(result))
