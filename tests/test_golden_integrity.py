"""The committed golden fixtures must equal the CURRENT oracle's output
on their stored inputs — guards against silent contract drift between
the fixtures and the oracle (the GPU is pinned against both)."""
import glob
import os

import numpy as np

import oracle

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


def test_fixtures_match_current_oracle():
    cases = sorted(glob.glob(os.path.join(GOLDEN, "*.npz")))
    assert len(cases) >= 3
    for path in cases:
        z = np.load(path)
        labels = np.asfortranarray(z["labels"])
        res = tuple(float(r) for r in z["resolution"])
        got = oracle.mesh_chunk(labels, resolution=res)
        want_labels = sorted(
            int(k[len("verts_"):]) for k in z.files if k.startswith("verts_"))
        assert sorted(got) == want_labels, os.path.basename(path)
        for lab in want_labels:
            assert np.array_equal(got[lab][0], z[f"verts_{lab}"]), \
                f"{os.path.basename(path)} label {lab} verts drifted"
            assert np.array_equal(got[lab][1], z[f"faces_{lab}"]), \
                f"{os.path.basename(path)} label {lab} faces drifted"
