"""embedding_search CLI end-to-end on CPU, using the REFERENCE flag
spellings (download_and_generate_embedding.py:16-37,
similarity_search.py:16-19) so a reference user's command lines work
verbatim."""
import pickle
import subprocess
import sys
from pathlib import Path

import numpy as np

ROOT = Path(__file__).resolve().parent.parent


def _run(script, *flags):
    return subprocess.run(
        [sys.executable, str(ROOT / "embedding_search" / script), *flags],
        capture_output=True, text=True, timeout=600)


def test_embedding_and_search_clis_reference_spellings(tmp_path):
    # 1) embed a synthetic LAION-shaped chunk with reference dash flags
    dump = tmp_path / "chunk0"
    r = _run("download_and_generate_embedding.py",
             "--dump-path", str(dump), "--pt-style", "sscd",
             "--batch-size", "16", "--workers", "0",
             "--synthetic_n", "24", "--skip-download", "--skip-image-delete")
    assert r.returncode == 0, r.stderr
    blob = pickle.loads((dump / "embedding.pkl").read_bytes())
    assert blob["features"].shape == (24, 512)
    assert blob["features"].dtype == np.float32
    assert len(blob["indexes"]) == 24

    # 2) build a 2-chunk LAION folder; chunk1 CONTAINS the query rows so the
    #    running-max merge must find score ~1 there
    laion = tmp_path / "laion"
    (laion / "c0").mkdir(parents=True)
    (laion / "c1").mkdir()
    qn = 6
    rest = {"features": blob["features"][qn:].copy(),
            "indexes": list(blob["indexes"][qn:])}
    (laion / "c0" / "embedding.pkl").write_bytes(pickle.dumps(rest))
    planted = {"features": blob["features"][:qn].copy(),
               "indexes": [f"laion_hit_{i}" for i in range(qn)]}
    (laion / "c1" / "embedding.pkl").write_bytes(pickle.dumps(planted))

    query = tmp_path / "query.pkl"
    query.write_bytes(pickle.dumps(
        {"features": blob["features"][:qn], "indexes": blob["indexes"][:qn]}))

    out = tmp_path / "matches.pkl"
    r = _run("similarity_search.py",
             "--generation-embedding-path", str(query),
             "--laion-embedding-folder", str(laion),
             "--dump-path", str(out), "--num-chunks", "2")
    assert r.returncode == 0, r.stderr
    matches = pickle.loads(out.read_bytes())
    assert matches["scores"].shape == (qn,)
    # queries are L2-normalized SSCD features planted in chunk1: cos sim ~1
    assert np.all(matches["scores"] > 0.999), matches["scores"]
    assert all(k.startswith("laion_hit_") for k in matches["keys"])
