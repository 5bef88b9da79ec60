"""LRC layered-composition tests (CPU, sub-plugin=oracle), mirroring the
reference's TestErasureCodeLrc.cc behaviours: kml expansion, layered
encode/decode, locality of minimum_to_decode, and the invalid-profile
rejections of parse_kml (ErasureCodeLrc.cc:292-395)."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HARNESS = os.path.join(ROOT, "ceph_amd", "harness")


def run_bench(*args):
    binp = os.path.join(HARNESS, "ec_benchmark")
    if not os.path.exists(binp):
        pytest.skip("harness not built")
    return subprocess.run([binp, "-d", HARNESS, *args],
                          capture_output=True, text=True)


LRC = ["-p", "lrc", "-P", "lrc-default-plugin=oracle"]


def test_kml_encode_roundtrip():
    # ceph docs example k=4 m=2 l=3: chunk_count = k+m+(k+m)/l = 8
    r = run_bench(*LRC, "-P", "k=4", "-P", "m=2", "-P", "l=3",
                  "-s", "65536", "-i", "5")
    assert r.returncode == 0, r.stderr


def test_single_erasure_exhaustive():
    """Any single erasure is recoverable by its local layer (the point of
    LRC); exhaustive sweep with byte verification."""
    r = run_bench(*LRC, "-P", "k=4", "-P", "m=2", "-P", "l=3",
                  "-s", "65536", "-i", "2", "-w", "decode", "-e", "1",
                  "-E", "exhaustive")
    assert r.returncode == 0, r.stderr


@pytest.mark.parametrize("erased", [(0, 4), (0, 1), (2, 6)])
def test_two_erasures_recoverable_patterns(erased):
    """Cross-group pairs and same-group data pairs are recoverable
    (locals or the global layer); verified byte-exact by the CLI."""
    args = LRC + ["-P", "k=4", "-P", "m=2", "-P", "l=3", "-s", "65536",
                  "-i", "2", "-w", "decode"]
    for e in erased:
        args += ["--erased", str(e)]
    r = run_bench(*args)
    assert r.returncode == 0, (erased, r.stderr)


def test_k9_m3_l4_near_baseline_shape():
    """BASELINE configs[3] names k=8 m=3 l=4, which the reference's own
    parse_kml rejects ((k+m) % l != 0). k=9 m=3 l=4 is the nearest valid
    shape: groups=3, chunk_count=15."""
    r = run_bench(*LRC, "-P", "k=9", "-P", "m=3", "-P", "l=4",
                  "-s", str(9 * 65536), "-i", "2", "-w", "decode",
                  "-e", "1", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr


@pytest.mark.parametrize("k,m,l", [(8, 3, 4),   # (k+m) % l != 0
                                   (4, 2, 0),   # l == 0
                                   (5, 1, 3)])  # k % groups != 0
def test_invalid_kml_rejected(k, m, l):
    r = run_bench(*LRC, "-P", f"k={k}", "-P", f"m={m}", "-P", f"l={l}",
                  "-s", "65536", "-i", "1")
    assert r.returncode != 0


def test_explicit_layers_profile():
    """Hand-written mapping+layers profile (the non-kml path,
    TestErasureCodeLrc.cc layer-parsing tests)."""
    r = run_bench("-p", "lrc", "-P", "lrc-default-plugin=oracle",
                  "-P", "mapping=DD__DD__",
                  "-P", 'layers=[ [ "DDc_DDc_", "" ], '
                        '[ "DDDc____", "" ], [ "____DDDc", "" ] ]',
                  "-s", "65536", "-i", "2", "-w", "decode", "-e", "1",
                  "-E", "exhaustive")
    assert r.returncode == 0, r.stderr


def test_layers_with_object_profiles():
    """The layers array's second element may be a JSON object
    (ErasureCodeLrc.cc:176-201); exercise the object form with per-layer
    plugin overrides."""
    r = run_bench("-p", "lrc",
                  "-P", "mapping=DD_DD_",
                  "-P", 'layers=[ [ "DDcDDc", {"plugin": "oracle"} ], '
                        '[ "DDc___", {"plugin": "oracle"} ], '
                        '[ "___DDc", {"plugin": "oracle"} ] ]',
                  "-s", "65536", "-i", "2", "-w", "decode", "-e", "1",
                  "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


@pytest.mark.parametrize("mapping,layers", [
    # mapping length != layer-string length (ErasureCodeLrc parse checks)
    ("DD__DD__", '[ [ "DDc_", "" ] ]'),
    # layers value not a JSON array
    ("DD__", '{"not": "an array"}'),
    # layer entry not a pair
    ("DD__", '[ [ "DD__" ] ]'),
    # unknown symbol in the layer spec
    ("DD__", '[ [ "DDxq", "" ] ]'),
])
def test_invalid_layer_profiles_rejected(mapping, layers):
    """Malformed mapping/layers profiles must be rejected at init
    (ErasureCodeLrc.cc layer parsing error paths)."""
    r = run_bench("-p", "lrc", "-P", "lrc-default-plugin=oracle",
                  "-P", f"mapping={mapping}", "-P", f"layers={layers}",
                  "-s", "65536", "-i", "1")
    assert r.returncode != 0, (mapping, layers, r.stdout)
