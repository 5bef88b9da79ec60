"""GPU harness tests: the mi355x plugin loaded through the registry's
dlopen path (the drop-in boundary), driven by the
reference-shaped ec_benchmark CLI. This is the end-to-end proof that a
Ceph-style host picks up the GPU backend unchanged."""
import os
import subprocess

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HARNESS = os.path.join(ROOT, "ceph_amd", "harness")


def run_bench(*args, timeout=300):
    return subprocess.run(
        [os.path.join(HARNESS, "ec_benchmark"), "-d", HARNESS, *args],
        capture_output=True, text=True, timeout=timeout)


def test_mi355x_plugin_encode_via_registry():
    r = run_bench("-p", "mi355x", "-P", "technique=reed_sol_van",
                  "-P", "k=8", "-P", "m=3", "-s", str(8 << 20), "-i", "5")
    assert r.returncode == 0, r.stderr
    secs, kib = r.stdout.split()
    assert float(secs) > 0 and int(kib) == 5 * (8 << 20) // 1024


def test_mi355x_plugin_decode_exhaustive_verifies():
    """Exhaustive erasure decode with byte verification against the
    encoded originals (benchmark.cc:211-258) — GPU plugin output checked
    chunk-for-chunk."""
    r = run_bench("-p", "mi355x", "-P", "technique=reed_sol_van",
                  "-P", "k=4", "-P", "m=2", "-s", str(256 * 1024),
                  "-i", "2", "-w", "decode", "-e", "2", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


def test_mi355x_vs_oracle_plugin_same_parity(tmp_path):
    """The GPU plugin and the oracle fixture plugin produce identical
    parity through the same harness path (same seed => same input)."""
    import ctypes
    import numpy as np
    import ceph_amd
    import oracle

    k, m, tech = 6, 2, "cauchy"
    C = 128 * 1024
    rng = np.random.default_rng(42)
    data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
    ctx = ceph_amd.EcContext(k, m, tech, device=0)
    try:
        got = ctx.encode_chunks(data)
        want = oracle.encode(tech, k, m, data)
        for j in range(m):
            assert np.array_equal(got[j], want[j])
    finally:
        ctx.close()


def test_mi355x_plugin_refuses_cleanly_on_bad_profile():
    r = run_bench("-p", "mi355x", "-P", "technique=no_such_technique",
                  "-P", "k=4", "-P", "m=2", "-s", "65536", "-i", "1")
    assert r.returncode != 0


def test_mi355x_reed_sol_r6():
    """RAID6 technique through the GPU plugin: exhaustive 2-erasure decode
    with byte verification."""
    r = run_bench("-p", "mi355x", "-P", "technique=reed_sol_r6_op",
                  "-P", "k=6", "-P", "m=2", "-s", str(6 * 65536), "-i", "2",
                  "-w", "decode", "-e", "2", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


def test_shec_flags_gpu():
    r = run_bench("-p", "shec", "-P", "k=4", "-P", "m=3", "-P", "c=2",
                  "--flags")
    assert r.returncode == 0, r.stderr
    assert r.stdout.strip() == ("partialread,partialwrite,zeroinout,"
                                "paritydelta")


def test_clay_flags_gpu():
    r = run_bench("-p", "clay", "-P", "k=4", "-P", "m=2", "-P", "d=5",
                  "--flags")
    assert r.returncode == 0, r.stderr
    assert r.stdout.strip() == "partialread,requiresubchunks"


def test_shec_decode_exhaustive_gpu():
    """SHEC end-to-end on the GPU path (plugin_shec drives the mi355x
    kernels via ecx_set_matrix): exhaustive single-erasure decode with
    byte verification."""
    r = run_bench("-p", "shec", "-P", "k=4", "-P", "m=3", "-P", "c=2",
                  "-s", str(4 * 65536), "-i", "2", "-w", "decode",
                  "-e", "1", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


def test_clay_decode_exhaustive_gpu():
    """Clay with the default scalar_mds=mi355x sub-codec: exhaustive
    m-erasure decode through decode_layered, verified byte-exact."""
    r = run_bench("-p", "clay", "-P", "k=4", "-P", "m=2", "-P", "d=5",
                  "-s", str(4 * 65536), "-i", "2", "-w", "decode",
                  "-e", "2", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


def test_lrc_decode_exhaustive_gpu():
    """LRC with the default mi355x sub-plugin for every layer: exhaustive
    single-erasure decode (each recovered by its local layer on GPU)."""
    r = run_bench("-p", "lrc", "-P", "k=4", "-P", "m=2", "-P", "l=3",
                  "-s", str(4 * 65536), "-i", "2", "-w", "decode",
                  "-e", "1", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


@pytest.mark.parametrize("profile,expected", [
    (["-P", "technique=reed_sol_van", "-P", "k=8", "-P", "m=3"],
     "partialread,partialwrite,zeroinout,paritydelta,optimizedsupport,"
     "crcencodedecode,directreads"),
    (["-P", "technique=cauchy", "-P", "k=4", "-P", "m=1"],
     "partialread,partialwrite,zeroinout,paritydelta,optimizedsupport,"
     "crcencodedecode,directreads"),
    (["-P", "technique=cauchy", "-P", "k=4", "-P", "m=2"],
     "partialread,partialwrite,zeroinout,paritydelta,optimizedsupport,"
     "directreads"),
    (["-P", "technique=jerasure_reed_sol_van", "-P", "k=4", "-P", "m=2"],
     "partialread,partialwrite,zeroinout,paritydelta,optimizedsupport,"
     "directreads"),
    (["-P", "technique=reed_sol_r6_op", "-P", "k=4", "-P", "m=2"],
     "partialread,partialwrite,zeroinout,paritydelta,crcencodedecode,"
     "directreads"),
    (["-P", "technique=cauchy_orig", "-P", "k=4", "-P", "m=2"],
     "partialread,partialwrite,zeroinout,paritydelta,directreads"),
    (["-P", "technique=cauchy_good", "-P", "k=4", "-P", "m=3"],
     "partialread,partialwrite,zeroinout,paritydelta,crcencodedecode,"
     "directreads"),
])
def test_mi355x_flags_mirror_reference(profile, expected):
    """Per-technique optimization flags mirror the owning reference
    plugin exactly (ErasureCodeIsa.h:66-79, ErasureCodeJerasure.h:52-63):
    isa techniques claim OPTIMIZED always and CRC for reed_sol_van /
    cauchy-at-m=1; jerasure techniques claim OPTIMIZED only for
    reed_sol_van and CRC for all but reed_sol_van and cauchy_orig;
    paritydelta is claimed for every technique (bitmatrix deltas are
    implemented via the schedule-delta kernel)."""
    r = run_bench("-p", "mi355x", *profile, "--flags")
    assert r.returncode == 0, r.stderr
    assert r.stdout.strip() == expected


def test_cauchy_good_cli_exhaustive_double_erasure():
    """cauchy_good through the dlopen plugin boundary at the headline
    cauchy width (k=10 m=4): exhaustive 2-erasure decode with byte
    verification by the CLI (benchmark.cc:211-258 style)."""
    r = run_bench("-p", "mi355x", "-P", "technique=cauchy_good",
                  "-P", "k=10", "-P", "m=4", "-P", "packetsize=512",
                  "-s", str(10 * 8 * 512 * 2), "-i", "1", "-w", "decode",
                  "-e", "2", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


def test_lrc_headline_shape_gpu():
    """LRC at the BASELINE configs[3]-adjacent shape (k=9 m=3 l=4 — the
    reference's parse_kml rejects k=8; see DESIGN) through the dlopen
    boundary: exhaustive single-erasure decode, byte-verified."""
    r = run_bench("-p", "lrc", "-P", "k=9", "-P", "m=3", "-P", "l=4",
                  "-s", str(9 * 65536), "-i", "1", "-w", "decode",
                  "-e", "1", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


def test_clay_wider_shape_gpu():
    """Clay at a wider shape (k=6 m=3 d=8): two-erasure exhaustive decode
    through decode_layered with the mi355x sub-codec."""
    r = run_bench("-p", "clay", "-P", "k=6", "-P", "m=3", "-P", "d=8",
                  "-s", str(6 * 65536), "-i", "1", "-w", "decode",
                  "-e", "2", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


def test_shec_wider_shape_gpu():
    """SHEC k=8 m=4 c=3: exhaustive single-erasure decode (the widest
    in-tree-documented shingle density at k=8)."""
    r = run_bench("-p", "shec", "-P", "k=8", "-P", "m=4", "-P", "c=3",
                  "-s", str(8 * 65536), "-i", "1", "-w", "decode",
                  "-e", "1", "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout
