"""Non-regression corpus tool (harness/ec_non_regression), mirroring
ceph_erasure_code_non_regression.cc: directory format, create/check
round trip, tamper detection — and (GPU) a cross-plugin check where a
corpus created by the CPU oracle plugin is verified byte-exact by the
mi355x GPU plugin, the same check a real ceph-erasure-code-corpus
directory would run."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HARNESS = os.path.join(ROOT, "ceph_amd", "harness")
TOOL = os.path.join(HARNESS, "ec_non_regression")


def run_tool(base, *args):
    if not os.path.exists(TOOL):
        pytest.skip("harness not built")
    return subprocess.run([TOOL, "-d", HARNESS, "--base", str(base), *args],
                          capture_output=True, text=True)


PROFILE = ["-P", "technique=reed_sol_van", "-P", "k=4", "-P", "m=2",
           "-s", "4096"]


def test_create_then_check_roundtrip(tmp_path):
    r = run_tool(tmp_path, "-p", "oracle", *PROFILE, "--create", "--check")
    assert r.returncode == 0, r.stderr
    d = tmp_path / "plugin=oracle stripe-width=4096 technique=reed_sol_van k=4 m=2"
    assert (d / "content").exists()
    # 6 chunk files named by shard id (reference chunk_path :297-302)
    assert sorted(p.name for p in d.iterdir()) == [
        "0", "1", "2", "3", "4", "5", "content"]


def test_check_detects_tampered_chunk(tmp_path):
    r = run_tool(tmp_path, "-p", "oracle", *PROFILE, "--create")
    assert r.returncode == 0, r.stderr
    d = tmp_path / "plugin=oracle stripe-width=4096 technique=reed_sol_van k=4 m=2"
    blob = bytearray((d / "4").read_bytes())
    blob[17] ^= 0xFF
    (d / "4").write_bytes(bytes(blob))
    r = run_tool(tmp_path, "-p", "oracle", *PROFILE, "--check")
    assert r.returncode != 0
    assert "encodes differently" in r.stderr


def test_check_unpadded_stripe_width(tmp_path):
    """Reference example uses stripe width 3181 (not chunk-aligned): the
    encode path pads per get_chunk_size; create/check must agree."""
    prof = ["-P", "technique=reed_sol_van", "-P", "k=3", "-P", "m=2",
            "-s", "3181"]
    r = run_tool(tmp_path, "-p", "oracle", *prof, "--create", "--check")
    assert r.returncode == 0, r.stderr


@pytest.mark.gpu
def test_cross_plugin_corpus_oracle_to_gpu(tmp_path):
    """Corpus created by the CPU oracle plugin verifies byte-exact under
    the mi355x GPU plugin: rename the directory so the plugin= segment
    matches, as the reference's corpus layout keys the directory name on
    the plugin that wrote it."""
    r = run_tool(tmp_path, "-p", "oracle", *PROFILE, "--create")
    assert r.returncode == 0, r.stderr
    src = tmp_path / "plugin=oracle stripe-width=4096 technique=reed_sol_van k=4 m=2"
    dst = tmp_path / "plugin=mi355x stripe-width=4096 technique=reed_sol_van k=4 m=2"
    src.rename(dst)
    r = run_tool(tmp_path, "-p", "mi355x", *PROFILE, "--check")
    assert r.returncode == 0, r.stderr + r.stdout
