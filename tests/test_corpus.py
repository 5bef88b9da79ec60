"""Committed non-regression corpus (tests/golden/corpus): chunk archives
generated once by the CPU oracle plugin via ec_non_regression, pinning
bit-exactness across future versions of this repo — the role the
ceph-erasure-code-corpus replay plays for the reference.
Every config is re-checked by the oracle plugin on CPU and by the mi355x
plugin on GPU."""
import os
import shutil
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HARNESS = os.path.join(ROOT, "ceph_amd", "harness")
CORPUS = os.path.join(ROOT, "tests", "golden", "corpus")
TOOL = os.path.join(HARNESS, "ec_non_regression")

ALL_CONFIGS = sorted(d for d in os.listdir(CORPUS)
                     if d.startswith("plugin="))
CONFIGS = [d for d in ALL_CONFIGS if d.startswith("plugin=oracle ")]


def plugin_of(dirname):
    return dirname.split(" ")[0].split("=", 1)[1]


def args_of(dirname):
    """Recover the tool invocation from the directory name (the layout
    embeds stripe width and profile params in order)."""
    args = []
    for tok in dirname.split(" ")[1:]:
        k, v = tok.split("=", 1)
        if k == "stripe-width":
            args += ["-s", v]
        else:
            args += ["-P", tok]
    return args


def run_check(base, plugin, dirname):
    if not os.path.exists(TOOL):
        pytest.skip("harness not built")
    return subprocess.run(
        [TOOL, "-d", HARNESS, "--base", str(base), "-p", plugin,
         *args_of(dirname), "--check"], capture_output=True, text=True)


@pytest.mark.parametrize("dirname", ALL_CONFIGS)
def test_cpu_plugins_match_committed_corpus(dirname):
    """Every committed corpus directory re-checks with the plugin that
    wrote it (oracle for the base techniques; lrc/clay compose their CPU
    sub-plugins)."""
    r = run_check(CORPUS, plugin_of(dirname), dirname)
    assert r.returncode == 0, (dirname, r.stderr)


@pytest.mark.gpu
@pytest.mark.parametrize("dirname", CONFIGS)
def test_gpu_plugin_matches_committed_corpus(dirname, tmp_path):
    renamed = dirname.replace("plugin=oracle", "plugin=mi355x", 1)
    shutil.copytree(os.path.join(CORPUS, dirname), tmp_path / renamed)
    r = run_check(tmp_path, "mi355x", renamed)
    assert r.returncode == 0, (dirname, r.stderr)


def test_corpus_check_script():
    """tools/corpus_check.sh (the one-command external parity pin,
    INTEGRATION.md) replays the whole committed corpus clean with the
    CPU oracle plugin."""
    r = subprocess.run(
        [os.path.join(ROOT, "tools", "corpus_check.sh"), CORPUS, "oracle"],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "0 failed" in r.stdout
