"""SCALE-run readiness: the driver launches bench.py
via torch.distributed.run with N ranks; these CPU tests pin the arg/env
parsing and the rank->device map with bench.py --dry-run, so an 8-GPU
SCALE run needs zero fixes."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(ROOT, "bench.py")


def dry(env_extra=None, *args):
    env = dict(os.environ)
    env.update(env_extra or {})
    r = subprocess.run([sys.executable, BENCH, "--dry-run", *args],
                       capture_output=True, text=True, env=env, cwd=ROOT)
    assert r.returncode == 0, r.stderr
    return json.loads(r.stdout.strip().splitlines()[-1])


def test_single_rank_defaults():
    d = dry()
    assert d["rank"] == 0 and d["world"] == 1 and d["device"] == 0
    assert d["k"] == 8 and d["m"] == 3 and d["chunk_bytes"] == 1 << 20
    assert d["stripes_per_gpu"] == 4096
    assert d["scaling"] == "weak"


def test_eight_rank_env_map():
    """One process per GPU: LOCAL_RANK i -> device i, per-rank work fixed
    (weak scaling), rendezvous env passed through."""
    for i in range(8):
        d = dry({"RANK": str(i), "WORLD_SIZE": "8", "LOCAL_RANK": str(i),
                 "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29511"})
        assert d["world"] == 8
        assert d["device"] == i
        assert d["backend"] == "nccl"
        assert d["master_addr"] == "127.0.0.1"
        assert d["buf_bytes"] == 4096 * 11 * (1 << 20)  # per-rank constant


def test_driver_flag_shapes():
    """The exact flag set the driver uses (--gpus N --steps K --warmup W)
    parses and shows up in the plan."""
    d = dry({"RANK": "3", "WORLD_SIZE": "4", "LOCAL_RANK": "3"},
            "--gpus", "4", "--steps", "10", "--warmup", "3")
    assert d["steps"] == 10 and d["warmup"] == 3 and d["rank"] == 3
