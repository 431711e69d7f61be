"""bench.py driver contract: one JSON line on stdout with the agreed
keys (the round driver parses exactly this — a key rename or a stray
print on stdout breaks the BENCH/SCALE records)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_emits_contract_json():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--tiny", "--gpus", "1", "--steps", "1", "--warmup", "0"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-800:]
    payloads = []
    for ln in out.stdout.splitlines():
        ln = ln.strip()
        if ln.startswith("{"):
            payloads.append(json.loads(ln))
    assert len(payloads) == 1, out.stdout
    j = payloads[0]
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in j, key
    assert j["n_gpus"] == 1 and j["steps"] == 1 and j["warmup"] == 0
    assert j["higher_is_better"] is True and j["scaling"] == "weak"
    assert j["data"] == "synthetic"
    for ck in ("model", "global_batch", "seq_len", "parallelism"):
        assert ck in j["config"], ck
    assert j["value"] > 0 and j["ms_per_step"] > 0


def test_bench_vit_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model", "vit",
         "--tiny", "--gpus", "1", "--steps", "1", "--warmup", "0"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-800:]
    payloads = [json.loads(l) for l in out.stdout.splitlines()
                if l.strip().startswith("{")]
    assert len(payloads) == 1
    assert payloads[0]["config"]["model"].startswith("vit")
