"""Compiled-resource regression guard (no GPU needed): the prepared
higher-occupancy kernel variants must KEEP their register/spill budgets
— an edit that pushes them over shows up here, before any GPU time is
spent (tools/dump_kernel_resources.py)."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _rows():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "dump_kernel_resources.py")],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-500:]
    rows = {}
    for ln in out.stdout.splitlines()[1:]:
        parts = ln.rsplit(None, 6)
        if len(parts) == 7:
            name, vgpr, agpr, sgpr, spill, lds, waves = parts
            rows[name.strip()] = {
                "vgpr": int(vgpr), "spill": int(spill),
                "lds": int(lds), "waves": int(waves),
            }
    return rows


@pytest.fixture(scope="module")
def rows():
    so = [f for f in os.listdir(os.path.join(REPO, "quintnet_amd"))
          if f.startswith("_C") and f.endswith(".so")]
    if not so:
        pytest.skip("extension not built")
    return _rows()


def test_occupancy_variants_stay_spill_free(rows):
    budgets = {
        # kernel-name fragment: (max vgpr, min waves/SIMD) with 0 spills
        "attn_bwd_dkv_np_kernel": (168, 3),
        "attn_bwd_dq_np_kernel": (128, 4),
        "wgrad_tn32_kernel": (128, 4),
    }
    for frag, (max_vgpr, min_waves) in budgets.items():
        match = [v for k, v in rows.items() if frag in k]
        assert match, f"{frag} missing from the build"
        r = match[0]
        assert r["spill"] == 0, (frag, r)
        assert r["vgpr"] <= max_vgpr, (frag, r)
        assert r["waves"] >= min_waves, (frag, r)


def test_default_kernels_stay_spill_free(rows):
    for frag in ("attn_fwd_kernel<3, true>", "attn_bwd_dq_kernel<3>",
                 "wgrad_tn_kernel"):
        match = [v for k, v in rows.items() if frag in k]
        assert match, frag
        assert match[0]["spill"] == 0, (frag, match[0])
    # the shipped dkv kernel: spill-free at its 2-wave point
    dkv = [v for k, v in rows.items()
           if "attn_bwd_dkv_kernel<2>" in k]
    assert dkv and dkv[0]["spill"] == 0
