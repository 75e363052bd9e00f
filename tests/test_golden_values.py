"""Golden-value loss-curve regression (reference functional_tests
pattern): deterministic fp32 CPU runs of the four architecture families
compared against committed curves — a broken backward fails here in
seconds instead of via convergence debugging on metered GPU time."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
GOLDEN = os.path.join(REPO, "tests", "golden", "golden_values_cpu.json")


def test_golden_loss_curves_match():
    # strip inherited rendezvous vars (earlier tests mutate os.environ)
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
                        "MASTER_PORT")}
    r = subprocess.run([sys.executable, "scripts/golden_values.py"],
                       capture_output=True, text=True, cwd=REPO,
                       timeout=600, env=env)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert "GOLDEN OK" in r.stdout


def test_golden_file_has_learning_curves():
    ref = json.load(open(GOLDEN))
    assert set(ref) == {"gpt", "llama", "mixtral", "fbdgrid"}
    for k, v in ref.items():
        assert v[-1] < v[0] - 0.5, (k, v[0], v[-1])
