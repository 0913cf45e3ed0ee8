"""Performance regression floors (MI355X): the microbench kernels must stay
within ~20% of their tuned round-1 numbers (profiles/microbench.txt).
Thresholds are deliberately loose to absorb DVFS/box variance (~5%) while
catching real regressions (the kind that halved bandwidth during tuning)."""
import json
import subprocess
import sys

import pytest

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(600)]

# measured (GB/s r+w) -> floor
FLOORS = {
    "gather f32->f32 (512B rows)": 3700,          # measured 4665-4717
    "gather f32->bf16 (fused cast)": 3700,        # measured 4696-4759
    "gather u8->f32 (fused expand)": 3200,        # measured 3988-4123
    "gather f16->bf16": 3200,                     # measured 4073-4213
    "gather f32 64B rows": 2100,                  # measured 2746-2768
}
CSR_FLOOR = 3500  # measured 4359


def test_gather_bandwidth_floors():
    out = subprocess.run(
        [sys.executable, "tools/microbench.py", "--json"],
        capture_output=True, text=True, timeout=500,
    )
    assert out.returncode == 0, out.stderr[-1500:]
    res = json.loads(out.stdout.strip().splitlines()[-1])
    failures = []
    for name, floor in FLOORS.items():
        got = res[name]["GBps"]
        if got < floor:
            failures.append(f"{name}: {got:.0f} < floor {floor}")
    csr = next(v for k, v in res.items() if k.startswith("gather_csr"))
    if csr["GBps"] < CSR_FLOOR:
        failures.append(f"csr: {csr['GBps']:.0f} < floor {CSR_FLOOR}")
    assert not failures, "\n".join(failures)
