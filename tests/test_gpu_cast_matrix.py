"""Full dtype-cast matrix for the gather kernels vs PyTorch's own casts:
every supported (store, out) pair on the GPU. Catches both wrong-width
vectorization and silent byte-reinterpretation (a same-size pair like
f16 -> bf16 must CONVERT)."""
import pytest
import torch

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(600)]

IN_DTYPES = [torch.uint8, torch.int32, torch.int64, torch.float32,
             torch.float64, torch.float16, torch.bfloat16,
             torch.float8_e4m3fn, torch.float8_e5m2]
OUT_DTYPES = [torch.uint8, torch.int32, torch.int64, torch.float32,
              torch.float64, torch.float16, torch.bfloat16]


def _mk(dtype, n, d):
    if dtype in (torch.uint8,):
        return torch.randint(0, 100, (n, d), dtype=torch.uint8)
    if dtype in (torch.int32, torch.int64):
        return torch.randint(-1000, 1000, (n, d)).to(dtype)
    # values well inside every float format's range, away from rounding ties
    return (torch.randn(n, d) * 4).to(dtype)


def test_cast_matrix():
    from ddstore_amd import DDStore

    s = DDStore(device="cuda:0")
    n, d = 64, 24
    idx = torch.randint(0, n, (32,), dtype=torch.int64)
    failures = []
    for it in IN_DTYPES:
        arr = _mk(it, n, d)
        name = f"m{str(it)}"
        s.add(name, arr)
        for ot in OUT_DTYPES:
            if it in (torch.float8_e4m3fn, torch.float8_e5m2) and ot in (
                torch.uint8, torch.int32, torch.int64
            ):
                continue  # float->int of fp8 noise: not a meaningful path
            out = s.get_batch(name, idx, dtype=ot)
            torch.cuda.synchronize()
            ref = arr[idx] if it == ot else arr[idx].to(ot)
            if not torch.equal(out.cpu(), ref):
                failures.append(f"{it} -> {ot}")
    s.free()
    assert not failures, "mismatched pairs: " + ", ".join(failures)
