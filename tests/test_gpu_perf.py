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


def _measure_overlap_frac():
    """Config-5 regression tripwire, EVENT-based: >=85% of
    the side-stream fetch busy time must lie inside train-step busy
    intervals on the main stream. (An end-to-end wall-clock delta cannot
    resolve a ~100-us fetch under a multi-ms train step against ~1.5%
    timing noise -- measured swings of +-90%; and sizing the fetch up to be
    resolvable makes both sides bandwidth-bound, where overlap cannot hide
    anything. Event timelines measure the overlap directly, like the r1
    rocprof-trace analysis but automated.)"""
    import torch

    from ddstore_amd import DDStore, PrefetchLoader

    dev = torch.device("cuda:0")
    torch.cuda.set_device(dev)
    rows, dim, batch, steps = 1 << 21, 128, 1 << 18, 24
    store = DDStore(device=dev)
    store.add("ov", torch.randn(rows, dim, device=dev))

    model = torch.nn.Sequential(
        torch.nn.Linear(dim, 1024), torch.nn.GELU(), torch.nn.Linear(1024, dim)
    ).to(device=dev, dtype=torch.bfloat16)
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)

    def train_step(b):
        opt.zero_grad(set_to_none=True)
        torch.nn.functional.mse_loss(model(b), b).backward()
        opt.step()

    g = torch.Generator().manual_seed(7)
    order = torch.cat([torch.randperm(rows, generator=g) for _ in range(
        (steps + 4) * batch // rows + 1)])[: (steps + 4) * batch]

    store.epoch_begin()
    loader = PrefetchLoader(store, "ov", order, batch, out_dtype=torch.bfloat16,
                            depth=3, drop_last=True, collect_events=True)
    it = iter(loader)
    train_step(next(it))  # warm (model lazy init, loader spin-up)
    step_events = []
    for _ in range(steps):
        b = next(it)
        e0 = torch.cuda.Event(enable_timing=True)
        e1 = torch.cuda.Event(enable_timing=True)
        e0.record()
        train_step(b)
        e1.record()
        step_events.append((e0, e1))
    torch.cuda.synchronize()
    store.epoch_end()

    # timelines relative to the first train step's start
    ref = step_events[0][0]
    train_iv = [(0.0 if es is ref else ref.elapsed_time(es),
                 ref.elapsed_time(ee)) for es, ee in step_events]
    fetch_iv = [(ref.elapsed_time(es), ref.elapsed_time(ee))
                for es, ee in loader.fetch_events]
    # only fetches that START inside the timed window (prefill ones ran
    # before any train step existed to hide them)
    fetch_iv = [(s0, e0_) for s0, e0_ in fetch_iv
                if s0 >= 0 and s0 <= train_iv[-1][1]]

    def intersect(a, ivs):
        return sum(max(0.0, min(a[1], i1) - max(a[0], i0)) for i0, i1 in ivs)

    fetch_total = sum(e - s for s, e in fetch_iv)
    hidden = sum(intersect(f, train_iv) for f in fetch_iv)
    store.free()
    assert fetch_total > 0, "no fetch events recorded"
    return hidden / fetch_total


def test_prefetch_overlap_floor():
    """See _measure_overlap_frac. Measured twice when needed: a run that
    directly follows the subprocess benchmarks occasionally reports low
    overlap (box state), while an isolated run measures ~100%; the floor
    applies to the better of two fresh measurements."""
    frac = _measure_overlap_frac()
    if frac < 0.85:
        frac = max(frac, _measure_overlap_frac())
    assert frac >= 0.85, (
        f"prefetch overlap regressed: only {frac:.1%} of side-stream fetch "
        f"busy time overlaps train-step execution (best of 2 runs)"
    )


def test_bench_mode_floors():
    """End-to-end per-step floors for the two headline bench modes (r2
    measured: fetch 5.96-5.97G, csr 3.21G pipelined; floors leave
    ~12% for box variance)."""
    for mode, floor in [("fetch", 5.2e9), ("csr", 2.8e9)]:
        out = subprocess.run(
            [sys.executable, "bench.py", "--steps", "300", "--warmup", "50",
             "--mode", mode],
            capture_output=True, text=True, timeout=300,
        )
        assert out.returncode == 0, out.stderr[-1500:]
        val = json.loads(out.stdout.strip().splitlines()[-1])["value"]
        assert val >= floor, f"{mode}: {val/1e9:.2f}G < floor {floor/1e9:.1f}G"
