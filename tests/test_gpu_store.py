"""Single-GPU DeviceStore tests: HBM shards, CDNA4 gather kernels, numerics
vs a plain PyTorch fp32/host reference. Run with ``pytest -m gpu`` on an
MI355X box."""
import numpy as np
import pytest
import torch

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(600)]


@pytest.fixture
def store():
    from ddstore_amd import DDStore

    s = DDStore(device="cuda:0")
    assert s.mode == "hip", "GPU test must exercise the native HIP path"
    yield s
    s.free()


def test_native_extension_loaded():
    import ddstore_amd._C as C

    assert C.__file__.endswith(".so")
    assert "ddstore_amd" in C.__file__


def test_add_gather_roundtrip(store):
    arr = torch.randn(1024, 64, dtype=torch.float32)
    store.add("x", arr)
    idx = torch.randint(0, 1024, (256,), dtype=torch.int64)
    out = store.get_batch("x", idx)
    torch.cuda.synchronize()
    assert out.is_cuda
    assert torch.equal(out.cpu(), arr[idx])


@pytest.mark.parametrize(
    "dtype",
    [torch.uint8, torch.int32, torch.int64, torch.float32, torch.float64,
     torch.float16, torch.bfloat16, torch.float8_e4m3fn, torch.float8_e5m2],
)
def test_gather_dtypes(store, dtype):
    if dtype.is_floating_point:
        arr = torch.randn(128, 16).to(dtype)
    else:
        arr = torch.randint(0, 100, (128, 16)).to(dtype)
    store.add(f"v{str(dtype)}", arr)
    idx = torch.randint(0, 128, (64,), dtype=torch.int64)
    out = store.get_batch(f"v{str(dtype)}", idx)
    torch.cuda.synchronize()
    assert torch.equal(out.cpu().view(torch.uint8), arr[idx].view(torch.uint8))


def test_gather_odd_row_bytes(store):
    # row bytes not a multiple of 16 -> elementwise kernel path
    arr = torch.randn(100, 3, dtype=torch.float32)  # 12 B rows
    store.add("odd", arr)
    idx = torch.randint(0, 100, (37,), dtype=torch.int64)
    out = store.get_batch("odd", idx)
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), arr[idx])


def test_fused_cast_vs_fp32_reference(store):
    # store u8, gather as f32: kernel cast must equal the plain PyTorch cast
    arr = torch.randint(0, 255, (512, 32), dtype=torch.uint8)
    store.add("u8", arr)
    idx = torch.randint(0, 512, (128,), dtype=torch.int64)
    out = store.get_batch("u8", idx, dtype=torch.float32)
    torch.cuda.synchronize()
    ref = arr[idx].to(torch.float32)
    assert torch.equal(out.cpu(), ref)
    # store f16, gather as f32
    h = torch.randn(256, 24, dtype=torch.float16)
    store.add("h16", h)
    out2 = store.get_batch("h16", idx[:64] % 256, dtype=torch.float32)
    torch.cuda.synchronize()
    ref2 = h[(idx[:64] % 256)].to(torch.float32)
    assert torch.equal(out2.cpu(), ref2)
    # store f32, gather as bf16 (compressed fetch): matches torch's cast
    f = torch.randn(256, 24, dtype=torch.float32)
    store.add("f32c", f)
    out3 = store.get_batch("f32c", idx[:64] % 256, dtype=torch.bfloat16)
    torch.cuda.synchronize()
    ref3 = f[(idx[:64] % 256)].to(torch.bfloat16)
    assert torch.equal(out3.cpu(), ref3)


def test_get_range_d2h(store):
    arr = torch.arange(64 * 8, dtype=torch.float64).reshape(64, 8)
    store.add("r", arr)
    out = np.zeros((5, 8), dtype=np.float64)
    store.get("r", out, start=30)
    assert np.array_equal(out, arr[30:35].numpy())


def test_get_range_d2d(store):
    arr = torch.randn(64, 8)
    store.add("r2", arr)
    out = torch.empty(5, 8, device="cuda:0")
    store.get("r2", out, start=10)
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), arr[10:15])


def test_init_update_gpu(store):
    store.init("y", 32, 8, dtype=torch.float32)
    upd = torch.full((4, 8), 9.0)
    store.update("y", upd, offset=10)
    out = store.get_batch("y", list(range(32)))
    torch.cuda.synchronize()
    ref = torch.zeros(32, 8)
    ref[10:14] = 9.0
    assert torch.equal(out.cpu(), ref)


def test_csr_gpu(store):
    rng = np.random.default_rng(0)
    lengths = rng.integers(1, 50, size=200)
    total = int(lengths.sum())
    vals = torch.randn(total, 4)
    store.add_csr("c", vals, lengths)
    idx = rng.integers(0, 200, size=64)
    v, off = store.get_csr("c", idx)
    torch.cuda.synchronize()
    goff = np.zeros(201, dtype=np.int64)
    np.cumsum(lengths, out=goff[1:])
    off_h = off.cpu().tolist()
    v_h = v.cpu()
    for k, g in enumerate(idx):
        seg = v_h[off_h[k] : off_h[k + 1]]
        ref = vals[goff[g] : goff[g + 1]]
        assert torch.equal(seg, ref)


def test_csr_odd_elem_bytes(store):
    # 12-byte elements -> uint32 chunk path
    lengths = [5, 1, 9, 3]
    vals = torch.randn(18, 3)
    store.add_csr("c3", vals, lengths)
    v, off = store.get_csr("c3", [3, 1, 0, 2])
    torch.cuda.synchronize()
    goff = [0, 5, 6, 15, 18]
    off_h = off.cpu().tolist()
    for k, g in enumerate([3, 1, 0, 2]):
        assert torch.equal(v.cpu()[off_h[k] : off_h[k + 1]], vals[goff[g] : goff[g + 1]])


def test_local_shard_view_gpu(store):
    arr = torch.randn(16, 4)
    store.add("ls", arr)
    sh = store.local_shard("ls")
    assert sh.is_cuda and sh.shape == (16, 4)
    assert torch.equal(sh.cpu(), arr)


def test_reshuffle_gpu_single(store):
    arr = torch.arange(256, dtype=torch.float32).repeat_interleave(8).reshape(256, 8)
    store.add("x", arr)
    store.reshuffle("x", seed=11)
    from ddstore_amd.reshuffle import expected_perm

    perm = expected_perm(256, 11, store.device).cpu()
    out = store.get_batch("x", list(range(256)))
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), arr[perm])


def test_prefetch_loader_gpu(store):
    arr = torch.arange(512, dtype=torch.float32).reshape(512, 1).repeat(1, 16)
    store.add("p", arr)
    from ddstore_amd import PrefetchLoader

    order = np.random.default_rng(2).permutation(512)
    got = []
    for batch in PrefetchLoader(store, "p", order, batch_size=64, depth=3):
        assert batch.is_cuda and batch.shape == (64, 16)
        got.append(batch[:, 0].cpu().clone())
    torch.cuda.synchronize()
    got = torch.cat(got)
    assert torch.equal(got, torch.from_numpy(order).to(torch.float32))


def test_epoch_fsm_gpu(store):
    store.epoch_begin()
    with pytest.raises(RuntimeError, match="epoch already began"):
        store.epoch_begin()
    store.epoch_end()


def test_oob_index_skipped_and_counted(store):
    # device gathers must never read out of bounds: bad indices are skipped
    # and counted (the reference's host path throws; a device tensor of
    # indices cannot be checked host-side without a sync)
    arr = torch.randn(32, 8)
    store.add("ob", arr)
    idx = torch.tensor([0, 31, 32, -1, 5], dtype=torch.int64)
    out = store.get_batch("ob", idx)
    torch.cuda.synchronize()
    ok = [0, 31, 5]
    for k, g in enumerate(idx.tolist()):
        if g in ok:
            assert torch.equal(out[k].cpu(), arr[g])
    q = store.query("ob")
    assert q["oob_skipped"] == 2, q


def test_dump_load_gpu(tmp_path, store):
    arr = torch.randn(64, 16)
    store.add("ckg", arr)
    store.dump("ckg", str(tmp_path / "ckg.pt"))
    from ddstore_amd import DDStore

    s2 = DDStore(device="cuda:0")
    s2.load("ckg", str(tmp_path / "ckg.pt"))
    out = s2.get_batch("ckg", list(range(64)))
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), arr)
    s2.free()


@pytest.mark.parametrize("disp", [1, 2, 4, 7])
def test_csr_elem_widths_gpu(store, disp):
    # covers the dwordx4-realigned dword path (4/8 B elems), the uint4 path
    # (16 B elems) and the byte path (28 B elems? no: 7*4=28 -> dword path).
    rng = np.random.default_rng(disp)
    lengths = rng.integers(0, 64, size=300)
    total = int(lengths.sum())
    vals = torch.randn(total, disp)
    store.add_csr(f"cw{disp}", vals, lengths)
    idx = rng.integers(0, 300, size=128)
    v, off = store.get_csr(f"cw{disp}", idx)
    torch.cuda.synchronize()
    goff = np.concatenate([[0], np.cumsum(lengths)])
    off_h = off.cpu().tolist()
    v_h = v.cpu()
    for k, g in enumerate(idx):
        assert torch.equal(v_h[off_h[k] : off_h[k + 1]], vals[goff[g] : goff[g + 1]])


def test_reshuffle_csr_gpu(store):
    from ddstore_amd.reshuffle import expected_perm

    lengths = [3, 7, 2, 5, 1, 4]
    vals = torch.cat([torch.full((l, 2), float(i)) for i, l in enumerate(lengths)])
    store.add_csr("rcg", vals, lengths)
    store.reshuffle("rcg", seed=31)
    perm = expected_perm(6, 31, store.device).cpu().tolist()
    v, off = store.get_csr("rcg", list(range(6)))
    torch.cuda.synchronize()
    off = off.cpu().tolist()
    for j in range(6):
        seg = v.cpu()[off[j] : off[j + 1]]
        assert seg.shape[0] == lengths[perm[j]] and (seg == float(perm[j])).all()


@pytest.mark.parametrize("fp8", [torch.float8_e4m3fn, torch.float8_e5m2])
def test_fp8_fused_expand_gpu(store, fp8):
    # store fp8 (half the fetch bytes), gather expanded to bf16/f32 in-kernel;
    # must match torch's own fp8 decode exactly (OCP e4m3fn/e5m2 on gfx950)
    arr = torch.randn(512, 32).to(fp8)
    store.add(f"g8{str(fp8)[-4:]}", arr)
    idx = torch.randint(0, 512, (128,), dtype=torch.int64)
    out = store.get_batch(f"g8{str(fp8)[-4:]}", idx, dtype=torch.float32)
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), arr[idx].to(torch.float32))
    outb = store.get_batch(f"g8{str(fp8)[-4:]}", idx, dtype=torch.bfloat16)
    torch.cuda.synchronize()
    assert torch.equal(outb.cpu(), arr[idx].to(torch.bfloat16))


def test_add_update_from_device_tensors(store):
    # zero-copy device ingest path (new capability vs the reference's
    # host-only arrays)
    arr = torch.randn(64, 8, device="cuda:0")
    store.add("dev", arr)
    upd = torch.full((4, 8), 3.5, device="cuda:0")
    store.update("dev", upd, offset=10)
    out = store.get_batch("dev", list(range(64)))
    torch.cuda.synchronize()
    ref = arr.cpu().clone()
    ref[10:14] = 3.5
    assert torch.equal(out.cpu(), ref)


def test_prefetch_with_labels_gpu(store):
    from ddstore_amd import PrefetchLoader

    data = torch.arange(256, dtype=torch.float32).unsqueeze(1).repeat(1, 8)
    labels = (torch.arange(256, dtype=torch.int64) * 10).unsqueeze(1)
    store.add("pd", data)
    store.add("pl", labels)
    order = torch.randperm(256)
    for xb, yb in PrefetchLoader(store, "pd", order, 32, label_name="pl"):
        torch.cuda.synchronize()
        assert torch.equal(yb.view(-1).cpu(), (xb[:, 0].cpu() * 10).long())


def test_affine_gather_gpu(store):
    # fused normalization: u8 pixels -> normalized f32/bf16, one kernel
    arr = torch.randint(0, 255, (256, 32), dtype=torch.uint8)
    store.add("afg", arr)
    idx = torch.randint(0, 256, (64,), dtype=torch.int64)
    out = store.get_batch("afg", idx, dtype=torch.float32, affine=(1 / 255.0, -0.5))
    torch.cuda.synchronize()
    ref = arr[idx].to(torch.float32) * (1 / 255.0) - 0.5
    assert torch.allclose(out.cpu(), ref)
    outb = store.get_batch("afg", idx, dtype=torch.bfloat16, affine=(1 / 255.0, -0.5))
    torch.cuda.synchronize()
    refb = (arr[idx].to(torch.float32) * (1 / 255.0) - 0.5).to(torch.bfloat16)
    # kernel uses fmaf in f32 then casts; allow 1-ulp bf16 differences
    assert (outb.cpu().to(torch.float32) - refb.to(torch.float32)).abs().max() < 1e-2
    # f16 store with odd row width -> scalar affine fallback
    h = torch.randn(100, 5, dtype=torch.float16)
    store.add("afh", h)
    out5 = store.get_batch("afh", idx[:32] % 100, dtype=torch.float32, affine=(3.0, 1.0))
    torch.cuda.synchronize()
    ref5 = h[(idx[:32] % 100)].to(torch.float32) * 3.0 + 1.0
    assert torch.allclose(out5.cpu(), ref5, atol=1e-3, rtol=1e-3)


@pytest.mark.parametrize("dtype,dim", [(torch.float16, 4), (torch.float32, 3),
                                        (torch.float8_e4m3fn, 16)])
def test_reshuffle_odd_rows_gpu(store, dtype, dim):
    # exercises the scatter byte path (row_bytes % 16 != 0) and 1-byte dtypes
    from ddstore_amd.reshuffle import expected_perm

    n = 128
    arr = (torch.arange(n, dtype=torch.float32) % 13).unsqueeze(1).repeat(1, dim).to(dtype)
    store.add(f"ro{dim}{str(dtype)[-4:]}", arr)
    store.reshuffle(f"ro{dim}{str(dtype)[-4:]}", seed=77)
    perm = expected_perm(n, 77, store.device).cpu()
    out = store.get_batch(f"ro{dim}{str(dtype)[-4:]}", list(range(n)))
    torch.cuda.synchronize()
    assert torch.equal(out.cpu().view(torch.uint8), arr[perm].view(torch.uint8))


def test_prefetch_csr_gpu(store):
    from ddstore_amd import PrefetchLoader

    rng = np.random.default_rng(4)
    lengths = rng.integers(1, 12, size=300)
    vals = torch.cat(
        [torch.full((int(l), 2), float(i)) for i, l in enumerate(lengths)]
    )
    store.add_csr("pcg", vals, lengths)
    store.add("pcl", torch.arange(300, dtype=torch.int64).unsqueeze(1))
    order = rng.permutation(300)
    seen = 0
    for (v, off), y in PrefetchLoader(store, "pcg", order, batch_size=64,
                                      label_name="pcl", depth=3):
        torch.cuda.synchronize()
        off_h = off.cpu().tolist()
        y_h = y.view(-1).cpu().tolist()
        for k in range(len(off_h) - 1):
            g = int(order[seen + k])
            seg = v.cpu()[off_h[k] : off_h[k + 1]]
            assert y_h[k] == g
            assert seg.shape[0] == lengths[g] and (seg == float(g)).all()
        seen += len(off_h) - 1
    assert seen == 300


def test_pyddstore_numpy_staging_on_gpu():
    # the reference-compatible NumPy surface backed by HBM shards
    import pyddstore

    p = pyddstore.PyDDStore()
    try:
        assert p.store.mode == "hip"
        arr = np.random.rand(64, 8)
        p.add("t", arr)
        out = np.zeros((5, 8))
        p.epoch_begin()
        p.get("t", out, start=10)
        p.epoch_end()
        assert np.array_equal(out, arr[10:15])
        p.init("u", 16, 4, itemsize=8)
        p.update("u", np.full((4, 4), 2.0), offset=3)
        ou = np.zeros((16, 4))
        p.get("u", ou, 0)
        assert (ou[3:7] == 2).all() and ou.sum() == 2.0 * 16
    finally:
        p.free()


@pytest.mark.parametrize("nidx", [1, 17, 4095, 4096, 4097, 12288, 70000])
def test_csr_plan_scan_matches_cumsum(store, nidx):
    # the fused 3-kernel plan scan must match torch's cumsum exactly,
    # including at the 4096-sample block boundaries
    rng = np.random.default_rng(nidx)
    lens = rng.integers(0, 30, size=5000)
    vals = torch.randn(int(lens.sum()), 1)
    store.add_csr(f"ps{nidx}", vals, lens)
    idx = torch.from_numpy(rng.integers(0, 5000, size=nidx)).cuda()
    cap = nidx * 30
    out = torch.empty(cap, 1, device="cuda:0")
    off = store._backend.gather_csr_fast(f"ps{nidx}", idx, out)
    torch.cuda.synchronize()
    goff = torch.from_numpy(np.concatenate([[0], np.cumsum(lens)])).cuda()
    ref_lens = goff[idx + 1] - goff[idx]
    ref_off = torch.zeros(nidx + 1, dtype=torch.int64, device="cuda:0")
    torch.cumsum(ref_lens, 0, out=ref_off[1:])
    assert torch.equal(off, ref_off)
    # and the gathered payload for a few samples
    off_h = off.cpu().tolist()
    for k in [0, nidx // 2, nidx - 1]:
        g = int(idx[k])
        seg = out[off_h[k] : off_h[k + 1]].cpu()
        assert torch.equal(seg, vals[int(goff[g]) : int(goff[g + 1])])


def test_csr_fused_large_multitile(store):
    # many lookback tiles (600k samples = ~2344 tiles) + payload check:
    # stresses the decoupled-lookback chain across grid-stride tile rounds
    rng = np.random.default_rng(99)
    lens = rng.integers(0, 9, size=20000)
    vals = torch.arange(int(lens.sum()), dtype=torch.float32).reshape(-1, 1)
    store.add_csr("big", vals, lens)
    nidx = 600_000
    idx = torch.from_numpy(rng.integers(0, 20000, size=nidx)).cuda()
    out = torch.empty(nidx * 9, 1, device="cuda:0")
    off = store._backend.gather_csr_fast("big", idx, out)
    torch.cuda.synchronize()
    goff = torch.from_numpy(np.concatenate([[0], np.cumsum(lens)])).cuda()
    ref_off = torch.zeros(nidx + 1, dtype=torch.int64, device="cuda:0")
    torch.cumsum(goff[idx + 1] - goff[idx], 0, out=ref_off[1:])
    assert torch.equal(off, ref_off)
    # spot-check payloads at tile boundaries and random positions
    off_h = off.cpu()
    goff_h = goff.cpu()
    for k in [0, 255, 256, 257, 65535, 65536, nidx - 1, 300_001]:
        g = int(idx[k])
        seg = out[int(off_h[k]) : int(off_h[k + 1])].cpu()
        assert torch.equal(seg, vals[int(goff_h[g]) : int(goff_h[g + 1])]), k


def test_csr_capacity_clamp_gpu(store):
    lens = [4, 4, 4, 4]
    vals = torch.arange(16, dtype=torch.float32).reshape(16, 1)
    store.add_csr("clamp", vals, lens)
    guard = torch.full((64,), -7.0, device="cuda:0")  # heap canary after out
    out = torch.zeros(8, 1, device="cuda:0")
    v, off = store.get_csr("clamp", [0, 1, 2, 3], out=out)
    torch.cuda.synchronize()
    q = store.query("clamp")
    assert q["cap_skipped"] == 2, q
    assert torch.equal(out.cpu(), vals[:8])
    assert (guard == -7.0).all()  # nothing wrote past the capacity buffer
    assert q["bytes_gathered"] == 8 * 4  # true bytes, not 4-sample total


def test_csr_fused_true_byte_stats(store):
    lens = [3, 1]
    vals = torch.arange(4, dtype=torch.float32).reshape(4, 1)
    store.add_csr("tb", vals, lens)
    big = torch.zeros(4096, 1, device="cuda:0")  # heavily oversized capacity
    store.get_csr("tb", [0, 1], out=big)
    torch.cuda.synchronize()
    assert store.query("tb")["bytes_gathered"] == 4 * 4


def test_misaligned_output_view_gpu(store):
    # an offset view of a buffer is contiguous but not 16-B aligned: the
    # dispatch must fall back to scalar kernels, not emit misaligned vector
    # stores
    arr = torch.randn(64, 8)
    store.add("mis", arr)
    back = torch.empty(65 * 8 + 1, device="cuda:0")
    out = back[1 : 1 + 64 * 8].view(64, 8)  # 4-B-aligned only
    idx = torch.arange(64, device="cuda:0")
    store.gather_into("mis", idx, out)
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), arr)
    # same-dtype byte-move path with a bf16 store (2-B elements)
    arr16 = torch.randn(32, 16, dtype=torch.bfloat16)
    store.add("mis16", arr16)
    back16 = torch.empty(33 * 16, dtype=torch.bfloat16, device="cuda:0")
    out16 = back16[1 : 1 + 32 * 16].view(32, 16)  # 2-B-aligned only
    store.gather_into("mis16", torch.arange(32, device="cuda:0"), out16)
    torch.cuda.synchronize()
    assert torch.equal(out16.cpu(), arr16)


def test_csr_init_update_gpu(store):
    lens = torch.tensor([3, 1, 4, 2])
    store.init_csr("icu", lens, disp=2, dtype=torch.float32)
    v, off = store.get_csr("icu", [0, 1, 2, 3])
    torch.cuda.synchronize()
    assert v.abs().sum() == 0 and off.cpu().tolist() == [0, 3, 4, 8, 10]
    part0 = torch.arange(8, dtype=torch.float32).reshape(4, 2)
    part1 = 100 + torch.arange(12, dtype=torch.float32).reshape(6, 2)
    store.update_csr("icu", part0, offset=0)
    store.update_csr("icu", part1, offset=2)
    v, _ = store.get_csr("icu", [0, 1, 2, 3])
    torch.cuda.synchronize()
    assert torch.equal(v.cpu(), torch.cat([part0, part1]))


def test_strict_mode_gpu_subprocess():
    import os
    import subprocess
    import sys

    code = """
import torch
from ddstore_amd import DDStore
s = DDStore(device="cuda:0")
s.add("x", torch.randn(8, 8))
s.epoch_begin()
out = s.get_batch("x", [0, 99])  # 99 out of range
torch.cuda.synchronize()
try:
    s.epoch_end()
    print("NO-RAISE")
except RuntimeError as e:
    assert "DDSTORE_STRICT" in str(e), e
    print("RAISED-OK")
s._backend.free_all()
"""
    env = dict(os.environ, DDSTORE_STRICT="1")
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300,
                       cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr
    assert "RAISED-OK" in r.stdout


def test_chunked_reshuffle_gpu(store):
    from ddstore_amd.reshuffle import expected_perm

    n = 4096
    arr = torch.arange(n, dtype=torch.float32).unsqueeze(1).repeat(1, 16)
    store.add("chk", arr)
    store.reshuffle("chk", seed=5, max_chunk_bytes=16 * 64 * 100)  # many chunks
    perm = expected_perm(n, 5, store.device).cpu()
    out = store.get_batch("chk", list(range(n)))
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), arr[perm])
