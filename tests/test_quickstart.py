"""The docs/QUICKSTART.md flows, executable (CPU-adapted so the snippets in
the docs stay true to the code)."""
import numpy as np
import pytest
import torch

from ddstore_amd import DDStore, PrefetchLoader

pytestmark = pytest.mark.timeout(300)


def test_single_device_flow():
    store = DDStore(device="cpu")
    data = torch.randn(10_000, 32)
    store.add("train", data)
    idx = torch.randint(0, 10_000, (2_048,))
    batch = store.get_batch("train", idx, dtype=torch.bfloat16)
    assert batch.shape == (2048, 32) and batch.dtype == torch.bfloat16
    assert torch.equal(batch, data[idx].to(torch.bfloat16))
    store.free()


def test_epoch_loader_reshuffle_flow():
    from ddstore_amd.reshuffle import expected_perm

    store = DDStore(device="cpu")
    n = 4096
    store.add("train", torch.arange(n, dtype=torch.float32).unsqueeze(1))
    ntotal = store.query("train")["nrows_total"]
    seen = []
    perm = torch.randperm(ntotal, generator=torch.Generator().manual_seed(0))
    mine = perm[0::1]
    with store.epoch():
        for batch in PrefetchLoader(store, "train", mine, 512):
            seen.append(batch.view(-1))
    assert torch.equal(torch.cat(seen), mine.to(torch.float32))
    store.reshuffle("train", seed=0)
    p = expected_perm(ntotal, 0, store.device)
    out = store.get_batch("train", list(range(n)))
    assert torch.equal(out.view(-1), p.to(torch.float32))
    store.free()


def test_csr_and_compressed_flow():
    store = DDStore(device="cpu")
    lens = [3, 5, 2]
    feats = torch.randn(10, 4)
    store.add_csr("graphs", feats, lens)
    values, offsets = store.get_csr("graphs", [1, 0])
    assert offsets.tolist() == [0, 5, 8]
    pix = torch.randint(0, 255, (100, 16), dtype=torch.uint8)
    store.add("imgs", pix)
    b = store.get_batch("imgs", [7], dtype=torch.float32, affine=(1 / 255.0, -0.5))
    assert torch.allclose(b, pix[7:8].to(torch.float32) / 255.0 - 0.5)
    store.free()


def test_pyddstore_flow():
    import pyddstore

    s = pyddstore.PyDDStore(None, device="cpu")
    arr = np.random.rand(128, 64)
    s.add("traindata", arr)
    out = np.zeros((16, 64))
    s.epoch_begin()
    s.get("traindata", out, start=100)
    s.epoch_end()
    assert np.array_equal(out, arr[100:116])
    s.free()
