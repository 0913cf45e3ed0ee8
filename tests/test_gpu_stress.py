"""Integration stress: prefetch-fed epochs interleaved with reshuffles and a
second (CSR) variable, contents verified through permutation composition."""
import numpy as np
import pytest
import torch

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(600)]


def test_epochs_prefetch_reshuffle_interleaved():
    from ddstore_amd import DDStore, PrefetchLoader
    from ddstore_amd.reshuffle import expected_perm

    s = DDStore(device="cuda:0")
    N, D = 4096, 32
    base = torch.arange(N, dtype=torch.float32).unsqueeze(1).repeat(1, D)
    s.add("x", base)
    lens = np.random.default_rng(0).integers(1, 9, size=256)
    vals = torch.cat([torch.full((int(l), 2), float(i)) for i, l in enumerate(lens)])
    s.add_csr("c", vals, lens)

    # slot_content[j] = original row currently at slot j
    slot_content = torch.arange(N)
    for epoch in range(4):
        order = torch.randperm(N)[: 2048]
        s.epoch_begin()
        got = []
        for batch in PrefetchLoader(s, "x", order, batch_size=256, depth=3):
            got.append(batch[:, 0].cpu().clone())
        # interleave a CSR read inside the epoch
        v, off = s.get_csr("c", [3, 100, 255])
        torch.cuda.synchronize()
        s.epoch_end()
        got = torch.cat(got)
        assert torch.equal(got, slot_content[order].to(torch.float32)), epoch
        off = off.cpu().tolist()
        assert (v.cpu()[off[0] : off[1]] == 3.0).all()
        # reshuffle between epochs; track composition
        s.reshuffle("x", seed=50 + epoch)
        p = expected_perm(N, 50 + epoch, s.device).cpu()
        slot_content = slot_content[p]
    st = s.query("x")
    assert st["oob_skipped"] == 0
    s.free()
