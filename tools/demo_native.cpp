// Native C++ smoke demo -- the MI355X analog of the reference's test/demo.cxx
// (C++ API, 2 rows x 2 cols, neighbor-shard read): exercises
// ddstore::DeviceStore directly from C++, no Python. Single rank (the
// multi-rank IPC path needs a collective plane; that lives in the Python
// layer and is covered by tests/test_gpu_multiproc.py).
//
// Build + run (on a ROCm box):
//   bash tools/build_demo_native.sh && ./tools/demo_native
#include "../ddstore_amd/csrc/ddstore.hpp"

#include <iostream>

int main() {
    if (!torch::cuda::is_available()) {
        std::cerr << "demo_native: no GPU visible\n";
        return 77;
    }
    ddstore::DeviceStore store(/*device=*/0, /*rank=*/0, /*nparts=*/1);
    auto opts = at::TensorOptions().dtype(at::kFloat);
    at::Tensor shard = at::arange(64, opts).reshape({16, 4});
    store.add("demo", shard, 16, 4, {16});
    store.open_peers("demo", {std::string()});

    store.epoch_begin();
    at::Tensor idx = at::tensor({15L, 0L, 7L},
                                at::TensorOptions().dtype(at::kLong).device(at::kCUDA));
    at::Tensor out = at::empty({3, 4}, opts.device(at::kCUDA));
    store.gather("demo", idx, out);
    store.epoch_end();

    at::Tensor expect = shard.index_select(0, at::tensor({15L, 0L, 7L}));
    TORCH_CHECK(at::equal(out.cpu(), expect), "gather mismatch");

    at::Tensor range_out = at::empty({4, 4}, opts);
    store.get_range("demo", 6, 4, range_out);
    TORCH_CHECK(at::equal(range_out, shard.slice(0, 6, 10)), "range mismatch");

    auto fail = [&](auto fn) {
        try { fn(); return false; } catch (const c10::Error&) { return true; }
    };
    TORCH_CHECK(fail([&] { store.epoch_end(); }), "double-end must throw");

    store.free_all();
    std::cout << "demo_native OK: C++ DeviceStore add/gather/get_range/epoch verified\n";
    return 0;
}
