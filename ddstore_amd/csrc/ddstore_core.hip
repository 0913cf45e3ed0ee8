// ddstore_amd core: native store classes + Python bindings.
//
// MI355X-native re-design of the reference DDStore core
// (reference: include/ddstore.hpp:26-258, src/ddstore.cxx:19-96):
//
//   * DeviceStore -- each rank (one process per GPU) owns its shard in local
//     HBM3E via hipMalloc. Peer shards are mapped with hipIpcOpenMemHandle so
//     a `get` is a direct xGMI read: either a batched in-kernel gather
//     (ddstore_kernels.hip) or hipMemcpyAsync D2D for contiguous ranges.
//     This replaces both reference transports (MPI_Win RMA, ddstore.hpp:56-62/
//     222-237, and libfabric fi_read, common.cxx:13-376) with ONE transport.
//   * HostStore -- CPU compatibility path (BASELINE config 1): shards live in
//     POSIX shared memory; every rank on the node maps every peer's segment,
//     so a remote get is a one-sided memcpy, preserving the reference's
//     "no receiver-side code" property without MPI windows.
//
// The shard directory is the reference's prefix-sum lenlist
// (ddstore.hpp:75-89) replicated host-side and, for DeviceStore, mirrored to
// device memory for in-kernel owner lookup. Owner lookup on the host is a
// binary search (reference `sortedsearch` is a linear scan, ddstore.cxx:5-17).
//
// Epoch semantics (reference ddstore.cxx:51-77): epoch_begin/epoch_end keep
// the double-begin / end-without-begin misuse throws; the collective barrier
// is issued by the Python layer (torch.distributed), while this class does
// the local fence (hipStreamSynchronize of the store's stream).

#include "ddstore.hpp"

using ddstore::DeviceStore;
using ddstore::HostStore;

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "ddstore_amd native core (MI355X / gfx950)";
    m.attr("MAX_PARTS") = DDS_MAX_PARTS;
    m.def("cycle_order", &ddstore::cycle_order,
          py::call_guard<py::gil_scoped_release>(),
          "Concatenated-cycle traversal of a permutation (order, starts)");

    py::class_<DeviceStore>(m, "DeviceStore")
        .def(py::init<int, int, int>(), py::arg("device"), py::arg("rank"),
             py::arg("nparts"))
        .def("add", &DeviceStore::add, py::call_guard<py::gil_scoped_release>())
        .def("init", &DeviceStore::init, py::call_guard<py::gil_scoped_release>())
        .def("add_csr", &DeviceStore::add_csr, py::call_guard<py::gil_scoped_release>())
        .def("init_csr", &DeviceStore::init_csr, py::call_guard<py::gil_scoped_release>())
        .def("ipc_handle", &DeviceStore::ipc_handle)
        .def("open_peers", &DeviceStore::open_peers)
        .def("update", &DeviceStore::update, py::call_guard<py::gil_scoped_release>())
        .def("update_elems", &DeviceStore::update_elems, py::call_guard<py::gil_scoped_release>())
        .def("reset_counters", &DeviceStore::reset_counters)
        .def("check_strict", &DeviceStore::check_strict)
        .def("get_range", &DeviceStore::get_range, py::call_guard<py::gil_scoped_release>())
        .def("gather", &DeviceStore::gather, py::call_guard<py::gil_scoped_release>())
        .def("gather_affine", &DeviceStore::gather_affine, py::call_guard<py::gil_scoped_release>())
        .def("gather_csr", &DeviceStore::gather_csr, py::call_guard<py::gil_scoped_release>())
        .def("csr_lens", &DeviceStore::csr_lens, py::call_guard<py::gil_scoped_release>())
        .def("gather_csr_fast", &DeviceStore::gather_csr_fast, py::call_guard<py::gil_scoped_release>())
        .def("scatter_local", &DeviceStore::scatter_local, py::call_guard<py::gil_scoped_release>())
        .def("local_shard", &DeviceStore::local_shard)
        .def("epoch_begin", &DeviceStore::epoch_begin, py::call_guard<py::gil_scoped_release>())
        .def("epoch_end", &DeviceStore::epoch_end, py::call_guard<py::gil_scoped_release>())
        .def("epoch_active", &DeviceStore::epoch_active)
        .def("query", &DeviceStore::query)
        .def("has", &DeviceStore::has)
        .def("free_var", &DeviceStore::free_var)
        .def("free_all", &DeviceStore::free_all);

    py::class_<HostStore>(m, "HostStore")
        .def(py::init<const std::string&, int, int>(), py::arg("session"),
             py::arg("rank"), py::arg("nparts"))
        .def("shm_name", &HostStore::shm_name)
        .def("add", &HostStore::add, py::call_guard<py::gil_scoped_release>())
        .def("init", &HostStore::init, py::call_guard<py::gil_scoped_release>())
        .def("add_csr", &HostStore::add_csr, py::call_guard<py::gil_scoped_release>())
        .def("init_csr", &HostStore::init_csr, py::call_guard<py::gil_scoped_release>())
        .def("open_peers", &HostStore::open_peers)
        .def("update", &HostStore::update, py::call_guard<py::gil_scoped_release>())
        .def("update_elems", &HostStore::update_elems, py::call_guard<py::gil_scoped_release>())
        .def("reset_counters", &HostStore::reset_counters)
        .def("check_strict", &HostStore::check_strict)
        .def("get_range", &HostStore::get_range, py::call_guard<py::gil_scoped_release>())
        .def("gather", &HostStore::gather, py::call_guard<py::gil_scoped_release>())
        .def("gather_csr", &HostStore::gather_csr, py::call_guard<py::gil_scoped_release>())
        .def("scatter_local", &HostStore::scatter_local, py::call_guard<py::gil_scoped_release>())
        .def("local_shard", &HostStore::local_shard)
        .def("epoch_begin", &HostStore::epoch_begin, py::call_guard<py::gil_scoped_release>())
        .def("epoch_end", &HostStore::epoch_end, py::call_guard<py::gil_scoped_release>())
        .def("epoch_active", &HostStore::epoch_active)
        .def("query", &HostStore::query)
        .def("has", &HostStore::has)
        .def("free_var", &HostStore::free_var)
        .def("free_all", &HostStore::free_all);
}
