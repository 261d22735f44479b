// pybind11 bindings for libhipshuffle — torch-free; Python passes raw
// device pointers (tensor.data_ptr()) and stream handles
// (torch.cuda.current_stream().cuda_stream).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "common.h"
#include "hipshuffle.h"

namespace py = pybind11;
namespace hs = hipshuffle;

PYBIND11_MODULE(_hipshuffle, m) {
  m.doc() = "MI355X-native shuffle data plane: HBM slabs, ROCm IPC, xGMI "
            "peer copies, CDNA4 radix partition/sort kernels";

  m.def("device_count", &hs::device_count);
  m.def("set_device", [](int d) { HIP_CHECK(hipSetDevice(d)); });
  m.def("device_synchronize", [] { HIP_CHECK(hipDeviceSynchronize()); });

  // slabs + IPC
  m.def("slab_alloc", &hs::slab_alloc_id, py::arg("bytes"));
  m.def("slab_free", &hs::slab_free_id);
  m.def("slab_base", &hs::slab_base_id);
  m.def("slab_handle",
        [](int id) { return py::bytes(hs::slab_handle_id(id)); });
  m.def("ipc_open", [](py::bytes handle) {
    return hs::ipc_open(std::string(handle));
  });
  m.def("ipc_close", &hs::ipc_close);
  m.def("enable_peer_access", &hs::enable_peer_access);

  // copy engine
  m.def("read_batch", &hs::read_batch_ids, py::arg("peer"), py::arg("dsts"),
        py::arg("srcs"), py::arg("sizes"),
        py::call_guard<py::gil_scoped_release>());
  m.def("poll_event", &hs::poll_event,
        py::call_guard<py::gil_scoped_release>());
  m.def("wait_event", &hs::wait_event,
        py::call_guard<py::gil_scoped_release>());
  m.def("tcp_recv_chunks", &hs::tcp_recv_chunks,
        py::call_guard<py::gil_scoped_release>());
  m.def("tcp_send_all", &hs::tcp_send_all,
        py::call_guard<py::gil_scoped_release>());
  m.def("host_alloc_pinned", &hs::host_alloc_pinned);
  m.def("host_free_pinned", &hs::host_free_pinned);
  m.def("memcpy_h2d", &hs::memcpy_h2d,
        py::call_guard<py::gil_scoped_release>());
  m.def("memcpy_d2h", &hs::memcpy_d2h,
        py::call_guard<py::gil_scoped_release>());

  // kernels
  m.def("radix_hist_bytes", &hs::radix_hist_bytes);
  m.def("radix_scan_ws_bytes", &hs::radix_scan_ws_bytes);
  m.def("radix_hist", &hs::radix_hist, py::arg("keys"), py::arg("n"),
        py::arg("shift"), py::arg("nbits"), py::arg("hist"),
        py::arg("stream") = 0, py::arg("func") = 0, py::arg("in_stride") = 1,
        py::arg("nparts") = 0);
  m.def("radix_scan", &hs::radix_scan, py::arg("hist"), py::arg("n"),
        py::arg("nbits"), py::arg("totals"), py::arg("scan_ws"),
        py::arg("stream") = 0);
  m.def("radix_scatter", &hs::radix_scatter, py::arg("keys"), py::arg("vals"),
        py::arg("n"), py::arg("shift"), py::arg("nbits"), py::arg("hist"),
        py::arg("key_dst"), py::arg("val_dst"), py::arg("stream") = 0,
        py::arg("func") = 0, py::arg("aos_out") = 0, py::arg("in_stride") = 1,
        py::arg("nparts") = 0);
  m.def("extract_pairs", &hs::extract_pairs, py::arg("recs"), py::arg("n"),
        py::arg("rec_bytes"), py::arg("key_bytes"), py::arg("pairs"),
        py::arg("stream") = 0, py::arg("pid_func") = -1,
        py::arg("pid_shift") = 0, py::arg("pid_mask") = 0,
        py::arg("pid_nparts") = 0, py::arg("idx_base") = 0);
  m.def("gather_records", &hs::gather_records, py::arg("recs"),
        py::arg("pairs"), py::arg("n"), py::arg("rec_bytes"),
        py::arg("dst_mode"), py::arg("out_base"), py::arg("dst_addr") = 0,
        py::arg("dstart") = 0, py::arg("shift") = 0, py::arg("mask") = 0,
        py::arg("func") = 0, py::arg("nparts") = 0, py::arg("stream") = 0);
  m.def("onesweep_sort_aos_word_u64", &hs::onesweep_sort_aos_word_u64,
        py::arg("pairs"), py::arg("tmp_pairs"), py::arg("n"),
        py::arg("start_bit"), py::arg("end_bit"), py::arg("ws"),
        py::arg("stream") = 0, py::arg("sort_word") = 0,
        py::call_guard<py::gil_scoped_release>());
  m.def("onesweep_sort_aos_fused_u64", &hs::onesweep_sort_aos_fused_u64,
        py::arg("pairs"), py::arg("tmp_pairs"), py::arg("n"),
        py::arg("start_bit"), py::arg("end_bit"), py::arg("ws"),
        py::arg("stream") = 0, py::arg("sort_word") = 0,
        py::arg("recs") = 0, py::arg("out") = 0, py::arg("rec_bytes") = 0,
        py::call_guard<py::gil_scoped_release>());
  m.def("sort_workspace_bytes", &hs::sort_workspace_bytes);
  m.def("sort_pairs_u64", &hs::sort_pairs_u64, py::arg("keys"),
        py::arg("vals"), py::arg("tmp_keys"), py::arg("tmp_vals"),
        py::arg("n"), py::arg("start_bit"), py::arg("end_bit"), py::arg("ws"),
        py::arg("stream") = 0,
        py::call_guard<py::gil_scoped_release>());
  m.def("probe_scatter_write", &hs::probe_scatter_write);
  m.def("join_count", &hs::join_count, py::arg("a_keys"), py::arg("na"),
        py::arg("b_keys"), py::arg("nb"), py::arg("counts"),
        py::arg("lo_idx"), py::arg("stream") = 0);
  m.def("join_emit", &hs::join_emit, py::arg("a_keys"), py::arg("a_vals"),
        py::arg("na"), py::arg("b_vals"), py::arg("counts"),
        py::arg("lo_idx"), py::arg("offsets"), py::arg("out_key"),
        py::arg("out_a"), py::arg("out_b"), py::arg("stream") = 0);
  m.def("onesweep_workspace_bytes", &hs::onesweep_workspace_bytes);
  m.def("set_aos_tile", &hs::set_aos_tile);
  m.def("set_pass_stage", &hs::set_pass_stage);
  m.def("set_timing_buf", &hs::set_timing_buf);
  m.def("set_sort_mode", &hs::set_sort_mode);
  m.def("set_lookback_mode", &hs::set_lookback_mode);
  m.def("set_split_exchange", &hs::set_split_exchange);
  m.def("set_lean_pass", &hs::set_lean_pass);
  m.def("onesweep_sort_aos7_u64", &hs::onesweep_sort_aos7_u64,
        py::arg("pairs"), py::arg("tmp_pairs"), py::arg("n"),
        py::arg("start_bit"), py::arg("end_bit"), py::arg("ws"),
        py::arg("stream") = 0, py::call_guard<py::gil_scoped_release>());
  m.def("onesweep_sort_aos_u64", &hs::onesweep_sort_aos_u64,
        py::arg("pairs"), py::arg("tmp_pairs"), py::arg("n"),
        py::arg("start_bit"), py::arg("end_bit"), py::arg("ws"),
        py::arg("stream") = 0, py::call_guard<py::gil_scoped_release>());
  m.def("onesweep_sort_pairs_u64", &hs::onesweep_sort_pairs_u64,
        py::arg("keys"), py::arg("vals"), py::arg("tmp_keys"),
        py::arg("tmp_vals"), py::arg("n"), py::arg("start_bit"),
        py::arg("end_bit"), py::arg("ws"), py::arg("stream") = 0,
        py::call_guard<py::gil_scoped_release>());
}
