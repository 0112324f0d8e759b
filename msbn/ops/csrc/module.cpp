// msbn._C pybind module: gfx950 BatchNorm kernels + C++ DDP reducer.
#include <torch/extension.h>

#include <torch/csrc/distributed/c10d/ProcessGroup.hpp>

#include "bn_ops.hpp"
#include "reducer.hpp"

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "msbn MI355X-native kernels + reducer";

  // ---- BatchNorm op set (gfx950 HIP kernels) ----
  m.def("batch_norm_stats", &msbn::batch_norm_stats, py::arg("input"),
        py::arg("eps"));
  m.def("batch_norm_stats_packed", &msbn::batch_norm_stats_packed,
        py::arg("input"), py::arg("eps"), py::arg("out"));
  m.def("batch_norm_gather_stats_with_counts",
        &msbn::batch_norm_gather_stats_with_counts, py::arg("mean_all"),
        py::arg("invstd_all"), py::arg("running_mean"), py::arg("running_var"),
        py::arg("momentum"), py::arg("eps"), py::arg("counts"));
  m.def("batch_norm_gather_stats_packed",
        &msbn::batch_norm_gather_stats_packed, py::arg("packed_all"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("momentum"),
        py::arg("eps"));
  m.def("batch_norm_gather_stats_packed_coefs",
        &msbn::batch_norm_gather_stats_packed_coefs, py::arg("packed_all"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("momentum"),
        py::arg("eps"), py::arg("weight"), py::arg("bias"),
        py::arg("want_coefs"));
  m.def("batch_norm_stats_local", &msbn::batch_norm_stats_local,
        py::arg("input"), py::arg("eps"), py::arg("running_mean"),
        py::arg("running_var"), py::arg("momentum"), py::arg("weight"),
        py::arg("bias"), py::arg("want_coefs"));
  m.def("batch_norm_elemt", &msbn::batch_norm_elemt, py::arg("input"),
        py::arg("weight"), py::arg("bias"), py::arg("mean"), py::arg("invstd"),
        py::arg("eps"));
  m.def("batch_norm_backward_reduce", &msbn::batch_norm_backward_reduce,
        py::arg("grad_out"), py::arg("input"), py::arg("mean"),
        py::arg("invstd"), py::arg("weight"), py::arg("input_g"),
        py::arg("weight_g"), py::arg("bias_g"));
  m.def("batch_norm_backward_elemt", &msbn::batch_norm_backward_elemt,
        py::arg("grad_out"), py::arg("input"), py::arg("mean"),
        py::arg("invstd"), py::arg("weight"), py::arg("sum_dy"),
        py::arg("sum_dy_xmu"), py::arg("count"));
  m.def("bn_make_coefs", &msbn::bn_make_coefs, py::arg("mean"),
        py::arg("invstd"), py::arg("weight"), py::arg("bias"));
  m.def("batch_norm_elemt_act", &msbn::batch_norm_elemt_act, py::arg("input"),
        py::arg("residual"), py::arg("weight"), py::arg("bias"),
        py::arg("mean"), py::arg("invstd"), py::arg("relu"),
        py::arg("coefs") = py::none());
  m.def("batch_norm_backward_reduce_act",
        &msbn::batch_norm_backward_reduce_act, py::arg("grad_out"),
        py::arg("input"), py::arg("residual"), py::arg("mean"),
        py::arg("invstd"), py::arg("weight"), py::arg("bias"),
        py::arg("relu_mask"), py::arg("input_g"), py::arg("weight_g"),
        py::arg("bias_g"), py::arg("coefs") = py::none(),
        py::arg("gm_out") = py::none());
  m.def("batch_norm_backward_elemt_act", &msbn::batch_norm_backward_elemt_act,
        py::arg("grad_out"), py::arg("input"), py::arg("residual"),
        py::arg("mean"), py::arg("invstd"), py::arg("weight"), py::arg("bias"),
        py::arg("sum_dy"), py::arg("sum_dy_xmu"), py::arg("count"),
        py::arg("relu_mask"), py::arg("want_res_grad"),
        py::arg("coefs") = py::none());

  m.def("bn_fused_local_eligible", &msbn::bn_fused_local_eligible,
        py::arg("input"), py::arg("weight"), py::arg("bias"),
        py::arg("running_mean"), py::arg("running_var"));
  m.def("batch_norm_fwd_fused_local", &msbn::batch_norm_fwd_fused_local,
        py::arg("input"), py::arg("residual"), py::arg("weight"),
        py::arg("bias"), py::arg("eps"), py::arg("momentum"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("relu"));
  m.def("batch_norm_bwd_fused_local", &msbn::batch_norm_bwd_fused_local,
        py::arg("grad_out"), py::arg("input"), py::arg("residual"),
        py::arg("mean"), py::arg("invstd"), py::arg("weight"),
        py::arg("coefs"), py::arg("relu_mask"), py::arg("want_res_grad"),
        py::arg("weight_g"), py::arg("bias_g"));

  // ---- DDP machinery ----
  m.def("compute_bucket_assignment_by_size",
        &msbn::compute_bucket_assignment_by_size, py::arg("tensors"),
        py::arg("bucket_size_limits"));
  m.def("broadcast_coalesced", &msbn::broadcast_coalesced,
        py::call_guard<py::gil_scoped_release>(), py::arg("process_group"),
        py::arg("tensors"), py::arg("buffer_bytes"), py::arg("src_rank") = 0);
  m.def("verify_params_across_processes",
        &msbn::verify_params_across_processes,
        py::call_guard<py::gil_scoped_release>(), py::arg("process_group"),
        py::arg("params"));

  // module_local: the A/B build (msbn._C_nont) registers the same C++ type
  py::class_<msbn::Reducer, std::shared_ptr<msbn::Reducer>>(
      m, "Reducer", py::module_local())
      .def(py::init<std::vector<at::Tensor>, std::vector<std::vector<int64_t>>,
                    c10::intrusive_ptr<c10d::ProcessGroup>, bool, size_t,
                    size_t>(),
           py::arg("params"), py::arg("bucket_indices"),
           py::arg("process_group"), py::arg("gradient_as_bucket_view") = false,
           py::arg("first_bucket_bytes") = (size_t)1024 * 1024,
           py::arg("bucket_bytes") = (size_t)25 * 1024 * 1024)
      .def("prepare_for_backward", &msbn::Reducer::prepare_for_backward,
           py::arg("unused_params") = std::vector<int64_t>())
      .def("set_grad_sync_enabled", &msbn::Reducer::set_grad_sync_enabled)
      .def("set_comm_dtype", &msbn::Reducer::set_comm_dtype)
      .def("set_nan_check", &msbn::Reducer::set_nan_check)
      .def("set_div_factor", &msbn::Reducer::set_div_factor)
      .def("set_python_comm_hook", &msbn::Reducer::set_python_comm_hook,
           py::arg("state"), py::arg("hook"), py::arg("bucket_cls"))
      .def("div_factor", &msbn::Reducer::div_factor)
      .def("find_unused", &msbn::Reducer::find_unused, py::arg("outputs"))
      .def("finalize_backward", &msbn::Reducer::finalize_backward,
           py::call_guard<py::gil_scoped_release>())
      .def("rebuild_buckets", &msbn::Reducer::rebuild_buckets,
           py::call_guard<py::gil_scoped_release>())
      .def("get_bucket_indices", &msbn::Reducer::get_bucket_indices)
      .def("get_backward_stats", &msbn::Reducer::get_backward_stats)
      .def("iterations", &msbn::Reducer::iterations)
      .def("rebuilt", &msbn::Reducer::rebuilt);
}
