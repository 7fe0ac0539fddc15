// Python bindings for the mi355x_ddp native layer (kernels + RCCL comm).
#include <torch/extension.h>

#include "ops.h"
#include "p2p_mesh.h"
#include "rccl_comm.h"
#include "reducer_core.h"

namespace mi355x {
// autograd_ops.hip
torch::Tensor linear_autograd(torch::Tensor x, torch::Tensor w,
                              c10::optional<torch::Tensor> b);
torch::Tensor mse_autograd(torch::Tensor y, torch::Tensor t);
torch::Tensor ce_autograd(torch::Tensor y, torch::Tensor t);
}  // namespace mi355x

namespace py = pybind11;

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native kernels and RCCL communicator for mi355x_ddp";

  // pop HIP's latched per-thread error (e.g. hipErrorStreamCaptureInvalidated
  // after a failed graph capture) so eager fallback paths can proceed
  m.def("clear_hip_errors", [] { (void)hipGetLastError(); });

  m.def("linear_fwd", &mi355x::linear_fwd, py::arg("x"), py::arg("w"),
        py::arg("bias") = c10::nullopt);
  m.def("linear_bwd_weight", &mi355x::linear_bwd_weight, py::arg("x"),
        py::arg("dy"), py::arg("dw"), py::arg("db"),
        py::arg("accumulate") = false);
  m.def("linear_bwd_input", &mi355x::linear_bwd_input);
  m.def("gemm_bf16", &mi355x::gemm_bf16, py::arg("x"), py::arg("w"),
        py::arg("bias") = c10::nullopt);
  m.def("ce_fwd", &mi355x::ce_fwd);
  m.def("ce_bwd", &mi355x::ce_bwd, py::arg("probs"), py::arg("t"),
        py::arg("tsum"), py::arg("grad_scale"),
        py::arg("gout") = c10::nullopt);
  m.def("mse_fwd", &mi355x::mse_fwd);
  m.def("mse_bwd", &mi355x::mse_bwd, py::arg("y"), py::arg("t"),
        py::arg("grad_scale"), py::arg("gout") = c10::nullopt);
  m.def("sgd_flat", &mi355x::sgd_flat, py::arg("param_flat"),
        py::arg("grad_flat"), py::arg("lr"), py::arg("zero_grad") = true);
  m.def("build_copy_plan",
        [](const std::vector<torch::Tensor>& tensors,
           const std::vector<int64_t>& offsets, int device) {
          return mi355x::build_copy_plan(tensors, offsets,
                                         torch::Device(torch::kCUDA, device));
        });
  m.def("flatten_into", &mi355x::flatten_into, py::arg("bucket"),
        py::arg("plan"), py::arg("total_blocks"), py::arg("zero_src") = false);
  m.def("unflatten_from", &mi355x::unflatten_from);
  m.def("toy_fused_fwd_bwd", &mi355x::toy_fused_fwd_bwd, py::arg("x"),
        py::arg("t"), py::arg("param_flat"), py::arg("grad_flat"),
        py::arg("loss_out"), py::arg("use_mse") = true,
        py::arg("w_off") = 0, py::arg("b_off") = 0, py::arg("lr") = 0.0);
  m.def("toy_multistep_mesh", &mi355x::toy_multistep_mesh, py::arg("x"),
        py::arg("t"), py::arg("param_flat"), py::arg("loss_out"),
        py::arg("use_mse"), py::arg("w_off"), py::arg("b_off"),
        py::arg("lr"), py::arg("batch"), py::arg("mesh"));
  m.def("epoch_shard", &mi355x::epoch_shard, py::arg("X"), py::arg("T"),
        py::arg("seed"), py::arg("rank"), py::arg("world"));
  m.def("epoch_shard_multi", &mi355x::epoch_shard_multi, py::arg("X"),
        py::arg("T"), py::arg("seed0"), py::arg("epochs"), py::arg("rank"),
        py::arg("world"));
  m.def("toy_multistep", &mi355x::toy_multistep, py::arg("x"), py::arg("t"),
        py::arg("param_flat"), py::arg("loss_out"), py::arg("use_mse") = true,
        py::arg("w_off") = 0, py::arg("b_off") = 0, py::arg("lr") = 0.0,
        py::arg("batch") = 32);

  py::class_<mi355x::P2pMesh>(m, "P2pMesh")
      .def(py::init<int, int, int>(), py::arg("rank"), py::arg("world"),
           py::arg("device"))
      .def("handle_bytes",
           [](const mi355x::P2pMesh& m_) { return py::bytes(m_.handle_bytes()); })
      .def("connect",
           [](mi355x::P2pMesh& m_, const std::vector<py::bytes>& hs) {
             std::vector<std::string> v(hs.begin(), hs.end());
             m_.connect(v);
           })
      .def("all_reduce_avg_inline", &mi355x::P2pMesh::all_reduce_avg_inline)
      .def("check", &mi355x::P2pMesh::check)
      .def_property_readonly("rank", &mi355x::P2pMesh::rank)
      .def_property_readonly("world", &mi355x::P2pMesh::world);

  // C++ autograd Functions (whole fwd/bwd chain stays out of Python)
  m.def("linear_autograd", &mi355x::linear_autograd, py::arg("x"),
        py::arg("w"), py::arg("bias") = c10::nullopt);
  m.def("mse_autograd", &mi355x::mse_autograd);
  m.def("ce_autograd", &mi355x::ce_autograd);

  py::class_<mi355x::ReducerCore, std::shared_ptr<mi355x::ReducerCore>>(
      m, "ReducerCore")
      .def(py::init([](std::vector<std::vector<torch::Tensor>> params,
                       std::vector<std::vector<torch::Tensor>> views,
                       std::vector<torch::Tensor> flats,
                       mi355x::RcclComm* comm) {
             return std::make_shared<mi355x::ReducerCore>(
                 std::move(params), std::move(views), std::move(flats), comm);
           }),
           py::arg("bucket_params"), py::arg("bucket_views"),
           py::arg("bucket_flat_grads"), py::arg("comm") = nullptr,
           // the core holds a non-owning RcclComm*: keep the comm alive as
           // long as the core (nurse the adapter's lifetime from C++ side)
           py::keep_alive<1, 5>())
      .def("attach_hooks", &mi355x::ReducerCore::attach_hooks)
      .def("detach_hooks", &mi355x::ReducerCore::detach_hooks)
      .def("finalize", &mi355x::ReducerCore::finalize,
           py::call_guard<py::gil_scoped_release>())
      .def("set_skip_comm", &mi355x::ReducerCore::set_skip_comm)
      .def_property_readonly("steps", &mi355x::ReducerCore::steps)
      .def_property_readonly("unfenced", &mi355x::ReducerCore::unfenced);

  py::class_<mi355x::RcclComm>(m, "RcclComm")
      .def(py::init<const std::string&, int, int, int>(), py::arg("unique_id"),
           py::arg("rank"), py::arg("world"), py::arg("device"),
           py::call_guard<py::gil_scoped_release>())
      .def_static("make_unique_id", [] {
        return py::bytes(mi355x::RcclComm::make_unique_id());
      })
      .def_property_readonly("rank", &mi355x::RcclComm::rank)
      .def_property_readonly("world", &mi355x::RcclComm::world)
      .def("all_reduce_avg", &mi355x::RcclComm::all_reduce_avg)
      .def("all_reduce_avg_inline", &mi355x::RcclComm::all_reduce_avg_inline)
      .def("broadcast", &mi355x::RcclComm::broadcast)
      .def("join_compute", &mi355x::RcclComm::join_compute)
      .def("barrier", &mi355x::RcclComm::barrier,
           py::call_guard<py::gil_scoped_release>());
}
