// torchmpi_amd._C — pybind11 bindings.
//
// Re-designs TorchMPI's Lua registration layer (lib/torch_mpi.cpp, SURVEY.md
// section 2.1): instead of macro-stamped per-type C functions registered into
// Lua, ATen tensors carry their dtype and a single entry point per verb
// dispatches internally. Sync and async verbs both enqueue onto the
// communicator's worker thread (per-communicator ordering — reference
// lib/thread_pool.h semantics); sync callers immediately wait.
#include <torch/extension.h>

#include <memory>

#include "common.h"
#include "cpu_comm.h"
#include "ps.h"

namespace py = pybind11;

namespace tma {

void register_gpu(py::module_& m);     // defined in rccl_engine.hip
void register_ipc_ps(py::module_& m);  // defined in ipc_ps.hip
void register_batchnorm(py::module_& m);  // defined in kernels/batchnorm.hip
void register_rmsnorm(py::module_& m);    // defined in kernels/rmsnorm.hip

namespace {

DType dtype_of_cpu(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return DType::F32;
    case at::kDouble: return DType::F64;
    case at::kHalf: return DType::F16;
    case at::kBFloat16: return DType::BF16;
    case at::kInt: return DType::I32;
    case at::kLong: return DType::I64;
    case at::kByte: return DType::U8;
    case at::kChar: return DType::I8;
    default:
      throw std::runtime_error("torchmpi_amd: unsupported tensor dtype");
  }
}

void check_cpu_tensor(const at::Tensor& t) {
  TMA_CHECK(t.device().is_cpu(), "this communicator path handles CPU tensors");
  TMA_CHECK(t.is_contiguous(), "collectives require contiguous tensors");
}

}  // namespace

PYBIND11_MODULE(_C, m) {
  m.doc() = "torchmpi_amd native core (TCP transport, RCCL engine, gfx950 kernels)";

  py::enum_<RedOp>(m, "RedOp")
      .value("SUM", RedOp::SUM)
      .value("MAX", RedOp::MAX)
      .value("MIN", RedOp::MIN)
      .value("PROD", RedOp::PROD);

  py::class_<Work, std::shared_ptr<Work>>(m, "Work")
      .def("wait", &Work::wait, py::call_guard<py::gil_scoped_release>())
      .def("ready", &Work::ready);

  py::class_<CpuComm, std::shared_ptr<CpuComm>>(m, "CpuComm")
      .def(py::init<int, const std::string&>(), py::arg("rank"), py::arg("bind_addr"))
      .def("listen_port", &CpuComm::listen_port)
      .def("connect", &CpuComm::connect, py::call_guard<py::gil_scoped_release>())
      .def("rank", &CpuComm::rank)
      .def("size", &CpuComm::size)
      .def("allreduce",
           [](CpuComm& c, at::Tensor t, RedOp op) {
             check_cpu_tensor(t);
             DType d = dtype_of_cpu(t);
             return c.enqueue([&c, t, d, op] {
               c.allreduce(t.data_ptr(), static_cast<size_t>(t.numel()), d, op);
             });
           })
      .def("reduce_scatter",
           [](CpuComm& c, at::Tensor t, at::Tensor out, RedOp op) {
             check_cpu_tensor(t);
             check_cpu_tensor(out);
             DType d = dtype_of_cpu(t);
             return c.enqueue([&c, t, out, d, op] {
               c.reduce_scatter(t.data_ptr(), out.data_ptr(),
                                static_cast<size_t>(t.numel()), d, op);
             });
           })
      .def("broadcast",
           [](CpuComm& c, at::Tensor t, int root) {
             check_cpu_tensor(t);
             size_t bytes = static_cast<size_t>(t.numel()) * t.element_size();
             return c.enqueue([&c, t, bytes, root] {
               c.broadcast(t.data_ptr(), bytes, root);
             });
           })
      .def("reduce",
           [](CpuComm& c, at::Tensor t, RedOp op, int root) {
             check_cpu_tensor(t);
             DType d = dtype_of_cpu(t);
             return c.enqueue([&c, t, d, op, root] {
               c.reduce(t.data_ptr(), static_cast<size_t>(t.numel()), d, op, root);
             });
           })
      .def("allgather",
           [](CpuComm& c, at::Tensor input, at::Tensor output) {
             check_cpu_tensor(input);
             check_cpu_tensor(output);
             size_t bytes = static_cast<size_t>(input.numel()) * input.element_size();
             TMA_CHECK(static_cast<size_t>(output.numel()) * output.element_size() ==
                           bytes * c.size(),
                       "allgather output must be world_size x input");
             return c.enqueue([&c, input, output, bytes] {
               c.allgather(input.data_ptr(), output.data_ptr(), bytes);
             });
           })
      .def("gather",
           [](CpuComm& c, at::Tensor input, at::Tensor output, int root) {
             check_cpu_tensor(input);
             size_t bytes = static_cast<size_t>(input.numel()) * input.element_size();
             return c.enqueue([&c, input, output, bytes, root] {
               c.gather(input.data_ptr(),
                        output.defined() ? output.data_ptr() : nullptr, bytes, root);
             });
           })
      .def("scatter",
           [](CpuComm& c, at::Tensor input, at::Tensor output, int root) {
             check_cpu_tensor(output);
             size_t bytes = static_cast<size_t>(output.numel()) * output.element_size();
             return c.enqueue([&c, input, output, bytes, root] {
               c.scatter(input.defined() ? input.data_ptr() : nullptr,
                         output.data_ptr(), bytes, root);
             });
           })
      .def("alltoall",
           [](CpuComm& c, at::Tensor input, at::Tensor output) {
             check_cpu_tensor(input);
             check_cpu_tensor(output);
             TMA_CHECK(input.numel() % c.size() == 0 &&
                           output.numel() == input.numel(),
                       "alltoall: numel must be divisible by world_size");
             size_t bytes = static_cast<size_t>(input.numel()) * input.element_size() /
                            c.size();
             return c.enqueue([&c, input, output, bytes] {
               c.alltoall(input.data_ptr(), output.data_ptr(), bytes);
             });
           })
      .def("sendrecv",
           [](CpuComm& c, at::Tensor send_t, int dst, at::Tensor recv_t, int src) {
             check_cpu_tensor(send_t);
             check_cpu_tensor(recv_t);
             size_t sb = static_cast<size_t>(send_t.numel()) * send_t.element_size();
             size_t rb = static_cast<size_t>(recv_t.numel()) * recv_t.element_size();
             return c.enqueue([&c, send_t, recv_t, sb, rb, dst, src] {
               c.sendrecv(send_t.data_ptr(), sb, dst, recv_t.data_ptr(), rb, src);
             });
           })
      .def("send",
           [](CpuComm& c, at::Tensor t, int dst) {
             check_cpu_tensor(t);
             size_t bytes = static_cast<size_t>(t.numel()) * t.element_size();
             return c.enqueue([&c, t, bytes, dst] { c.send(t.data_ptr(), bytes, dst); });
           })
      .def("recv",
           [](CpuComm& c, at::Tensor t, int src) {
             check_cpu_tensor(t);
             size_t bytes = static_cast<size_t>(t.numel()) * t.element_size();
             return c.enqueue([&c, t, bytes, src] { c.recv(t.data_ptr(), bytes, src); });
           })
      .def("barrier", [](CpuComm& c) { return c.enqueue([&c] { c.barrier(); }); });

  py::class_<PsContext, std::shared_ptr<PsContext>>(m, "PsContext")
      .def(py::init<int, const std::string&>(), py::arg("rank"), py::arg("bind_addr"))
      .def("listen_port", &PsContext::listen_port)
      .def("start", &PsContext::start, py::call_guard<py::gil_scoped_release>())
      .def("stop", &PsContext::stop, py::call_guard<py::gil_scoped_release>())
      .def("register_tensor",
           [](PsContext& p, int tid, at::Tensor t) {
             check_cpu_tensor(t);
             p.register_tensor(tid, static_cast<size_t>(t.numel()), dtype_of_cpu(t),
                               t.data_ptr());
           })
      .def("free_tensor", &PsContext::free_tensor)
      .def("send",
           [](PsContext& p, int tid, at::Tensor t, int rule) {
             check_cpu_tensor(t);
             py::gil_scoped_release nogil;
             return p.send(tid, t.data_ptr(), rule);
           })
      .def("receive",
           [](PsContext& p, int tid, at::Tensor t) {
             check_cpu_tensor(t);
             py::gil_scoped_release nogil;
             return p.receive(tid, t.data_ptr());
           });

  // GPU engine + gfx950 kernels + xGMI IPC parameter server.
  register_gpu(m);
  register_ipc_ps(m);
  register_batchnorm(m);
  register_rmsnorm(m);
}

}  // namespace tma
