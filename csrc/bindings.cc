// pybind11 bindings (reference: horovod/torch/mpi_ops_v2.cc N24).
//
// One templated entry point per op instead of per-dtype symbol names; the
// Python layer (horovod_amd/torch/mpi_ops.py) builds the hvd.* surface on
// top of these.
#include <torch/extension.h>

#include "core.h"
#include "timeline.h"
#include "gpu.h"

namespace {

using namespace hvd;

ReduceOp OpFromInt(int op) { return (ReduceOp)op; }

py::tuple WaitHandle(int handle) {
  std::vector<at::Tensor> outputs;
  at::Tensor extra;
  int32_t result_int = -1;
  Status s;
  {
    py::gil_scoped_release nogil;
    s = State().handles.Wait(handle, outputs, extra, &result_int);
  }
  if (!s.ok()) {
    if (s.type == StatusType::ABORTED)
      throw std::runtime_error("HorovodInternalError: " + s.reason);
    throw std::runtime_error(s.reason);
  }
  py::list outs;
  for (auto& t : outputs) outs.append(t);
  py::object extra_obj = py::none();
  if (extra.defined()) extra_obj = py::cast(extra);
  return py::make_tuple(outs, extra_obj, result_int);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "horovod_amd core (MI355X-native: TCP controller + RCCL/xGMI)";

  m.def("init",
        [](int rank, int size, int local_rank, int local_size, int cross_rank,
           int cross_size, const std::string& addr, int port,
           int64_t fusion_threshold, double cycle_time_ms, int cache_capacity,
           double stall_warning_sec, double stall_shutdown_sec, bool timeline) {
          ControllerConfig cfg;
          cfg.fusion_threshold_bytes = fusion_threshold;
          cfg.cycle_time_ms = cycle_time_ms;
          cfg.cache_capacity = (size_t)cache_capacity;
          cfg.stall_warning_sec = stall_warning_sec;
          cfg.stall_shutdown_sec = stall_shutdown_sec;
          cfg.timeline_enabled = timeline;
          py::gil_scoped_release nogil;
          InitHorovod(rank, size, local_rank, local_size, cross_rank, cross_size,
                      addr, port, cfg);
        },
        py::arg("rank"), py::arg("size"), py::arg("local_rank"),
        py::arg("local_size"), py::arg("cross_rank"), py::arg("cross_size"),
        py::arg("addr"), py::arg("port"), py::arg("fusion_threshold"),
        py::arg("cycle_time_ms"), py::arg("cache_capacity"),
        py::arg("stall_warning_sec"), py::arg("stall_shutdown_sec"),
        py::arg("timeline"));

  m.def("interrupt", [](const std::string& why) {
    py::gil_scoped_release nogil;
    InterruptHorovod(why);
  });
  m.def("shutdown", [] {
    py::gil_scoped_release nogil;
    ShutdownHorovod();
  });
  m.def("is_initialized", &IsInitialized);

  m.def("rank", [] { return State().rank; });
  m.def("size", [] { return State().size; });
  m.def("local_rank", [] { return State().local_rank; });
  m.def("local_size", [] { return State().local_size; });
  m.def("cross_rank", [] { return State().cross_rank; });
  m.def("cross_size", [] { return State().cross_size; });
  m.def("rccl_used", [] { return hvd::gpu::RcclUsed(); });

  m.def("set_fusion_threshold",
        [](int64_t b) { if (State().controller) State().controller->set_fusion_threshold(b); });
  m.def("get_fusion_threshold",
        [] { return State().controller ? State().controller->fusion_threshold() : 0; });
  m.def("set_cycle_time_ms",
        [](double ms) { if (State().controller) State().controller->set_cycle_time_ms(ms); });
  m.def("get_cycle_time_ms",
        [] { return State().controller ? State().controller->cycle_time_ms() : 0.0; });

  // ---- enqueue ops --------------------------------------------------------
  m.def("allreduce_async",
        [](std::vector<at::Tensor> tensors, std::vector<at::Tensor> outputs,
           std::vector<std::string> names, int op, double prescale,
           double postscale, int process_set, int wire_dtype) {
          return EnqueueAllreduceMulti(std::move(tensors), std::move(outputs),
                                       std::move(names), OpFromInt(op), prescale,
                                       postscale, process_set,
                                       (DataType)wire_dtype);
        });
  m.def("allgather_async", [](at::Tensor t, const std::string& name, int ps) {
    return EnqueueAllgather(t, name, ps);
  });
  m.def("broadcast_async",
        [](at::Tensor t, at::Tensor out, int root, const std::string& name,
           int ps) { return EnqueueBroadcast(t, out, root, name, ps); });
  m.def("alltoall_async",
        [](at::Tensor t, at::Tensor splits, const std::string& name, int ps) {
          return EnqueueAlltoall(t, splits, name, ps);
        });
  m.def("reducescatter_async",
        [](at::Tensor t, const std::string& name, int op, double prescale,
           double postscale, int ps) {
          return EnqueueReducescatter(t, name, OpFromInt(op), prescale, postscale,
                                      ps);
        });
  m.def("join_async", [](int device, int ps) { return EnqueueJoin(device, ps); });
  m.def("barrier_async", [](int ps) { return EnqueueBarrier(ps); });

  // ---- completion ---------------------------------------------------------
  m.def("poll", [](int handle) { return State().handles.Poll(handle); });
  m.def("flush", [] { FlushCycle(); });
  m.def("wait", &WaitHandle);

  m.def("adasum_combine_", [](std::vector<at::Tensor> a,
                              std::vector<at::Tensor> b) {
    hvd::gpu::AdasumCombine(a, b);
  });

  // ---- fused optimizer kernels -------------------------------------------
  m.def("fused_sgd_step",
        [](std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
           std::vector<at::Tensor> momenta, double lr, double momentum,
           double weight_decay, double dampening, bool nesterov) {
          hvd::gpu::FusedSgdStep(params, grads, momenta, lr, momentum,
                                 weight_decay, dampening, nesterov);
        });

  m.def("fused_adamw_step",
        [](std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
           std::vector<at::Tensor> exp_avgs,
           std::vector<at::Tensor> exp_avg_sqs, double lr, double beta1,
           double beta2, double eps, double weight_decay, int64_t step) {
          hvd::gpu::FusedAdamwStep(params, grads, exp_avgs, exp_avg_sqs, lr,
                                   beta1, beta2, eps, weight_decay, step);
        });

  m.def("fused_bn_relu_forward",
        [](at::Tensor x, py::object residual, at::Tensor gamma, at::Tensor beta,
           at::Tensor rm, at::Tensor rv, double momentum, double eps) {
          at::Tensor res;
          if (!residual.is_none()) res = residual.cast<at::Tensor>();
          return hvd::gpu::FusedBnReluForward(x, res, gamma, beta, rm, rv,
                                              momentum, eps);
        });
  m.def("fused_bn_relu_backward",
        [](at::Tensor x, at::Tensor y, at::Tensor dy, at::Tensor mean,
           at::Tensor invstd, at::Tensor gamma, bool need_res) {
          return hvd::gpu::FusedBnReluBackward(x, y, dy, mean, invstd, gamma,
                                               need_res);
        });

  // ---- timeline -----------------------------------------------------------
  m.def("start_timeline", [](const std::string& path, bool mark_cycles) {
    auto& st = State();
    if (!st.initialized) throw std::runtime_error("not initialized");
    SetTimeline(st, std::make_shared<Timeline>(path, st.rank));
    (void)mark_cycles;
  });
  m.def("stop_timeline", [] {
    auto tl = GetTimeline(State());
    SetTimeline(State(), nullptr);
    if (tl) {
      py::gil_scoped_release nogil;
      tl->Finalize();  // file is complete, valid JSON on return
    }
  });

  // ---- process sets -------------------------------------------------------
  m.def("add_process_set", [](std::vector<int32_t> ranks) {
    if (!State().controller) throw std::runtime_error("not initialized");
    return State().controller->AddProcessSet(ranks);
  });
  m.def("remove_process_set", [](int id) {
    if (State().controller) State().controller->RemoveProcessSet(id);
  });
  m.def("process_set_ranks", [](int id) {
    if (!State().controller || !State().controller->has_process_set(id))
      throw std::runtime_error("unknown process set");
    return State().controller->process_set(id).ranks;
  });
}
