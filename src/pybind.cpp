// Python bindings for parsec_amd (_core extension).
//
// The Python layer is the driver surface (bench.py, tests, tools); the
// runtime, DAG execution, kernels and comm are all native C++/HIP. Python
// task bodies (insert_py) exist for runtime-semantics tests only, mirroring
// the reference's tests/dsl/dtd/* programs.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <chrono>
#include <map>

#include "data.hpp"
#include "device_gpu.hpp"
#include "comm.hpp"
#include "dtd.hpp"
#include "gpu_graph.hpp"
#include "kernels.hpp"
#include "pins.hpp"
#include "profiling.hpp"
#include "runtime.hpp"

namespace pa {
std::string rccl_get_unique_id();
void rccl_set_unique_id(const std::string&);
void test_dgemm_nt_hip(int, int, int, const double*, int, const double*, int,
                       double*, int);
void test_potf2_hip(double*, int);
double bench_dgemm_hip(int, int, int, int, int);
double bench_dgemm_rocblas(int, int, int, int);
double bench_gemm_bf16(int, int, int, int);
double bench_qr_factor(int, int, int, int, int);
void test_gemm_bf16_hip(int, int, int, const uint16_t*, const uint16_t*,
                        float*);
}  // namespace pa

namespace py = pybind11;
using namespace pa;

namespace {

// ---- Python CPU task bodies (for DTD semantics tests) ----
struct PyTaskPayload {
  PyObject* fn;
  int with_data;  // pass per-flow host buffers as memoryviews
};

void py_task_hook(Task& t) {
  const PyTaskPayload& pl = t.arg<PyTaskPayload>();
  // materialize host buffers OUTSIDE the GIL (may sync/copy from device)
  void* bufs[MAX_FLOWS] = {};
  if (pl.with_data) {
    for (int i = 0; i < t.nflows; i++) {
      Data* d = t.flows[i].data;
      if (!d) continue;
      if (t.flows[i].mode & ACCESS_IN) {
        bufs[i] = d->pull_to_host();
      } else {
        d->begin_host_overwrite();
        bufs[i] = d->ensure_host();
      }
    }
  }
  py::gil_scoped_acquire gil;
  py::handle fn(pl.fn);
  try {
    if (pl.with_data) {
      py::tuple args(t.nflows);
      for (int i = 0; i < t.nflows; i++) {
        Data* d = t.flows[i].data;
        if (d)
          args[i] = py::memoryview::from_memory(bufs[i], (ssize_t)d->bytes);
        else
          args[i] = py::none();
      }
      fn(*args);
    } else {
      fn();
    }
  } catch (py::error_already_set& e) {
    fprintf(stderr, "[parsec_amd] python task raised: %s\n", e.what());
  }
  // OUT flows were (re)written on the host by the body
  for (int i = 0; i < t.nflows && pl.with_data; i++)
    if (t.flows[i].data && (t.flows[i].mode & ACCESS_OUT))
      t.flows[i].data->written_on(false);
}

void py_task_destruct(Task& t) {
  py::gil_scoped_acquire gil;
  Py_XDECREF(t.arg<PyTaskPayload>().fn);
}

TaskClass& py_task_class() {
  static TaskClass tc = [] {
    TaskClass c;
    c.name = "py_task";
    c.kind = TaskKind::CPU;
    c.cpu_hook = py_task_hook;
    c.destruct = py_task_destruct;
    c.id = 100;
    return c;
  }();
  return tc;
}

Context::Options options_from_env(int nworkers, int rank, int world,
                                  std::string comm, int gpu) {
  Context::Options o;
  o.nworkers = nworkers;
  if (rank < 0) {
    const char* r = getenv("RANK");
    const char* w = getenv("WORLD_SIZE");
    o.rank = r ? atoi(r) : 0;
    o.world = w ? atoi(w) : 1;
  } else {
    o.rank = rank;
    o.world = world;
  }
  o.comm = std::move(comm);
  o.gpu_device = gpu;
  return o;
}

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "parsec_amd: MI355X-native task-dataflow runtime (PaRSEC-class)";

  py::class_<Context>(m, "Context")
      .def(py::init([](int nworkers, int rank, int world, std::string comm,
                       int gpu) {
             return new Context(
                 options_from_env(nworkers, rank, world, std::move(comm), gpu));
           }),
           py::arg("nworkers") = -1, py::arg("rank") = -1,
           py::arg("world") = -1, py::arg("comm") = std::string(),
           py::arg("gpu") = -1,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("_handle", [](Context& c) { return (uintptr_t)&c; })
      .def_property_readonly("rank", &Context::rank)
      .def_property_readonly("world", &Context::world)
      .def_property_readonly("nworkers", &Context::nworkers)
      .def_property_readonly("has_gpu", &Context::has_gpu)
      .def("barrier", &Context::barrier,
           py::call_guard<py::gil_scoped_release>())
      .def("gpu_sync", [](Context& c) {
        if (c.gpu()) c.gpu()->sync_all();
      }, py::call_guard<py::gil_scoped_release>())
      .def("gpu_stats", [](Context& c) {
        py::dict d;
        if (c.gpu()) {
          d["tasks"] = c.gpu()->stats.tasks.load();
          d["bytes_h2d"] = c.gpu()->stats.bytes_h2d.load();
          d["bytes_d2h"] = c.gpu()->stats.bytes_d2h.load();
          d["evictions"] = c.gpu()->stats.evictions.load();
          d["bytes_required"] = c.gpu()->stats.bytes_required.load();
        }
        return d;
      })
      .def("send_ctl", [](Context& c, int dst, uint32_t tag, py::bytes b) {
        PA_CHECK(c.comm(), "send_ctl: no comm engine");
        std::string s(b);
        py::gil_scoped_release rel;
        c.comm()->send_ctl(dst, tag, s.data(), s.size());
      }, py::arg("dst"), py::arg("tag"), py::arg("payload"))
      .def("set_ctl_handler", [](Context& c, py::function fn) {
        PA_CHECK(c.comm(), "set_ctl_handler: no comm engine");
        PyObject* f = fn.ptr();
        Py_XINCREF(f);
        c.comm()->set_ctl_handler(
            [f](int src, uint32_t tag, const std::string& pl) {
              py::gil_scoped_acquire gil;
              try {
                py::handle h(f);
                h(src, tag, py::bytes(pl));
              } catch (py::error_already_set& e) {
                fprintf(stderr, "[parsec_amd] ctl handler raised: %s\n",
                        e.what());
              }
            });
      })
      .def("info", [](Context& c) {
        py::dict d;
        for (auto& [k, v] : c.info_all()) d[py::str(k)] = v;
        return d;
      })
      .def("info_set", [](Context& c, std::string k, std::string v) {
        c.info_set(k, v);
      })
      .def("comm_stats", [](Context& c) {
        // per-peer traffic table (device-stats analog for the comm engine)
        py::list out;
        if (c.comm()) {
          for (auto& ps : c.comm()->peer_stats()) {
            py::dict d;
            d["sent_msgs"] = ps.sent_msgs.load();
            d["sent_bytes"] = ps.sent_bytes.load();
            d["recv_msgs"] = ps.recv_msgs.load();
            d["recv_bytes"] = ps.recv_bytes.load();
            out.append(d);
          }
        }
        return out;
      })
      .def("counters", [](Context&) {
        // PINS/papi_sde-style software counters (process-wide)
        auto& c = counters();
        py::dict d;
        d["tasks_executed_cpu"] = c.tasks_executed_cpu.load();
        d["tasks_executed_gpu"] = c.tasks_executed_gpu.load();
        d["tasks_scheduled"] = c.tasks_scheduled.load();
        d["steals"] = c.steals.load();
        d["comm_msgs"] = c.comm_msgs.load();
        d["comm_bytes"] = c.comm_bytes.load();
        d["renames"] = c.renames.load();
        return d;
      });

  py::class_<Data>(m, "Data")
      .def_property_readonly("home_rank", [](Data& d) { return d.home_rank; })
      .def_property_readonly("owner_rank", [](Data& d) { return d.owner_rank; })
      .def_property_readonly("version", [](Data& d) { return d.version; })
      .def_property_readonly("nbytes", [](Data& d) { return d.bytes; });

  py::class_<TiledMatrix>(m, "TiledMatrix")
      .def(py::init<Context*, int64_t, int64_t, int, int, int, int, size_t,
                    bool>(),
           py::arg("ctx"), py::arg("m"), py::arg("n"), py::arg("mb"),
           py::arg("nb"), py::arg("p") = 1, py::arg("q") = 1,
           py::arg("elem_size") = 8, py::arg("sym") = false,
           py::keep_alive<1, 2>())
      .def_property_readonly("_handle", [](TiledMatrix& a) { return (uintptr_t)&a; })
      .def_property_readonly("m", &TiledMatrix::m)
      .def_property_readonly("n", &TiledMatrix::n)
      .def_property_readonly("mt", &TiledMatrix::mt)
      .def_property_readonly("nt", &TiledMatrix::nt)
      .def_property_readonly("mb", &TiledMatrix::mb)
      .def_property_readonly("nb", &TiledMatrix::nb)
      .def("rank_of", &TiledMatrix::rank_of)
      .def("set_kcyclic", &TiledMatrix::set_kcyclic)
      .def("set_rank_table", &TiledMatrix::set_rank_table)
      .def("set_band", &TiledMatrix::set_band)
      .def("in_band", &TiledMatrix::in_band)
      .def("is_local", &TiledMatrix::is_local)
      .def("tile", &TiledMatrix::tile, py::return_value_policy::reference_internal)
      .def("tile_rows", &TiledMatrix::tile_rows)
      .def("tile_cols", &TiledMatrix::tile_cols)
      .def("tile_numpy", [](TiledMatrix& A, int tm, int tn) {
        // Readback of a LOCAL tile as a numpy array (column-major view
        // copied to a C-order array of shape [rows, cols]).
        PA_CHECK(A.is_local(tm, tn), "tile_numpy: tile is not local");
        Data* d = A.tile(tm, tn);
        double* p;
        {
          py::gil_scoped_release rel;
          p = (double*)d->pull_to_host();
        }
        int rows = A.tile_rows(tm), cols = A.tile_cols(tn);
        py::array_t<double> out({rows, cols});
        auto r = out.mutable_unchecked<2>();
        for (int j = 0; j < cols; j++)
          for (int i = 0; i < rows; i++) r(i, j) = p[(size_t)j * A.mb() + i];
        return out;
      })
      .def("tile_numpy_set", [](TiledMatrix& A, int tm, int tn,
                                py::array_t<double> v) {
        PA_CHECK(A.is_local(tm, tn), "tile_numpy_set: tile is not local");
        Data* d = A.tile(tm, tn);
        d->begin_host_overwrite();
        double* p = (double*)d->ensure_host();
        auto r = v.unchecked<2>();
        int rows = A.tile_rows(tm), cols = A.tile_cols(tn);
        PA_CHECK(r.shape(0) == rows && r.shape(1) == cols);
        for (int j = 0; j < cols; j++)
          for (int i = 0; i < rows; i++) p[(size_t)j * A.mb() + i] = r(i, j);
        d->written_on(false);
      })
      .def("tile_bytes", [](TiledMatrix& A, int tm, int tn) {
        PA_CHECK(A.is_local(tm, tn));
        Data* d = A.tile(tm, tn);
        void* p;
        {
          py::gil_scoped_release rel;
          p = d->pull_to_host();
        }
        return py::bytes((const char*)p, d->bytes);
      })
      .def("tile_bytes_set", [](TiledMatrix& A, int tm, int tn, py::bytes b) {
        PA_CHECK(A.is_local(tm, tn));
        Data* d = A.tile(tm, tn);
        d->begin_host_overwrite();
        std::string s(b);
        PA_CHECK(s.size() <= d->bytes);
        memcpy(d->ensure_host(), s.data(), s.size());
        d->written_on(false);
      });

  // CPU-testable view of the GPU slab allocator (zone_malloc analog).
  py::class_<ZoneAlloc>(m, "ZoneAlloc")
      .def(py::init([](size_t bytes) {
             auto* z = new ZoneAlloc();
             z->init(bytes);
             return z;
           }),
           py::arg("bytes"))
      .def("alloc",
           [](ZoneAlloc& z, size_t sz) -> py::object {
             size_t off = z.alloc(sz);
             if (off == ZoneAlloc::NPOS) return py::none();
             return py::int_(off);
           })
      .def("free", &ZoneAlloc::free, py::arg("off"), py::arg("size"))
      .def_property_readonly("in_use", &ZoneAlloc::in_use)
      .def_property_readonly("capacity", &ZoneAlloc::capacity)
      .def_property_readonly("largest_free", &ZoneAlloc::largest_free)
      .def_property_readonly("free_blocks", &ZoneAlloc::free_blocks);

  py::class_<IrregularCollection>(m, "IrregularCollection")
      .def(py::init<Context*>(), py::arg("ctx"), py::keep_alive<1, 2>())
      .def("add", &IrregularCollection::add, py::arg("key"), py::arg("rank"),
           py::arg("bytes"), py::return_value_policy::reference_internal)
      .def("at", &IrregularCollection::at, py::arg("key"),
           py::return_value_policy::reference_internal)
      .def_property_readonly("size", &IrregularCollection::size)
      .def("bytes_get", [](IrregularCollection& c, uint64_t key) {
        Data* d = c.at(key);
        void* p;
        {
          py::gil_scoped_release rel;
          p = d->pull_to_host();
        }
        return py::bytes((const char*)p, d->bytes);
      })
      .def("bytes_set", [](IrregularCollection& c, uint64_t key, py::bytes b) {
        Data* d = c.at(key);
        d->begin_host_overwrite();
        std::string s2(b);
        PA_CHECK(s2.size() <= d->bytes);
        memcpy(d->ensure_host(), s2.data(), s2.size());
        d->written_on(false);
      });

  py::class_<Taskpool>(m, "Taskpool")
      .def("wait_dynamic", &Taskpool::wait_dynamic,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("_handle", [](Taskpool& t) { return (uintptr_t)&t; })
      .def("wait", &Taskpool::wait, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("pending", &Taskpool::pending)
      // parsec_compose analog (compound.c:17-135): chain a callback to run
      // when this pool drains (e.g. to start inserting the next pool).
      .def("on_complete", [](Taskpool& t, py::function fn) {
        // deleter must hold the GIL: the callback object may be destroyed
        // on a worker thread when the pool drains
        auto holder = std::shared_ptr<py::function>(
            new py::function(std::move(fn)), [](py::function* p) {
              py::gil_scoped_acquire gil;
              delete p;
            });
        t.on_complete([holder] {
          py::gil_scoped_acquire gil;
          try {
            (*holder)();
          } catch (py::error_already_set& e) {
            fprintf(stderr, "[parsec_amd] on_complete raised: %s\n", e.what());
          }
        });
      });

  py::class_<Dtd, Taskpool>(m, "Dtd")
      .def(py::init<Context*, std::string>(), py::arg("ctx"),
           py::arg("name") = std::string("dtd"), py::keep_alive<1, 2>())
      .def("insert_py",
           [](Dtd& tp, py::function fn, py::list flows, int priority,
              int rank, bool with_data) {
             PyTaskPayload pl{fn.ptr(), with_data ? 1 : 0};
             Py_XINCREF(pl.fn);
             std::vector<Dtd::FlowSpec> fs;
             for (auto h : flows) {
               auto t2 = h.cast<py::tuple>();
               Dtd::FlowSpec f{t2[0].cast<Data*>(),
                               (AccessMode)t2[1].cast<int>(),
                               Reshape::NONE};
               if (t2.size() > 2) f.reshape = (Reshape)t2[2].cast<int>();
               fs.push_back(f);
             }
             bool local;
             {
               py::gil_scoped_release rel;
               local = tp.insert(&py_task_class(), &pl, sizeof(pl),
                                 fs.data(), (int)fs.size(), priority, rank);
             }
             // remote tasks never run here: release the body's ref
             if (!local) Py_XDECREF(pl.fn);
           },
           py::arg("fn"), py::arg("flows") = py::list(),
           py::arg("priority") = 0, py::arg("rank") = -1,
           py::arg("with_data") = false)
      .def("flush", &Dtd::flush, py::call_guard<py::gil_scoped_release>())
      .def("flush_all", &Dtd::flush_all,
           py::call_guard<py::gil_scoped_release>())
      .def("capture_begin", &Dtd::capture_begin)
      .def("capture_end", &Dtd::capture_end,
           py::call_guard<py::gil_scoped_release>(),
           py::keep_alive<0, 1>());  // graph keeps the pool (and ctx) alive

  // hipGraph replay handle (gpu_graph.hpp): launch-bound steady-state DAGs
  // re-run with ONE hipGraphLaunch per iteration.
  py::class_<GpuGraph>(m, "GpuGraph")
      .def("launch", &GpuGraph::launch, py::arg("iters") = 1,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("nodes", &GpuGraph::nodes)
      .def_property_readonly("n_tasks", &GpuGraph::n_tasks);

  m.attr("ACCESS_IN") = (int)ACCESS_IN;
  m.attr("ACCESS_UNTRACKED") = (int)ACCESS_UNTRACKED;
  m.attr("ACCESS_OUT") = (int)ACCESS_OUT;
  m.attr("ACCESS_INOUT") = (int)ACCESS_INOUT;
  m.attr("RESHAPE_TRANSPOSE") = (int)Reshape::TRANSPOSE;
  m.attr("RESHAPE_TO_BF16") = (int)Reshape::TO_BF16;
  m.attr("RESHAPE_FROM_BF16") = (int)Reshape::FROM_BF16;
  m.attr("RESHAPE_TRIL") = (int)Reshape::TRIL;
  m.attr("RESHAPE_TRIU") = (int)Reshape::TRIU;

  m.def("bench_qr_factor", &bench_qr_factor, py::arg("m"), py::arg("k"),
        py::arg("ts_split"), py::arg("iters"), py::arg("mode"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_spd_fill", &insert_spd_fill, py::arg("tp"), py::arg("A"),
        py::arg("seed") = 42u, py::call_guard<py::gil_scoped_release>());
  m.def("insert_full_fill", &insert_full_fill, py::arg("tp"), py::arg("A"),
        py::arg("seed") = 42u, py::call_guard<py::gil_scoped_release>());
  m.def("insert_potrf", &insert_potrf, py::arg("tp"), py::arg("A"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_getrf_nopiv", &insert_getrf_nopiv, py::arg("tp"),
        py::arg("A"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_geqrf", &insert_geqrf, py::arg("tp"), py::arg("A"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_geqrf_bcgs", &insert_geqrf_bcgs, py::arg("tp"),
        py::arg("A"), py::arg("R"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_redistribute", &insert_redistribute, py::arg("tp"),
        py::arg("src"), py::arg("dst"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_reduce_axis", &insert_reduce_axis, py::arg("tp"),
        py::arg("A"), py::arg("R"), py::arg("axis"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_band_to_rect", &insert_band_to_rect, py::arg("tp"),
        py::arg("S"), py::arg("D"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_subtile_extract", &insert_subtile_extract, py::arg("tp"),
        py::arg("A"), py::arg("tm"), py::arg("tn"), py::arg("S"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_subtile_insert", &insert_subtile_insert, py::arg("tp"),
        py::arg("S"), py::arg("A"), py::arg("tm"), py::arg("tn"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_potrs", &insert_potrs, py::arg("tp"), py::arg("A"),
        py::arg("B"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_posv", &insert_posv, py::arg("tp"), py::arg("A"),
        py::arg("B"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_getrs_nopiv", &insert_getrs_nopiv, py::arg("tp"),
        py::arg("A"), py::arg("B"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_gesv_nopiv", &insert_gesv_nopiv, py::arg("tp"),
        py::arg("A"), py::arg("B"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_gels_bcgs", &insert_gels_bcgs, py::arg("tp"), py::arg("A"),
        py::arg("R"), py::arg("B"), py::arg("X"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_advise_prefetch", &insert_advise_prefetch, py::arg("tp"),
        py::arg("tile"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_gemm_fp64", &insert_gemm_fp64, py::arg("tp"), py::arg("A"),
        py::arg("B"), py::arg("C"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_apply_scale", &insert_apply_scale, py::arg("tp"), py::arg("A"),
        py::arg("alpha"), py::arg("beta"),
        py::call_guard<py::gil_scoped_release>());
  m.def("insert_reduce_sum_tree", &insert_reduce_sum_tree, py::arg("tp"),
        py::arg("A"), py::arg("R"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_reduce_sum", &insert_reduce_sum, py::arg("tp"), py::arg("A"),
        py::arg("R"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_stencil_1d", &insert_stencil_1d, py::arg("tp"),
        py::arg("src"), py::arg("dst"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_panel_fill", &insert_panel_fill, py::arg("tp"), py::arg("A"),
        py::arg("seed") = 42u, py::call_guard<py::gil_scoped_release>());
  m.def("insert_potrf_panel", &insert_potrf_panel, py::arg("tp"),
        py::arg("A"), py::call_guard<py::gil_scoped_release>());
  m.def("insert_fill_bf16", &insert_fill_bf16, py::arg("tp"), py::arg("A"),
        py::arg("seed") = 1u, py::call_guard<py::gil_scoped_release>());
  m.def("insert_gemm_bf16", &insert_gemm_bf16, py::arg("tp"), py::arg("At"),
        py::arg("B"), py::arg("C"), py::call_guard<py::gil_scoped_release>());
  m.def("bench_gemm_bf16", [](int m, int n, int k, int iters) {
    return pa::bench_gemm_bf16(m, n, k, iters);
  }, py::call_guard<py::gil_scoped_release>());
  m.def("gemm_bf16_hip", [](py::array_t<uint16_t> A, py::array_t<uint16_t> B,
                            py::array_t<float> C, int m, int n, int k) {
    pa::test_gemm_bf16_hip(m, n, k, A.data(), B.data(), C.mutable_data());
  });

  // PINS callback chain (mca/pins analog). Python callbacks fire from
  // worker threads: the wrapper takes the GIL. Returns a handle for
  // pins_remove. events: list of names from PINS_EVENTS.
  m.def("pins_add", [](py::function fn, std::vector<std::string> events) {
    static const std::map<std::string, PinsEv> names = {
        {"exec_begin", PinsEv::EXEC_BEGIN}, {"exec_end", PinsEv::EXEC_END},
        {"schedule", PinsEv::SCHEDULE},     {"complete", PinsEv::COMPLETE},
        {"gpu_submit", PinsEv::GPU_SUBMIT}, {"gpu_retire", PinsEv::GPU_RETIRE},
        {"create", PinsEv::CREATE},
        {"release_deps", PinsEv::RELEASE_DEPS},
        {"stage_in", PinsEv::STAGE_IN},     {"comm_post", PinsEv::COMM_POST},
        {"comm_done", PinsEv::COMM_DONE},   {"steal", PinsEv::STEAL}};
    uint32_t mask = 0;
    for (auto& e : events) {
      auto it = names.find(e);
      if (it == names.end())
        throw std::invalid_argument("unknown PINS event: " + e);
      mask |= 1u << (int)it->second;
    }
    // keep the function alive via shared_ptr with GIL-acquiring deleter
    auto keep = std::shared_ptr<py::function>(
        new py::function(std::move(fn)), [](py::function* f) {
          py::gil_scoped_acquire g;
          delete f;
        });
    static const char* evname[] = {"exec_begin", "exec_end",   "schedule",
                                   "complete",   "gpu_submit", "gpu_retire",
                                   "create",     "release_deps", "stage_in",
                                   "comm_post",  "comm_done",  "steal"};
    return Pins::inst().add(
        [keep](PinsEv e, const Task* t, int worker) {
          py::gil_scoped_acquire g;
          (*keep)(evname[(int)e], t ? t->tc->name : std::string(), worker);
        },
        mask);
  });
  m.def("pins_remove", [](int id) { Pins::inst().remove(id); });

  // Fatal-error callback (error-callback analog). The Python callable is
  // invoked with the message right before abort(); use it to flush logs.
  m.def("set_fatal_handler", [](py::function fn) {
    static py::function* g_fn = nullptr;
    if (g_fn) { delete g_fn; }
    g_fn = new py::function(std::move(fn));
    set_fatal_handler(+[](const char* msg) {
      py::gil_scoped_acquire g;
      try { (*reinterpret_cast<py::function*>(g_fn))(msg); } catch (...) {}
    });
  });

  m.def("param_set", &param_set);
  m.def("param_dump", &param_dump);
  m.def("hip_device_count", [] {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) n = 0;
    return n;
  });
  // Standalone kernel harnesses for GPU numerics tests/debugging.
  m.def("dgemm_nt_hip", [](py::array_t<double> A, py::array_t<double> B,
                           py::array_t<double> C, int m, int n, int k) {
    // arrays are column-major buffers flattened 1-D, ld == rows
    pa::test_dgemm_nt_hip(m, n, k, A.data(), m, B.data(), n,
                          C.mutable_data(), m);
  });
  m.def("potf2_hip", [](py::array_t<double> A, int n) {
    pa::test_potf2_hip(A.mutable_data(), n);
  });
  m.def("bench_dgemm", [](int m, int n, int k, int iters, std::string impl) {
    double dt = impl == "rocblas" ? pa::bench_dgemm_rocblas(m, n, k, iters)
                                  : pa::bench_dgemm_hip(m, n, k, iters,
                                                        impl == "v1" ? 1 : 0);
    return 2.0 * m * n * k * iters / dt / 1e12;  // TFLOP/s
  }, py::call_guard<py::gil_scoped_release>());
  // PCIe/xGMI link probe (reference tools/gpu/testbandwidth analog):
  // pinned-host H2D/D2H and on-device D2D copy bandwidth in GB/s.
  m.def("hip_bandwidth", [](size_t nbytes, int iters) {
    void *hbuf = nullptr, *d0 = nullptr, *d1 = nullptr;
    if (hipHostMalloc(&hbuf, nbytes) != hipSuccess)
      throw std::runtime_error("hipHostMalloc failed");
    if (hipMalloc(&d0, nbytes) != hipSuccess ||
        hipMalloc(&d1, nbytes) != hipSuccess)
      throw std::runtime_error("hipMalloc failed");
    hipStream_t s;
    (void)hipStreamCreate(&s);
    auto run = [&](void* dst, const void* src, hipMemcpyKind kind) {
      (void)hipMemcpyAsync(dst, src, nbytes, kind, s);  // warmup
      (void)hipStreamSynchronize(s);
      auto t0 = std::chrono::steady_clock::now();
      for (int i = 0; i < iters; i++)
        (void)hipMemcpyAsync(dst, src, nbytes, kind, s);
      (void)hipStreamSynchronize(s);
      double dt = std::chrono::duration<double>(
                      std::chrono::steady_clock::now() - t0).count();
      return (double)nbytes * iters / dt / 1e9;
    };
    py::dict r;
    r["h2d_gbs"] = run(d0, hbuf, hipMemcpyHostToDevice);
    r["d2h_gbs"] = run(hbuf, d0, hipMemcpyDeviceToHost);
    r["d2d_gbs"] = run(d1, d0, hipMemcpyDeviceToDevice);
    r["bytes"] = nbytes;
    (void)hipStreamDestroy(s);
    (void)hipFree(d0);
    (void)hipFree(d1);
    (void)hipHostFree(hbuf);
    return r;
  }, py::arg("nbytes") = (size_t)256 << 20, py::arg("iters") = 20);
  m.def("nccl_unique_id", [] { return py::bytes(pa::rccl_get_unique_id()); });
  m.def("set_nccl_unique_id",
        [](py::bytes b) { pa::rccl_set_unique_id(std::string(b)); });
}
