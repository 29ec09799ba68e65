/* pybind11 bindings: ucc_amd._core
 *
 * Thin Python surface over the public C API (src/api/ucc.h) used by the
 * test jig, bench.py and the torch integration. Buffers are passed as raw
 * pointers (numpy .ctypes.data / torch .data_ptr()). Two OOB flavors:
 *  - LocalOob: in-process multi-rank jig (the gtest UccJob design,
 *    reference test/gtest/common/test_ucc.h:121-224) — memcpy allgather.
 *  - PyOob: any Python allgather(bytes)->list[bytes] (e.g. torch gloo).
 */
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <map>
#include <memory>
#include <mutex>
#include <vector>

#include "../api/ucc.h"
#include "../topo/topo.h"
#include "../mc/mc.h"

namespace py = pybind11;

extern "C" ucc_status_t ucc_amd_coll_from_name_c(const char *,
                                                 ucc_coll_type_t *);
extern "C" int ucc_amd_hip_device_count_c();
extern "C" uint64_t ucc_amd_cdna4_memh_uses();

/* ------------------------------------------------------------- LocalOob */
struct LocalOobShared {
    explicit LocalOobShared(int n) : n_ranks(n) {}
    struct Round {
        std::vector<std::vector<uint8_t>> contrib;
        int                               count = 0;
        std::vector<uint8_t>              result;
        bool                              done = false;
    };
    int                        n_ranks;
    std::map<uint64_t, Round>  rounds;
    std::mutex                 mtx;
};

struct LocalOobEp {
    std::shared_ptr<LocalOobShared> shared;
    int                             rank;
    uint64_t                        next_round = 0;
};

struct LocalOobReq {
    LocalOobEp *ep;
    uint64_t    round;
    void       *recv;
    size_t      size;
};

static ucc_status_t local_oob_allgather(void *src, void *recv, size_t size,
                                        void *info, void **request)
{
    auto *ep = (LocalOobEp *)info;
    auto *sh = ep->shared.get();
    std::lock_guard<std::mutex> lk(sh->mtx);
    uint64_t round = ep->next_round++;
    auto    &r     = sh->rounds[round];
    if (r.contrib.empty()) {
        r.contrib.resize(sh->n_ranks);
    }
    r.contrib[ep->rank].assign((uint8_t *)src, (uint8_t *)src + size);
    r.count++;
    auto *req    = new LocalOobReq{ep, round, recv, size};
    *request     = req;
    return UCC_OK;
}

static ucc_status_t local_oob_test(void *request)
{
    auto *req = (LocalOobReq *)request;
    auto *sh  = req->ep->shared.get();
    std::lock_guard<std::mutex> lk(sh->mtx);
    auto &r = sh->rounds[req->round];
    if (r.count < sh->n_ranks) {
        return UCC_INPROGRESS;
    }
    uint8_t *dst = (uint8_t *)req->recv;
    for (int i = 0; i < sh->n_ranks; i++) {
        memcpy(dst + i * req->size, r.contrib[i].data(), req->size);
    }
    return UCC_OK;
}

static ucc_status_t local_oob_free(void *request)
{
    delete (LocalOobReq *)request;
    return UCC_OK;
}

/* --------------------------------------------------------------- PyOob  */
struct PyOobInfo {
    py::function allgather; /* (bytes) -> list[bytes] (len == n_ranks) */
    int          n_ranks;
    int          rank;
};

struct PyOobReq {
    int dummy;
};

static ucc_status_t py_oob_allgather(void *src, void *recv, size_t size,
                                     void *info, void **request)
{
    auto *pi = (PyOobInfo *)info;
    try {
        py::bytes arg((const char *)src, size);
        py::list  out = pi->allgather(arg);
        uint8_t  *dst = (uint8_t *)recv;
        for (int i = 0; i < pi->n_ranks; i++) {
            std::string s = py::cast<std::string>(out[i]);
            if (s.size() != size) {
                return UCC_ERR_INVALID_PARAM;
            }
            memcpy(dst + i * size, s.data(), size);
        }
    } catch (const std::exception &e) {
        PyErr_Clear();
        fprintf(stderr, "ucc_amd: python OOB allgather failed: %s\n",
                e.what());
        return UCC_ERR_NO_MESSAGE;
    }
    *request = new PyOobReq{0};
    return UCC_OK;
}

static ucc_status_t py_oob_test(void *request)
{
    (void)request;
    return UCC_OK;
}
static ucc_status_t py_oob_free(void *request)
{
    delete (PyOobReq *)request;
    return UCC_OK;
}

/* ------------------------------------------------------------- wrappers */
struct CoreLib {
    ucc_lib_h lib = nullptr;
    ~CoreLib()
    {
        if (lib) {
            ucc_finalize(lib);
        }
    }
};

struct CoreCtx {
    ucc_context_h               ctx = nullptr;
    std::shared_ptr<CoreLib>      lib;
    std::unique_ptr<LocalOobEp> local_ep; /* keep alive */
    std::unique_ptr<PyOobInfo>  py_info;
    ~CoreCtx()
    {
        if (ctx) {
            ucc_context_destroy(ctx);
        }
    }
    void progress() { ucc_context_progress(ctx); }
};

struct CoreTeam {
    ucc_team_h                  team = nullptr;
    std::shared_ptr<CoreCtx>  ctx;
    std::unique_ptr<LocalOobEp> local_ep;
    std::unique_ptr<PyOobInfo>  py_info;
    ~CoreTeam()
    {
        if (team) {
            ucc_team_destroy(team);
        }
    }
};

struct CoreReq {
    ucc_coll_req_h              req = nullptr;
    std::shared_ptr<CoreTeam>     team;
    std::vector<uint64_t>       counts_s, counts_d; /* v-coll storage     */
    std::vector<uint64_t>       displs_s, displs_d;
    ~CoreReq()
    {
        if (req) {
            ucc_collective_finalize(req);
        }
    }
};

static void check(ucc_status_t st, const char *what)
{
    if (st < 0) {
        throw std::runtime_error(std::string(what) + ": " +
                                 ucc_status_string(st));
    }
}

static ucc_oob_coll_t make_local_oob(LocalOobEp *ep)
{
    ucc_oob_coll_t oob{};
    oob.allgather = local_oob_allgather;
    oob.req_test  = local_oob_test;
    oob.req_free  = local_oob_free;
    oob.coll_info = ep;
    oob.n_oob_eps = ep->shared->n_ranks;
    oob.oob_ep    = ep->rank;
    return oob;
}

static ucc_oob_coll_t make_py_oob(PyOobInfo *pi)
{
    ucc_oob_coll_t oob{};
    oob.allgather = py_oob_allgather;
    oob.req_test  = py_oob_test;
    oob.req_free  = py_oob_free;
    oob.coll_info = pi;
    oob.n_oob_eps = pi->n_ranks;
    oob.oob_ep    = pi->rank;
    return oob;
}

PYBIND11_MODULE(_core, m)
{
    m.doc() = "ucc_amd native core (MI355X collective communication)";

    py::class_<LocalOobShared, std::shared_ptr<LocalOobShared>>(m,
                                                                "LocalOob")
        .def(py::init<int>(), py::arg("n_ranks"));

    py::class_<CoreLib, std::shared_ptr<CoreLib>>(m, "Lib").def(
        py::init([](const std::string &thread_mode) {
            auto             l = std::make_shared<CoreLib>();
            ucc_lib_params_t p{};
            p.mask        = UCC_LIB_PARAM_FIELD_THREAD_MODE;
            p.thread_mode = thread_mode == "multiple" ? UCC_THREAD_MULTIPLE
                                                      : UCC_THREAD_SINGLE;
            check(ucc_init(&p, nullptr, &l->lib), "ucc_init");
            return l;
        }),
        py::arg("thread_mode") = "single");

    py::class_<CoreCtx, std::shared_ptr<CoreCtx>>(m, "Context")
        .def(py::init([](std::shared_ptr<CoreLib> lib) {
                 auto c = std::make_shared<CoreCtx>();
                 c->lib = lib;
                 ucc_context_params_t p{};
                 check(ucc_context_create(lib->lib, &p, nullptr, &c->ctx),
                       "ucc_context_create");
                 return c;
             }),
             py::arg("lib"))
        .def("progress", &CoreCtx::progress);

    py::class_<CoreTeam, std::shared_ptr<CoreTeam>>(m, "Team")
        .def_property_readonly("rank",
                               [](CoreTeam &t) {
                                   uint64_t ep;
                                   ucc_team_get_my_ep(t.team, &ep);
                                   return (int)ep;
                               })
        .def_property_readonly("size", [](CoreTeam &t) {
            uint32_t s;
            ucc_team_get_size(t.team, &s);
            return (int)s;
        });

    /* team create: nonblocking post; returns a CoreTeam plus a test fn */
    m.def(
        "team_create_post",
        [](std::shared_ptr<CoreCtx> ctx,
           std::shared_ptr<LocalOobShared> local, int rank,
           py::object py_allgather, int n_ranks) {
            auto t = std::make_shared<CoreTeam>();
            t->ctx = ctx;
            ucc_team_params_t tp{};
            if (local) {
                t->local_ep = std::make_unique<LocalOobEp>();
                t->local_ep->shared = local;
                t->local_ep->rank   = rank;
                tp.mask = UCC_TEAM_PARAM_FIELD_OOB;
                tp.oob  = make_local_oob(t->local_ep.get());
            } else if (!py_allgather.is_none()) {
                t->py_info = std::make_unique<PyOobInfo>();
                t->py_info->allgather = py::cast<py::function>(py_allgather);
                t->py_info->n_ranks   = n_ranks;
                t->py_info->rank      = rank;
                tp.mask = UCC_TEAM_PARAM_FIELD_OOB;
                tp.oob  = make_py_oob(t->py_info.get());
            } else {
                tp.mask      = UCC_TEAM_PARAM_FIELD_TEAM_SIZE;
                tp.team_size = 1;
            }
            ucc_context_h ch = ctx->ctx;
            check(ucc_team_create_post(&ch, 1, &tp, &t->team),
                  "ucc_team_create_post");
            return t;
        },
        py::arg("ctx"), py::arg("local_oob") = nullptr, py::arg("rank") = 0,
        py::arg("py_allgather") = py::none(), py::arg("n_ranks") = 1);

    m.def("team_create_test", [](std::shared_ptr<CoreTeam> t) {
        return (int)ucc_team_create_test(t->team);
    });

    m.def("team_create_from_parent",
          [](std::shared_ptr<CoreTeam> parent, uint64_t my_ep,
             uint32_t included) {
              auto t  = std::make_shared<CoreTeam>();
              t->ctx  = parent->ctx;
              check(ucc_team_create_from_parent(my_ep, included,
                                                parent->team, &t->team),
                    "team_create_from_parent");
              return t;
          });

    /* ---------------------------------------------------- collectives */
    m.def(
        "coll_init",
        [](std::shared_ptr<CoreTeam> team, const std::string &coll,
           uintptr_t src, uintptr_t dst, uint64_t count, int dt, int op,
           int mem_type, uint64_t root, uint64_t flags,
           std::vector<uint64_t> src_counts, std::vector<uint64_t> src_displs,
           std::vector<uint64_t> dst_counts,
           std::vector<uint64_t> dst_displs, double timeout,
           std::vector<int64_t> active_set, int tag, int src_mem_type,
           uintptr_t src_memh, uintptr_t dst_memh) {
            int s_mt = src_mem_type >= 0 ? src_mem_type : mem_type;
            auto r  = std::make_shared<CoreReq>();
            r->team = team;
            ucc_coll_args_t a{};
            a.mask  = UCC_COLL_ARGS_FIELD_FLAGS;
            a.flags = flags | UCC_COLL_ARGS_FLAG_COUNT_64BIT |
                      UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT;
            ucc_coll_type_t ct;
            if (ucc_amd_coll_from_name_c(coll.c_str(), &ct) != UCC_OK) {
                throw std::runtime_error("unknown coll: " + coll);
            }
            a.coll_type = ct;
            a.op        = (ucc_reduction_op_t)op;
            a.root      = root;
            if (timeout > 0) {
                a.flags |= UCC_COLL_ARGS_FLAG_TIMEOUT;
                a.timeout = timeout;
            }
            if (active_set.size() == 3) {
                a.mask |= UCC_COLL_ARGS_FIELD_ACTIVE_SET;
                a.active_set.start  = active_set[0];
                a.active_set.stride = active_set[1];
                a.active_set.size   = active_set[2];
            }
            if (tag >= 0) {
                a.mask |= UCC_COLL_ARGS_FIELD_TAG;
                a.tag = (uint16_t)tag;
            }
            if (src_memh) {
                a.mask |= UCC_COLL_ARGS_FIELD_MEM_MAP_SRC_MEMH;
                a.src_memh = (ucc_mem_map_mem_h)src_memh;
            }
            if (dst_memh) {
                a.mask |= UCC_COLL_ARGS_FIELD_MEM_MAP_DST_MEMH;
                a.dst_memh = (ucc_mem_map_mem_h)dst_memh;
            }
            bool sv = ct == UCC_COLL_TYPE_ALLTOALLV ||
                      ct == UCC_COLL_TYPE_SCATTERV;
            bool dv = ct == UCC_COLL_TYPE_ALLTOALLV ||
                      ct == UCC_COLL_TYPE_ALLGATHERV ||
                      ct == UCC_COLL_TYPE_GATHERV ||
                      ct == UCC_COLL_TYPE_REDUCE_SCATTERV;
            r->counts_s = std::move(src_counts);
            r->displs_s = std::move(src_displs);
            r->counts_d = std::move(dst_counts);
            r->displs_d = std::move(dst_displs);
            if (sv && !r->counts_s.empty()) {
                a.src.info_v.buffer        = (void *)src;
                a.src.info_v.counts        = r->counts_s.data();
                a.src.info_v.displacements = r->displs_s.data();
                a.src.info_v.datatype      = (ucc_datatype_t)dt;
                a.src.info_v.mem_type      = (ucc_memory_type_t)s_mt;
            } else {
                a.src.info.buffer   = (void *)src;
                a.src.info.count    = count;
                a.src.info.datatype = (ucc_datatype_t)dt;
                a.src.info.mem_type = (ucc_memory_type_t)s_mt;
            }
            if (dv && !r->counts_d.empty()) {
                a.dst.info_v.buffer        = (void *)dst;
                a.dst.info_v.counts        = r->counts_d.data();
                a.dst.info_v.displacements = r->displs_d.data();
                a.dst.info_v.datatype      = (ucc_datatype_t)dt;
                a.dst.info_v.mem_type      = (ucc_memory_type_t)mem_type;
            } else {
                a.dst.info.buffer   = (void *)dst;
                a.dst.info.count    = count;
                a.dst.info.datatype = (ucc_datatype_t)dt;
                a.dst.info.mem_type = (ucc_memory_type_t)mem_type;
            }
            check(ucc_collective_init(&a, &r->req, team->team),
                  "ucc_collective_init");
            return r;
        },
        py::arg("team"), py::arg("coll"), py::arg("src"), py::arg("dst"),
        py::arg("count"), py::arg("dt"), py::arg("op") = 0,
        py::arg("mem_type") = 0, py::arg("root") = 0, py::arg("flags") = 0,
        py::arg("src_counts") = std::vector<uint64_t>(),
        py::arg("src_displs") = std::vector<uint64_t>(),
        py::arg("dst_counts") = std::vector<uint64_t>(),
        py::arg("dst_displs") = std::vector<uint64_t>(),
        py::arg("timeout") = 0.0,
        py::arg("active_set") = std::vector<int64_t>(),
        py::arg("tag") = -1, py::arg("src_mem_type") = -1,
        py::arg("src_memh") = 0, py::arg("dst_memh") = 0);

    /* ------------------------------------------------------ mem_map */
    m.def("mem_map_export", [](uintptr_t addr, size_t len) {
        ucc_mem_map_t        seg{(void *)addr, len};
        ucc_mem_map_params_t p{&seg, 1};
        size_t               sz = 0;
        ucc_mem_map_mem_h    h  = nullptr;
        check(ucc_mem_map(nullptr, UCC_MEM_MAP_MODE_EXPORT, &p, &sz, &h),
              "ucc_mem_map export");
        py::bytes out((const char *)h, sz);
        ucc_mem_unmap(&h);
        return out;
    });
    m.def("mem_map_import", [](py::bytes blob) {
        std::string s = blob;
        /* keep the blob alive for unmap: heap copy owned by caller via
         * the returned handle pair (ptr to mapped base) */
        void  *copy = malloc(s.size());
        memcpy(copy, s.data(), s.size());
        size_t            sz = s.size();
        ucc_mem_map_mem_h h  = copy;
        check(ucc_mem_map(nullptr, UCC_MEM_MAP_MODE_IMPORT, nullptr, &sz,
                          &h),
              "ucc_mem_map import");
        /* first segment's mapped VA (test helper) */
        struct SegView {
            uint32_t magic, n;
            int32_t  imported, pad;
            uint64_t addr, len, base_off;
            int32_t  mt, has_ipc;
            uint8_t  handle[64];
            uint64_t mapped;
        };
        auto *v = (SegView *)copy;
        return (uintptr_t)(v->mapped + v->base_off);
    });
    /* live exported handle for coll-args src_memh/dst_memh (the
     * bytes-returning export above frees its handle immediately) */
    m.def("mem_map_export_keep", [](uintptr_t addr, size_t len) {
        ucc_mem_map_t        seg{(void *)addr, len};
        ucc_mem_map_params_t p{&seg, 1};
        size_t               sz = 0;
        ucc_mem_map_mem_h    h  = nullptr;
        check(ucc_mem_map(nullptr, UCC_MEM_MAP_MODE_EXPORT, &p, &sz, &h),
              "ucc_mem_map export");
        return (uintptr_t)h;
    });
    m.def("mem_unmap_handle", [](uintptr_t h) {
        ucc_mem_map_mem_h hh = (ucc_mem_map_mem_h)h;
        ucc_mem_unmap(&hh);
    });
    m.def("cdna4_memh_uses",
          []() { return (uint64_t)ucc_amd_cdna4_memh_uses(); });
    m.def("mem_map_close", [](uintptr_t mapped) {
        ucc::mc::ipc_close((void *)mapped);
    });
    m.def("hip_memcpy_d2d", [](uintptr_t dst, uintptr_t src, size_t len) {
        check(ucc::mc::copy((void *)dst, UCC_MEMORY_TYPE_CUDA,
                            (const void *)src, UCC_MEMORY_TYPE_CUDA, len),
              "d2d copy");
    });

    m.def("score_map_str", [](std::shared_ptr<CoreTeam> t) {
        auto *team = reinterpret_cast<ucc::Team *>(t->team);
        return team->score_map.to_string();
    });
    m.def("topo_sbgps", [](std::shared_ptr<CoreTeam> t) {
        auto *team = reinterpret_cast<ucc::Team *>(t->team);
        auto  node = ucc::topo::build_sbgp(team, ucc::topo::SbgpType::NODE);
        auto  ldr =
            ucc::topo::build_sbgp(team, ucc::topo::SbgpType::NODE_LEADERS);
        py::dict d;
        d["node_size"]    = node.ranks.size();
        d["node_idx"]     = node.my_idx;
        d["leaders_size"] = ldr.ranks.size();
        d["leaders_idx"]  = ldr.my_idx;
        auto sock = ucc::topo::build_sbgp(team,
                                          ucc::topo::SbgpType::SOCKET);
        auto numa = ucc::topo::build_sbgp(team,
                                          ucc::topo::SbgpType::NUMA);
        auto sldr = ucc::topo::build_sbgp(
            team, ucc::topo::SbgpType::SOCKET_LEADERS);
        auto nldr = ucc::topo::build_sbgp(
            team, ucc::topo::SbgpType::NUMA_LEADERS);
        d["socket_size"]         = sock.ranks.size();
        d["socket_idx"]          = sock.my_idx;
        d["numa_size"]           = numa.ranks.size();
        d["numa_idx"]            = numa.my_idx;
        d["socket_leaders_size"] = sldr.ranks.size();
        d["numa_leaders_size"]   = nldr.ranks.size();
        d["same_cpu"] = ucc::topo::team_same_cpu(team);
        return d;
    });
    m.def("gpu_link_hops", []() {
        auto &g = ucc::topo::gpu_links();
        return g.hops;
    });

    /* ------------------------------------------------- EE / triggered */
    m.def("ee_create", [](std::shared_ptr<CoreTeam> t, uintptr_t stream) {
        ucc_ee_params_t p{};
        p.ee_type         = UCC_EE_ROCM_STREAM;
        p.ee_context      = (void *)stream;
        p.ee_context_size = sizeof(void *);
        ucc_ee_h ee;
        check(ucc_ee_create(t->team, &p, &ee), "ucc_ee_create");
        return (uintptr_t)ee;
    });
    m.def("ee_destroy",
          [](uintptr_t ee) { ucc_ee_destroy((ucc_ee_h)ee); });
    /* pop one event from the EE queue; returns ev_type or -1 if empty
     * (reference ucc.h event flow: POST at launch, COLLECTIVE_COMPLETE
     * when the stream work finishes) */
    m.def("ee_pop_event", [](uintptr_t ee) {
        ucc_ev_t *ev = nullptr;
        if (ucc_ee_get_event((ucc_ee_h)ee, &ev) != UCC_OK) {
            return -1;
        }
        int t = (int)ev->ev_type;
        ucc_ee_ack_event((ucc_ee_h)ee, ev);
        return t;
    });
    m.attr("EVENT_COLLECTIVE_POST") = (int)UCC_EVENT_COLLECTIVE_POST;
    m.attr("EVENT_COLLECTIVE_COMPLETE") =
        (int)UCC_EVENT_COLLECTIVE_COMPLETE;
    m.def("triggered_post", [](uintptr_t ee, std::shared_ptr<CoreReq> r) {
        ucc_ev_t ev{};
        ev.ev_type = UCC_EVENT_COMPUTE_COMPLETE;
        ev.req     = r->req;
        check(ucc_collective_triggered_post((ucc_ee_h)ee, &ev),
              "triggered_post");
    });

    py::class_<CoreReq, std::shared_ptr<CoreReq>>(m, "Request")
        .def("post",
             [](CoreReq &r) { check(ucc_collective_post(r.req), "post"); })
        .def("test", [](CoreReq &r) { return (int)r.req->status; })
        .def("wait", [](CoreReq &r) {
            /* single-process convenience: progress own context */
            while (r.req->status == UCC_INPROGRESS) {
                ucc_context_progress(r.team->ctx->ctx);
            }
            check(r.req->status, "collective");
        });

    m.attr("OK")         = (int)UCC_OK;
    m.attr("INPROGRESS") = (int)UCC_INPROGRESS;
    m.attr("FLAG_IN_PLACE")   = (uint64_t)UCC_COLL_ARGS_FLAG_IN_PLACE;
    m.attr("FLAG_PERSISTENT") = (uint64_t)UCC_COLL_ARGS_FLAG_PERSISTENT;
    m.attr("FLAG_MEM_MAPPED") =
        (uint64_t)UCC_COLL_ARGS_FLAG_MEM_MAPPED_BUFFERS;
    m.def("dt_size", [](int dt) { return ucc_dt_size((ucc_datatype_t)dt); });
    m.def("version", []() { return std::string(ucc_get_version_string()); });
    m.def("hip_device_count",
          []() { return ucc_amd_hip_device_count_c(); });
    /* raw (non-cache-hit) allocations done by the mc scratch mpool —
     * regression hook: repeated asymmetric-staging colls must hit the
     * pool, not hipMalloc (ref mc_rocm.c:97-108) */
    m.def("scratch_raw_allocs",
          []() { return (size_t)ucc::mc::scratch_raw_allocs(); });
}
