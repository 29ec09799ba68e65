/* Generic user-defined datatype unit test (reference gtest dt coverage):
 * create/query/destroy, CPU reduction via the user callback, and a full
 * 2-rank in-process allreduce through the public C API with a generic
 * contiguous dtype. Exits 0 on success. */
#include <cstdio>
#include <cstring>
#include <map>
#include <vector>

#include "../../src/api/ucc.h"
#include "../../src/ec/ec_cpu.h"

/* element: pair of floats reduced as (min, max) */
struct MinMax {
    float mn, mx;
};

static ucc_status_t mm_reduce(const void *s1, const void *s2, void *dst,
                              size_t count, void *cookie)
{
    (void)cookie;
    auto *a = (const MinMax *)s1;
    auto *b = (const MinMax *)s2;
    auto *d = (MinMax *)dst;
    for (size_t i = 0; i < count; i++) {
        d[i].mn = a[i].mn < b[i].mn ? a[i].mn : b[i].mn;
        d[i].mx = a[i].mx > b[i].mx ? a[i].mx : b[i].mx;
    }
    return UCC_OK;
}

/* round-keyed in-process OOB (same design as tools/perftest.cc) */
struct Loob {
    struct Round {
        std::vector<std::vector<uint8_t>> blobs;
        int                               arrived = 0;
    };
    std::map<uint64_t, Round> rounds;
    uint64_t                  next[2] = {0, 0};
};
static Loob g_loob;

struct LoobReq {
    uint64_t round;
    void    *recv;
    size_t   size;
};

static ucc_status_t lo_ag(void *src, void *recv, size_t size, void *info,
                          void **req)
{
    int      rank  = (int)(intptr_t)info;
    uint64_t round = g_loob.next[rank]++;
    auto    &r     = g_loob.rounds[round];
    if (r.blobs.empty()) {
        r.blobs.resize(2);
    }
    r.blobs[rank].assign((uint8_t *)src, (uint8_t *)src + size);
    r.arrived++;
    *req = new LoobReq{round, recv, size};
    return UCC_OK;
}
static ucc_status_t lo_test(void *req)
{
    auto *r  = (LoobReq *)req;
    auto &rd = g_loob.rounds[r->round];
    if (rd.arrived < 2) {
        return UCC_INPROGRESS;
    }
    for (int i = 0; i < 2; i++) {
        memcpy((uint8_t *)r->recv + i * r->size, rd.blobs[i].data(),
               r->size);
    }
    return UCC_OK;
}
static ucc_status_t lo_free(void *req)
{
    delete (LoobReq *)req;
    return UCC_OK;
}

struct Strided {
    static void *start_pack(void *ck, const void *b, size_t c)
    {
        auto *st = new std::pair<void *, size_t>((void *)b, c);
        (void)ck;
        return st;
    }
    static void *start_unpack(void *ck, void *b, size_t c)
    {
        auto *st = new std::pair<void *, size_t>(b, c);
        (void)ck;
        return st;
    }
    static size_t packed_size(void *o)
    {
        return ((std::pair<void *, size_t> *)o)->second *
               sizeof(float);
    }
    static ucc_status_t pack(void *o, size_t off, void *dst,
                             size_t *len)
    {
        auto  *st  = (std::pair<void *, size_t> *)o;
        float *src = (float *)st->first;
        size_t tot = st->second * sizeof(float);
        size_t l   = *len < tot - off ? *len : tot - off;
        size_t e0  = off / sizeof(float);
        for (size_t e = 0; e < l / sizeof(float); e++) {
            ((float *)dst)[e] = src[(e0 + e) * 2];
}
        *len = l;
        return UCC_OK;
    }
    static ucc_status_t unpack(void *o, size_t off,
                               const void *src, size_t len)
    {
        auto  *st  = (std::pair<void *, size_t> *)o;
        float *dst = (float *)st->first;
        size_t e0  = off / sizeof(float);
        for (size_t e = 0; e < len / sizeof(float); e++) {
            dst[(e0 + e) * 2] = ((const float *)src)[e];
}
        return UCC_OK;
    }
    static void finish(void *o)
    {
        delete (std::pair<void *, size_t> *)o;
    }
};


#define CHECK(x)                                                             \
    do {                                                                     \
        if (!(x)) {                                                          \
            fprintf(stderr, "FAILED: %s (line %d)\n", #x, __LINE__);         \
            return 1;                                                        \
        }                                                                    \
    } while (0)

int main()
{
    /* 1. create/query/destroy */
    ucc_generic_dt_ops_t ops{};
    ops.flags       = UCC_GENERIC_DT_OPS_FLAG_CONTIG |
                      UCC_GENERIC_DT_OPS_FLAG_REDUCE;
    ops.contig_size = sizeof(MinMax);
    ops.reduce      = mm_reduce;
    ucc_datatype_t gdt;
    CHECK(ucc_dt_create_generic(&ops, nullptr, &gdt) == UCC_OK);
    CHECK(!ucc_dt_is_predefined(gdt));
    CHECK(ucc_dt_size(gdt) == sizeof(MinMax));
    CHECK(ucc_dt_generic_ops(gdt) != nullptr);
    CHECK(ucc_dt_is_predefined(UCC_DT_FLOAT32));

    /* 2. CPU reduce via the callback */
    MinMax a[4] = {{1, 1}, {5, 5}, {-3, -3}, {0, 0}};
    MinMax b[4] = {{2, 2}, {4, 4}, {-1, -9}, {7, 7}};
    MinMax out[4];
    const void *srcs[2] = {a, b};
    CHECK(ucc::ec_cpu::reduce(out, srcs, 2, 4, gdt, UCC_OP_SUM) == UCC_OK);
    CHECK(out[0].mn == 1 && out[0].mx == 2);
    CHECK(out[2].mn == -3 && out[2].mx == -3);

    /* 3. 2-rank allreduce through the public API */
    ucc_lib_h     libs[2];
    ucc_context_h ctxs[2];
    ucc_team_h    teams[2];
    for (int r = 0; r < 2; r++) {
        ucc_lib_params_t lp{};
        CHECK(ucc_init(&lp, nullptr, &libs[r]) == UCC_OK);
        ucc_context_params_t cp{};
        CHECK(ucc_context_create(libs[r], &cp, nullptr, &ctxs[r]) ==
              UCC_OK);
        ucc_team_params_t tp{};
        tp.mask          = UCC_TEAM_PARAM_FIELD_OOB;
        tp.oob.allgather = lo_ag;
        tp.oob.req_test  = lo_test;
        tp.oob.req_free  = lo_free;
        tp.oob.coll_info = (void *)(intptr_t)r;
        tp.oob.n_oob_eps = 2;
        tp.oob.oob_ep    = r;
        CHECK(ucc_team_create_post(&ctxs[r], 1, &tp, &teams[r]) == UCC_OK);
    }
    while (true) {
        ucc_status_t s0 = ucc_team_create_test(teams[0]);
        ucc_status_t s1 = ucc_team_create_test(teams[1]);
        CHECK(s0 >= 0 && s1 >= 0);
        if (s0 == UCC_OK && s1 == UCC_OK) {
            break;
        }
    }
    const size_t        n = 1000;
    std::vector<MinMax> src0(n), src1(n), dst0(n), dst1(n);
    for (size_t i = 0; i < n; i++) {
        src0[i] = {(float)i, (float)i};
        src1[i] = {(float)(n - i), (float)(n - i)};
    }
    ucc_coll_req_h reqs[2];
    std::vector<MinMax> *sb[2] = {&src0, &src1}, *db[2] = {&dst0, &dst1};
    for (int r = 0; r < 2; r++) {
        ucc_coll_args_t args{};
        args.mask              = UCC_COLL_ARGS_FIELD_FLAGS;
        args.coll_type         = UCC_COLL_TYPE_ALLREDUCE;
        args.op                = UCC_OP_SUM; /* mapped to user callback */
        args.src.info.buffer   = sb[r]->data();
        args.src.info.count    = n;
        args.src.info.datatype = gdt;
        args.src.info.mem_type = UCC_MEMORY_TYPE_HOST;
        args.dst.info          = args.src.info;
        args.dst.info.buffer   = db[r]->data();
        CHECK(ucc_collective_init(&args, &reqs[r], teams[r]) == UCC_OK);
        CHECK(ucc_collective_post(reqs[r]) == UCC_OK);
    }
    while (ucc_collective_test(reqs[0]) == UCC_INPROGRESS ||
           ucc_collective_test(reqs[1]) == UCC_INPROGRESS) {
        ucc_context_progress(ctxs[0]);
        ucc_context_progress(ctxs[1]);
    }
    CHECK(ucc_collective_test(reqs[0]) == UCC_OK);
    CHECK(ucc_collective_test(reqs[1]) == UCC_OK);
    for (size_t i = 0; i < n; i++) {
        float mn = src0[i].mn < src1[i].mn ? src0[i].mn : src1[i].mn;
        float mx = src0[i].mx > src1[i].mx ? src0[i].mx : src1[i].mx;
        CHECK(dst0[i].mn == mn && dst0[i].mx == mx);
        CHECK(dst1[i].mn == mn && dst1[i].mx == mx);
    }
    for (int r = 0; r < 2; r++) {
        ucc_collective_finalize(reqs[r]);
    }

    /* 4. NON-CONTIGUOUS generic dtype: element = 1 float at stride 2;
     * bcast moves the packed image over tl/tcp (pack at root, unpack at
     * receivers — shm declines non-contig generics). */
    {
                ucc_generic_dt_ops_t nops{};
        nops.flags            = 0; /* non-contig */
        nops.ops.start_pack   = Strided::start_pack;
        nops.ops.start_unpack = Strided::start_unpack;
        nops.ops.packed_size  = Strided::packed_size;
        nops.ops.pack         = Strided::pack;
        nops.ops.unpack       = Strided::unpack;
        nops.ops.finish       = Strided::finish;
        ucc_datatype_t ndt;
        CHECK(ucc_dt_create_generic(&nops, nullptr, &ndt) == UCC_OK);

        const size_t       ecount = 500;
        std::vector<float> b0(ecount * 2), b1(ecount * 2, -1.0f);
        for (size_t i = 0; i < ecount; i++) {
            b0[i * 2]     = (float)i * 0.5f; /* payload lanes  */
            b0[i * 2 + 1] = -7.0f;           /* gap lanes      */
        }
        ucc_coll_req_h breqs[2];
        std::vector<float> *bb[2] = {&b0, &b1};
        for (int r = 0; r < 2; r++) {
            ucc_coll_args_t args{};
            args.mask              = UCC_COLL_ARGS_FIELD_FLAGS;
            args.coll_type         = UCC_COLL_TYPE_BCAST;
            args.root              = 0;
            args.src.info.buffer   = bb[r]->data();
            args.src.info.count    = ecount;
            args.src.info.datatype = ndt;
            args.src.info.mem_type = UCC_MEMORY_TYPE_HOST;
            CHECK(ucc_collective_init(&args, &breqs[r], teams[r]) ==
                  UCC_OK);
            CHECK(ucc_collective_post(breqs[r]) == UCC_OK);
        }
        while (ucc_collective_test(breqs[0]) == UCC_INPROGRESS ||
               ucc_collective_test(breqs[1]) == UCC_INPROGRESS) {
            ucc_context_progress(ctxs[0]);
            ucc_context_progress(ctxs[1]);
        }
        CHECK(ucc_collective_test(breqs[0]) == UCC_OK);
        CHECK(ucc_collective_test(breqs[1]) == UCC_OK);
        for (size_t i = 0; i < ecount; i++) {
            CHECK(b1[i * 2] == (float)i * 0.5f);  /* payload arrived */
            CHECK(b1[i * 2 + 1] == -1.0f);        /* gaps untouched  */
        }
        for (int r = 0; r < 2; r++) {
            ucc_collective_finalize(breqs[r]);
        }
        ucc_dt_destroy(ndt);
    }

    /* 5. non-contig generic through the core packed-image wrapper:
     * allgather / alltoall / gather / scatter run on packed bytes with
     * pack at post and unpack at completion (core ucc_collective_init
     * wrapper; reference generic-dt coverage beyond bcast). */
    {
        ucc_generic_dt_ops_t nops{};
        nops.flags            = 0;
        nops.ops.start_pack   = Strided::start_pack;
        nops.ops.start_unpack = Strided::start_unpack;
        nops.ops.packed_size  = Strided::packed_size;
        nops.ops.pack         = Strided::pack;
        nops.ops.unpack       = Strided::unpack;
        nops.ops.finish       = Strided::finish;
        ucc_datatype_t ndt;
        CHECK(ucc_dt_create_generic(&nops, nullptr, &ndt) == UCC_OK);
        const size_t per = 300; /* elements per block */

        auto run2 = [&](ucc_coll_args_t args0, ucc_coll_args_t args1) {
            ucc_coll_req_h rq[2];
            /* one rank uses the combined init_and_post entry point */
            if (ucc_collective_init_and_post(&args0, &rq[0], teams[0]) <
                0) {
                return false;
            }
            if (ucc_collective_init(&args1, &rq[1], teams[1]) != UCC_OK) {
                ucc_collective_finalize(rq[0]);
                return false;
            }
            if (ucc_collective_post(rq[1]) != UCC_OK) {
                return false;
            }
            while (ucc_collective_test(rq[0]) == UCC_INPROGRESS ||
                   ucc_collective_test(rq[1]) == UCC_INPROGRESS) {
                ucc_context_progress(ctxs[0]);
                ucc_context_progress(ctxs[1]);
            }
            bool ok = ucc_collective_test(rq[0]) == UCC_OK &&
                      ucc_collective_test(rq[1]) == UCC_OK;
            ucc_collective_finalize(rq[0]);
            ucc_collective_finalize(rq[1]);
            return ok;
        };
        auto set_elem = [](std::vector<float> &b, size_t i, float v) {
            b[i * 2] = v;
        };
        auto get_elem = [](const std::vector<float> &b, size_t i) {
            return b[i * 2];
        };

        { /* allgather */
            std::vector<float> s0(per * 2), s1(per * 2);
            std::vector<float> d0(per * 2 * 2, -1.f), d1(per * 2 * 2, -1.f);
            for (size_t i = 0; i < per; i++) {
                set_elem(s0, i, (float)i);
                set_elem(s1, i, (float)i + 1000.f);
            }
            ucc_coll_args_t a0{}, a1{};
            for (auto *a : {&a0, &a1}) {
                a->coll_type         = UCC_COLL_TYPE_ALLGATHER;
                a->src.info.count    = per;
                a->src.info.datatype = ndt;
                a->src.info.mem_type = UCC_MEMORY_TYPE_HOST;
                a->dst.info          = a->src.info;
                a->dst.info.count    = per * 2;
            }
            a0.src.info.buffer = s0.data();
            a0.dst.info.buffer = d0.data();
            a1.src.info.buffer = s1.data();
            a1.dst.info.buffer = d1.data();
            CHECK(run2(a0, a1));
            for (size_t i = 0; i < per; i++) {
                CHECK(get_elem(d0, i) == (float)i);
                CHECK(get_elem(d0, per + i) == (float)i + 1000.f);
                CHECK(get_elem(d1, per + i) == (float)i + 1000.f);
                CHECK(d0[i * 2 + 1] == -1.f); /* gaps untouched */
            }
        }
        { /* alltoall */
            std::vector<float> s0(per * 2 * 2), s1(per * 2 * 2);
            std::vector<float> d0(per * 2 * 2, -1.f), d1(per * 2 * 2, -1.f);
            for (size_t i = 0; i < 2 * per; i++) {
                set_elem(s0, i, (float)i);
                set_elem(s1, i, (float)i + 5000.f);
            }
            ucc_coll_args_t a0{}, a1{};
            for (auto *a : {&a0, &a1}) {
                a->coll_type         = UCC_COLL_TYPE_ALLTOALL;
                a->src.info.count    = per * 2;
                a->src.info.datatype = ndt;
                a->src.info.mem_type = UCC_MEMORY_TYPE_HOST;
                a->dst.info          = a->src.info;
            }
            a0.src.info.buffer = s0.data();
            a0.dst.info.buffer = d0.data();
            a1.src.info.buffer = s1.data();
            a1.dst.info.buffer = d1.data();
            CHECK(run2(a0, a1));
            for (size_t i = 0; i < per; i++) {
                CHECK(get_elem(d0, i) == (float)i);              /* 0->0 */
                CHECK(get_elem(d0, per + i) == (float)i + 5000.f);
                CHECK(get_elem(d1, i) == (float)(per + i));      /* 0->1 */
                CHECK(get_elem(d1, per + i) == (float)(per + i) + 5000.f);
            }
        }
        { /* gather to root 1 + scatter from root 1 */
            std::vector<float> s0(per * 2), s1(per * 2);
            std::vector<float> gd(per * 2 * 2, -1.f);
            for (size_t i = 0; i < per; i++) {
                set_elem(s0, i, (float)i * 3.f);
                set_elem(s1, i, (float)i * 7.f);
            }
            ucc_coll_args_t a0{}, a1{};
            a0.coll_type         = UCC_COLL_TYPE_GATHER;
            a0.root              = 1;
            a0.src.info.buffer   = s0.data();
            a0.src.info.count    = per;
            a0.src.info.datatype = ndt;
            a0.src.info.mem_type = UCC_MEMORY_TYPE_HOST;
            a0.dst.info          = a0.src.info;
            a0.dst.info.buffer   = nullptr;
            a1                   = a0;
            a1.src.info.buffer   = s1.data();
            a1.dst.info.buffer   = gd.data();
            a1.dst.info.count    = per * 2;
            a1.src.info.count    = per;
            CHECK(run2(a0, a1));
            for (size_t i = 0; i < per; i++) {
                CHECK(get_elem(gd, i) == (float)i * 3.f);
                CHECK(get_elem(gd, per + i) == (float)i * 7.f);
            }
            /* scatter it back out */
            std::vector<float> r0(per * 2, -2.f), r1(per * 2, -2.f);
            ucc_coll_args_t b0{}, b1{};
            b0.coll_type         = UCC_COLL_TYPE_SCATTER;
            b0.root              = 1;
            b0.dst.info.buffer   = r0.data();
            b0.dst.info.count    = per;
            b0.dst.info.datatype = ndt;
            b0.dst.info.mem_type = UCC_MEMORY_TYPE_HOST;
            b0.src.info          = b0.dst.info;
            b0.src.info.buffer   = nullptr;
            b1                   = b0;
            b1.src.info.buffer   = gd.data();
            b1.src.info.count    = per * 2;
            b1.dst.info.buffer   = r1.data();
            CHECK(run2(b0, b1));
            for (size_t i = 0; i < per; i++) {
                CHECK(get_elem(r0, i) == (float)i * 3.f);
                CHECK(get_elem(r1, i) == (float)i * 7.f);
                CHECK(r0[i * 2 + 1] == -2.f);
            }
        }
        ucc_dt_destroy(ndt);
    }

    for (int r = 0; r < 2; r++) {
        ucc_team_destroy(teams[r]);
        ucc_context_destroy(ctxs[r]);
        ucc_finalize(libs[r]);
    }
    ucc_dt_destroy(gdt);
    printf("GENERIC_DT_OK\n");
    return 0;
}
