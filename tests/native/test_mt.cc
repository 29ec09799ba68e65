/* ThreadSanitizer-targeted MT test: THREAD_MULTIPLE contexts with
 * dedicated progress threads pumping concurrently with posting threads,
 * two teams driven at once (lock-free progress queue + shm slot-counter
 * protocol under tsan). Exits 0 on success. */
#include <atomic>
#include <cstdio>
#include <cstring>
#include <map>
#include <mutex>
#include <thread>
#include <vector>

#include "../../src/api/ucc.h"

/* thread-safe round-keyed in-process OOB */
struct Loob {
    struct Round {
        std::vector<std::vector<uint8_t>> blobs;
        std::atomic<int>                  arrived{0};
    };
    std::mutex                 mu;
    std::map<uint64_t, Round>  rounds;
    uint64_t                   next[2] = {0, 0};
    int                        n = 2;

    Round *get(uint64_t round)
    {
        std::lock_guard<std::mutex> g(mu);
        auto &r = rounds[round];
        if (r.blobs.empty()) {
            r.blobs.resize(n);
        }
        return &r;
    }
};
static Loob g_loob[2]; /* one OOB per team */

struct LoobReq {
    Loob    *lo;
    uint64_t round;
    void    *recv;
    size_t   size;
};

template <int TEAM>
static ucc_status_t lo_ag(void *src, void *recv, size_t size, void *info,
                          void **req)
{
    Loob    *lo   = &g_loob[TEAM];
    int      rank = (int)(intptr_t)info;
    uint64_t round;
    {
        std::lock_guard<std::mutex> g(lo->mu);
        round = lo->next[rank]++;
    }
    Loob::Round *r = lo->get(round);
    r->blobs[rank].assign((uint8_t *)src, (uint8_t *)src + size);
    r->arrived.fetch_add(1, std::memory_order_release);
    *req = new LoobReq{lo, round, recv, size};
    return UCC_OK;
}
static ucc_status_t lo_test(void *req)
{
    auto        *r  = (LoobReq *)req;
    Loob::Round *rd = r->lo->get(r->round);
    if (rd->arrived.load(std::memory_order_acquire) < r->lo->n) {
        return UCC_INPROGRESS;
    }
    for (int i = 0; i < r->lo->n; i++) {
        memcpy((uint8_t *)r->recv + i * r->size, rd->blobs[i].data(),
               r->size);
    }
    return UCC_OK;
}
static ucc_status_t lo_free(void *req)
{
    delete (LoobReq *)req;
    return UCC_OK;
}

#define CHECK(x)                                                           \
    do {                                                                   \
        if (!(x)) {                                                        \
            fprintf(stderr, "FAILED: %s (line %d)\n", #x, __LINE__);       \
            exit(1);                                                       \
        }                                                                  \
    } while (0)

int main()
{
    const int     n = 2;
    ucc_lib_h     libs[2];
    ucc_context_h ctxs[2];
    ucc_team_h    t1[2], t2[2];
    for (int r = 0; r < n; r++) {
        ucc_lib_params_t lp{};
        lp.mask        = UCC_LIB_PARAM_FIELD_THREAD_MODE;
        lp.thread_mode = UCC_THREAD_MULTIPLE;
        CHECK(ucc_init(&lp, nullptr, &libs[r]) == UCC_OK);
        ucc_context_params_t cp{};
        CHECK(ucc_context_create(libs[r], &cp, nullptr, &ctxs[r]) ==
              UCC_OK);
    }
    auto mk_team = [&](ucc_team_h *out, int which) {
        for (int r = 0; r < n; r++) {
            ucc_team_params_t tp{};
            tp.mask          = UCC_TEAM_PARAM_FIELD_OOB;
            tp.oob.allgather = which == 0 ? lo_ag<0> : lo_ag<1>;
            tp.oob.req_test  = lo_test;
            tp.oob.req_free  = lo_free;
            tp.oob.coll_info = (void *)(intptr_t)r;
            tp.oob.n_oob_eps = n;
            tp.oob.oob_ep    = r;
            CHECK(ucc_team_create_post(&ctxs[r], 1, &tp, &out[r]) ==
                  UCC_OK);
        }
        while (true) {
            ucc_status_t s0 = ucc_team_create_test(out[0]);
            ucc_status_t s1 = ucc_team_create_test(out[1]);
            CHECK(s0 >= 0 && s1 >= 0);
            if (s0 == UCC_OK && s1 == UCC_OK) {
                break;
            }
        }
    };
    mk_team(t1, 0);
    mk_team(t2, 1);

    /* dedicated progress threads (concurrent with posting threads) */
    std::atomic<bool> stop{false};
    std::thread pumps[2];
    for (int r = 0; r < n; r++) {
        pumps[r] = std::thread([&, r] {
            while (!stop.load(std::memory_order_acquire)) {
                ucc_context_progress(ctxs[r]);
            }
        });
    }

    auto run2 = [&](ucc_coll_args_t (&aa)[2], ucc_team_h *teams) {
        ucc_coll_req_h rq[2];
        for (int r = 0; r < n; r++) {
            CHECK(ucc_collective_init(&aa[r], &rq[r], teams[r]) ==
                  UCC_OK);
            CHECK(ucc_collective_post(rq[r]) >= 0);
        }
        while (ucc_collective_test(rq[0]) == UCC_INPROGRESS ||
               ucc_collective_test(rq[1]) == UCC_INPROGRESS) {
            std::this_thread::yield();
        }
        CHECK(ucc_collective_test(rq[0]) == UCC_OK);
        CHECK(ucc_collective_test(rq[1]) == UCC_OK);
        ucc_collective_finalize(rq[0]);
        ucc_collective_finalize(rq[1]);
    };
    /* rotate collective types and sizes so the MT run crosses score
     * bands (knomial/ring/SRA/etc.) instead of hammering one path */
    auto drive = [&](ucc_team_h *teams, float base) {
        for (int it = 0; it < 40; it++) {
            const size_t cnt = (it % 3 == 2) ? 20000 : 700;
            std::vector<float> s0(cnt, base + it), s1(cnt, base - it);
            std::vector<float> d0(cnt), d1(cnt);
            std::vector<float> *sb[2] = {&s0, &s1}, *db[2] = {&d0, &d1};
            ucc_coll_args_t aa[2]{};
            switch (it % 4) {
            case 0:
            case 2: { /* allreduce (small + SRA-band sizes) */
                for (int r = 0; r < n; r++) {
                    aa[r].coll_type         = UCC_COLL_TYPE_ALLREDUCE;
                    aa[r].op                = UCC_OP_SUM;
                    aa[r].src.info.buffer   = sb[r]->data();
                    aa[r].src.info.count    = cnt;
                    aa[r].src.info.datatype = UCC_DT_FLOAT32;
                    aa[r].src.info.mem_type = UCC_MEMORY_TYPE_HOST;
                    aa[r].dst.info          = aa[r].src.info;
                    aa[r].dst.info.buffer   = db[r]->data();
                }
                run2(aa, teams);
                for (size_t i = 0; i < cnt; i++) {
                    CHECK(d0[i] == 2 * base && d1[i] == 2 * base);
                }
                break;
            }
            case 1: { /* bcast from rank 0 */
                for (int r = 0; r < n; r++) {
                    aa[r].coll_type         = UCC_COLL_TYPE_BCAST;
                    aa[r].root              = 0;
                    aa[r].src.info.buffer   = sb[r]->data();
                    aa[r].src.info.count    = cnt;
                    aa[r].src.info.datatype = UCC_DT_FLOAT32;
                    aa[r].src.info.mem_type = UCC_MEMORY_TYPE_HOST;
                }
                run2(aa, teams);
                for (size_t i = 0; i < cnt; i++) {
                    CHECK(s1[i] == base + it); /* got rank 0's value */
                }
                break;
            }
            case 3: { /* allgather (own half + peer half) */
                size_t per = cnt / 2;
                for (int r = 0; r < n; r++) {
                    aa[r].coll_type         = UCC_COLL_TYPE_ALLGATHER;
                    aa[r].src.info.buffer   = sb[r]->data();
                    aa[r].src.info.count    = per;
                    aa[r].src.info.datatype = UCC_DT_FLOAT32;
                    aa[r].src.info.mem_type = UCC_MEMORY_TYPE_HOST;
                    aa[r].dst.info          = aa[r].src.info;
                    aa[r].dst.info.count    = per * 2;
                    aa[r].dst.info.buffer   = db[r]->data();
                }
                run2(aa, teams);
                for (size_t i = 0; i < per; i++) {
                    CHECK(d0[i] == base + it &&
                          d0[per + i] == base - it);
                    CHECK(d1[i] == base + it &&
                          d1[per + i] == base - it);
                }
                break;
            }
            }
        }
    };
    std::thread a([&] { drive(t1, 100.f); });
    std::thread b([&] { drive(t2, 900.f); });
    a.join();
    b.join();
    stop.store(true, std::memory_order_release);
    pumps[0].join();
    pumps[1].join();

    for (int r = 0; r < n; r++) {
        ucc_team_destroy(t1[r]);
        ucc_team_destroy(t2[r]);
        ucc_context_destroy(ctxs[r]);
        ucc_finalize(libs[r]);
    }
    printf("MT_OK\n");
    return 0;
}
