/* Task engine: dependency-counted tasks + schedules driven by the context
 * progress queue. Parity: reference src/schedule/ (ucc_coll_task_t event
 * manager + ucc_schedule_t) and core/ucc_progress_queue*. */
#include "../core/core.h"

#include <algorithm>
#include <ctime>

namespace ucc {

double time_sec()
{
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (double)ts.tv_sec + 1e-9 * (double)ts.tv_nsec;
}

void Context::pq_push(Task *t)
{
    /* atomic claim: two concurrent pushers must not both enqueue (a
     * double enqueue means double progress()/task_completed()) */
    if (t->in_pq.exchange(true, std::memory_order_acq_rel)) {
        return;
    }
    if (mt && lock_free) {
        if (lf_pq.push(t)) {
            return;
        }
        /* ring full: overflow to the locked deque */
        std::lock_guard<std::recursive_mutex> lk(pq_mtx);
        pq.push_back(t);
        return;
    }
    if (mt) {
        std::lock_guard<std::recursive_mutex> lk(pq_mtx);
        pq.push_back(t);
    } else {
        pq.push_back(t);
    }
}

void task_start(Task *t)
{
    t->start_time = time_sec();
    ucc_status_t st = t->post();
    t->status       = st;
    if (st == UCC_INPROGRESS) {
        t->ctx_->pq_push(t);
    } else {
        task_completed(t);
    }
}

void task_completed(Task *t)
{
    t->on_complete();
    if (t->req_ && t->status == UCC_OK && t->req_->post_complete) {
        ucc_status_t fs = t->req_->post_complete();
        if (fs != UCC_OK) {
            t->status = fs;
        }
    }
    /* everything that dereferences t must happen BEFORE completion is
     * published: the moment super.status flips, a THREAD_MULTIPLE user
     * may call ucc_collective_finalize and free the task under us
     * (tsan-caught lifetime race). */
    if (t->sched) {
        t->sched->subtask_completed(t);
    }
    for (Task *d : t->dependents) {
        d->n_satisfied++;
        if (d->n_satisfied == d->n_deps) {
            task_start(d);
        }
    }
    if (t->req_) {
        ucc_coll_req_t *pub = &t->req_->super;
        ucc_status_t    st  = t->status.load();
        ucc_coll_callback_t cb{};
        bool has_cb = (t->req_->args.mask & UCC_COLL_ARGS_FIELD_CB) &&
                      t->req_->args.cb.cb;
        if (has_cb) {
            cb = t->req_->args.cb;
        }
        req_status_store(pub, st); /* last access to t/req             */
        if (has_cb) {
            cb.cb(cb.data, st);
        }
    }
}

ucc_status_t Schedule::post()
{
    status       = UCC_INPROGRESS;
    n_completed_ = 0;
    posting_     = true;
    /* reset dep counters for persistent re-post */
    for (auto *t : tasks_) {
        t->n_satisfied = 0;
        t->status      = UCC_OPERATION_INITIALIZED;
    }
    for (auto *t : tasks_) {
        if (t->n_deps == 0) {
            task_start(t);
        }
    }
    posting_ = false;
    if (tasks_.empty() || n_completed_ == tasks_.size()) {
        if (status == UCC_INPROGRESS) {
            status = UCC_OK;
        }
    }
    /* task_start(this) handles the terminal-status notification */
    return status;
}

void Schedule::subtask_completed(Task *t)
{
    if (t->status < 0 && status >= 0) {
        status = t->status.load(); /* propagate first error */
        if (!posting_) {
            task_completed(this);
        }
        return;
    }
    n_completed_++;
    if (n_completed_ == tasks_.size() && status == UCC_INPROGRESS) {
        if (!posting_) {
            status = UCC_OK;
            task_completed(this);
        }
        /* else: Schedule::post finalizes the status itself */
    }
}

/* ----------------------------------------------------- PipelineTask */
PipelineTask::~PipelineTask()
{
    for (auto &f : fl_) {
        if (f.req) {
            ucc_collective_finalize(f.req);
        }
    }
}

ucc_status_t PipelineTask::post()
{
    done_ = 0;
    next_ = 0;
    ord_.assign(n_stages, 0);
    size_t d = pdepth < n_frags ? pdepth : n_frags;
    if (d == 0) {
        status = UCC_OK;
        return UCC_OK;
    }
    fl_.assign(d, Flight{});
    for (size_t i = 0; i < d; i++) {
        fl_[i].frag   = next_++;
        fl_[i].active = true;
    }
    status = UCC_INPROGRESS;
    return drive();
}

ucc_status_t PipelineTask::drive()
{
    bool moved = true;
    while (moved) {
        moved = false;
        for (auto &f : fl_) {
            if (!f.active) {
                continue;
            }
            if (!f.posted) {
                if (ord_[f.stage] != f.frag) {
                    continue; /* stage s posts in fragment order */
                }
                if (f.stage == pair_late &&
                    ord_[pair_early] <
                        std::min(f.frag + 2, n_frags)) {
                    continue; /* canonical shared-team interleave */
                }
                ucc_coll_req_h r  = nullptr;
                ucc_status_t   st = stage_post(f.frag, f.stage, &r);
                if (st != UCC_OK) {
                    return st;
                }
                f.req    = r;
                f.posted = true;
                ord_[f.stage]++;
                if (trace) {
                    trace('P', f.frag, f.stage);
                }
                moved = true;
            }
            ucc_status_t st = f.req ? ucc_collective_test(f.req) : UCC_OK;
            if (st == UCC_INPROGRESS) {
                continue;
            }
            if (f.req) {
                ucc_collective_finalize(f.req);
                f.req = nullptr;
            }
            if (st != UCC_OK) {
                return st;
            }
            if (stage_done) {
                st = stage_done(f.frag, f.stage);
                if (st != UCC_OK) {
                    return st;
                }
            }
            if (trace) {
                trace('C', f.frag, f.stage);
            }
            f.posted = false;
            f.stage++;
            moved = true;
            if (f.stage == n_stages) {
                done_++;
                if (next_ < n_frags) {
                    f.frag  = next_++;
                    f.stage = 0;
                } else {
                    f.active = false;
                }
            }
        }
    }
    return done_ == n_frags ? UCC_OK : UCC_INPROGRESS;
}

ucc_status_t Context::progress()
{
    n_progress_calls++;
    Task *t = nullptr;
    if (mt && lock_free) {
        if (!lf_pq.pop(&t)) {
            std::lock_guard<std::recursive_mutex> lk(pq_mtx);
            if (!pq.empty()) {
                t = pq.front();
                pq.pop_front();
            }
        }
    } else {
        if (mt) {
            pq_mtx.lock();
        }
        if (!pq.empty()) {
            t = pq.front();
            pq.pop_front();
        }
        if (mt) {
            pq_mtx.unlock();
        }
    }
    if (!t) {
        return UCC_OK;
    }
    t->in_pq        = false;
    ucc_status_t st = t->progress();
    t->status       = st;
    if (st == UCC_INPROGRESS) {
        if (t->timeout > 0 && time_sec() - t->start_time > t->timeout) {
            ucc_warn("collective task %p timed out after %.1fs", (void *)t,
                     t->timeout);
            t->status = UCC_ERR_TIMED_OUT;
            task_completed(t);
            return UCC_ERR_TIMED_OUT;
        }
        pq_push(t);
        return UCC_OK;
    }
    task_completed(t);
    return st < 0 ? st : UCC_OK;
}

} // namespace ucc
