/* Fork-mode bootstrap OOB for ucc_perftest: a POSIX-shm allgather shared
 * by N forked children (the role MPI plays for the reference's perftest,
 * tools/perf — re-derived MPI-free for single-node MI355X boxes).
 *
 * Protocol: monotone per-rank produced/consumed round counters + a fixed
 * per-rank blob area. Rank r: wait all consumed >= round (area free),
 * write blob, produced[r] = round+1 (release); reader waits all
 * produced >= round+1, copies all blobs, consumed[r] = round+1. */
#ifndef UCC_AMD_TOOLS_SHM_OOB_H_
#define UCC_AMD_TOOLS_SHM_OOB_H_

#include <atomic>
#include <cstdint>
#include <cstring>
#include <string>

#include <fcntl.h>
#include <sched.h>
#include <sys/mman.h>
#include <unistd.h>

namespace uccperf {

constexpr size_t kMaxBlob = 1 << 20; /* 1 MiB per rank per round */

struct ShmOobSeg {
    std::atomic<uint64_t> produced[64];
    std::atomic<uint64_t> consumed[64];
    /* data[rank * kMaxBlob] follows */
    uint8_t data[];
};

class ShmOob {
  public:
    /* parent calls create() before forking; children call open_(). */
    bool create(const std::string &name, int nranks)
    {
        name_ = name;
        n_    = nranks;
        shm_unlink(name.c_str());
        int fd = shm_open(name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
        if (fd < 0) {
            return false;
        }
        size_ = sizeof(ShmOobSeg) + (size_t)nranks * kMaxBlob;
        if (ftruncate(fd, (off_t)size_) != 0) {
            close(fd);
            return false;
        }
        seg_ = (ShmOobSeg *)mmap(nullptr, size_, PROT_READ | PROT_WRITE,
                                 MAP_SHARED, fd, 0);
        close(fd);
        if (seg_ == MAP_FAILED) {
            seg_ = nullptr;
            return false;
        }
        memset((void *)seg_, 0, sizeof(ShmOobSeg));
        return true;
    }

    bool open_(const std::string &name, int nranks, int rank)
    {
        name_ = name;
        n_    = nranks;
        rank_ = rank;
        size_ = sizeof(ShmOobSeg) + (size_t)nranks * kMaxBlob;
        int fd = shm_open(name.c_str(), O_RDWR, 0600);
        if (fd < 0) {
            return false;
        }
        seg_ = (ShmOobSeg *)mmap(nullptr, size_, PROT_READ | PROT_WRITE,
                                 MAP_SHARED, fd, 0);
        close(fd);
        return seg_ != MAP_FAILED;
    }

    void set_rank(int rank) { rank_ = rank; }

    /* blocking allgather: send size bytes, receive n*size into recv */
    void allgather(const void *src, void *recv, size_t size)
    {
        uint64_t r = round_++;
        for (int i = 0; i < n_; i++) {
            while (seg_->consumed[i].load(std::memory_order_acquire) < r) {
                sched_yield();
            }
        }
        memcpy(seg_->data + (size_t)rank_ * kMaxBlob, src, size);
        seg_->produced[rank_].store(r + 1, std::memory_order_release);
        for (int i = 0; i < n_; i++) {
            while (seg_->produced[i].load(std::memory_order_acquire) <
                   r + 1) {
                sched_yield();
            }
        }
        for (int i = 0; i < n_; i++) {
            memcpy((uint8_t *)recv + (size_t)i * size,
                   seg_->data + (size_t)i * kMaxBlob, size);
        }
        seg_->consumed[rank_].store(r + 1, std::memory_order_release);
    }

    double max_double(double v)
    {
        double all[64];
        allgather(&v, all, sizeof(double));
        double m = v;
        for (int i = 0; i < n_; i++) {
            m = all[i] > m ? all[i] : m;
        }
        return m;
    }

    void unlink_()
    {
        if (!name_.empty()) {
            shm_unlink(name_.c_str());
        }
    }

    int n() const { return n_; }
    int rank() const { return rank_; }

  private:
    std::string name_;
    ShmOobSeg  *seg_  = nullptr;
    size_t      size_ = 0;
    int         n_ = 0, rank_ = 0;
    uint64_t    round_ = 0;
};

} // namespace uccperf

#endif
