/* Slotted shared-memory synchronization segment: per-slot per-rank
 * monotonic step counters + optional staging data/result areas. Shared by
 * tl/shm (host data plane) and tl/cdna4 (control plane for the xGMI data
 * plane). Protocol documented in tl_shm.cc; design lineage: reference
 * tl_cuda_coll.h slot/sync structures re-derived without sense reversal. */
#ifndef UCC_AMD_SLOT_SEG_H_
#define UCC_AMD_SLOT_SEG_H_

#include <atomic>
#include <string>
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include "../../api/ucc.h"

namespace ucc {

constexpr uint64_t kCap       = 1ull << 20; /* max steps per slot use   */
constexpr size_t   kLine      = 64;
constexpr uint32_t kMagic     = 0x55cca3d1;

struct SegHeader {
    uint32_t magic;
    uint32_t nranks;
    uint32_t nslots;
    uint64_t chunk;
};

class ShmSeg {
  public:
    ucc_status_t create(const std::string &name, uint32_t nranks,
                        uint32_t nslots, size_t chunk)
    {
        name_  = name;
        owner_ = true;
        size_  = layout_size(nranks, nslots, chunk);
        shm_unlink(name.c_str()); /* stale cleanup */
        int fd = shm_open(name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
        if (fd < 0) {
            return UCC_ERR_NO_RESOURCE;
        }
        if (ftruncate(fd, (off_t)size_) != 0) {
            close(fd);
            shm_unlink(name.c_str());
            return UCC_ERR_NO_MEMORY;
        }
        base_ = mmap(nullptr, size_, PROT_READ | PROT_WRITE, MAP_SHARED, fd,
                     0);
        close(fd);
        if (base_ == MAP_FAILED) {
            base_ = nullptr;
            shm_unlink(name.c_str());
            return UCC_ERR_NO_MEMORY;
        }
        auto *h   = (SegHeader *)base_;
        h->nranks = nranks;
        h->nslots = nslots;
        h->chunk  = chunk;
        set_geom(nranks, nslots, chunk);
        ((std::atomic<uint32_t> *)&h->magic)
            ->store(kMagic, std::memory_order_release);
        return UCC_OK;
    }

    ucc_status_t attach(const std::string &name, uint32_t nranks,
                        uint32_t nslots, size_t chunk)
    {
        name_  = name;
        owner_ = false;
        size_  = layout_size(nranks, nslots, chunk);
        int fd = shm_open(name.c_str(), O_RDWR, 0600);
        if (fd < 0) {
            return UCC_INPROGRESS; /* creator not there yet */
        }
        struct stat st;
        if (fstat(fd, &st) != 0 || (size_t)st.st_size < size_) {
            close(fd);
            return UCC_INPROGRESS;
        }
        base_ = mmap(nullptr, size_, PROT_READ | PROT_WRITE, MAP_SHARED, fd,
                     0);
        close(fd);
        if (base_ == MAP_FAILED) {
            base_ = nullptr;
            return UCC_ERR_NO_MEMORY;
        }
        set_geom(nranks, nslots, chunk);
        return UCC_OK;
    }

    bool ready() const
    {
        if (!base_) {
            return false;
        }
        auto *h = (const SegHeader *)base_;
        return ((const std::atomic<uint32_t> *)&h->magic)
                   ->load(std::memory_order_acquire) == kMagic;
    }

    ~ShmSeg()
    {
        if (base_) {
            munmap(base_, size_);
        }
        if (owner_) {
            shm_unlink(name_.c_str());
        }
    }

    std::atomic<uint64_t> *step(uint32_t slot, uint32_t rank)
    {
        return (std::atomic<uint64_t> *)((uint8_t *)base_ + steps_off_ +
                                         ((size_t)slot * nranks_ + rank) *
                                             kLine);
    }
    uint8_t *data(uint32_t slot, uint32_t parity, uint32_t rank)
    {
        return (uint8_t *)base_ + data_off_ +
               (((size_t)slot * 2 + parity) * nranks_ + rank) * chunk_;
    }
    uint8_t *result(uint32_t slot, uint32_t parity)
    {
        return (uint8_t *)base_ + result_off_ +
               ((size_t)slot * 2 + parity) * chunk_;
    }
    size_t chunk() const { return chunk_; }

  private:
    static size_t layout_size(uint32_t nranks, uint32_t nslots, size_t chunk)
    {
        size_t sz = kLine;                              /* header      */
        sz += (size_t)nslots * nranks * kLine;          /* steps       */
        sz += (size_t)nslots * 2 * nranks * chunk;      /* data        */
        sz += (size_t)nslots * 2 * chunk;               /* result      */
        return sz;
    }
    void set_geom(uint32_t nranks, uint32_t nslots, size_t chunk)
    {
        nranks_     = nranks;
        nslots_     = nslots;
        chunk_      = chunk;
        steps_off_  = kLine;
        data_off_   = steps_off_ + (size_t)nslots * nranks * kLine;
        result_off_ = data_off_ + (size_t)nslots * 2 * nranks * chunk;
    }

    void       *base_  = nullptr;
    size_t      size_  = 0;
    bool        owner_ = false;
    std::string name_;
    uint32_t    nranks_ = 0, nslots_ = 0;
    size_t      chunk_ = 0;
    size_t      steps_off_ = 0, data_off_ = 0, result_off_ = 0;
};


} // namespace ucc

#endif
