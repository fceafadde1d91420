// lmdb_reader.cpp — from-scratch read-only LMDB walker.
//
// The reference reads training data from LMDB through liblmdb
// (src/caffe/util/db_lmdb.cpp); this environment has no liblmdb, no
// python-lmdb and no datasets, so the build implements the published
// LMDB file format directly (symas.com/lmdb, lmdb.h/mdb.c layout,
// version 1 / magic 0xBEEFC0DE, 4096-byte pages):
//   page 0/1: meta pages (MDB_meta after the page header); the live one
//     has the larger txnid; its main-DB record carries the B+tree root.
//   branch pages (flags & 0x01): nodes point at child pages;
//   leaf pages  (flags & 0x02): nodes carry key + value, or key +
//     overflow pgno when F_BIGDATA (0x01) is set — value bytes then live
//     in consecutive overflow pages (flags & 0x04) after their headers.
// Iteration = leftmost-descent then leaf-by-leaf cursor walk, exactly
// the record order liblmdb's MDB_NEXT yields (keys are sorted, which is
// why convert_imageset zero-pads its "%08d_" key prefix).
// Values are caffe::Datum wire messages (decoded in data layer code).
#include "lmdb_reader.hpp"

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstring>

namespace camd {

namespace {
constexpr uint32_t MDB_MAGIC = 0xBEEFC0DE;
constexpr uint16_t P_BRANCH = 0x01;
constexpr uint16_t P_LEAF = 0x02;
constexpr uint16_t P_OVERFLOW = 0x04;
constexpr uint16_t F_BIGDATA = 0x01;

#pragma pack(push, 1)
struct PageHeader {       // MDB_page header (64-bit build)
  uint64_t pgno;          // also mp_next/overflow count union
  uint16_t pad;
  uint16_t flags;
  union {
    struct {
      uint16_t lower;     // offset of end-of-ptrs
      uint16_t upper;     // offset of start-of-nodes
    } b;
    uint32_t pages;       // P_OVERFLOW: number of overflow pages
  } u;
};
struct Node {             // MDB_node
  uint16_t lo;            // data size low / child pgno low
  uint16_t hi;            // data size high / child pgno mid
  uint16_t flags;         // child pgno high sits here on branch pages
  uint16_t ksize;
  // key bytes follow, then value bytes (leaf)
};
struct MetaDB {           // MDB_db
  uint32_t pad;
  uint16_t flags;
  uint16_t depth;
  uint64_t branch_pages, leaf_pages, overflow_pages, entries, root;
};
struct Meta {             // MDB_meta (after the page header)
  uint32_t magic;
  uint32_t version;
  uint64_t fixup;         // mm_address
  uint64_t mapsize;
  MetaDB dbs[2];          // free DB, main DB
  uint64_t last_pg;
  uint64_t txnid;
};
#pragma pack(pop)

inline uint64_t branch_child(const Node* n) {
  // branch node: child pgno packed into lo | hi<<16 | flags<<32
  return (uint64_t)n->lo | ((uint64_t)n->hi << 16) |
         ((uint64_t)n->flags << 32);
}
inline size_t leaf_dsize(const Node* n) {
  return (size_t)n->lo | ((size_t)n->hi << 16);
}
}  // namespace

LmdbReader::~LmdbReader() {
  if (map_) munmap((void*)map_, map_size_);
  if (fd_ >= 0) close(fd_);
}

void LmdbReader::open(const std::string& source) {
  // caffe convention: `source` is the LMDB directory holding data.mdb
  std::string path = source;
  struct stat st {};
  if (stat(path.c_str(), &st) == 0 && S_ISDIR(st.st_mode))
    path += "/data.mdb";
  fd_ = ::open(path.c_str(), O_RDONLY);
  CHECK_GE_(fd_, 0) << "cannot open LMDB " << path;
  CHECK_EQ_(fstat(fd_, &st), 0);
  map_size_ = (size_t)st.st_size;
  map_ = (const uint8_t*)mmap(nullptr, map_size_, PROT_READ, MAP_SHARED,
                              fd_, 0);
  CHECK_(map_ != MAP_FAILED) << "mmap failed for " << path;

  // pick the live meta page (larger txnid)
  const Meta* m0 = (const Meta*)(map_ + sizeof(PageHeader));
  const Meta* m1 = (const Meta*)(map_ + page_size_ + sizeof(PageHeader));
  CHECK_EQ_(m0->magic, MDB_MAGIC) << "not an LMDB file: " << path;
  CHECK_EQ_(m0->version, 1u) << "unsupported LMDB format version";
  const Meta* m = (m1->magic == MDB_MAGIC && m1->txnid > m0->txnid) ? m1
                                                                    : m0;
  root_ = m->dbs[1].root;
  entries_ = m->dbs[1].entries;
  CHECK_(root_ != ~0ull) << "empty LMDB " << path;

  // collect leaf pages left-to-right (the sorted cursor order)
  leaves_.clear();
  collect_leaves(root_);
  CHECK_GT_((long)leaves_.size(), 0);
}

const uint8_t* LmdbReader::page(uint64_t pgno) const {
  const uint64_t off = pgno * page_size_;
  CHECK_LT_(off, map_size_);
  return map_ + off;
}

void LmdbReader::collect_leaves(uint64_t pgno) {
  const PageHeader* ph = (const PageHeader*)page(pgno);
  if (ph->flags & P_LEAF) {
    leaves_.push_back(pgno);
    return;
  }
  CHECK_(ph->flags & P_BRANCH) << "unexpected page flags "
                               << ph->flags;
  const uint16_t* ptrs =
      (const uint16_t*)(page(pgno) + sizeof(PageHeader));
  // node count the way mdb.c's NUMKEYS does: (mp_lower - PAGEHDRSZ) / 2
  const int nn = (int)((ph->u.b.lower - 16) / 2);
  for (int i = 0; i < nn; ++i) {
    const Node* nd = (const Node*)(page(pgno) + ptrs[i]);
    collect_leaves(branch_child(nd));
  }
}

long LmdbReader::size() const { return (long)entries_; }

// record i (global sorted order) -> value bytes
std::pair<const uint8_t*, size_t> LmdbReader::at(long idx) {
  // walk the cached leaf list; per-leaf node counts are cheap to read
  CHECK_GE_(idx, 0);
  long rem = idx;
  for (uint64_t pgno : leaves_) {
    const PageHeader* ph = (const PageHeader*)page(pgno);
    const int nn = (int)((ph->u.b.lower - 16) / 2);
    if (rem >= nn) {
      rem -= nn;
      continue;
    }
    const uint16_t* ptrs =
        (const uint16_t*)(page(pgno) + sizeof(PageHeader));
    const Node* nd = (const Node*)(page(pgno) + ptrs[rem]);
    const uint8_t* kv = (const uint8_t*)nd + 8 + nd->ksize;
    const size_t dsz = leaf_dsize(nd);
    if (nd->flags & F_BIGDATA) {
      // value = overflow pgno; bytes start after that page's header
      uint64_t ovpg;
      memcpy(&ovpg, kv, 8);
      return {page(ovpg) + sizeof(PageHeader), dsz};
    }
    return {kv, dsz};
  }
  CAMD_FATAL << "LMDB record index " << idx << " out of range";
}

}  // namespace camd
