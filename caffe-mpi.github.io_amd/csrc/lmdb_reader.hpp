// lmdb_reader.hpp — read-only LMDB walker (no liblmdb in this
// environment; the published format is implemented directly — see
// lmdb_reader.cpp).  Replaces the reference's util/db_lmdb.cpp cursor for
// the DataLayer's LMDB backend (data_param { source backend: LMDB }).
#pragma once

#include <string>
#include <utility>
#include <vector>

#include "core.hpp"

namespace camd {

class LmdbReader {
 public:
  LmdbReader() = default;
  ~LmdbReader();
  LmdbReader(const LmdbReader&) = delete;

  void open(const std::string& source);  // dir with data.mdb, or the file
  long size() const;                     // record count (main DB entries)
  // record idx in global sorted-key order -> (value bytes, length);
  // the pointer stays valid for the reader's lifetime (mmap)
  std::pair<const uint8_t*, size_t> at(long idx);

 private:
  const uint8_t* page(uint64_t pgno) const;
  void collect_leaves(uint64_t pgno);

  int fd_ = -1;
  const uint8_t* map_ = nullptr;
  size_t map_size_ = 0;
  size_t page_size_ = 4096;
  uint64_t root_ = 0;
  uint64_t entries_ = 0;
  std::vector<uint64_t> leaves_;
};

}  // namespace camd
