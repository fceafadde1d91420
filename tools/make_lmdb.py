"""Pure-python LMDB writer producing caffe-style image databases.

No liblmdb / py-lmdb exists in this environment, so fixtures for the
engine's from-scratch LMDB reader (csrc/lmdb_reader.cpp) are written by
implementing the same published format this writes: 4096-byte pages, two
meta pages (magic 0xBEEFC0DE, version 1), a B+tree whose leaf nodes carry
key -> caffe::Datum wire messages (channels=1, height=2, width=3,
data=4 bytes, label=5), values spilling into overflow pages when they
exceed the in-leaf maximum — the layout liblmdb 0.9.x emits for the
reference's convert_imageset output (keys "%08d_x", sorted).

Usage:
  python tools/make_lmdb.py OUT_DIR N C H W [seed]
Records are deterministic uint8 images keyed by (seed, index) and labels
index % 10 — tests regenerate and independently predict the bytes.
"""
import os
import struct
import sys

PAGE = 4096
P_BRANCH, P_LEAF, P_OVERFLOW = 0x01, 0x02, 0x04
F_BIGDATA = 0x01


def varint(v):
    out = b""
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out += bytes([b | 0x80])
        else:
            return out + bytes([b])


def datum(c, h, w, data, label):
    msg = b""
    msg += b"\x08" + varint(c)      # field 1 varint
    msg += b"\x10" + varint(h)      # field 2
    msg += b"\x18" + varint(w)      # field 3
    msg += b"\x22" + varint(len(data)) + data  # field 4 bytes
    msg += b"\x28" + varint(label)  # field 5
    return msg


def record_bytes(seed, idx, c, h, w):
    # deterministic bytes: splitmix64 keyed by (seed, idx) — mirrored in
    # tests for independent prediction
    n = c * h * w
    out = bytearray(n)
    x = (seed * 0x9E3779B97F4A7C15 + idx) & 0xFFFFFFFFFFFFFFFF

    def nxt(x):
        x = (x + 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
        z = x
        z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
        z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
        return x, z ^ (z >> 31)

    i = 0
    while i < n:
        x, z = nxt(x)
        for b in struct.pack("<Q", z):
            if i < n:
                out[i] = b
                i += 1
    return bytes(out)


class Writer:
    def __init__(self):
        self.pages = {}   # pgno -> bytearray(PAGE)
        self.next_pg = 2  # 0,1 are meta

    def alloc(self, count=1):
        pg = self.next_pg
        self.next_pg += count
        return pg

    def page_hdr(self, pgno, flags, lower=16, upper=PAGE, pages=None):
        b = bytearray(PAGE)
        struct.pack_into("<QHH", b, 0, pgno, 0, flags)
        if pages is not None:
            struct.pack_into("<I", b, 12, pages)
        else:
            struct.pack_into("<HH", b, 12, lower, upper)
        return b

    def write_leaf(self, pgno, items):
        """items: list of (key, value_bytes_or_(ovpgno,size))."""
        b = self.page_hdr(pgno, P_LEAF)
        ptrs = []
        upper = PAGE
        for key, val in items:
            if isinstance(val, tuple):
                ovpg, dsz = val
                node = struct.pack("<HHHH", dsz & 0xFFFF, dsz >> 16,
                                   F_BIGDATA, len(key)) + key + \
                    struct.pack("<Q", ovpg)
            else:
                dsz = len(val)
                node = struct.pack("<HHHH", dsz & 0xFFFF, dsz >> 16, 0,
                                   len(key)) + key + val
            if len(node) & 1:
                node += b"\x00"
            upper -= len(node)
            b[upper:upper + len(node)] = node
            ptrs.append(upper)
        lower = 16 + 2 * len(ptrs)
        assert lower <= upper, "leaf overflow"
        struct.pack_into("<HH", b, 12, lower, upper)
        for i, p in enumerate(ptrs):
            struct.pack_into("<H", b, 16 + 2 * i, p)
        self.pages[pgno] = b

    def write_branch(self, pgno, children):
        """children: list of (first_key, child_pgno); first key may be b''"""
        b = self.page_hdr(pgno, P_BRANCH)
        ptrs = []
        upper = PAGE
        for i, (key, child) in enumerate(children):
            k = b"" if i == 0 else key  # leftmost key is implicit
            node = struct.pack("<HHHH", child & 0xFFFF,
                               (child >> 16) & 0xFFFF,
                               (child >> 32) & 0xFFFF, len(k)) + k
            if len(node) & 1:
                node += b"\x00"
            upper -= len(node)
            b[upper:upper + len(node)] = node
            ptrs.append(upper)
        lower = 16 + 2 * len(ptrs)
        struct.pack_into("<HH", b, 12, lower, upper)
        for i, p in enumerate(ptrs):
            struct.pack_into("<H", b, 16 + 2 * i, p)
        self.pages[pgno] = b

    def write_overflow(self, data):
        npages = (16 + len(data) + PAGE - 1) // PAGE
        pg = self.alloc(npages)
        blob = bytearray(npages * PAGE)
        hdr = self.page_hdr(pg, P_OVERFLOW, pages=npages)
        blob[:16] = hdr[:16]
        blob[16:16 + len(data)] = data
        # store as one multi-page blob under the first pgno
        self.pages[pg] = blob
        return pg, npages

    def finish(self, path, root, entries, depth, branch_pages, leaf_pages,
               overflow_pages):
        last_pg = self.next_pg - 1
        out = bytearray(self.next_pg * PAGE)
        # meta pages (live = larger txnid; mapsize generous)
        for mp, txnid in ((0, 1), (1, 0)):
            hdr = self.page_hdr(mp, 0x08)  # P_META
            out[mp * PAGE:mp * PAGE + 16] = hdr[:16]
            meta = struct.pack(
                "<IIQQ" + "IHHQQQQQ" * 2 + "QQ",
                0xBEEFC0DE, 1, 0, max(1 << 20, self.next_pg * PAGE),
                # free DB (empty)
                0, 0, 0, 0, 0, 0, 0, 0xFFFFFFFFFFFFFFFF,
                # main DB
                0, 0, depth, branch_pages, leaf_pages, overflow_pages,
                entries, root,
                last_pg, txnid)
            out[mp * PAGE + 16:mp * PAGE + 16 + len(meta)] = meta
        for pg, blob in self.pages.items():
            out[pg * PAGE:pg * PAGE + len(blob)] = blob
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, "data.mdb"), "wb") as f:
            f.write(out)


def make_lmdb(path, n, c, h, w, seed=1234):
    w_ = Writer()
    items = []  # (key, payload) in sorted key order
    for i in range(n):
        key = b"%08d" % i
        val = datum(c, h, w, record_bytes(seed, i, c, h, w), i % 10)
        items.append((key, val))
    # leaf capacity: node 8 + klen + dlen (pad to even); spill big values
    leaves = []
    cur, cur_sz = [], 0
    ovpages = 0
    for key, val in items:
        node_sz = 8 + len(key) + len(val)
        if node_sz > PAGE // 2:  # overflow value
            pg, np_ = w_.write_overflow(val)
            ovpages += np_
            entry = (key, (pg, len(val)))
            node_sz = 8 + len(key) + 8
        else:
            entry = (key, val)
        node_sz = (node_sz + 1) & ~1
        if cur and cur_sz + node_sz + 2 > PAGE - 16:
            leaves.append(cur)
            cur, cur_sz = [], 0
        cur.append(entry)
        cur_sz += node_sz + 2
    if cur:
        leaves.append(cur)
    leaf_pgs = [w_.alloc() for _ in leaves]
    for pg, it in zip(leaf_pgs, leaves):
        w_.write_leaf(pg, it)
    if len(leaves) == 1:
        root, depth, nbranch = leaf_pgs[0], 1, 0
    else:
        root = w_.alloc()
        w_.write_branch(root, [(it[0][0], pg)
                               for it, pg in zip(leaves, leaf_pgs)])
        depth, nbranch = 2, 1
    w_.finish(path, root, n, depth, nbranch, len(leaves), ovpages)
    return path


if __name__ == "__main__":
    out = sys.argv[1]
    n, c, h, w = (int(a) for a in sys.argv[2:6])
    seed = int(sys.argv[6]) if len(sys.argv) > 6 else 1234
    make_lmdb(out, n, c, h, w, seed)
    print(f"wrote {out}/data.mdb: {n} records {c}x{h}x{w} seed {seed}")
