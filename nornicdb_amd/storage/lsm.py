"""LSM key-value store: the on-disk backbone of the DiskEngine.

This replaces the role BadgerDB plays in the reference
(reference pkg/storage/badger.go:1-35 — LSM with 1-byte key prefixes,
50KB inline cap, online backup via badger_backup.go). It is a clean-room
design, not a Badger port:

  - memtable (dict) + per-memtable commit log with CRC32 records and
    batched fsync; replay is STREAMING (bounded by memtable size, never
    the full history — fixes reference-rebuild round-1's full-file replay).
  - sorted immutable SSTables with 32 KiB blocks, per-block CRC32,
    a sparse first-key index and a bloom filter (10 bits/key) in the
    footer; point reads touch one block, scans stream block by block.
  - size-tiered compaction in a background thread: tables are bucketed
    by log4(size); >=4 tables in a bucket merge into the next bucket.
    Tombstones are dropped only when the merge includes the oldest table.
  - MANIFEST (atomic rename) lists live tables; orphan files from a
    crash mid-compaction are deleted on open.
  - optional at-rest encryption: every block and every log payload is
    sealed with utils.encryption (PBKDF2 key, AEAD with AAD binding the
    file identity), so files on disk are ciphertext end to end
    (reference pkg/nornicdb/db.go:775-808 Badger at-rest encryption).
  - online backup: a consistent merged snapshot streamed to one file,
    restorable with LSMStore.restore() (reference badger_backup.go).

Memory stays bounded by (memtable + block cache + footers); datasets can
exceed RAM, and restart cost is O(active log) not O(history).
"""

from __future__ import annotations

import heapq
import io
import os
import struct
import threading
import zlib
from typing import Dict, Iterator, List, Optional, Tuple

import msgpack

TOMBSTONE = None  # memtable tombstone marker

_LOG_HDR = struct.Struct("<HII")  # magic, payload_len, crc32
_LOG_MAGIC = 0x4C44  # "DL"
_FOOTER = struct.Struct("<IQ")  # footer_len, magic
_SST_MAGIC = 0x4E444253535431  # "NDBSST1"
_BLOCK_TARGET = 32 * 1024
_INDEX_EVERY = 1  # one index entry per block (first key)

# record inside a block: klen u32 | vlen i32 (-1 = tombstone) | key | value
_REC = struct.Struct("<Ii")


def _bloom_build(keys: List[bytes], bits_per_key: int = 10) -> bytes:
    m = max(64, len(keys) * bits_per_key)
    m = (m + 7) & ~7
    arr = bytearray(m // 8)
    for k in keys:
        h1 = zlib.crc32(k) & 0xFFFFFFFF
        h2 = zlib.crc32(k, 0x9747B28C) | 1
        for i in range(6):
            b = (h1 + i * h2) % m
            arr[b >> 3] |= 1 << (b & 7)
    return bytes(arr)


def _bloom_maybe(bloom: bytes, k: bytes) -> bool:
    if not bloom:
        return True
    m = len(bloom) * 8
    h1 = zlib.crc32(k) & 0xFFFFFFFF
    h2 = zlib.crc32(k, 0x9747B28C) | 1
    for i in range(6):
        b = (h1 + i * h2) % m
        if not (bloom[b >> 3] >> (b & 7)) & 1:
            return False
    return True


def _pack_block(records: List[Tuple[bytes, Optional[bytes]]]) -> bytes:
    out = io.BytesIO()
    for k, v in records:
        if v is None:
            out.write(_REC.pack(len(k), -1))
            out.write(k)
        else:
            out.write(_REC.pack(len(k), len(v)))
            out.write(k)
            out.write(v)
    return out.getvalue()


def _unpack_block(data: bytes) -> List[Tuple[bytes, Optional[bytes]]]:
    out = []
    off = 0
    n = len(data)
    while off < n:
        klen, vlen = _REC.unpack_from(data, off)
        off += _REC.size
        k = data[off:off + klen]
        off += klen
        if vlen < 0:
            out.append((k, None))
        else:
            out.append((k, data[off:off + vlen]))
            off += vlen
    return out


class SSTable:
    """Immutable sorted table. Blocks of ~32 KiB, CRC32 per block,
    sparse index (first key, offset, length) + bloom in the footer."""

    def __init__(self, path: str, crypt=None):
        self.path = path
        self.crypt = crypt
        self._f = open(path, "rb")
        self._fd = self._f.fileno()
        size = os.fstat(self._fd).st_size
        if size < _FOOTER.size:
            raise CorruptTable(f"{path}: too small")
        tail = os.pread(self._fd, _FOOTER.size, size - _FOOTER.size)
        flen, magic = _FOOTER.unpack(tail)
        if magic != _SST_MAGIC:
            raise CorruptTable(f"{path}: bad magic")
        raw = os.pread(self._fd, flen, size - _FOOTER.size - flen)
        if self.crypt is not None:
            raw = self.crypt.decrypt(raw, aad=b"footer")
        meta = msgpack.unpackb(raw, raw=False)
        self.index: List[Tuple[bytes, int, int]] = [
            (e[0], e[1], e[2]) for e in meta["index"]]
        self.bloom: bytes = meta["bloom"]
        self.count: int = meta["count"]
        self.min_key: bytes = meta["min"]
        self.max_key: bytes = meta["max"]
        self.size = size
        self._cache_lock = threading.Lock()

    def close(self):
        try:
            self._f.close()
        except OSError:
            pass

    # ---- block access ----
    def _read_block(self, bi: int, cache=None) -> List[Tuple[bytes, Optional[bytes]]]:
        key = (id(self), bi)
        if cache is not None:
            blk = cache.get(key)
            if blk is not None:
                return blk
        _, off, ln = self.index[bi]
        raw = os.pread(self._fd, ln, off)
        stored_crc = struct.unpack_from("<I", raw, 0)[0]
        body = raw[4:]
        if zlib.crc32(body) != stored_crc:
            raise CorruptTable(f"{self.path}: block {bi} CRC mismatch")
        if self.crypt is not None:
            body = self.crypt.decrypt(body, aad=b"blk%d" % bi)
        blk = _unpack_block(body)
        if cache is not None:
            cache.put(key, blk, len(raw))
        return blk

    def _block_for(self, key: bytes) -> int:
        # last block whose first key <= key
        lo, hi = 0, len(self.index) - 1
        if key < self.index[0][0]:
            return -1
        while lo < hi:
            mid = (lo + hi + 1) // 2
            if self.index[mid][0] <= key:
                lo = mid
            else:
                hi = mid - 1
        return lo

    def get(self, key: bytes, cache=None):
        """Returns (found, value_or_None-tombstone)."""
        if key < self.min_key or key > self.max_key:
            return False, None
        if not _bloom_maybe(self.bloom, key):
            return False, None
        bi = self._block_for(key)
        if bi < 0:
            return False, None
        for k, v in self._read_block(bi, cache):
            if k == key:
                return True, v
            if k > key:
                break
        return False, None

    def iter_from(self, start: bytes = b"", cache=None):
        """Yield (key, value) from the first key >= start."""
        if self.index and start > self.max_key:
            return
        bi = self._block_for(start) if start else 0
        if bi < 0:
            bi = 0
        for i in range(bi, len(self.index)):
            for k, v in self._read_block(i, cache):
                if k >= start:
                    yield k, v

    @staticmethod
    def write(path: str, items: Iterator[Tuple[bytes, Optional[bytes]]],
              crypt=None, drop_tombstones: bool = False) -> Optional["SSTable"]:
        """Write sorted (key, value|None) items. Returns the opened table,
        or None if no records were written."""
        tmp = path + ".tmp"
        index = []
        keys = []
        count = 0
        min_key = max_key = None
        with open(tmp, "wb") as f:
            block: List[Tuple[bytes, Optional[bytes]]] = []
            bsz = 0

            def flush_block():
                nonlocal bsz
                if not block:
                    return
                body = _pack_block(block)
                if crypt is not None:
                    body = crypt.encrypt(body, aad=b"blk%d" % len(index))
                off = f.tell()
                f.write(struct.pack("<I", zlib.crc32(body)))
                f.write(body)
                index.append((block[0][0], off, len(body) + 4))
                block.clear()
                bsz = 0

            for k, v in items:
                if v is None and drop_tombstones:
                    continue
                if min_key is None:
                    min_key = k
                max_key = k
                keys.append(k)
                count += 1
                block.append((k, v))
                bsz += len(k) + (len(v) if v is not None else 0) + _REC.size
                if bsz >= _BLOCK_TARGET:
                    flush_block()
            flush_block()
            if count == 0:
                f.close()
                os.remove(tmp)
                return None
            meta = msgpack.packb(
                {"index": index, "bloom": _bloom_build(keys), "count": count,
                 "min": min_key, "max": max_key}, use_bin_type=True)
            if crypt is not None:
                meta = crypt.encrypt(meta, aad=b"footer")
            f.write(meta)
            f.write(_FOOTER.pack(len(meta), _SST_MAGIC))
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, path)
        return SSTable(path, crypt)


class CorruptTable(Exception):
    pass


class _BlockCache:
    """LRU over decoded blocks, bounded by encoded bytes."""

    def __init__(self, max_bytes: int = 32 << 20):
        self._d: Dict = {}
        self._order: List = []
        self._bytes = 0
        self._max = max_bytes
        self._lock = threading.Lock()

    def get(self, key):
        with self._lock:
            v = self._d.get(key)
            return v[0] if v else None

    def put(self, key, blk, nbytes):
        with self._lock:
            if key in self._d:
                return
            self._d[key] = (blk, nbytes)
            self._order.append(key)
            self._bytes += nbytes
            while self._bytes > self._max and self._order:
                old = self._order.pop(0)
                ent = self._d.pop(old, None)
                if ent:
                    self._bytes -= ent[1]


class LSMStore:
    MANIFEST = "MANIFEST"

    def __init__(self, path: str, sync_on_write: bool = False,
                 memtable_bytes: int = 8 << 20, max_tables: int = 12,
                 cache_bytes: int = 32 << 20, crypt=None,
                 compact_interval: float = 2.0,
                 sync_interval: float = 0.05):
        self.path = path
        os.makedirs(path, exist_ok=True)
        self.crypt = crypt
        self._sync_on_write = sync_on_write
        self._memtable_bytes = memtable_bytes
        self._max_tables = max_tables
        self._lock = threading.RLock()
        self._cache = _BlockCache(cache_bytes)
        self._mem: Dict[bytes, Optional[bytes]] = {}
        self._mem_sz = 0
        self._tables: List[SSTable] = []  # newest first
        self._next_file = 0
        self._log = None
        self._log_path = None
        self._log_dirty = False
        self._open()
        self._stop = threading.Event()
        self._compactor = threading.Thread(target=self._compact_loop,
                                           args=(compact_interval,), daemon=True)
        self._compactor.start()
        # batched-fsync loop (reference wal.go:377): commits are pushed to
        # the OS on every batch (flush) and to disk every sync_interval.
        self._syncer = None
        if not sync_on_write and sync_interval > 0:
            self._syncer = threading.Thread(target=self._sync_loop,
                                            args=(sync_interval,), daemon=True)
            self._syncer.start()

    # ------------------------------------------------------------------
    # open / manifest / log replay
    # ------------------------------------------------------------------
    def _manifest_path(self):
        return os.path.join(self.path, self.MANIFEST)

    def _open(self):
        man = {"tables": [], "next_file": 0}
        mp = self._manifest_path()
        if os.path.exists(mp):
            with open(mp, "rb") as f:
                man = msgpack.unpackb(f.read(), raw=False)
        self._next_file = man.get("next_file", 0)
        live = []
        for name in man.get("tables", []):  # newest first
            p = os.path.join(self.path, name)
            if os.path.exists(p):
                live.append(SSTable(p, self.crypt))
        self._tables = live
        # delete orphans from crashed compactions/flushes
        referenced = set(man.get("tables", []))
        for fn in os.listdir(self.path):
            if fn.startswith("sst.") and fn not in referenced:
                try:
                    os.remove(os.path.join(self.path, fn))
                except OSError:
                    pass
            if fn.endswith(".tmp"):
                try:
                    os.remove(os.path.join(self.path, fn))
                except OSError:
                    pass
        # replay any active logs (streaming, oldest first), then reopen log
        logs = sorted(
            (fn for fn in os.listdir(self.path) if fn.startswith("log.")),
            key=lambda fn: int(fn.split(".")[1]))
        for fn in logs:
            self._replay_log(os.path.join(self.path, fn))
        # seal replayed state into a table if substantial, else keep in mem
        if self._mem_sz > self._memtable_bytes:
            self._flush_memtable_locked()
        else:
            # keep ops in a fresh log so they stay durable
            pass
        self._new_log(keep_mem=True)
        for fn in logs:
            try:
                os.remove(os.path.join(self.path, fn))
            except OSError:
                pass

    def _replay_log(self, path: str):
        """Streaming replay: reads records incrementally, never the whole
        file at once. Torn tails and CRC mismatches stop replay."""
        try:
            f = open(path, "rb")
        except OSError:
            return
        with f:
            while True:
                hdr = f.read(_LOG_HDR.size)
                if len(hdr) < _LOG_HDR.size:
                    break
                magic, plen, crc = _LOG_HDR.unpack(hdr)
                if magic != _LOG_MAGIC:
                    break
                payload = f.read(plen)
                if len(payload) < plen or zlib.crc32(payload) != crc:
                    break
                if self.crypt is not None:
                    try:
                        payload = self.crypt.decrypt(payload, aad=b"log")
                    except Exception:
                        break
                puts, dels = msgpack.unpackb(payload, raw=False, use_list=True)
                for k, v in puts:
                    self._mem_put(bytes(k), bytes(v))
                for k in dels:
                    self._mem_put(bytes(k), None)

    def _new_log(self, keep_mem: bool = False):
        if self._log:
            try:
                self._log.close()
            except OSError:
                pass
            if self._log_path and not keep_mem:
                try:
                    os.remove(self._log_path)
                except OSError:
                    pass
        n = self._next_file
        self._next_file += 1
        self._log_path = os.path.join(self.path, f"log.{n}")
        self._log = open(self._log_path, "ab")
        self._log_dirty = False

    def _write_manifest_locked(self):
        man = {"tables": [os.path.basename(t.path) for t in self._tables],
               "next_file": self._next_file}
        tmp = self._manifest_path() + ".tmp2"
        with open(tmp, "wb") as f:
            f.write(msgpack.packb(man, use_bin_type=True))
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, self._manifest_path())

    # ------------------------------------------------------------------
    # writes
    # ------------------------------------------------------------------
    def _mem_put(self, k: bytes, v: Optional[bytes]):
        old = self._mem.get(k, b"")
        self._mem[k] = v
        self._mem_sz += len(k) + (len(v) if v else 0) + 24
        if old != b"" and old is not None:
            self._mem_sz -= len(old)

    def write_batch(self, puts: List[Tuple[bytes, bytes]] = (),
                    dels: List[bytes] = ()):
        """Atomic batch: logged as ONE record, applied to the memtable."""
        payload = msgpack.packb(
            [[(k, v) for k, v in puts], list(dels)], use_bin_type=True)
        if self.crypt is not None:
            payload = self.crypt.encrypt(payload, aad=b"log")
        rec = _LOG_HDR.pack(_LOG_MAGIC, len(payload), zlib.crc32(payload)) + payload
        with self._lock:
            self._log.write(rec)
            self._log_dirty = True
            if self._sync_on_write:
                self._log.flush()
                os.fsync(self._log.fileno())
                self._log_dirty = False
            else:
                # push the record out of the PYTHON buffer on every batch:
                # a killed process must lose at most the fsync window, not
                # whole buffered batches (serve-integration regression)
                self._log.flush()
            for k, v in puts:
                self._mem_put(k, v)
            for k in dels:
                self._mem_put(k, None)
            if self._mem_sz >= self._memtable_bytes:
                self._flush_memtable_locked()

    def put(self, k: bytes, v: bytes):
        self.write_batch([(k, v)])

    def delete(self, k: bytes):
        self.write_batch(dels=[k])

    def _sync_loop(self, interval: float):
        while not self._stop.wait(interval):
            try:
                self.sync()
            except Exception:
                pass

    def sync(self):
        with self._lock:
            if self._log_dirty and self._log and not self._log.closed:
                self._log.flush()
                os.fsync(self._log.fileno())
                self._log_dirty = False

    def _flush_memtable_locked(self):
        if not self._mem:
            return
        n = self._next_file
        self._next_file += 1
        p = os.path.join(self.path, f"sst.{n}")
        items = sorted(self._mem.items())
        t = SSTable.write(p, iter(items), self.crypt)
        if t is not None:
            self._tables.insert(0, t)
        self._mem = {}
        self._mem_sz = 0
        self._write_manifest_locked()
        self._new_log()

    def flush(self):
        with self._lock:
            self._flush_memtable_locked()

    # ------------------------------------------------------------------
    # reads
    # ------------------------------------------------------------------
    def get(self, key: bytes) -> Optional[bytes]:
        with self._lock:
            if key in self._mem:
                return self._mem[key]
            tables = list(self._tables)
        for t in tables:
            found, v = t.get(key, self._cache)
            if found:
                return v  # may be None (tombstone)
        return None

    def scan(self, prefix: bytes = b"") -> Iterator[Tuple[bytes, bytes]]:
        """Merged sorted scan of keys with the given prefix (newest wins,
        tombstones elided). Streams — bounded memory."""
        hi = prefix[:-1] + bytes([prefix[-1] + 1]) if prefix else None
        return self.scan_range(prefix, hi)

    def scan_range(self, lo: bytes, hi: Optional[bytes]) -> Iterator[Tuple[bytes, bytes]]:
        with self._lock:
            mem_items = sorted(
                (k, v) for k, v in self._mem.items()
                if k >= lo and (hi is None or k < hi))
            tables = list(self._tables)

        def ranked(src, rank):
            for k, v in src:
                if hi is not None and k >= hi:
                    return
                yield k, rank, v

        # rank 0 = memtable (newest), then tables newest..oldest; ranks are
        # distinct so ties on k resolve newest-first and never compare v.
        sources = [ranked(iter(mem_items), 0)]
        sources += [ranked(t.iter_from(lo, self._cache), r + 1)
                    for r, t in enumerate(tables)]
        prev = None
        for k, rank, v in heapq.merge(*sources):
            if k == prev:
                continue
            prev = k
            if v is None:
                continue
            yield k, v

    # ------------------------------------------------------------------
    # compaction (size-tiered)
    # ------------------------------------------------------------------
    def _compact_loop(self, interval: float):
        while not self._stop.wait(interval):
            try:
                self.maybe_compact()
            except Exception:
                pass

    def maybe_compact(self):
        with self._lock:
            tables = list(self._tables)
        if len(tables) < 4:
            return False
        # bucket by log4 of size
        import math
        buckets: Dict[int, List[SSTable]] = {}
        for t in tables:
            b = int(math.log(max(t.size, 4096), 4))
            buckets.setdefault(b, []).append(t)
        group = None
        for b in sorted(buckets):
            if len(buckets[b]) >= 4 or len(tables) > self._max_tables:
                group = buckets[b]
                break
        if group is None:
            return False
        self._merge_tables(group)
        return True

    def compact_all(self):
        """Full merge of every table (drops all shadowed data + tombstones)."""
        with self._lock:
            self._flush_memtable_locked()
            tables = list(self._tables)
        if len(tables) <= 1:
            return
        self._merge_tables(tables)

    def _merge_tables(self, group: List[SSTable]):
        """Merge `group` (subset of tables, keeping newest-wins semantics)
        into one new table placed at the position of the group's newest."""
        with self._lock:
            order = {id(t): i for i, t in enumerate(self._tables)}
            group = sorted(group, key=lambda t: order[id(t)])  # newest first
            includes_oldest = order[id(group[-1])] == len(self._tables) - 1
            n = self._next_file
            self._next_file += 1
        p = os.path.join(self.path, f"sst.{n}")

        def ranked(t, rank):
            for k, v in t.iter_from(b"", None):
                yield k, rank, v

        def merged():
            srcs = [ranked(t, r) for r, t in enumerate(group)]
            prev = None
            for k, rank, v in heapq.merge(*srcs):
                if k == prev:
                    continue
                prev = k
                yield k, v

        t_new = SSTable.write(p, merged(), self.crypt,
                              drop_tombstones=includes_oldest)
        with self._lock:
            ids = {id(t) for t in group}
            pos = min(i for i, t in enumerate(self._tables) if id(t) in ids)
            rest = [t for t in self._tables if id(t) not in ids]
            if t_new is not None:
                rest.insert(pos, t_new)
            self._tables = rest
            self._write_manifest_locked()
        for t in group:
            try:
                os.remove(t.path)
            except OSError:
                pass
            # fds stay open for in-flight scans; closed by GC

    # ------------------------------------------------------------------
    # backup / restore (reference badger_backup.go, /admin/backup)
    # ------------------------------------------------------------------
    def backup(self, dest: str):
        """Online, consistent backup: merged live view streamed into a
        single SSTable-format file (restorable; scans keep running)."""
        tmp = dest + ".tmp"
        t = SSTable.write(tmp, self.scan_range(b"", None), self.crypt,
                          drop_tombstones=True)
        if t is None:  # empty store -> write an empty marker file
            with open(tmp, "wb") as f:
                meta = msgpack.packb({"index": [], "bloom": b"", "count": 0,
                                      "min": b"", "max": b""}, use_bin_type=True)
                if self.crypt is not None:
                    meta = self.crypt.encrypt(meta, aad=b"footer")
                f.write(meta)
                f.write(_FOOTER.pack(len(meta), _SST_MAGIC))
        else:
            t.close()
        os.replace(tmp, dest)

    @staticmethod
    def restore(backup_path: str, target_dir: str, crypt=None) -> "LSMStore":
        """Create a fresh store at target_dir seeded from a backup file."""
        os.makedirs(target_dir, exist_ok=True)
        if os.listdir(target_dir):
            raise RuntimeError(f"restore target {target_dir} not empty")
        src = SSTable(backup_path, crypt)
        dst = os.path.join(target_dir, "sst.0")
        if src.count:
            SSTable.write(dst, src.iter_from(b""), crypt)
        src.close()
        man = {"tables": ["sst.0"] if src.count else [], "next_file": 1}
        with open(os.path.join(target_dir, LSMStore.MANIFEST), "wb") as f:
            f.write(msgpack.packb(man, use_bin_type=True))
            f.flush()
            os.fsync(f.fileno())
        return LSMStore(target_dir, crypt=crypt)

    # ------------------------------------------------------------------
    def stats(self) -> dict:
        with self._lock:
            return {
                "memtable_bytes": self._mem_sz,
                "memtable_keys": len(self._mem),
                "tables": len(self._tables),
                "table_bytes": sum(t.size for t in self._tables),
                "table_keys": sum(t.count for t in self._tables),
            }

    def close(self):
        self._stop.set()
        self._compactor.join(timeout=5)
        with self._lock:
            self._flush_memtable_locked()
            if self._log:
                try:
                    self._log.flush()
                    os.fsync(self._log.fileno())
                    self._log.close()
                except (OSError, ValueError):
                    pass
            for t in self._tables:
                t.close()
