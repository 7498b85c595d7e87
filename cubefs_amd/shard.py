"""cubefs_amd.shard — blobnode on-disk shard image codec over the gfrs
C ABI (core/shard.go:42-111, datafile.go:342-445).

Images produced here are byte-identical to what blobnode's write path
persists: 32 B big-endian header (crc|magic|bid|vuid|size|reserved),
crc32block-framed body, 8 B footer (magic | whole-shard CRC32 of the raw
data).  The footer CRC is derived on-device by GF(2)-combining the frame
header CRCs — the payload crosses HBM exactly once.
"""
import ctypes

from .runtime import GfrsError, Tactic, check, lib

HEADER_SIZE = 32
FOOTER_SIZE = 8
DEFAULT_BLOCK = 64 * 1024


def disk_size(size, block_len=DEFAULT_BLOCK):
    return check(lib().gfrs_shard_disk_size(size, block_len), "disk_size")


class ShardCodec:
    def __init__(self, device=-1):
        t = Tactic(4, 2, 0, 1, 5, 0, 2048)
        self._ctx = lib().gfrs_create(ctypes.byref(t), device)
        if not self._ctx:
            raise GfrsError(-100, lib().gfrs_last_error().decode())
        self._destroy = lib().gfrs_destroy

    def __del__(self):
        # interpreter shutdown can clear module globals before __del__
        # runs, so use the function handle captured at construction
        ctx = getattr(self, "_ctx", None)
        destroy = getattr(self, "_destroy", None)
        if ctx and destroy is not None:
            try:
                destroy(ctx)
            except TypeError:
                pass
            self._ctx = None

    def write_batch(self, dst, src, bids, vuids, block_len=DEFAULT_BLOCK):
        """src: [nshards, size] device tensor; dst: [nshards, disk_size]."""
        ns, n = src.shape
        assert dst.shape[1] >= disk_size(n, block_len)
        ab = (ctypes.c_uint64 * ns)(*bids)
        av = (ctypes.c_uint64 * ns)(*vuids)
        check(lib().gfrs_shard_write_batch(
            self._ctx, dst.data_ptr(), dst.stride(0), src.data_ptr(),
            src.stride(0), n, block_len, ab, av, ns), "shard_write")

    def parse_batch(self, img, size, block_len=DEFAULT_BLOCK):
        """Returns list of dicts {bid, vuid, size, err, bad_block}."""
        ns = img.shape[0]
        meta = (ctypes.c_uint64 * (4 * ns))()
        bad = (ctypes.c_int64 * ns)()
        check(lib().gfrs_shard_parse_batch(
            self._ctx, img.data_ptr(), img.stride(0), size, block_len,
            meta, bad, ns), "shard_parse")
        out = []
        for j in range(ns):
            err = ctypes.c_int64(meta[4 * j + 3]).value
            out.append({"bid": meta[4 * j], "vuid": meta[4 * j + 1],
                        "size": meta[4 * j + 2], "err": err,
                        "bad_block": bad[j]})
        return out

    def synchronize(self):
        check(lib().gfrs_synchronize(self._ctx), "synchronize")
