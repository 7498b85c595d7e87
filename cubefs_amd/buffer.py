"""cubefs_amd.buffer — ec.Buffer (blobstore/common/ec/buf.go:36-208).

One EC blob's reused contiguous buffer:

    | data | align padding | parity | local |
    |    DataBuf   |
    |       ECDataBuf      |
    |                ECBuf                  |

shard_size = max(ceil(data_size/N), MinShardSize); the padding between
DataSize and ECDataSize is zeroed (buf.go:113).  Backed by one contiguous
torch tensor (device or CPU) so `shards()` views feed the batch kernels
directly; a simple size-classed pool mirrors resourcepool.MemPool
(mempool.go:51-142) to keep HBM allocations reused across blobs.
"""
import threading

from . import codemode
from .runtime import GfrsError


class MemPool:
    """Size-classed buffer pool (resourcepool/mempool.go:51-142): Get
    returns a tensor of the smallest class >= size; Put recycles it."""

    def __init__(self, classes=None, device="cuda"):
        # default classes follow the reference pool's spirit: 2^n sweep
        self.classes = sorted(classes or [1 << n for n in range(12, 31)])
        self.device = device
        self._free = {c: [] for c in self.classes}
        self._lock = threading.Lock()

    def _cls(self, size):
        for c in self.classes:
            if c >= size:
                return c
        raise GfrsError(-7, "no suitable size class for %d" % size)

    def get(self, size):
        import torch
        c = self._cls(size)
        with self._lock:
            if self._free[c]:
                return self._free[c].pop()
        return torch.empty(c, dtype=torch.uint8, device=self.device)

    def put(self, buf):
        n = buf.numel()
        if n in self._free:
            with self._lock:
                self._free[n].append(buf)

    def zero(self, view):
        view.zero_()


class Buffer:
    """ec.Buffer (buf.go:36-133)."""

    def __init__(self, data_size, tactic, pool=None, device="cuda"):
        if isinstance(tactic, (str, int)):
            tactic = codemode.get_tactic(tactic)
        if data_size <= 0:
            raise GfrsError(-7, "short data")
        if tactic.N <= 0:
            raise GfrsError(-1, "invalid code mode")
        self.tactic = tactic
        self.pool = pool
        shard = max(-(-data_size // tactic.N), tactic.MinShardSize)
        self.shard_size = shard
        self.data_size = data_size
        self.ec_data_size = shard * tactic.N
        self.ec_size = shard * tactic.total
        if pool is not None:
            self._buf = pool.get(self.ec_size)
        else:
            import torch
            self._buf = torch.empty(self.ec_size, dtype=torch.uint8,
                                    device=device)
        # zero the padding bytes of the data section (buf.go:113)
        self._buf[self.data_size:self.ec_data_size].zero_()

    @property
    def data_buf(self):
        """DataBuf: the real-data view."""
        return self._buf[:self.data_size]

    @property
    def ec_data_buf(self):
        """ECDataBuf: the view to Split (data + zero padding)."""
        return self._buf[:self.ec_data_size]

    def shards(self):
        """No-copy equal-size shard views over the contiguous buffer —
        exactly what encoder.Split produces on ECDataBuf plus the
        parity/local regions (stream_put.go:120-146)."""
        s = self.shard_size
        return [self._buf[i * s:(i + 1) * s] for i in range(self.tactic.total)]

    def batch_view(self):
        """[1, total, shard_size] view for the batch kernels."""
        return self._buf[:self.ec_size].view(1, self.tactic.total,
                                             self.shard_size)

    def resize(self, data_size):
        """Resize (buf.go:155-181): reuse when capacity suffices."""
        if data_size == self.data_size:
            return
        t = self.tactic
        shard = max(-(-data_size // t.N), t.MinShardSize)
        ec_size = shard * t.total
        if ec_size <= self._buf.numel():
            self.shard_size = shard
            self.data_size = data_size
            self.ec_data_size = shard * t.N
            self.ec_size = ec_size
            self._buf[self.data_size:self.ec_data_size].zero_()
            return
        old = self._buf
        self.__init__(data_size, t, pool=self.pool,
                      device=str(old.device))
        if self.pool is not None:
            self.pool.put(old)

    def release(self):
        """Release (buf.go:185-204): recycle into the pool."""
        if self._buf is not None and self.pool is not None:
            self.pool.put(self._buf)
        self._buf = None
