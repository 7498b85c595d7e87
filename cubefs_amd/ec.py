"""cubefs_amd.ec — the blobstore/common/ec.Encoder surface over the gfrs
C ABI (include/gfrs.h).

Mirrors the ten-method Encoder interface (encoder.go:41-62) with the same
semantics: caller-owned equal-length shards, LRC layering handled inside,
sentinel errors surfaced as GfrsError.  Shards are 1-D uint8 torch tensors
(CUDA tensors run in-place on the GPU; CPU tensors/numpy arrays are staged
through pinned memory by the C library).

The compute path is hand-written HIP — torch supplies memory and streams
only.  If libgfrs.so is missing or no GPU is present, calls raise.
"""
import ctypes
import io

import numpy as np

from . import codemode
from .runtime import GfrsError, MEM_DEVICE, MEM_HOST, Tactic, check, lib


def _tactic_struct(t: codemode.Tactic) -> Tactic:
    return Tactic(t.N, t.M, t.L, t.AZCount, t.PutQuorum, t.GetQuorum,
                  t.MinShardSize)


def _shard_view(s):
    """Returns (pointer:int, nbytes:int, is_device:bool)."""
    if isinstance(s, np.ndarray):
        assert s.dtype == np.uint8 and s.flags["C_CONTIGUOUS"]
        return s.ctypes.data, s.nbytes, False
    # torch tensor
    assert s.dtype.itemsize == 1 and s.is_contiguous()
    return s.data_ptr(), s.numel() * s.element_size(), s.is_cuda


def buffer_sizes(t: codemode.Tactic, data_size: int):
    """ec.GetBufferSizes (buf.go:146-153)."""
    ts = _tactic_struct(t)
    ss, eds, es = (ctypes.c_int64(), ctypes.c_int64(), ctypes.c_int64())
    check(lib().gfrs_buffer_sizes(ctypes.byref(ts), data_size,
                                  ctypes.byref(ss), ctypes.byref(eds),
                                  ctypes.byref(es)), "buffer_sizes")
    return ss.value, eds.value, es.value


class Encoder:
    """ec.Encoder (encoder.go:41-62) on MI355X."""

    def __init__(self, tactic, device=-1, enable_verify=False):
        if isinstance(tactic, (str, int)):
            tactic = codemode.get_tactic(tactic)
        # Replicate tactics (M == 0) are valid per the reference
        # (codemode.Tactic.IsValid, encoder.go:78): Encode/Verify become
        # no-ops with zero parity and Reconstruct can only succeed when
        # nothing is missing — all handled by the C layer (m==0 paths).
        if not tactic.is_valid():
            raise GfrsError(-1, "invalid code mode for EC encoder")
        self.tactic = tactic
        self.enable_verify = enable_verify
        ts = _tactic_struct(tactic)
        self._ctx = lib().gfrs_create(ctypes.byref(ts), device)
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

    # ---- helpers ----

    def _ptrs(self, shards, expect=None):
        expect = expect if expect is not None else self.tactic.total
        if len(shards) != expect:
            raise GfrsError(-6, "want %d shards, got %d" % (expect, len(shards)))
        views = [_shard_view(s) for s in shards]
        ln = views[0][1]
        dev = views[0][2]
        for p, n, d in views:
            if n != ln:
                raise GfrsError(-3, "unequal shard sizes")
            if d != dev:
                raise GfrsError(-6, "mixed host/device shards")
        arr = (ctypes.c_void_p * len(shards))(*[v[0] for v in views])
        return arr, ln, (MEM_DEVICE if dev else MEM_HOST)

    # ---- the Encoder interface ----

    def encode(self, shards):
        """Encode (encoder.go:114, lrcencoder.go:35)."""
        arr, ln, loc = self._ptrs(shards)
        check(lib().gfrs_encode(self._ctx, arr, ln, len(shards), loc), "encode")
        if self.enable_verify and not self.verify(shards):
            raise GfrsError(-5, "verify after encode failed")

    def verify(self, shards):
        """Verify (encoder.go:133, lrcencoder.go:89).  Accepts a full set
        or, for LRC, one local stripe set."""
        t = self.tactic
        expect = len(shards) if (
            t.L > 0 and len(shards) == t.total // t.AZCount) else t.total
        arr, ln, loc = self._ptrs(shards, expect)
        ok = ctypes.c_int(0)
        check(lib().gfrs_verify(self._ctx, arr, ln, len(shards), loc,
                                ctypes.byref(ok)), "verify")
        return bool(ok.value)

    def reconstruct(self, shards, bad_idx):
        """Reconstruct (encoder.go:139, lrcencoder.go:133).  Accepts a full
        set or, for LRC, one local stripe set."""
        self._reconstruct(shards, bad_idx, data_only=0)

    def reconstruct_data(self, shards, bad_idx):
        """ReconstructData (encoder.go:146, lrcencoder.go:190)."""
        self._reconstruct(shards, bad_idx, data_only=1)

    def _reconstruct(self, shards, bad_idx, data_only):
        t = self.tactic
        expect = len(shards) if (
            t.L > 0 and len(shards) == t.total // t.AZCount) else t.total
        arr, ln, loc = self._ptrs(shards, expect)
        bad = (ctypes.c_int32 * max(1, len(bad_idx)))(*bad_idx)
        check(lib().gfrs_reconstruct(self._ctx, arr, ln, len(shards), loc,
                                     bad, len(bad_idx), data_only),
              "reconstruct")

    def split(self, data):
        """Split (reedsolomon.go:1574-1632 + lrcencoder.go:209-228): no-copy
        slicing into ceil(len/N) shards, zero-padded; parity/local shards
        allocated.  `data` is a 1-D uint8 tensor/array; returns a list of
        views over a (possibly re-allocated) contiguous buffer."""
        t = self.tactic
        n = int(data.shape[0]) if hasattr(data, "shape") else len(data)
        if n == 0:
            raise GfrsError(-7, "short data")
        per = (n + t.N - 1) // t.N
        total = per * t.total
        if isinstance(data, np.ndarray):
            buf = np.zeros(total, dtype=np.uint8)
            buf[:n] = data
        else:
            import torch
            buf = torch.zeros(total, dtype=torch.uint8, device=data.device)
            buf[:n] = data
        return [buf[i * per:(i + 1) * per] for i in range(t.total)]

    def get_data_shards(self, shards):
        return shards[:self.tactic.N]

    def get_parity_shards(self, shards):
        return shards[self.tactic.N:self.tactic.N + self.tactic.M]

    def get_local_shards(self, shards):
        if self.tactic.L == 0:
            return []
        return shards[self.tactic.N + self.tactic.M:]

    def get_shards_in_idc(self, shards, idx):
        """GetShardsInIdc (encoder.go:172-179 / lrcencoder.go:244-251)."""
        t = self.tactic
        if t.L > 0:
            locals_, _, _ = t.local_stripe_in_az(idx)
            return [shards[i] for i in locals_]
        n, m = t.N, t.M
        ln, lm = n // t.AZCount, m // t.AZCount
        return list(shards[idx * ln:(idx + 1) * ln]) + \
            list(shards[n + lm * idx:n + lm * (idx + 1)])

    def join(self, dst: io.RawIOBase, shards, out_size: int):
        """Join (reedsolomon.go:1646-1684): concatenates the data shards'
        first out_size bytes into dst."""
        t = self.tactic
        shards = shards[:t.N]
        have = sum(int(s.shape[0]) for s in shards)
        if have < out_size:
            raise GfrsError(-7, "short data")
        written = 0
        for s in shards:
            take = min(out_size - written, int(s.shape[0]))
            if take <= 0:
                break
            chunk = s[:take]
            if not isinstance(chunk, np.ndarray):
                chunk = chunk.cpu().numpy()
            dst.write(chunk.tobytes())
            written += take

    # ---- batched stripes (repair/migrate bulk path) ----
    # base: contiguous device tensor [nstripes, total, shard_len] uint8

    def _base(self, batch):
        assert batch.is_cuda and batch.is_contiguous()
        nstripes, total, ln = batch.shape
        assert total == self.tactic.total
        return batch.data_ptr(), ln, total * ln, nstripes

    def encode_batch(self, batch):
        p, ln, stride, ns = self._base(batch)
        check(lib().gfrs_encode_batch(self._ctx, p, ln, stride, ns),
              "encode_batch")

    def encode_frame_batch(self, framed, batch, block_len=65536):
        """Fused PUT pipeline: parity + crc32block framed images in one
        pass (stream_put.go:146 + datafile.go:342).  framed:
        [nstripes*(n+m), encode_size(shard_len)] device tensor."""
        p, ln, stride, ns = self._base(batch)
        check(lib().gfrs_encode_frame_batch(self._ctx, framed.data_ptr(),
                                            framed.stride(0), p, ln, stride,
                                            ns, block_len),
              "encode_frame_batch")

    def verify_batch(self, batch):
        p, ln, stride, ns = self._base(batch)
        nwords = (ns + 63) // 64
        bm = (ctypes.c_uint64 * nwords)()
        check(lib().gfrs_verify_batch(self._ctx, p, ln, stride, ns, bm),
              "verify_batch")
        import numpy as np
        bmn = np.ctypeslib.as_array(bm)
        bits = np.unpackbits(bmn.view(np.uint8), bitorder="little")[:ns]
        return bits.astype(bool).tolist()

    def reconstruct_batch(self, batch, bad_idx, data_only=False):
        p, ln, stride, ns = self._base(batch)
        bad = (ctypes.c_int32 * max(1, len(bad_idx)))(*bad_idx)
        check(lib().gfrs_reconstruct_batch(self._ctx, p, ln, stride, ns, bad,
                                           len(bad_idx), int(data_only)),
              "reconstruct_batch")

    def encode_idx(self, data_shard, idx, parity):
        """EncodeIdx (reedsolomon.go:631): parity[r] ^= coeff[r][idx]*data.
        Parity must start zeroed; call once per data shard."""
        t = self.tactic
        views = [_shard_view(s) for s in parity]
        arr = (ctypes.c_void_p * len(parity))(*[v[0] for v in views])
        ln = _shard_view(data_shard)[1]
        check(lib().gfrs_encode_idx(self._ctx, data_shard.data_ptr(), idx,
                                    arr, ln, len(parity)), "encode_idx")

    def update_idx(self, old_shard, new_shard, idx, parity):
        """Update (reedsolomon.go:676): patch parity for a replaced data
        shard: parity[r] ^= coeff[r][idx]*(old^new)."""
        views = [_shard_view(s) for s in parity]
        arr = (ctypes.c_void_p * len(parity))(*[v[0] for v in views])
        ln = _shard_view(old_shard)[1]
        check(lib().gfrs_update_idx(self._ctx, old_shard.data_ptr(),
                                    new_shard.data_ptr(), idx, arr, ln,
                                    len(parity)), "update_idx")

    def reconstruct_verify_batch(self, batch, bad_idx):
        """Fused Reconstruct + mandatory Verify (worker_slice_recover.go:
        865-874) in one data pass.  Returns the per-stripe verify-fail
        list."""
        p, ln, stride, ns = self._base(batch)
        bad = (ctypes.c_int32 * max(1, len(bad_idx)))(*bad_idx)
        nwords = (ns + 63) // 64
        bm = (ctypes.c_uint64 * nwords)()
        check(lib().gfrs_reconstruct_verify_batch(self._ctx, p, ln, stride,
                                                  ns, bad, len(bad_idx), bm),
              "reconstruct_verify")
        import numpy as np
        bmn = np.ctypeslib.as_array(bm)
        bits = np.unpackbits(bmn.view(np.uint8), bitorder="little")[:ns]
        return bits.astype(bool).tolist()

    def repair_batch(self, batch, bad_idx, disk_dst, bids, vuids,
                     block_len=65536):
        """Fused repair tasklet (worker_slice_recover.go:804-888 +
        datafile.go:342): reconstruct bad shards, verify every stripe,
        frame each repaired shard into a pwrite-able disk image.
        disk_dst: [nstripes*len(bad_idx), disk_size] device tensor;
        bids/vuids: flat per (stripe, bad) row-major.
        Returns the per-stripe verify-fail list."""
        import numpy as np
        p, ln, stride, ns = self._base(batch)
        nb = len(bad_idx)
        bad = (ctypes.c_int32 * nb)(*bad_idx)
        # large batches: keep the id arrays and fail bitmap vectorized
        abn = np.ascontiguousarray(bids, dtype=np.uint64)
        avn = np.ascontiguousarray(vuids, dtype=np.uint64)
        ab = abn.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64))
        av = avn.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64))
        nwords = (ns + 63) // 64
        bmn = np.zeros(nwords, dtype=np.uint64)
        bm = bmn.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64))
        check(lib().gfrs_repair_batch(self._ctx, p, ln, stride, ns, bad, nb,
                                      disk_dst.data_ptr(), disk_dst.stride(0),
                                      block_len, ab, av, bm), "repair_batch")
        bits = np.unpackbits(bmn.view(np.uint8), bitorder="little")[:ns]
        return bits.astype(bool).tolist()

    def synchronize(self):
        check(lib().gfrs_synchronize(self._ctx), "synchronize")

    def encode_matrix(self):
        t = self.tactic
        out = np.zeros(((t.N + t.M), t.N), dtype=np.uint8)
        check(lib().gfrs_encode_matrix(
            self._ctx, out.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))),
            "encode_matrix")
        return out


def new_encoder(cfg_codemode, device=-1, enable_verify=False):
    """ec.NewEncoder (encoder.go:78)."""
    return Encoder(cfg_codemode, device=device, enable_verify=enable_verify)
