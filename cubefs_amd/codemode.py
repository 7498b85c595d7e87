"""cubefs_amd.codemode — CodeMode registry mirroring
blobstore/common/codemode (codemode.go:28-99,156-190,399-441).

The registry is host-side configuration: the same N/M/L/AZCount tactics,
names and numeric codes as the reference, plus Extend() for user-defined
modes >= 240.
"""
from dataclasses import dataclass

ALIGN_0B = 0
ALIGN_512B = 512
ALIGN_2KB = 2048


@dataclass(frozen=True)
class Tactic:
    """codemode.Tactic (codemode.go:156-190)."""
    N: int
    M: int
    L: int = 0
    AZCount: int = 1
    PutQuorum: int = 0
    GetQuorum: int = 0
    MinShardSize: int = ALIGN_2KB

    def is_valid(self):
        # codemode.go:291-299 (EC portion; replicate modes have M == 0)
        if self.is_replicate():
            return (self.N > 0 and self.AZCount > 0 and self.N % self.AZCount == 0
                    and self.PutQuorum > 0 and self.GetQuorum >= 0)
        return (self.N > 0 and self.M > 0 and self.L >= 0 and self.AZCount > 0
                and self.PutQuorum > 0 and self.GetQuorum >= 0
                and self.MinShardSize >= 0
                and self.N % self.AZCount == 0 and self.M % self.AZCount == 0
                and self.L % self.AZCount == 0)

    def is_replicate(self):
        # codemode.go:375-377
        return self.M == 0 and self.L == 0

    @property
    def total(self):
        return self.N + self.M + self.L

    # ---- stripe layout (codemode.go:301-372) ----

    def ec_layout_by_az(self):
        """GetECLayoutByAZ (codemode.go:301-318)."""
        n, m, l = self.N // self.AZCount, self.M // self.AZCount, self.L // self.AZCount
        stripes = []
        for idx in range(self.AZCount):
            stripe = [idx * n + i for i in range(n)]
            stripe += [self.N + idx * m + i for i in range(m)]
            stripe += [self.N + self.M + idx * l + i for i in range(l)]
            stripes.append(stripe)
        return stripes

    def global_stripe(self):
        """GlobalStripe (codemode.go:321-327)."""
        return list(range(self.N + self.M)), self.N, self.M

    def all_local_stripe(self):
        """AllLocalStripe (codemode.go:330-338)."""
        if self.L == 0:
            return [], 0, 0
        n, m, l = self.N // self.AZCount, self.M // self.AZCount, self.L // self.AZCount
        return self.ec_layout_by_az(), n + m, l

    def local_stripe(self, index):
        """LocalStripe (codemode.go:341-358)."""
        if self.L == 0:
            return [], 0, 0
        n, m, l = self.N // self.AZCount, self.M // self.AZCount, self.L // self.AZCount
        if index < self.N:
            az = index // n
        elif index < self.N + self.M:
            az = (index - self.N) // m
        elif index < self.N + self.M + self.L:
            az = (index - self.N - self.M) // l
        else:
            return [], 0, 0
        return self.local_stripe_in_az(az)

    def local_stripe_in_az(self, az_index):
        """LocalStripeInAZ (codemode.go:361-372)."""
        if self.L == 0:
            return [], 0, 0
        n, m, l = self.N // self.AZCount, self.M // self.AZCount, self.L // self.AZCount
        stripes = self.ec_layout_by_az()
        if az_index < 0 or az_index >= len(stripes):
            return [], 0, 0
        return stripes[az_index], n + m, l


# pre-defined modes (codemode.go:28-52,65-94)
_TACTICS = {
    "EC15P12": (1, Tactic(15, 12, 0, 3, 24, 0, ALIGN_2KB)),
    "EC6P6": (2, Tactic(6, 6, 0, 3, 11, 0, ALIGN_2KB)),
    "EC16P20L2": (3, Tactic(16, 20, 2, 2, 34, 0, ALIGN_2KB)),
    "EC6P10L2": (4, Tactic(6, 10, 2, 2, 14, 0, ALIGN_2KB)),
    "EC6P3L3": (5, Tactic(6, 3, 3, 3, 9, 0, ALIGN_2KB)),
    "EC6P6Align0": (6, Tactic(6, 6, 0, 3, 11, 0, ALIGN_0B)),
    "EC6P6Align512": (7, Tactic(6, 6, 0, 3, 11, 0, ALIGN_512B)),
    "EC4P4L2": (8, Tactic(4, 4, 2, 2, 6, 0, ALIGN_2KB)),
    "EC12P4": (9, Tactic(12, 4, 0, 1, 15, 0, ALIGN_2KB)),
    "EC16P4": (10, Tactic(16, 4, 0, 1, 19, 0, ALIGN_2KB)),
    "EC3P3": (11, Tactic(3, 3, 0, 1, 5, 0, ALIGN_2KB)),
    "EC10P4": (12, Tactic(10, 4, 0, 1, 13, 0, ALIGN_2KB)),
    "EC6P3": (13, Tactic(6, 3, 0, 1, 8, 0, ALIGN_2KB)),
    "EC12P9": (14, Tactic(12, 9, 0, 3, 20, 0, ALIGN_2KB)),
    "EC24P8": (15, Tactic(24, 8, 0, 1, 30, 0, ALIGN_2KB)),
    "Replica3": (100, Tactic(3, 0, 0, 3, 3, 0, 0)),
    "Replica3OneAZ": (101, Tactic(3, 0, 0, 1, 3, 0, 0)),
    "EC6P6L9": (200, Tactic(6, 6, 9, 3, 11, 0, ALIGN_2KB)),
    "EC6P8L10": (201, Tactic(6, 8, 10, 2, 13, 0, ALIGN_0B)),
    "Replica4TwoAZ": (202, Tactic(4, 0, 0, 2, 3, 0, 0)),
}

_BY_CODE = {code: (name, t) for name, (code, t) in _TACTICS.items()}

EXTEND_START = 256 - 16  # codemode.go:399


def get_tactic(name_or_code):
    """CodeMode.Tactic / CodeModeName.Tactic (codemode.go:234-241,287)."""
    if isinstance(name_or_code, str):
        if name_or_code not in _TACTICS:
            raise KeyError("codemode: %s is invalid" % name_or_code)
        return _TACTICS[name_or_code][1]
    if name_or_code not in _BY_CODE:
        raise KeyError("Invalid codemode:%d" % name_or_code)
    return _BY_CODE[name_or_code][1]


def get_code(name):
    return _TACTICS[name][0]


def get_name(code):
    return _BY_CODE[code][0]


def is_valid(name_or_code):
    try:
        get_tactic(name_or_code)
        return True
    except KeyError:
        return False


def all_code_modes():
    return sorted(_BY_CODE)


def ec_code_modes():
    """GetECCodeModes (codemode.go:380-388)."""
    return [c for c in all_code_modes() if not _BY_CODE[c][1].is_replicate()]


def extend(code, name, tactic):
    """codemode.Extend (codemode.go:411-441)."""
    if code < EXTEND_START:
        raise ValueError("codemode:%d not in extend [%d-255]" % (code, EXTEND_START))
    if not tactic.is_valid():
        raise ValueError("codemode:%d invalid:%r" % (code, tactic))
    if tactic.PutQuorum < tactic.N:
        raise ValueError("codemode:%d too small put quorum" % code)
    if code in _BY_CODE:
        old_name, old = _BY_CODE[code]
        if old != tactic:
            raise ValueError("codemode:%d code conflicted" % code)
        if old_name != name:
            raise ValueError("codemode:%d name mismatch" % code)
        return
    if name in _TACTICS:
        raise ValueError("codemode:%d name conflicted %s" % (code, name))
    _TACTICS[name] = (code, tactic)
    _BY_CODE[code] = (name, tactic)
