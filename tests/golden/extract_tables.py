#!/usr/bin/env python3
"""Extract the literal GF(2^8) tables from the reference's vendored
klauspost/reedsolomon v1.11.7 source (galois.go) into a binary fixture.

The fixture pins the oracle: oracle/gf_ref.c GENERATES the same tables from
the field definition (poly 29 / 0x11D, galois.go:25) and the test suite
asserts byte-identity against this file.  This is the strongest pinning the
reference offers for RS parity bytes — CubeFS/klauspost ship no golden parity
vectors, only the literal tables + algorithm.

Source of truth parsed here (read-only, never copied as code):
  /root/reference/vendor/github.com/klauspost/reedsolomon/galois.go
    logTable      [256]byte     (galois.go:28)
    expTable      []byte  (510) (galois.go:70; two 255-element cycles)
    mulTable      [256][256]    (galois.go:81)
    mulTableLow   [256][16]     (galois.go:340)
    mulTableHigh  [256][16]     (galois.go:596)

Output layout (tests/golden/gf_tables.bin, 74,494 bytes):
  [0:256)        logTable
  [256:768)      expTable (510)
  [766:66302)    mulTable row-major
  [66302:70398)  mulTableLow row-major
  [70398:74494)  mulTableHigh row-major

Run (in the build container only; the fixture is committed so the GPU box
never needs /root/reference):
  python3 tests/golden/extract_tables.py
"""
import re
import sys
import hashlib

SRC = "/root/reference/vendor/github.com/klauspost/reedsolomon/galois.go"
OUT = __file__.rsplit("/", 1)[0] + "/gf_tables.bin"


def parse_var(text, name, expect_len):
    # Capture from "var <name>" up to the next top-level declaration.
    m = re.search(r"var %s\b(.*?)(?:\nvar |\nfunc |\n/\*)" % name, text, re.S)
    assert m, name
    body = m.group(1)
    body = body[body.index("{") + 1 :]
    body = re.sub(r"//[^\n]*", "", body)  # strip line comments
    nums = [int(x, 0) for x in re.findall(r"0x[0-9a-fA-F]+|\d+", body)]
    assert len(nums) == expect_len, (name, len(nums))
    assert all(0 <= v < 256 for v in nums), name
    return bytes(nums)


def main():
    with open(SRC) as f:
        text = f.read()
    log_t = parse_var(text, "logTable", 256)
    exp_t = parse_var(text, "expTable", 510)
    mul_t = parse_var(text, "mulTable", 256 * 256)
    mul_lo = parse_var(text, "mulTableLow", 256 * 16)
    mul_hi = parse_var(text, "mulTableHigh", 256 * 16)
    blob = log_t + exp_t + mul_t + mul_lo + mul_hi
    with open(OUT, "wb") as f:
        f.write(blob)
    print("wrote %s (%d bytes) sha256=%s" % (OUT, len(blob), hashlib.sha256(blob).hexdigest()))


if __name__ == "__main__":
    sys.exit(main())
