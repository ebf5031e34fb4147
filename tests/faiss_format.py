"""faiss_format — independent pure-Python parser of the faiss 1.7.x index
container, used by the tests to validate dg_save_faiss output against the
published byte layout (faiss index_read.cpp semantics, restated; the
reference ships these files between nodes via
src/vector/vector_index_snapshot_manager.cc:583-599).

Test infrastructure only — never imported by the product path.
"""
import struct

import numpy as np


def fourcc(s):
    return struct.unpack("<I", s.encode())[0]


class Reader:
    def __init__(self, data):
        self.b = data
        self.o = 0

    def take(self, n):
        if self.o + n > len(self.b):
            raise ValueError("truncated faiss container")
        v = self.b[self.o:self.o + n]
        self.o += n
        return v

    def u8(self):
        return self.take(1)[0]

    def i32(self):
        return struct.unpack("<i", self.take(4))[0]

    def u32(self):
        return struct.unpack("<I", self.take(4))[0]

    def i64(self):
        return struct.unpack("<q", self.take(8))[0]

    def u64(self):
        return struct.unpack("<Q", self.take(8))[0]

    def vec(self, dtype, count=None):
        n = self.u64() if count is None else count
        a = np.frombuffer(self.take(n * np.dtype(dtype).itemsize), dtype)
        return a


def read_index_header(r):
    h = {"d": r.i32(), "ntotal": r.i64()}
    r.i64()  # dummy
    r.i64()  # dummy
    h["is_trained"] = r.u8()
    h["metric"] = r.i32()
    if h["metric"] > 1:
        h["metric_arg"] = struct.unpack("<f", r.take(4))[0]
    return h


def read_flat(r):
    h4 = r.u32()
    assert h4 in (fourcc("IxF2"), fourcc("IxFI")), hex(h4)
    h = read_index_header(r)
    h["fourcc"] = "IxF2" if h4 == fourcc("IxF2") else "IxFI"
    nfloat = r.u64()  # xb-vector encoding: count in floats
    assert nfloat == h["ntotal"] * h["d"], (nfloat, h)
    h["xb"] = r.vec(np.float32, nfloat).reshape(h["ntotal"], h["d"])
    return h


def read_direct_map(r):
    t = r.u8()
    arr = r.vec(np.int64)
    return {"type": t, "array": arr}


def read_invlists(r):
    assert r.u32() == fourcc("ilar")
    nlist = r.u64()
    code_size = r.u64()
    lt = r.u32()
    sizes = np.zeros(nlist, np.uint64)
    if lt == fourcc("full"):
        sizes = r.vec(np.uint64)
        assert len(sizes) == nlist
    elif lt == fourcc("sprs"):
        pairs = r.vec(np.uint64)
        for j in range(0, len(pairs), 2):
            sizes[int(pairs[j])] = pairs[j + 1]
    else:
        raise ValueError("unknown list format")
    codes, ids = [], []
    for n in sizes:
        n = int(n)
        codes.append(np.frombuffer(r.take(n * code_size), np.uint8)
                     .reshape(n, code_size) if n else
                     np.empty((0, code_size), np.uint8))
        ids.append(r.vec(np.int64, n) if n else np.empty(0, np.int64))
    return {"nlist": nlist, "code_size": code_size, "sizes": sizes,
            "codes": codes, "ids": ids}


def read_index(path_or_bytes):
    """Parse IxM2{IndexFlat} / IwFl / IwPQ containers (the three shapes the
    reference constructs)."""
    data = (path_or_bytes if isinstance(path_or_bytes, (bytes, bytearray))
            else open(path_or_bytes, "rb").read())
    r = Reader(data)
    h4 = r.u32()
    if h4 in (fourcc("IxM2"), fourcc("IxMp")):
        out = {"kind": "idmap2" if h4 == fourcc("IxM2") else "idmap"}
        out["header"] = read_index_header(r)
        out["inner"] = read_flat(r)
        out["id_map"] = r.vec(np.int64)
        assert len(out["id_map"]) == out["inner"]["ntotal"]
    elif h4 in (fourcc("IwFl"), fourcc("IwPQ")):
        out = {"kind": "ivfflat" if h4 == fourcc("IwFl") else "ivfpq"}
        out["header"] = read_index_header(r)
        out["nlist"] = r.u64()
        out["nprobe"] = r.u64()
        out["quantizer"] = read_flat(r)
        out["direct_map"] = read_direct_map(r)
        if out["kind"] == "ivfpq":
            out["by_residual"] = r.u8()
            out["code_size"] = r.u64()
            out["pq"] = {"d": r.u64(), "M": r.u64(), "nbits": r.u64()}
            out["pq"]["centroids"] = r.vec(np.float32)
        out["invlists"] = read_invlists(r)
        assert out["invlists"]["nlist"] == out["nlist"]
    else:
        raise ValueError(f"unknown fourcc {h4:#x}")
    assert r.o == len(data), f"trailing bytes: {len(data) - r.o}"
    return out
