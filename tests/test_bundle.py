"""TF-V2 bundle codec tests (capability C12).

Round-trip via the native codec, plus an INDEPENDENT pure-Python parse
of the leveldb-table index (magic, block layout, crc32c masking, proto
fields) so writer bugs can't hide behind a matching reader.
"""

import os
import struct

import numpy as np
import pytest
import torch

from rocps.checkpoint.bundle import (read_bundle, write_bundle, list_bundle)
from rocps import _bundle as C

MAGIC = 0xDB4775248B80FB57


def _tensors():
    g = torch.Generator().manual_seed(7)
    return {
        "layer1/kernel": torch.randn(17, 31, generator=g),
        "layer1/bias": torch.randn(31, generator=g),
        "bf16w": torch.randn(8, 9, generator=g).bfloat16(),
        "step": torch.tensor(123, dtype=torch.int64),
        "small_int": torch.arange(10, dtype=torch.int32),
        "scalar": torch.tensor(2.5),
        "halfp": torch.randn(5, generator=g).half(),
        "mask": torch.tensor([True, False, True]),
    }


def test_roundtrip(tmp_path):
    prefix = str(tmp_path / "m.ckpt-1")
    t = _tensors()
    write_bundle(prefix, t)
    assert os.path.exists(prefix + ".index")
    assert os.path.exists(prefix + ".data-00000-of-00001")
    back = read_bundle(prefix)
    assert set(back) == set(t)
    for k in t:
        assert back[k].dtype == t[k].dtype, k
        assert torch.equal(back[k], t[k]), k


def test_large_tensor_multiblock_index(tmp_path):
    """>4KB of index entries forces multiple data blocks in the table."""
    prefix = str(tmp_path / "big.ckpt")
    t = {f"var_{i:04d}/weight": torch.full((4,), float(i)) for i in range(300)}
    write_bundle(prefix, t)
    back = read_bundle(prefix)
    assert len(back) == 300
    for i in range(300):
        assert back[f"var_{i:04d}/weight"][0].item() == float(i)


def test_crc_corruption_detected(tmp_path):
    prefix = str(tmp_path / "c.ckpt")
    write_bundle(prefix, {"w": torch.randn(128)})
    data_path = prefix + ".data-00000-of-00001"
    raw = bytearray(open(data_path, "rb").read())
    raw[10] ^= 0xFF
    open(data_path, "wb").write(bytes(raw))
    r = C.BundleReader(prefix)
    with pytest.raises(RuntimeError, match="crc32c"):
        r.read("w")


def test_index_corruption_detected(tmp_path):
    prefix = str(tmp_path / "c2.ckpt")
    write_bundle(prefix, {"w": torch.randn(4)})
    idx_path = prefix + ".index"
    raw = bytearray(open(idx_path, "rb").read())
    raw[2] ^= 0x01  # inside first data block
    open(idx_path, "wb").write(bytes(raw))
    with pytest.raises(RuntimeError):
        C.BundleReader(prefix)


def test_masked_crc32c_known_vectors():
    # crc32c("123456789") = 0xe3069283 (Castagnoli standard test vector)
    crc = 0xE3069283
    masked = ((crc >> 15 | (crc << 17 & 0xFFFFFFFF)) + 0xA282EAD8) & 0xFFFFFFFF
    assert C.masked_crc32c(np.frombuffer(b"123456789", dtype=np.uint8)) == masked


# ---------------------------------------------------------------------------
# independent structural parse of the index file
# ---------------------------------------------------------------------------

def _get_varint(buf, p):
    shift, out = 0, 0
    while True:
        b = buf[p]
        p += 1
        out |= (b & 0x7F) << shift
        if not b & 0x80:
            return out, p
        shift += 7


def _parse_block(buf):
    n_restarts = struct.unpack("<I", buf[-4:])[0]
    data_end = len(buf) - 4 - 4 * n_restarts
    p, key, out = 0, b"", []
    while p < data_end:
        shared, p = _get_varint(buf, p)
        non_shared, p = _get_varint(buf, p)
        vlen, p = _get_varint(buf, p)
        key = key[:shared] + buf[p:p + non_shared]
        p += non_shared
        out.append((key, buf[p:p + vlen]))
        p += vlen
    return out


def test_index_structure_independent_parse(tmp_path):
    prefix = str(tmp_path / "s.ckpt")
    w = torch.arange(6, dtype=torch.float32).reshape(2, 3)
    write_bundle(prefix, {"a/b": w, "z": torch.tensor(1, dtype=torch.int64)})
    raw = open(prefix + ".index", "rb").read()
    # footer: 48 bytes, magic at the end
    assert struct.unpack("<Q", raw[-8:])[0] == MAGIC
    footer = raw[-48:]
    meta_off, p = _get_varint(footer, 0)
    meta_sz, p = _get_varint(footer, p)
    idx_off, p = _get_varint(footer, p)
    idx_sz, p = _get_varint(footer, p)
    # index block → first data block handle
    idx_entries = _parse_block(raw[idx_off:idx_off + idx_sz])
    assert len(idx_entries) >= 1
    h_off, q = _get_varint(idx_entries[0][1], 0)
    h_sz, q = _get_varint(idx_entries[0][1], q)
    entries = _parse_block(raw[h_off:h_off + h_sz])
    keys = [k.decode() for k, _ in entries]
    assert keys == ["", "a/b", "z"]  # header first, then sorted
    # block trailer crc over contents+type byte (crc32c, not zlib crc32)
    block = raw[h_off:h_off + h_sz]
    typ = raw[h_off + h_sz:h_off + h_sz + 1]
    stored = struct.unpack("<I", raw[h_off + h_sz + 1:h_off + h_sz + 5])[0]
    assert C.masked_crc32c(np.frombuffer(block + typ, dtype=np.uint8)) == stored
    # BundleEntryProto of "a/b": dtype=1 (DT_FLOAT), shape [2,3]
    entry = entries[1][1]
    # field1 varint dtype
    assert entry[0] == (1 << 3) | 0 and entry[1] == 1
    # shape submessage: field 2, containing two Dim submessages (sizes 2,3)
    assert entry[2] == (2 << 3) | 2
    shape_len = entry[3]
    shape = entry[4:4 + shape_len]
    dims = []
    p = 0
    while p < len(shape):
        assert shape[p] == (2 << 3) | 2
        dlen = shape[p + 1]
        dim = shape[p + 2:p + 2 + dlen]
        assert dim[0] == (1 << 3) | 0
        dims.append(dim[1])
        p += 2 + dlen
    assert dims == [2, 3]
    # raw data bytes are the LE float bytes at the recorded offset
    data = open(prefix + ".data-00000-of-00001", "rb").read()
    assert np.frombuffer(data[:24], dtype="<f4").tolist() == [0, 1, 2, 3, 4, 5]


def test_edge_tensors_roundtrip(tmp_path):
    """0-dim scalars, empty tensors, and single-element bf16."""
    prefix = str(tmp_path / "e.ckpt")
    t = {
        "scalar0d": torch.tensor(3.25),
        "empty": torch.empty(0, 4),
        "one_bf16": torch.tensor([1.5]).bfloat16(),
        "int64_scalar": torch.tensor(-7, dtype=torch.int64),
    }
    write_bundle(prefix, t)
    back = read_bundle(prefix)
    for k in t:
        assert back[k].dtype == t[k].dtype, k
        assert back[k].shape == t[k].shape, k
        assert torch.equal(back[k], t[k]), k


def test_empty_bundle(tmp_path):
    prefix = str(tmp_path / "none.ckpt")
    write_bundle(prefix, {})
    assert read_bundle(prefix) == {}


# ---------------------------------------------------------------------------
# property-based fuzz: arbitrary name/shape/dtype sets round-trip
# ---------------------------------------------------------------------------

from hypothesis import given, settings, strategies as st  # noqa: E402

_DTYPES = [torch.float32, torch.float64, torch.float16, torch.bfloat16,
           torch.int32, torch.int64, torch.bool]
_name = st.text(
    alphabet=st.sampled_from("abcdefgh_/0123456789"), min_size=1,
    max_size=24).filter(lambda s: not s.startswith("/") and "//" not in s)
_entry = st.tuples(
    _name,
    st.lists(st.integers(0, 7), min_size=0, max_size=3),  # shape, 0-dim ok
    st.sampled_from(range(len(_DTYPES))),
)


@settings(deadline=None, max_examples=30)
@given(st.lists(_entry, min_size=0, max_size=12,
                unique_by=lambda e: e[0]))
def test_bundle_fuzz_roundtrip(entries):
    import tempfile

    g = torch.Generator().manual_seed(0)
    tensors = {}
    for name, shape, di in entries:
        dt = _DTYPES[di]
        if dt.is_floating_point:
            t = torch.randn(shape, generator=g).to(dt)
        elif dt == torch.bool:
            t = torch.rand(shape, generator=g) > 0.5
        else:
            t = torch.randint(-1000, 1000, shape, generator=g, dtype=dt)
        tensors[name] = t
    with tempfile.TemporaryDirectory() as d:
        prefix = os.path.join(d, "f.ckpt")
        write_bundle(prefix, tensors)
        back = read_bundle(prefix)
    assert set(back) == set(tensors)
    for k, t in tensors.items():
        assert back[k].dtype == t.dtype, k
        assert back[k].shape == t.shape, k
        assert torch.equal(back[k], t), k


def test_sharded_bundle_roundtrip(tmp_path):
    """num_shards > 1 writes .data-SSSSS-of-NNNNN files; every tensor
    reads back from its shard; the header carries the shard count."""
    import os

    from rocps.checkpoint.bundle import read_bundle, write_bundle

    g = torch.Generator().manual_seed(3)
    tensors = {f"v{i}": torch.randn((17 + i, 9), generator=g)
               for i in range(7)}
    tensors["scalar"] = torch.tensor(42, dtype=torch.int64)
    prefix = str(tmp_path / "model.ckpt-5")
    write_bundle(prefix, tensors, num_shards=3)
    files = sorted(os.listdir(tmp_path))
    assert "model.ckpt-5.data-00000-of-00003" in files
    assert "model.ckpt-5.data-00001-of-00003" in files
    assert "model.ckpt-5.data-00002-of-00003" in files
    assert "model.ckpt-5.data-00000-of-00001" not in files
    back = read_bundle(prefix)
    for k, v in tensors.items():
        assert torch.equal(back[k], v), k
    # entries really spread across shards (greedy least-loaded)
    from rocps.checkpoint.bundle import list_bundle
    shards = {e["shard_id"] for e in list_bundle(prefix)}
    assert shards == {0, 1, 2}


def test_large_offset_tensor_roundtrip(tmp_path):
    """Offsets past 2 GiB exercise the int64 offset path (the classic
    failure is an int32 truncation).  Zero-filled tensors keep the
    wall cost at disk-write speed."""
    from rocps.checkpoint.bundle import list_bundle, read_bundle, write_bundle

    big = torch.zeros(550_000_000, dtype=torch.int32)   # 2.2 GB
    tail = torch.arange(7, dtype=torch.int64)
    prefix = str(tmp_path / "big.ckpt")
    write_bundle(prefix, {"a_big": big, "z_tail": tail})
    ents = {e["name"]: e for e in list_bundle(prefix)}
    assert ents["z_tail"]["offset"] == 2_200_000_000  # > 2**31
    back = read_bundle(prefix, names=["z_tail"])
    assert torch.equal(back["z_tail"], tail)
    del big
