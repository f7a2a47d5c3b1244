"""Multi-GPU exchange: partial aggregation states and
repartitioned rows move as DEVICE byte buffers over torch.distributed
collectives — RCCL over xGMI on an MI355X node (the "nccl" backend IS RCCL
on ROCm), gloo CPU tensors in multi-process CPU tests.

The byte format on the wire is the reference's chunk wire codec
(util/chunk/codec.go:41-141, via gx_chunk_encode/gx_chunk_decode) — i.e. the
same bytes TiDB's exchange operators put on the wire — so the exchange is
the device-buffer analog of ShuffleExec's hash fan-out
(/root/reference/pkg/executor/shuffle.go:459) and the partial/final agg
hand-off (aggregate/agg_hash_partial_worker.go -> final_worker).
"""
import ctypes
import struct

from tidb_amd.chunkpy import PyChunk

_DECLARED = set()


def _decl(lib):
    if id(lib) not in _DECLARED:
        lib.gx_chunk_encode.restype = ctypes.c_int64
        lib.gx_chunk_encode.argtypes = [ctypes.c_void_p,
                                        ctypes.POINTER(ctypes.c_uint8),
                                        ctypes.c_int64]
        lib.gx_chunk_decode.restype = ctypes.c_int64
        lib.gx_chunk_decode.argtypes = [ctypes.POINTER(ctypes.c_uint8),
                                        ctypes.c_int64, ctypes.c_void_p]
        _DECLARED.add(id(lib))
    return lib


def encode_chunk(lib, chunk):
    """Chunk -> wire bytes (codec.go:41-75 Encode)."""
    _decl(lib)
    g = chunk.as_gx()
    need = lib.gx_chunk_encode(ctypes.byref(g), None, 0)
    assert need < 0, "encode size probe failed"
    buf = (ctypes.c_uint8 * -need)()
    n = lib.gx_chunk_encode(ctypes.byref(g), buf, -need)
    assert n == -need
    return bytes(buf)


def decode_chunk(lib, data, types, fracs, max_rows=65536, data_caps=None):
    """Wire bytes -> PyChunk (codec.go:77-141 Decode)."""
    _decl(lib)
    if data_caps is None:
        data_caps = [len(data) if t == 4 else None for t in types]
    chunk = PyChunk(types, max_rows, fracs, data_caps)
    g = chunk.as_gx()
    buf = (ctypes.c_uint8 * len(data)).from_buffer_copy(data)
    n = lib.gx_chunk_decode(buf, len(data), ctypes.byref(g))
    assert n >= 0, "chunk decode failed"
    for c, col in enumerate(chunk.columns):
        col.length = g.cols[c].length
    return chunk, g.n_rows


def rows_to_wire(lib, types, fracs, rows, str_cap=1 << 20):
    """Row tuples (decoded values: int/str/decimal-display/None) -> wire
    bytes. Decimal displays re-encode to the canonical 40-byte struct."""
    from tidb_amd.decimals import str_to_decimal_bytes
    caps = [str_cap if t == 4 else None for t in types]
    ch = PyChunk(types, max(len(rows), 1), fracs, caps)
    for r in rows:
        vals = []
        for v, t in zip(r, types):
            if v is not None and t == 2:  # decimal display -> canonical bytes
                vals.append(str_to_decimal_bytes(lib, v))
            else:
                vals.append(v)
        ch.append_row(vals)
    return encode_chunk(lib, ch)


def wire_to_rows(lib, data, types, fracs):
    if not data:
        return []
    ch, n = decode_chunk(lib, data, types, fracs)
    return ch.rows(n)


def _exchange_device(dist):
    """Collective buffers live on the GPU for the nccl(=RCCL) backend —
    the bytes cross xGMI device-to-device; gloo uses host tensors."""
    import torch
    if dist.get_backend() == "nccl" and torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def all_gather_bytes(dist, payload):
    """All-gather one bytes payload per rank as device tensors (two
    collectives: u64 sizes, then max-padded uint8 buffers)."""
    import torch
    dev = _exchange_device(dist)
    world = dist.get_world_size()
    size = torch.tensor([len(payload)], dtype=torch.int64, device=dev)
    sizes = [torch.zeros(1, dtype=torch.int64, device=dev) for _ in range(world)]
    dist.all_gather(sizes, size)
    sizes = [int(s.item()) for s in sizes]
    cap = max(max(sizes), 1)
    buf = torch.zeros(cap, dtype=torch.uint8, device=dev)
    if payload:
        buf[:len(payload)] = torch.frombuffer(bytearray(payload),
                                              dtype=torch.uint8).to(dev)
    out = [torch.zeros(cap, dtype=torch.uint8, device=dev)
           for _ in range(world)]
    dist.all_gather(out, buf)
    return [bytes(out[r][:sizes[r]].cpu().numpy().tobytes())
            for r in range(world)]


def all_to_all_bytes(dist, parts):
    """parts[j] goes to rank j; returns the payloads received from every
    rank. nccl(=RCCL): all_to_all_single on device buffers (the xGMI
    point-to-point pattern ShuffleExec's fan-out maps to). gloo has no
    all_to_all — emulated with the gather path (correct, more traffic)."""
    import torch
    world = dist.get_world_size()
    rank = dist.get_rank()
    assert len(parts) == world
    dev = _exchange_device(dist)
    if dist.get_backend() == "nccl":
        lens = torch.tensor([len(p) for p in parts], dtype=torch.int64,
                            device=dev)
        rlens = torch.zeros(world, dtype=torch.int64, device=dev)
        dist.all_to_all_single(rlens, lens)
        send = torch.frombuffer(bytearray(b"".join(parts)) or bytearray(1),
                                dtype=torch.uint8).to(dev)
        in_splits = [len(p) for p in parts]
        out_splits = [int(x) for x in rlens.cpu()]
        recv = torch.zeros(max(sum(out_splits), 1), dtype=torch.uint8,
                           device=dev)
        dist.all_to_all_single(recv, send[:max(sum(in_splits), 1)],
                               out_splits, in_splits)
        out = []
        off = 0
        for n in out_splits:
            out.append(bytes(recv[off:off + n].cpu().numpy().tobytes()))
            off += n
        return out
    # gloo emulation: every rank gathers every (src -> dst) payload and
    # keeps the column addressed to it
    packed = struct.pack("<i", len(parts)) + b"".join(
        struct.pack("<q", len(p)) + p for p in parts)
    gathered = all_gather_bytes(dist, packed)
    out = []
    for src in range(world):
        data = gathered[src]
        (n,) = struct.unpack_from("<i", data, 0)
        off = 4
        part = b""
        for j in range(n):
            (ln,) = struct.unpack_from("<q", data, off)
            off += 8
            if j == rank:
                part = data[off:off + ln]
            off += ln
        out.append(part)
    return out


def gather_partial_rows(dist, lib, types, fracs, rows):
    """The PARTIAL -> FINAL hand-off: each rank's partial-state rows travel
    as one wire-encoded chunk in a device buffer; every rank receives all
    shards (all-gather; the KB-scale Q1 payload is latency-bound,
    SURVEY §8e)."""
    payload = rows_to_wire(lib, types, fracs, rows) if rows else b""
    gathered = all_gather_bytes(dist, payload)
    out = []
    for data in gathered:
        out.extend(wire_to_rows(lib, data, types, fracs))
    return out


def repartition_rows(dist, lib, types, fracs, rows, part_fn):
    """ShuffleExec's hash fan-out (shuffle.go:459): route each row to
    part_fn(row) % world over the all-to-all path; returns this rank's
    rows from every sender."""
    world = dist.get_world_size()
    buckets = [[] for _ in range(world)]
    for r in rows:
        buckets[part_fn(r) % world].append(r)
    parts = [rows_to_wire(lib, types, fracs, b) if b else b""
             for b in buckets]
    received = all_to_all_bytes(dist, parts)
    out = []
    for data in received:
        out.extend(wire_to_rows(lib, data, types, fracs))
    return out
