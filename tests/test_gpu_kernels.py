"""Native CDNA4 kernel numerics vs the pure-torch fp32 references
(bagua_amd/ops/quant.py and torch ops). GPU-only."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.float16,
                                   torch.bfloat16])
@pytest.mark.parametrize("n", [1024, 4097, 1 << 20])
def test_elementwise(dtype, n):
    from bagua_amd import _C

    torch.manual_seed(0)
    x = torch.randn(n, device="cuda").to(dtype)
    y = torch.randn(n, device="cuda").to(dtype)

    atol = 1e-6 if dtype == torch.float32 else 2e-2

    a = x.clone()
    _C.average_inplace(a, y)
    ref = ((x.float() + y.float()) / 2).to(dtype)
    assert torch.allclose(a.float(), ref.float(), atol=atol)

    a = x.clone()
    _C.add_inplace(a, y)
    assert torch.allclose(a.float(), (x.float() + y.float()).to(dtype).float(),
                          atol=atol)

    a = x.clone()
    _C.substract_inplace(a, y)
    assert torch.allclose(a.float(), (x.float() - y.float()).to(dtype).float(),
                          atol=atol)

    a = x.clone()
    _C.addmul_inplace(a, y, 0.37)
    assert torch.allclose(a.float(),
                          (x.float() + 0.37 * y.float()).to(dtype).float(),
                          atol=atol)

    a = x.clone()
    _C.divide_inplace(a, 3.0)
    assert torch.allclose(a.float(), (x.float() / 3.0).to(dtype).float(),
                          atol=atol)

    a = x.clone()
    r = torch.randn(n, device="cuda").to(dtype)
    c = torch.randn(n, device="cuda").to(dtype)
    _C.async_model_average(a, r, c, 4.0)
    ref = (x.float() + r.float() / 4.0 - c.float()).to(dtype)
    assert torch.allclose(a.float(), ref.float(), atol=atol * 2)


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.float16,
                                   torch.bfloat16])
@pytest.mark.parametrize("num_chunks,chunk", [(4, 1024), (8, 4099)])
def test_reduce_chunk(dtype, num_chunks, chunk):
    from bagua_amd import _C

    torch.manual_seed(1)
    x = torch.randn(num_chunks * chunk, device="cuda").to(dtype)
    target = 2 % num_chunks

    ref = x.clone()
    v = ref.view(num_chunks, -1)
    red = v.float().mean(0).to(dtype)
    v[target].copy_(red)

    out = x.clone()
    _C.reduce_chunk_inplace(out, num_chunks, target, True)
    atol = 1e-5 if dtype == torch.float32 else 2e-2
    assert torch.allclose(out.float(), ref.float(), atol=atol)


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.float16,
                                   torch.bfloat16])
@pytest.mark.parametrize("num_chunks,chunk", [(1, 4096), (4, 1024),
                                              (8, 32768)])
def test_compress_matches_golden(dtype, num_chunks, chunk):
    """uint8 payload must match the pure-torch golden quantizer bitwise
    (modulo rint ties at .5, which both sides resolve to even)."""
    from bagua_amd import _C
    from bagua_amd.ops import quant

    torch.manual_seed(2)
    x = (torch.randn(num_chunks * chunk, device="cuda") * 3).to(dtype)

    stride = quant.compressed_chunk_bytes(chunk)
    out = torch.zeros(stride * num_chunks, dtype=torch.uint8, device="cuda")
    _C.compress_chunked(x, out, num_chunks, -1)
    torch.cuda.synchronize()

    gold = quant.compress_chunked(x.float(), num_chunks)
    # headers: min/max stored as f32 from the input dtype's values
    for c in range(num_chunks):
        hdr = out[c * stride:c * stride + 8].view(torch.float32)
        ghdr = gold[c * stride:c * stride + 8].view(torch.float32)
        assert torch.allclose(hdr, ghdr, atol=1e-6), "chunk %d header" % c
        payload = out[c * stride + 32:c * stride + 32 + chunk]
        gpayload = gold[c * stride + 32:c * stride + 32 + chunk]
        diff = (payload.int() - gpayload.int()).abs()
        # float->dtype->float rounding can move rint by 1 level for
        # f16/bf16; f32 must be exact
        if dtype == torch.float32:
            assert int(diff.max()) == 0
        else:
            assert int(diff.max()) <= 1

    # roundtrip through native decompress
    y = torch.zeros_like(x)
    _C.decompress_chunked(out, y, num_chunks, -1)
    torch.cuda.synchronize()
    v = x.float().view(num_chunks, -1)
    step = ((v.max(1).values - v.min(1).values) / 255.0).max()
    err = (x.float() - y.float()).abs().max()
    assert err <= step + step * 0.5 + 1e-5


@requires_gpu
def test_single_rank_communicator():
    """1-GPU RCCL communicator: allreduce/broadcast/allgather are
    identity; exercises unique-id exchange and stream discipline."""
    import os

    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29581")

    import bagua_amd
    from bagua_amd.communication import ReduceOp

    torch.cuda.set_device(0)
    bagua_amd.init_process_group()
    comm = bagua_amd.communication._get_default_group() \
        .get_global_communicator()
    comm.ensure_native()
    assert comm.is_native, "native RCCL communicator must be active on GPU"

    x = torch.randn(1024, device="cuda")
    ref = x.clone()
    comm.allreduce_inplace(x, ReduceOp.SUM)
    torch.cuda.synchronize()
    assert torch.allclose(x, ref)

    comm.broadcast(x, 0)
    torch.cuda.synchronize()
    assert torch.allclose(x, ref)


@requires_gpu
def test_c_abi_roundtrip_single_rank():
    """Drive the C ABI exactly as a non-Python host would (ctypes):
    create -> rank/nranks -> allreduce/alltoall/barrier -> destroy.
    World 1, so collectives are identities / local copies, but the calling
    conventions and RCCL/hip plumbing are fully exercised
    (reference surface: bagua-core-c/src/lib.rs:23-347)."""
    import ctypes

    from bagua_amd import _C

    lib = ctypes.CDLL(_C.__file__)
    lib.bagua_comm_create.restype = ctypes.c_void_p
    lib.bagua_comm_create.argtypes = [
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_size_t,
        ctypes.c_char_p, ctypes.c_size_t]

    torch.cuda.set_device(0)
    uid = ctypes.create_string_buffer(256)
    n = lib.bagua_nccl_unique_id(uid, 256)
    assert n > 0
    comm = lib.bagua_comm_create(0, 1, 0, 0, uid.raw, n)
    assert comm, "bagua_comm_create failed"
    c = ctypes.c_void_p(comm)
    assert lib.bagua_comm_rank(c) == 0
    assert lib.bagua_comm_nranks(c) == 1

    x = torch.randn(512, device="cuda")
    ref = x.clone()
    assert lib.bagua_comm_allreduce_inplace(
        c, ctypes.c_void_p(x.data_ptr()), 512, 0, 0, 0) == 0
    y = torch.empty_like(x)
    assert lib.bagua_comm_alltoall(
        c, ctypes.c_void_p(x.data_ptr()), ctypes.c_void_p(y.data_ptr()),
        512, 0, 0) == 0
    one = torch.ones(1, device="cuda")
    assert lib.bagua_comm_barrier(
        c, ctypes.c_void_p(one.data_ptr()), 0) == 0
    torch.cuda.synchronize()
    assert torch.equal(x, ref)
    assert torch.equal(y, ref), "alltoall self-copy mismatch"
    lib.bagua_comm_destroy(c)


@requires_gpu
def test_p2p_alltoall_world1():
    """P2PAlltoAll degenerate path at world 1: allocation, flag barrier
    kernel, self pull. The 8-GPU one-hop exchange reuses exactly this
    machinery with IPC-opened peer pointers (ops/p2p.py)."""
    import os

    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29582")

    import bagua_amd
    from bagua_amd.ops import p2p

    torch.cuda.set_device(0)
    bagua_amd.init_process_group()
    comm = bagua_amd.communication._get_default_group() \
        .get_global_communicator()
    comm.ensure_native()

    impl = p2p.get_for_communicator(comm, 1 << 20)
    x = torch.randn(1 << 18, device="cuda")
    ref = x.clone()
    out = torch.empty_like(x)
    with torch.cuda.stream(comm.stream):
        impl.alltoall(x, out)
        impl.alltoall(out, out)  # second round exercises seq progression
    comm.stream.synchronize()
    assert torch.equal(out, ref)
    # growth path: larger message forces a new epoch/instance
    big = torch.randn(1 << 21, device="cuda")
    impl2 = p2p.get_for_communicator(comm, big.numel() * 4)
    with torch.cuda.stream(comm.stream):
        impl2.alltoall(big, big)
    comm.stream.synchronize()
    assert torch.isfinite(big).all()
    # allgather leg (one-hop pull of each peer's own chunk; identity at
    # world 1)
    g = torch.randn(1 << 18, device="cuda")
    gref = g.clone()
    with torch.cuda.stream(comm.stream):
        impl2.allgather_inplace(g)
    comm.stream.synchronize()
    assert torch.equal(g, gref)


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.float16,
                                   torch.bfloat16])
@pytest.mark.parametrize("num_chunks", [2, 8])
def test_dequant_reduce_matches_unfused(dtype, num_chunks):
    """Fused dequantize+reduce must be BITWISE identical to the
    decompress -> reduce_chunk chain (it rounds through T between
    dequantize and f32 accumulation by construction)."""
    from bagua_amd import ops

    torch.manual_seed(11)
    chunk = 4096
    flat = (torch.randn(num_chunks * chunk, device="cuda") * 3).to(dtype)
    wire = ops.compress_chunked(flat, num_chunks)

    a = flat.clone()
    ops.decompress_chunked_into(wire, a, num_chunks)
    ops.reduce_chunk_inplace(a, num_chunks, 1, True)

    b = flat.clone()
    ops.dequant_reduce(wire, b, num_chunks, 1, True)
    torch.cuda.synchronize()

    av = a.view(num_chunks, -1)[1]
    bv = b.view(num_chunks, -1)[1]
    assert torch.equal(av, bv), (
        "fused dequant_reduce deviates (max diff %g)"
        % (av.float() - bv.float()).abs().max().item())
