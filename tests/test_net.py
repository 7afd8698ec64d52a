"""Wire codec, RPC and DHT tests (loopback, CPU; mirrors the reference's
test_lossless_transport.py and the swarm-bootstrap layer it never unit-tested)."""
import asyncio
import time

import pytest
import torch

from bloombee_amd.net import (RpcClient, RpcError, RpcServer, pack_frame,
                              unpack_frame)
from bloombee_amd.net.dht import Dht
from bloombee_amd.data_structures import (ServerInfo, compute_spans,
                                          declare_active_modules,
                                          get_remote_module_infos, module_uids)


@pytest.mark.parametrize("codec", ["raw", "zlib", "bsplit+zlib"])
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32, torch.int64,
                                   torch.bool])
def test_frame_roundtrip(codec, dtype):
    if dtype == torch.bool:
        t = torch.rand(13, 7) > 0.5
    elif dtype.is_floating_point:
        t = torch.randn(13, 7).to(dtype)
    else:
        t = torch.randint(0, 1000, (13, 7), dtype=dtype)
    meta, tensors = unpack_frame(pack_frame({"x": 1, "s": "ok"}, [t], codec))
    assert meta == {"x": 1, "s": "ok"}
    assert tensors[0].dtype == dtype
    assert torch.equal(tensors[0], t)


def test_bsplit_compresses_smooth_bf16():
    # above the BBAMD_WIRE_MIN_SIZE gate (64 KiB): small tensors ride raw
    t = torch.linspace(0, 1, 65536).to(torch.bfloat16)
    raw = pack_frame({}, [t], "raw")
    comp = pack_frame({}, [t], "bsplit+zlib")
    assert len(comp) < len(raw) * 0.8
    _, (back,) = unpack_frame(comp)
    assert torch.equal(back, t)


def test_rpc_unary_and_stream():
    async def run():
        server = RpcServer()

        async def echo(meta, tensors):
            return {"echo": meta.get("x")}, [t * 2 for t in tensors]

        async def stream_sum(meta, tensors, stream):
            total = torch.zeros(())
            async for m, ts in stream:
                total = total + ts[0].sum()
                await stream.send({"partial": float(total)}, [total])
            await stream.send_end()

        server.register("echo", echo)
        server.register_stream("sum", stream_sum)
        host, port = await server.start()

        client = RpcClient(host, port)
        meta, tensors = await client.call("echo", {"x": 41},
                                          [torch.ones(3)], codec="zlib")
        assert meta["echo"] == 41
        assert torch.equal(tensors[0], torch.full((3,), 2.0))

        with pytest.raises(RpcError):
            await client.call("nope", {})

        s = await client.open_stream("sum")
        total = 0.0
        for i in range(3):
            await s.send({}, [torch.full((4,), float(i))])
            m, ts = await s.recv()
            total = m["partial"]
        await s.send_end()
        assert await s.recv() is None
        assert total == pytest.approx(4.0 * (0 + 1 + 2))

        await client.close()
        await server.stop()

    asyncio.run(run())


def test_dht_store_get_across_nodes():
    boot = Dht()
    n2 = Dht(initial_peers=[boot.endpoint])
    n3 = Dht(initial_peers=[boot.endpoint])
    try:
        exp = time.time() + 30
        n2.store("model.0", "peerA", {"throughput": 5.0}, exp)
        n3.store("model.0", "peerB", {"throughput": 7.0}, exp)
        found = boot.get("model.0")
        assert set(found) == {"peerA", "peerB"}
        # expired records disappear
        n2.store("model.1", "peerA", {}, time.time() - 1)
        assert "peerA" not in n3.get("model.1")
    finally:
        for d in (n3, n2, boot):
            d.shutdown()


def test_declare_and_spans():
    boot = Dht()
    n2 = Dht(initial_peers=[boot.endpoint])
    try:
        uids = module_uids("m", 4)
        exp = time.time() + 30
        declare_active_modules(n2, uids[:2], "p1",
                               ServerInfo(port=1111, throughput=3.0), exp)
        declare_active_modules(n2, uids[2:], "p2",
                               ServerInfo(port=2222, throughput=4.0), exp)
        infos = get_remote_module_infos(boot, uids)
        assert [sorted(i.servers) for i in infos] == [["p1"], ["p1"], ["p2"], ["p2"]]
        spans = compute_spans(infos)
        assert (spans["p1"].start, spans["p1"].end) == (0, 2)
        assert (spans["p2"].start, spans["p2"].end) == (2, 4)
        assert spans["p2"].server_info.port == 2222
    finally:
        n2.shutdown()
        boot.shutdown()


def test_wire_codec_native_python_interop():
    """The C++ wire codec (ops/hip/wire.h) and the pure-Python fallback must
    produce mutually decodable streams (identical format)."""
    import zlib as _zlib

    import torch

    from bloombee_amd.net import tensors as T

    if T._native is None:
        import pytest
        pytest.skip("native codec not built")
    # above the min-size gate so compression actually runs
    t = (torch.randn(330, 170) * 0.1).bfloat16()
    raw = T._raw_bytes(t)

    # native-compress -> python-decompress
    head, payload = T.serialize_tensor(t, codec="bsplit+zlib")
    assert head["codec"] == "bsplit+zlib"
    if True:
        dec = _zlib.decompress(payload)
        n = len(dec) // 2
        hi, lo = dec[:n], dec[n:]
        joined = bytes(b for pair in zip(lo, hi) for b in pair)
        assert joined == raw

    # python-compress -> native-decompress (same header format)
    hi, lo = raw[1::2], raw[0::2]
    py_payload = _zlib.compress(hi + lo, level=1)
    back = T.deserialize_tensor(
        {"dtype": "torch.bfloat16", "shape": list(t.shape),
         "codec": "bsplit+zlib", "nbytes": len(py_payload)}, py_payload)
    assert torch.equal(back, t)

    # scalar edge
    s0 = torch.tensor(1.5, dtype=torch.bfloat16)
    h0, p0 = T.serialize_tensor(s0, codec="bsplit+zlib")
    assert torch.equal(T.deserialize_tensor(h0, p0), s0)


def test_wire_mt_python_fallback_decode():
    """A peer without the native extension must still decode mt streams."""
    import torch

    from bloombee_amd.net import tensors as T

    if T._native is None:
        import pytest
        pytest.skip("native codec not built")
    t = (torch.randn(600, 66) * 0.2).bfloat16()  # > min-size gate
    head, payload = T.serialize_tensor(t, codec="bsplit+zlibmt")
    assert head["codec"] == "bsplit+zlibmt"  # gate must not have bypassed
    saved = T._native
    try:
        T._native = None
        back = T.deserialize_tensor(head, payload)
    finally:
        T._native = saved
    assert torch.equal(back, t)


def test_wire_min_size_gate_sends_small_raw():
    """Tensors below BBAMD_WIRE_MIN_SIZE skip compression entirely (ref
    lossless_transport min-size gate)."""
    from bloombee_amd.net.tensors import serialize_tensor

    t = torch.linspace(0, 1, 512).to(torch.bfloat16)  # 1 KiB << 64 KiB
    head, _ = serialize_tensor(t, codec="bsplit+zlib")
    assert head["codec"] == "raw"


def test_dht_record_expiry():
    """Expired announcements disappear from reads (ref DHT record TTL —
    dead servers age out of routing within 2x update_period)."""
    import time as _time

    from bloombee_amd.data_structures import (ServerInfo,
                                              get_remote_module_infos)
    from bloombee_amd.net.dht import Dht

    boot = Dht()
    try:
        info = ServerInfo(host="h", port=1).to_dict()
        boot.store("m.0", "peerA", info, _time.time() + 0.3)
        boot.store("m.0", "peerB", info, _time.time() + 30)
        out = get_remote_module_infos(boot, ["m.0"])[0]
        assert set(out.servers) == {"peerA", "peerB"}
        _time.sleep(0.4)
        out = get_remote_module_infos(boot, ["m.0"])[0]
        assert set(out.servers) == {"peerB"}
    finally:
        boot.shutdown()
